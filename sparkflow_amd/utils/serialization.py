"""Model (de)serialization: the MI355X-native replacement for TF MetaGraph JSON.

The reference's ``graph_utils.build_graph(func)`` (reference
sparkflow/graph_utils.py:6-15) runs a user function inside a fresh
``tf.Graph`` and returns the MetaGraphDef as a JSON string; that string is
the architecture+init-weights artifact carried by the ``tensorflowGraph``
Param and re-parsed on every worker and on the parameter server.

Here the same JSON-string contract carries a **PyTorch module**: a dill dump
of the user's builder function plus the initial ``state_dict`` captured at
build time, so every worker / the PS reconstructs an identically-initialized
module.  The user builder returns an ``nn.Module`` implementing:

    forward(x) -> output tensor
    compute_loss(output, target) -> scalar loss

where ``target`` is the label batch for supervised training and the input
batch itself for unsupervised training (``tfLabel=None`` — reference
examples/autoencoder_example.py semantics).
"""
from __future__ import annotations

import base64
import io
import json
from typing import Callable, List

import numpy as np
import torch

FORMAT_TAG = "sparkflow-amd/module-v1"


def _dumps_func(func: Callable) -> str:
    import dill

    return base64.b64encode(dill.dumps(func, recurse=True)).decode("ascii")


def _loads_func(payload: str) -> Callable:
    import dill

    return dill.loads(base64.b64decode(payload.encode("ascii")))


def _state_to_b64(module: torch.nn.Module) -> str:
    buf = io.BytesIO()
    torch.save(module.state_dict(), buf)
    return base64.b64encode(buf.getvalue()).decode("ascii")


def _state_from_b64(payload: str):
    buf = io.BytesIO(base64.b64decode(payload.encode("ascii")))
    return torch.load(buf, map_location="cpu", weights_only=False)


def module_to_json(func: Callable[[], torch.nn.Module], module: torch.nn.Module) -> str:
    """Serialize builder + captured initial weights to the JSON-string artifact."""
    return json.dumps(
        {
            "format": FORMAT_TAG,
            "builder": _dumps_func(func),
            "state": _state_to_b64(module),
        }
    )


def module_from_json(graph_json: str) -> torch.nn.Module:
    """Rebuild the module described by a ``build_graph`` JSON string.

    The restored module carries the exact weights captured at build time
    (replacing the reference's per-worker ``global_variables_initializer``
    + weight pull; our workers pull from the PS before the first step
    anyway, so determinism of init only matters for the PS).
    """
    spec = json.loads(graph_json)
    if spec.get("format") != FORMAT_TAG:
        raise ValueError(
            "Unrecognized model artifact (expected %s): %r"
            % (FORMAT_TAG, spec.get("format"))
        )
    builder = _loads_func(spec["builder"])
    module = builder()
    if not isinstance(module, torch.nn.Module):
        raise TypeError("model builder must return a torch.nn.Module")
    module.load_state_dict(_state_from_b64(spec["state"]))
    return module


def capture_module(module: torch.nn.Module) -> str:
    """Serialize an already-built module instance (checkpoint import path —
    no separate builder function available)."""
    return module_to_json(lambda: module, module)


# ---------------------------------------------------------------------------
# Flat weight/grad packing.
#
# The reference moves weights as a pickled list of per-variable ndarrays
# (reference HogwildSparkModel.py:22-35).  On MI355X one contiguous fp32
# buffer per message is the right shape for RCCL p2p over xGMI (single
# send/recv per pull/push, no per-tensor latency), so the engine speaks
# flat tensors and converts at the edges.
# ---------------------------------------------------------------------------


def param_list(module: torch.nn.Module) -> List[torch.nn.Parameter]:
    """Deterministic parameter order: registration order, like
    ``tf.trainable_variables()`` order in the reference."""
    return [p for p in module.parameters() if p.requires_grad]


def flat_size(module: torch.nn.Module) -> int:
    return sum(p.numel() for p in param_list(module))


def pack_params(module: torch.nn.Module, out: torch.Tensor = None) -> torch.Tensor:
    params = param_list(module)
    if out is None:
        out = torch.empty(
            sum(p.numel() for p in params),
            dtype=torch.float32,
            device=params[0].device if params else "cpu",
        )
    off = 0
    with torch.no_grad():
        for p in params:
            n = p.numel()
            out[off : off + n].copy_(p.detach().reshape(-1))
            off += n
    return out


def unpack_params(module: torch.nn.Module, flat: torch.Tensor) -> None:
    off = 0
    with torch.no_grad():
        for p in param_list(module):
            n = p.numel()
            p.copy_(flat[off : off + n].reshape(p.shape))
            off += n


def pack_grads(module: torch.nn.Module, out: torch.Tensor = None) -> torch.Tensor:
    params = param_list(module)
    if out is None:
        out = torch.empty(
            sum(p.numel() for p in params),
            dtype=torch.float32,
            device=params[0].device if params else "cpu",
        )
    off = 0
    for p in params:
        n = p.numel()
        if p.grad is None:
            out[off : off + n].zero_()
        else:
            out[off : off + n].copy_(p.grad.detach().reshape(-1))
        off += n
    return out


def flat_to_list(module: torch.nn.Module, flat: torch.Tensor) -> List[np.ndarray]:
    """Split a flat buffer back into per-variable ndarrays (reference shapes)."""
    arrays = []
    off = 0
    flat_cpu = flat.detach().cpu()
    for p in param_list(module):
        n = p.numel()
        arrays.append(flat_cpu[off : off + n].reshape(p.shape).numpy().copy())
        off += n
    return arrays


def list_to_flat(arrays: List[np.ndarray]) -> torch.Tensor:
    return torch.cat(
        [torch.as_tensor(np.asarray(a), dtype=torch.float32).reshape(-1) for a in arrays]
    )
