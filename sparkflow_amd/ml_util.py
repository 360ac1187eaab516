"""Weight / feature / batching utilities — parity with reference sparkflow/ml_util.py.

Checkpoint format compatibility: ``convert_weights_to_json`` /
``convert_json_to_weights`` keep the reference's JSON list-of-nested-lists
model-weights format byte-for-byte (ml_util.py:31-40) — this JSON string is
the model checkpoint carried in the ``modelWeights`` Param.
"""
from __future__ import annotations

import json
from typing import List, Optional

import numpy as np
import torch


def torch_get_weights(module: torch.nn.Module) -> List[np.ndarray]:
    """Trainable weights as ndarrays, registration order (parity:
    tensorflow_get_weights, ml_util.py:9-13)."""
    from sparkflow_amd.utils.serialization import param_list

    return [p.detach().cpu().numpy().copy() for p in param_list(module)]


def torch_set_weights(module: torch.nn.Module, weights: List[np.ndarray]) -> None:
    """Assign a weight list onto a module (parity: tensorflow_set_weights,
    ml_util.py:16-28) — device-side ``copy_`` instead of placeholder+assign."""
    from sparkflow_amd.utils.serialization import param_list

    params = param_list(module)
    if len(params) != len(weights):
        raise ValueError(
            "weight count mismatch: module has %d trainable tensors, got %d"
            % (len(params), len(weights))
        )
    with torch.no_grad():
        for p, w in zip(params, weights):
            p.copy_(torch.as_tensor(np.asarray(w), dtype=p.dtype, device=p.device))


def convert_weights_to_json(weights) -> str:
    weights = [np.asarray(w).tolist() for w in weights]
    return json.dumps(weights)


def convert_json_to_weights(json_weights: str) -> List[np.ndarray]:
    return [np.asarray(x) for x in json.loads(json_weights)]


def calculate_weights(collected_weights):
    """Elementwise mean of N weight sets (parity: ml_util.py:43-51 — the
    vestigial synchronous-averaging helper, kept as a utility)."""
    size = len(collected_weights)
    acc = [np.asarray(w, dtype=np.float64).copy() for w in collected_weights[0]]
    for ws in collected_weights[1:]:
        for i, w in enumerate(ws):
            acc[i] += np.asarray(w, dtype=np.float64)
    return [a / size for a in acc]


def handle_features(data, is_supervised=False):
    """Partition rows -> (features ndarray, labels ndarray|None).

    Parity: ml_util.py:86-101 — int/float labels are wrapped to ``[y]``.
    """
    features = []
    labels = []
    for item in data:
        if is_supervised:
            x, y = item
            labels.append([y] if isinstance(y, (int, float)) else y)
        else:
            x = item
        features.append(np.asarray(x, dtype=np.float32))
    feats = np.asarray(features, dtype=np.float32)
    labs = np.asarray(labels) if is_supervised else None
    return feats, labs


def handle_batch(features: np.ndarray, labels: Optional[np.ndarray],
                 mini_batch_size: int = -1, idx: Optional[int] = None,
                 rng: Optional[np.random.Generator] = None):
    """Select a batch: full-batch, sequential slice, or random minibatch.

    Parity with ml_util.handle_feed_dict (ml_util.py:104-127) including the
    quirk that ``mini_batch_size > n`` is clamped to ``n-1``.
    Returns ``(x, y)`` (y is None when unsupervised).
    """
    n = features.shape[0]
    if mini_batch_size > n:
        mini_batch_size = n - 1

    if mini_batch_size <= 0:
        return features, labels

    if idx is not None:
        sl = slice(idx, idx + mini_batch_size)
        return features[sl], (labels[sl] if labels is not None else None)

    rng = rng or np.random.default_rng()
    pick = rng.choice(n, mini_batch_size, replace=False)
    return features[pick], (labels[pick] if labels is not None else None)


def handle_shuffle(features: np.ndarray, labels: Optional[np.ndarray],
                   rng: Optional[np.random.Generator] = None):
    """Random permutation of the partition (parity: ml_util.py:130-134)."""
    rng = rng or np.random.default_rng()
    order = rng.permutation(features.shape[0])
    return features[order], (labels[order] if labels is not None else None)


def _best_device() -> torch.device:
    return torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")


def predict_batch(module: torch.nn.Module, x: np.ndarray,
                  device: Optional[torch.device] = None,
                  keep_dropout: bool = False) -> np.ndarray:
    """One batched inference forward on the best device.

    ``keep_dropout`` mirrors the reference's ``to_keep_dropout`` flag
    (ml_util.py:70-71) which only selects WHICH constant is fed to the
    dropout placeholder — ``1.0`` read as keep-prob when True, ``0.0``
    read as drop-rate when False.  Dropout is inert at prediction under
    both values, so the flag is accepted for signature parity but never
    activates Dropout modules: everything runs in eval mode."""
    device = device or _best_device()
    module = module.to(device).eval()
    with torch.no_grad():
        out = module(torch.as_tensor(x, dtype=torch.float32, device=device))
    return out.detach().cpu().numpy()


def predict_func(rows, graph_json, prediction, graph_weights, inp,
                 tf_output=None, tf_input=None, tf_dropout=None,
                 to_keep_dropout=False):
    """Per-partition inference for ``SparkAsyncDLModel.transform`` (parity:
    ml_util.predict_func, ml_util.py:54-83).

    Differences by design: the module is rebuilt once per partition and the
    whole partition runs as ONE batched forward on the GPU; dropout is
    inert at prediction for both ``to_keep_dropout`` values, exactly as
    in the reference (ml_util.py:70-71 feeds 1.0-as-keep-prob or
    0.0-as-drop-rate — both disable it).  Scalar outputs become floats,
    vector outputs become ``pyspark.ml.linalg.Vectors.dense`` — identical
    row-shape contract.
    """
    from pyspark.ml.linalg import Vectors
    from pyspark.sql import Row

    from sparkflow_amd.utils.serialization import module_from_json

    rows = [r.asDict() for r in rows]
    if not rows:
        return []

    module = module_from_json(graph_json)
    torch_set_weights(module, convert_json_to_weights(graph_weights))
    x = np.asarray([np.asarray(row[inp], dtype=np.float32) for row in rows])
    pred = predict_batch(module, x, keep_dropout=bool(to_keep_dropout))

    out_rows = []
    for i, row in enumerate(rows):
        p = pred[i]
        if np.ndim(p) == 0 or (hasattr(p, "size") and p.size == 1):
            row[prediction] = float(np.asarray(p).reshape(-1)[0])
        else:
            row[prediction] = Vectors.dense(np.asarray(p, dtype=np.float64))
        out_rows.append(Row(**row))
    return out_rows
