"""The parameter-server core: canonical weights + optimizer applies.

Replaces the reference's Flask-process PS (reference
HogwildSparkModel.py:175-244): the canonical weights are ONE contiguous
fp32 tensor resident in HBM on the PS GPU; each incoming gradient push is
applied by a fused single-launch optimizer kernel
(sparkflow_amd/ops/hip/optim.hip) on the service thread's own HIP stream.

Two concurrency modes, both preserved from the reference:

* **Hogwild** (default, ``acquire_lock=False``): service threads apply
  updates and serve pulls with no synchronization — concurrent racy applies
  are the intended stochasticity (reference HogwildSparkModel.py:104-108).
* **async-with-lock** (``acquire_lock=True``): pulls take the read side,
  applies the write side of a writer-priority RWLock
  (reference HogwildSparkModel.py:212-240).

Error budget parity: failed applies count toward ``max_errors`` (default =
``iters``); past the budget the PS raises "Too many failures during
training" (reference HogwildSparkModel.py:183,231-237 — implemented
correctly here; the reference's ``cont.next()`` is a latent py2-ism).
"""
from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

import torch

from sparkflow_amd.parallel.optimizers import make_optimizer
from sparkflow_amd.parallel.rwlock import RWLock
from sparkflow_amd.utils.serialization import (
    flat_size,
    module_from_json,
    pack_params,
)


@dataclass
class OptimizerSpec:
    """What the API layer's ``build_optimizer`` produces (reference
    tensorflow_async.py:17-42 returns a TF optimizer object; we carry the
    registry name + options so the PS can build the fused GPU optimizer)."""

    name: str = "adam"
    learning_rate: float = 0.001
    options: Optional[Dict] = None

    def build(self, numel: int, device):
        return make_optimizer(self.name, numel, device,
                              learning_rate=self.learning_rate,
                              options=self.options)


@dataclass
class TrainConfig:
    """Worker-loop hyperparameters (reference handle_model signature,
    HogwildSparkModel.py:38-42 + SparkAsyncDL defaults 176-182)."""

    iters: int = 1000
    mini_batch: int = -1
    mini_stochastic_iters: int = -1
    shuffle: bool = True
    verbose: int = 0
    acquire_lock: bool = False
    partition_shuffles: int = 1
    port: int = 5000
    seed: Optional[int] = None
    # Worker-side gradient accumulation (star-contention knob for large
    # worlds): push the MEAN of grad_accum minibatch grads per transport
    # round — cuts PS pull/push traffic by that factor.  1 = reference
    # semantics (push every minibatch).  Applies to remote (RCCL/TCP)
    # transports; the same-GPU full-step graph path has no wire to save.
    grad_accum: int = 1
    extra: Dict = field(default_factory=dict)


class PSCore:
    def __init__(self, graph_json: str, optimizer: OptimizerSpec,
                 device="cpu", acquire_lock: bool = False,
                 max_errors: Optional[int] = None):
        self.device = torch.device(device)
        module = module_from_json(graph_json)
        self.numel = flat_size(module)
        self.weights = pack_params(module).to(self.device)
        self.optimizer = optimizer.build(self.numel, self.device)
        self.acquire_lock = acquire_lock
        self.lock = RWLock()
        self.max_errors = max_errors
        self._err_lock = threading.Lock()
        self.error_count = 0
        self.update_count = 0
        self.pull_count = 0
        # Lock-mode stream-ordered consistency (SPARKFLOW_LOCK_EVENTS=0
        # falls back to device syncs): instead of torch.cuda.synchronize
        # inside every read/write critical section (full host
        # serialization), readers wait on the last apply's EVENT and
        # writers wait on all outstanding readers' events — the RWLock
        # orders the ENQUEUES, the event chain orders the device work,
        # and the host never blocks.
        import os

        self._lock_events = (self.device.type == "cuda"
                             and os.environ.get("SPARKFLOW_LOCK_EVENTS",
                                                "1") != "0")
        self._apply_ev = None
        self._read_evs = []
        self._ev_mu = threading.Lock()

    def _reader_sync_in(self):
        if self._apply_ev is not None:
            torch.cuda.current_stream(self.device).wait_event(self._apply_ev)

    def _reader_sync_out(self):
        ev = torch.cuda.Event()
        ev.record(torch.cuda.current_stream(self.device))
        with self._ev_mu:
            self._read_evs.append(ev)

    def _writer_sync_in(self):
        with self._ev_mu:
            evs, self._read_evs = self._read_evs, []
        s = torch.cuda.current_stream(self.device)
        for ev in evs:
            s.wait_event(ev)

    def _writer_sync_out(self):
        ev = torch.cuda.Event()
        ev.record(torch.cuda.current_stream(self.device))
        self._apply_ev = ev

    # -- pulls ---------------------------------------------------------------

    def read_weights(self) -> torch.Tensor:
        """A snapshot copy of the canonical weights."""
        self.pull_count += 1
        if self.acquire_lock:
            with self.lock.reading():
                if self._lock_events:
                    self._reader_sync_in()
                    w = self.weights.clone()
                    self._reader_sync_out()
                    return w
                w = self.weights.clone()
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                return w
        return self.weights.clone()

    def send_weights(self, send_fn) -> None:
        """Zero-copy pull path: ``send_fn`` transmits straight out of HBM.

        In Hogwild mode the buffer may be concurrently updated by another
        service thread's apply — the torn read is the same intended race as
        the reference's lock-free ``/parameters`` (HogwildSparkModel.py:210-217).
        """
        self.pull_count += 1
        if self.acquire_lock:
            with self.lock.reading():
                if self._lock_events:
                    self._reader_sync_in()
                    send_fn(self.weights)
                    self._reader_sync_out()
                    return
                send_fn(self.weights)
                if self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
        else:
            send_fn(self.weights)

    # -- pushes --------------------------------------------------------------

    def apply_gradients(self, grad_flat: torch.Tensor) -> None:
        try:
            if grad_flat.numel() != self.numel:
                raise ValueError(
                    f"gradient size {grad_flat.numel()} != model size {self.numel}")
            if self.acquire_lock:
                with self.lock.writing():
                    if self._lock_events:
                        self._writer_sync_in()
                        self.optimizer.apply(self.weights, grad_flat)
                        self._writer_sync_out()
                    else:
                        self.optimizer.apply(self.weights, grad_flat)
                        if self.device.type == "cuda":
                            torch.cuda.synchronize(self.device)
            else:
                self.optimizer.apply(self.weights, grad_flat)
            self.update_count += 1
        except Exception:
            with self._err_lock:
                self.error_count += 1
                budget = self.max_errors if self.max_errors is not None else 1
                if self.error_count >= budget:
                    raise Exception("Too many failures during training")

    # -- stats / state -------------------------------------------------------

    def state_dict(self):
        return {
            "weights": self.weights.detach().cpu(),
            "optimizer": {k: v.detach().cpu()
                          for k, v in self.optimizer.state_tensors().items()},
            "t": self.optimizer.step_count(),
            "update_count": self.update_count,
        }

    def save(self, path: str) -> None:
        """Mid-training PS checkpoint (weights + optimizer state + step) —
        an addition over the reference, which persists only final weights."""
        torch.save(self.state_dict(), path)

    def restore(self, path: str) -> None:
        self.load_state_dict(torch.load(path, map_location="cpu",
                                        weights_only=False))

    def load_state_dict(self, state):
        with torch.no_grad():
            self.weights.copy_(state["weights"].to(self.device))
            for k, v in state.get("optimizer", {}).items():
                self.optimizer.state_tensors()[k].copy_(v.to(self.device))
        self.optimizer.set_step_count(state.get("t", 0))
        self.update_count = state.get("update_count", 0)
