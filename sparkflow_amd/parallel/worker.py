"""Worker training loop — parity with reference ``handle_model``
(HogwildSparkModel.py:38-100), GPU-resident.

Per outer iteration: pull the flat weight buffer from the PS, load it into
the module, optionally shuffle, then one of three batching modes:

* stochastic minibatches (``mini_stochastic_iters >= 1``): msi random
  minibatches per epoch, one push each (weights pulled once per epoch) —
  reference HogwildSparkModel.py:62-71;
* sequential minibatches (``mini_batch >= 1``): walk the partition in
  order, RE-PULLING weights before every batch — reference 72-83 (that
  per-batch re-pull is reference behavior, preserved);
* full batch — reference 84-92.

Gradients are packed into one flat fp32 buffer and pushed; push failures
print and continue (reference 68-71).  ``verbose``/``loss_callback`` get a
full-partition loss per outer iteration (reference 94-100).
"""
from __future__ import annotations

import uuid
from typing import Callable, Optional

import numpy as np
import torch

from sparkflow_amd.parallel.ps import TrainConfig
from sparkflow_amd.parallel.transport import WorkerTransport
from sparkflow_amd.utils.serialization import (
    module_from_json,
    pack_params,
    param_list,
)


class Worker:
    def __init__(self, graph_json: str, transport: WorkerTransport,
                 config: TrainConfig, device="cpu",
                 loss_callback: Optional[Callable] = None):
        self.device = torch.device(device)
        self.module = module_from_json(graph_json).to(self.device).train()
        self.transport = transport
        self.config = config
        self.loss_callback = loss_callback
        self.partition_id = uuid.uuid4().hex
        self.rng = np.random.default_rng(config.seed)
        self.steps_done = 0  # pushes performed (for benchmarking)

        # Params live as VIEWS into one flat fp32 buffer: a weight pull is
        # ONE device copy (vs one copy per tensor per step in the
        # reference's set-weights loop, ml_util.py:16-28).  Grads are
        # packed by a single cat into a stable flat buffer after backward.
        self._params = param_list(self.module)
        self.flat_params = pack_params(self.module)
        self.flat_grads = torch.zeros_like(self.flat_params)
        off = 0
        with torch.no_grad():
            for p in self._params:
                n = p.numel()
                p.data = self.flat_params[off:off + n].view(p.shape)
                off += n

    # -- data staging --------------------------------------------------------

    def stage(self, features: np.ndarray, labels: Optional[np.ndarray]):
        """Move the partition to the device once (the reference re-feeds
        host ndarrays every session.run; HBM is 288 GB — keep it resident)."""
        self.x = torch.as_tensor(np.ascontiguousarray(features),
                                 dtype=torch.float32, device=self.device)
        if labels is not None:
            if np.issubdtype(np.asarray(labels).dtype, np.floating):
                self.y = torch.as_tensor(labels, dtype=torch.float32,
                                         device=self.device)
            else:
                self.y = torch.as_tensor(labels, device=self.device)
        else:
            self.y = None

    # -- core step pieces ----------------------------------------------------

    def _pull_and_load(self):
        if hasattr(self.transport, "pull_into"):
            # remote transports recv straight into the param buffer
            self.transport.pull_into(self.flat_params)
            return
        flat = self.transport.pull()
        if flat.data_ptr() != self.flat_params.data_ptr():
            self.flat_params.copy_(flat)

    def _grad_step(self, xb: torch.Tensor, yb: Optional[torch.Tensor]):
        from sparkflow_amd.ops.functional import grad_only

        self.module.zero_grad(set_to_none=True)
        out = self.module(xb)
        target = yb if yb is not None else xb
        # The loss VALUE is never read here (monitoring computes its own
        # loss) — grad_only() lets the loss ops skip the scalar kernels,
        # and a preallocated ones-root skips autograd's fill per step.
        with grad_only():
            loss = self.module.compute_loss(out, target)
        if getattr(self, "_ones_root", None) is None \
                or self._ones_root.shape != loss.shape:
            self._ones_root = torch.ones_like(loss)
        loss.backward(gradient=self._ones_root)
        from sparkflow_amd import ops as _ops
        from sparkflow_amd.ops.functional import _defer_records

        if _defer_records() is not None:
            # deferred-reduce mode: grads stay in the raw split-K
            # workspaces (consumed by the fused reduce+apply launch)
            return None
        if (self.device.type == "cuda" and _ops.available()
                and len(self._params) <= 16):
            _ops.hip.pack_segments([p.grad.reshape(-1)
                                    for p in self._params], self.flat_grads)
        else:
            torch.cat([p.grad.reshape(-1) for p in self._params],
                      out=self.flat_grads)
        return self.flat_grads

    def _push(self, grads: torch.Tensor):
        try:
            self.transport.push(grads)
            self.steps_done += 1
        except Exception:
            print("Timeout error from partition %s" % self.partition_id)

    def _shuffle_indices(self, n: int) -> torch.Tensor:
        return torch.as_tensor(self.rng.permutation(n), device=self.device)

    def _minibatch_indices(self, n: int, mbs: int) -> torch.Tensor:
        return torch.as_tensor(self.rng.choice(n, mbs, replace=False),
                               device=self.device)

    # -- the loop ------------------------------------------------------------

    def run(self, features: np.ndarray, labels: Optional[np.ndarray]) -> None:
        import time

        from sparkflow_amd.utils.metrics import WorkerStats

        self.stats = WorkerStats()
        self.stage(features, labels)
        cfg = self.config
        n = self.x.shape[0]
        mbs = cfg.mini_batch
        if mbs > n:
            mbs = n - 1  # reference clamp (ml_util.py:105-106)

        for it in range(cfg.iters):
            _iter_t0 = time.perf_counter()
            self._pull_and_load()

            if cfg.shuffle:
                order = self._shuffle_indices(n)
                self.x = self.x[order]
                if self.y is not None:
                    self.y = self.y[order]

            if cfg.mini_stochastic_iters >= 1 and mbs >= 1:
                for _ in range(cfg.mini_stochastic_iters):
                    pick = self._minibatch_indices(n, mbs)
                    yb = self.y[pick] if self.y is not None else None
                    self._push(self._grad_step(self.x[pick], yb))
            elif mbs >= 1:
                for r in range(0, n, mbs):
                    self._pull_and_load()  # reference re-pulls per batch
                    xb = self.x[r:r + mbs]
                    yb = self.y[r:r + mbs] if self.y is not None else None
                    self._push(self._grad_step(xb, yb))
            else:
                self._push(self._grad_step(self.x, self.y))

            samples_this_iter = (
                cfg.mini_stochastic_iters * mbs
                if (cfg.mini_stochastic_iters >= 1 and mbs >= 1)
                else n)
            self.stats.record_iter(time.perf_counter() - _iter_t0,
                                   samples_this_iter)

            if cfg.verbose or self.loss_callback:
                with torch.no_grad():
                    out = self.module(self.x)
                    target = self.y if self.y is not None else self.x
                    loss = float(self.module.compute_loss(out, target))
                if cfg.verbose:
                    print("Partition Id: %s, Iteration: %i, Loss: %f"
                          % (self.partition_id, it, loss))
                if cfg.verbose >= 2:
                    print("Partition Id: %s, samples/sec: %.1f"
                          % (self.partition_id,
                             self.stats.samples.rate()))
                if self.loss_callback:
                    self.loss_callback(loss, it, self.partition_id)


    def run_steps(self, k: int, mini_batch: int,
                  use_graph: Optional[bool] = None) -> int:
        """Benchmark-shaped loop: k steps of {pull -> one stochastic
        minibatch grad -> push}.  Returns samples processed.

        On GPU the forward/backward compute is captured once as a hipGraph
        and replayed per step (launch-bound inner loops belong in graphs on
        MI355X); pulls, minibatch selection and pushes stay eager so the
        async-PS semantics (fresh weights per step, racy applies) are
        unchanged.  Falls back to the eager loop if capture fails.
        """
        n = self.x.shape[0]
        mbs = min(mini_batch, n)
        explicit = use_graph is True
        if use_graph is None:
            use_graph = self.device.type == "cuda"
        if use_graph and self.device.type == "cuda":
            try:
                return self._run_steps_graphed(k, mbs)
            except Exception as e:
                if explicit:  # caller demanded the graph path: fail loudly
                    raise
                print(f"[sparkflow_amd] hipGraph capture failed ({e}); "
                      "running eager steps")
        A = max(1, getattr(self.config, "grad_accum", 1))
        done = 0
        while done < k:
            a = min(A, k - done)
            self._pull_and_load()
            acc = None
            for _ in range(a):
                if mbs < n:
                    pick = self._minibatch_indices(n, mbs)
                    xb = self.x[pick]
                    yb = self.y[pick] if self.y is not None else None
                else:
                    xb, yb = self.x, self.y
                g = self._grad_step(xb, yb)
                acc = g if a == 1 else (g.clone() if acc is None
                                        else acc.add_(g))
            self._push(acc if a == 1 else acc.div_(a))
            done += a
        return k * mbs

    _IDX_POOL_CAP = 8192

    def ensure_graph(self, mini_batch: int) -> None:
        """Capture (once) the hipGraph for steps of ``mini_batch`` and fill
        the device-side minibatch index pool.  Idempotent per config."""
        from sparkflow_amd.parallel.transport import LocalTransport

        n = self.x.shape[0]
        mbs = min(mini_batch, n)
        sub = mbs < n
        full_step = (isinstance(self.transport, LocalTransport)
                     and not self.transport.ps.acquire_lock
                     and self.transport.ps.device == self.device)
        key = (mbs, sub, full_step, n, self.x.data_ptr())
        if getattr(self, "_graph_key", None) != key:
            self._capture_graph(key, mbs, sub, full_step)
            if sub:
                self._fill_idx_pool(n, mbs)

    def _draw_pool_picks(self, n: int, mbs: int) -> np.ndarray:
        """CAP sample-without-replacement minibatches (the reference's
        np.random.choice(n, mbs, replace=False), ml_util.py:118).  The
        dedicated rng keeps background refills race-free vs the eager
        paths' self.rng."""
        rng = getattr(self, "_pool_rng", None)
        if rng is None:
            seed = None if self.config.seed is None \
                else self.config.seed + 7919
            rng = self._pool_rng = np.random.default_rng(seed)
        return np.stack([rng.choice(n, mbs, replace=False)
                         for _ in range(self._IDX_POOL_CAP)])

    def _install_pool(self, picks: np.ndarray) -> None:
        g = self._graph_state
        g["idx_pool"].copy_(torch.as_tensor(picks, device=self.device))
        g["ctr"].zero_()
        if g.get("done") is not None:
            g["done"].zero_()  # ticket is 0 at quiescence; re-assert
        self._pool_used = 0
        self._pool_next = None
        # kick the NEXT draw right away: it then has the whole pool
        # lifetime (~CAP steps of GPU work) to finish, so installs are
        # always a join-of-a-finished-thread + one H2D copy
        self._start_pool_draw()

    def _start_pool_draw(self) -> None:
        if getattr(self, "_pool_thread", None) is not None \
                or getattr(self, "_pool_next", None) is not None:
            return
        import threading

        n, mbs = self.x.shape[0], self._graph_state["idx_pool"].shape[1]

        def _draw():
            self._pool_next = self._draw_pool_picks(n, mbs)

        t = threading.Thread(target=_draw, daemon=True)
        t.start()
        self._pool_thread = t

    def _fill_idx_pool(self, n: int, mbs: int) -> None:
        """(Re)draw the device-side pool of minibatch index rows and
        rewind the captured cursor.  The pool is REDRAWN once consumed
        (see maybe_refresh_pool) so long runs don't cycle the same
        _IDX_POOL_CAP batches forever."""
        self._install_pool(self._draw_pool_picks(n, mbs))

    def _has_pool(self) -> bool:
        g = getattr(self, "_graph_state", None)
        return bool(g) and g.get("idx_pool") is not None

    def pool_prefetch(self, k: int) -> None:
        """Ensure a background draw of the next pool is running (it
        normally already is — _install_pool kicks one immediately, so
        the ~50-200 ms host RNG has a whole pool lifetime of GPU work to
        hide behind)."""
        if self._has_pool():
            self._start_pool_draw()

    def pool_would_wrap(self, k: int) -> bool:
        return self._has_pool() and \
            getattr(self, "_pool_used", 0) + k > self._IDX_POOL_CAP

    def pool_install_now(self) -> None:
        """Join the background draw and install it (one H2D copy + ctr
        rewind).  The caller must have drained in-flight replays of THIS
        worker first (a device sync covers all workers at once)."""
        t = getattr(self, "_pool_thread", None)
        if t is not None:
            t.join()
            self._pool_thread = None
        picks = getattr(self, "_pool_next", None)
        if picks is None:  # fallback: synchronous draw
            g = self._graph_state
            picks = self._draw_pool_picks(self.x.shape[0],
                                          g["idx_pool"].shape[1])
        self._install_pool(picks)

    def pool_account(self, k: int) -> None:
        if self._has_pool():
            self._pool_used = getattr(self, "_pool_used", 0) + k

    def pool_reset(self) -> None:
        """Install a fresh pool NOW (used by benchmarks right before the
        timed region so no wrap lands inside a short measurement)."""
        if not self._has_pool():
            return
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self.pool_install_now()
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)

    def maybe_refresh_pool(self, k: int) -> None:
        """Single-worker wrapper: prefetch at half consumption, install
        on wrap (with a device drain), account k replays.  Callers must
        invoke this while no replay is being enqueued concurrently (all
        run paths do)."""
        if not self._has_pool():
            return
        self.pool_prefetch(k)
        if self.pool_would_wrap(k):
            if self.device.type == "cuda":
                torch.cuda.synchronize(self.device)
            self.pool_install_now()
            if self.device.type == "cuda":
                torch.cuda.synchronize(self.device)
        self.pool_account(k)

    def replay_once(self) -> None:
        """Enqueue ONE training step on the current HIP stream (graph must
        be captured via ensure_graph).  Full-step mode: the replay IS the
        whole pull->gather->fwd/bwd->apply step.  Remote mode: eager pull,
        compute replay, eager push."""
        g = self._graph_state
        if g["full_step"]:
            g["graph"].replay()
            return
        n = self.x.shape[0]
        self._pull_and_load()
        if g["xb"] is not self.x and not g["gather_inside"]:
            pick = self._minibatch_indices(n, g["xb"].shape[0])
            torch.index_select(self.x, 0, pick, out=g["xb"])
            if g["yb"] is not None:
                torch.index_select(self.y, 0, pick, out=g["yb"])
        g["graph"].replay()
        self._push(self.flat_grads)

    def finish_steps(self, k: int) -> None:
        """Account k replayed full steps (host-side counters)."""
        g = self._graph_state
        if g["full_step"]:
            ps = self.transport.ps
            ps.update_count += k
            ps.pull_count += k
            self.steps_done += k

    def _run_steps_graphed(self, k: int, mbs: int) -> int:
        """One hipGraph replay per step.

        For a same-GPU Hogwild LocalTransport (rank 0 / single-GPU jobs)
        the ENTIRE step is captured — weight-pull copy, minibatch gather
        (indices come from a device-side pool walked by a captured step
        counter, preserving sample-without-replacement semantics), forward
        /backward, and the PS's fused optimizer apply (device-state Adam)
        — so a training step is a single graph launch.  For remote
        (RCCL/TCP) transports only the compute is captured; pull/push stay
        eager.
        """
        self.ensure_graph(mbs)
        g = self._graph_state
        A = max(1, getattr(self.config, "grad_accum", 1))
        if A > 1 and not g["full_step"]:
            self.maybe_refresh_pool(k)
            return self._run_steps_graphed_accum(k, mbs, A)
        if not g["full_step"] and hasattr(self.transport, "run_session") \
                and (g["gather_inside"] or g["xb"] is self.x) \
                and getattr(self.transport, "native_comm", None) is not None:
            # whole session in one GIL-free C++ call over the raw comm
            self.maybe_refresh_pool(k)
            try:
                exec_h = g["graph"].raw_cuda_graph_exec()
            except Exception:
                exec_h = None
            if exec_h is not None and self.transport.run_session(
                    k, self.flat_params, self.flat_grads, exec_h):
                self.steps_done += k
                return k * mbs
        if not g["full_step"] and hasattr(self.transport, "run_begin"):
            self.transport.run_begin(k)  # zero per-step control traffic
        # replay in half-pool chunks so the background pool draw always
        # has a chunk of GPU time to overlap with (maybe_refresh_pool)
        chunk = max(1, self._IDX_POOL_CAP // 2)
        done = 0
        while done < k:
            c = min(chunk, k - done)
            self.maybe_refresh_pool(c)
            for _ in range(c):
                self.replay_once()
            done += c
        self.finish_steps(k)
        return k * mbs

    def _run_steps_graphed_accum(self, k: int, mbs: int, A: int) -> int:
        """Remote-transport steps with gradient accumulation: each
        transport round is {pull weights; A x (gather+fwd/bwd replay,
        accumulate); push mean grad} — k compute steps become
        ceil(k/A) PS pulls+pushes (star-contention knob; see
        TrainConfig.grad_accum)."""
        g = self._graph_state
        n = self.x.shape[0]
        rounds = (k + A - 1) // A
        if hasattr(self.transport, "run_begin"):
            self.transport.run_begin(rounds)
        acc = torch.zeros_like(self.flat_grads)
        done = 0
        for _ in range(rounds):
            a = min(A, k - done)
            self._pull_and_load()
            acc.zero_()
            for _ in range(a):
                if g["xb"] is not self.x and not g["gather_inside"]:
                    pick = self._minibatch_indices(n, g["xb"].shape[0])
                    torch.index_select(self.x, 0, pick, out=g["xb"])
                    if g["yb"] is not None:
                        torch.index_select(self.y, 0, pick, out=g["yb"])
                g["graph"].replay()
                acc.add_(self.flat_grads)
            self._push(acc.div_(a) if a > 1 else acc)
            done += a
        self.steps_done += k
        return k * mbs

    def _capture_graph(self, key, mbs: int, sub: bool, full_step: bool):
        self._graph_ws = None
        n = self.x.shape[0]
        xb = torch.empty((mbs,) + tuple(self.x.shape[1:]),
                         dtype=self.x.dtype, device=self.device) \
            if sub else self.x
        yb = None
        if self.y is not None:
            yb = torch.empty((mbs,) + tuple(self.y.shape[1:]),
                             dtype=self.y.dtype, device=self.device) \
                if sub else self.y
        idx_pool = None
        ctr = None
        done = None
        if sub:
            idx_pool = torch.zeros(self._IDX_POOL_CAP, mbs,
                                   dtype=torch.int64, device=self.device)
            ctr = torch.zeros(1, dtype=torch.int64, device=self.device)
            done = torch.zeros(1, dtype=torch.int32, device=self.device)
            pick0 = self._minibatch_indices(n, mbs)
            torch.index_select(self.x, 0, pick0, out=xb)
            if yb is not None:
                torch.index_select(self.y, 0, pick0, out=yb)

        ps = self.transport.ps if full_step else None
        from sparkflow_amd import ops as _ops

        import os

        fused_gather = (sub and _ops.available()
                        and self.device.type == "cuda"
                        and (yb is None or yb.dtype == torch.float32)
                        and os.environ.get("SPARKFLOW_FUSED_GATHER",
                                           "1") != "0")

        # the device-pool gather is captured in the graph whenever the
        # fused kernel applies (full-step AND remote compute graphs);
        # the non-fused torch-op sequence is only captured full-step
        # (the remote fallback gathers eagerly in replay_once)
        gather_inside = sub and (fused_gather or full_step)

        # Fused reduce+apply: backward leaves each layer's wgrad in its
        # raw split-K workspace and ONE launch sums them all and runs the
        # device-state Adam on the PS weights — the 4 reduce launches,
        # pack_segments, and the adam launch collapse into it
        # (numerically identical; optim.hip reduce_apply_adam).  Probed
        # on the first warmup call: a model with non-Linear params (conv)
        # or a bias-less layer fails the coverage check and falls back.
        fused_apply = (full_step and _ops.available()
                       and self.device.type == "cuda"
                       and hasattr(ps.optimizer, "apply_from_workspaces")
                       and getattr(ps.optimizer, "dev_state", None)
                       is not None
                       and os.environ.get("SPARKFLOW_FUSED_APPLY",
                                          "1") != "0")

        def _fused_tail(xb_, yb_):
            from sparkflow_amd.ops.functional import defer_wgrad

            recs = []
            with defer_wgrad(recs):
                self._grad_step(xb_, yb_)
            recs.sort(key=lambda r: r[0])
            base = self.flat_params.data_ptr()
            offs = [(ptr - base) // 4 for ptr, _ in recs]
            ws_list = [w for _, w in recs]
            ps.optimizer.apply_from_workspaces(ps.weights, ws_list, offs)
            return ws_list

        def body():
            if full_step and not fused_gather:
                self.flat_params.copy_(ps.weights)  # the weight pull
            if gather_inside:
                if fused_gather:
                    # full-step mode: the weight pull rides the gather
                    # launch (one kernel stages the whole step's inputs)
                    _ops.hip.gather_batch(self.x, xb, idx_pool, ctr, done,
                                          self.y if yb is not None else None,
                                          yb,
                                          ps.weights if full_step else None,
                                          self.flat_params if full_step
                                          else None)
                else:
                    row = idx_pool.index_select(0, ctr).view(-1)
                    torch.index_select(self.x, 0, row, out=xb)
                    if yb is not None:
                        torch.index_select(self.y, 0, row, out=yb)
                    ctr.add_(1).remainder_(self._IDX_POOL_CAP)
            if fused_apply:
                self._graph_ws = _fused_tail(xb, yb)
                return
            self._grad_step(xb, yb)
            if full_step:
                ps.optimizer.apply(ps.weights, self.flat_grads)

        # allocator warmup on a side stream, then capture
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for i in range(3):
                # first warmup call doubles as the fused-apply probe so
                # the warmup step count (and with it the minibatch
                # cursor / Adam t) is identical on both paths
                try:
                    body()
                except RuntimeError:
                    if not (fused_apply and i == 0):
                        raise
                    fused_apply = False
                    body()
        torch.cuda.current_stream().wait_stream(s)

        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            body()

        # `done` (the gather's done-ticket word) MUST stay referenced:
        # the captured kernel keeps using its address forever, and if
        # the tensor were garbage-collected the allocator would hand the
        # block to ANY later allocation (an eval forward, a checkpoint
        # copy), whose writes then corrupt the ticket and silently
        # freeze the minibatch cursor — the round-2 time-to-accuracy
        # plateau bug (every replay re-training the same minibatch).
        self._graph_state = {"graph": graph, "xb": xb, "yb": yb,
                             "idx_pool": idx_pool, "ctr": ctr,
                             "done": done,
                             "full_step": full_step,
                             "workspaces": getattr(self, "_graph_ws", None),
                             "gather_inside": gather_inside}
        self._graph_key = key


def run_partition(data_iter, graph_json: str, transport: WorkerTransport,
                  config: TrainConfig, is_supervised: bool, device="cpu",
                  loss_callback=None):
    """Entry used per partition (Spark executor closure / local worker)."""
    from sparkflow_amd.ml_util import handle_features

    data = list(data_iter)
    if not data:
        return
    features, labels = handle_features(data, is_supervised)
    w = Worker(graph_json, transport, config, device=device,
               loss_callback=loss_callback)
    try:
        w.run(features, labels)
    finally:
        transport.close()
