"""Multi-GPU async-PS engine over torch.distributed (RCCL on MI355X).

Topology for a torchrun-launched job with world size N (one process per
GPU, ``backend="nccl"`` = RCCL over xGMI):

* rank 0 — the GPU parameter server (canonical flat fp32 weights in HBM,
  fused optimizer applies) PLUS a local worker sharing the same GPU via an
  in-process transport;
* ranks 1..N-1 — workers, each pulling/pushing over its own {0, w} pair
  communicator (its own xGMI link into the PS GPU — the star topology of
  SURVEY.md §2.3/§2.4).

On CPU (tests) the same code runs over gloo.
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch

from sparkflow_amd.parallel.ps import OptimizerSpec, PSCore, TrainConfig
from sparkflow_amd.parallel.transport import (
    DistPairs,
    DistPSService,
    DistTransport,
    LocalTransport,
)
from sparkflow_amd.parallel.worker import Worker
from sparkflow_amd.utils.serialization import flat_size, module_from_json


def init_from_env(backend: Optional[str] = None, device: Optional[str] = None):
    """init_process_group from torchrun env vars; returns (rank, world, device)."""
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if device is None:
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
            device = f"cuda:{local_rank}"
        else:
            device = "cpu"
    if backend is None:
        backend = "nccl" if device.startswith("cuda") else "gloo"
    if world > 1 and not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world, torch.device(device)


def _warmup_pairs(pairs: DistPairs, device) -> None:
    """Initialize every pair communicator SEQUENTIALLY from the main
    thread before the PS service threads start.  Lazy init from N-1
    concurrent service threads is where pairwise-NCCL designs deadlock;
    one deterministic ping per pair avoids it.  Pairs whose raw RCCL
    channel exists skip the torch payload ping entirely — that
    communicator would never be used, so it is never created."""
    import torch.distributed as dist

    t = torch.ones(1, device=device)
    c = torch.ones(1)
    if pairs.rank == 0:
        for w in range(1, pairs.world):
            for i in range(pairs.workers_per_gpu):
                if (w, i) not in pairs.native:
                    dist.send(t, dst=w, group=pairs.payload[(w, i)])
                dist.send(c, dst=w, group=pairs.ctrl[(w, i)])
    else:
        for i in range(pairs.workers_per_gpu):
            if (pairs.rank, i) not in pairs.native:
                dist.recv(t, src=0, group=pairs.payload[(pairs.rank, i)])
            dist.recv(c, src=0, group=pairs.ctrl[(pairs.rank, i)])
    if device.type == "cuda":
        torch.cuda.synchronize(device)


@dataclass
class EngineHandle:
    rank: int
    world: int
    device: torch.device
    worker: Worker
    ps: Optional[PSCore]
    service: Optional[DistPSService]
    workers: Optional[list] = None

    def __post_init__(self):
        if self.workers is None:
            self.workers = [self.worker]

    def barrier(self):
        import torch.distributed as dist

        if self.world > 1:
            dist.barrier()

    def run_steps(self, k: int, mini_batch: int) -> int:
        """k async-PS steps on EVERY worker of this rank.

        With several workers per GPU each worker replays its own captured
        hipGraph on its own HIP stream — W concurrent Hogwild workers
        racing pulls and fused-optimizer applies on the shared PS buffer,
        the GPU-resident equivalent of the reference's N partitions
        hitting the threaded Flask PS (HogwildSparkModel.py:244).  One
        host thread drives all streams (replays are async enqueues)."""
        if len(self.workers) == 1:
            return self.worker.run_steps(k, mini_batch)
        if self.device.type == "cuda":
            from sparkflow_amd import ops as _ops

            for w in self.workers:
                w.ensure_graph(mini_batch)
            # All-local full-step graphs + extension: the whole W x k
            # Hogwild loop runs as GIL-free C++ calls (round-robin graph
            # launches onto W pool streams), in half-pool chunks so each
            # worker's background pool draw overlaps the previous
            # chunk's GPU work (Worker.maybe_refresh_pool).
            if _ops.available() and all(
                    w._graph_state["full_step"] for w in self.workers):
                execs = [w._graph_state["graph"].raw_cuda_graph_exec()
                         for w in self.workers]
                chunk = max(1, self.workers[0]._IDX_POOL_CAP // 2)
                done = 0
                while done < k:
                    c = min(chunk, k - done)
                    for w in self.workers:
                        w.pool_prefetch(c)
                    need = [w for w in self.workers
                            if w.pool_would_wrap(c)]
                    if need:  # ONE device drain covers every worker
                        torch.cuda.synchronize(self.device)
                        for w in need:
                            w.pool_install_now()
                        torch.cuda.synchronize(self.device)
                    for w in self.workers:
                        w.pool_account(c)
                    _ops.hip.multi_replay(execs, c)
                    done += c
                total = 0
                for w in self.workers:
                    w.finish_steps(k)
                    total += k * min(mini_batch, w.x.shape[0])
                return total
            # Remote / lock-mode workers: each worker's whole k-step run
            # is enqueued under its own stream (run_steps uses the
            # GIL-free C++ session when the raw comm exists, else the
            # per-step replay loop — either way enqueues are async, so
            # the W workers overlap on the device).
            streams = getattr(self, "_streams", None)
            if streams is None or len(streams) != len(self.workers):
                streams = [torch.cuda.Stream(device=self.device)
                           for _ in self.workers]
                self._streams = streams
            cur = torch.cuda.current_stream(self.device)
            for s in streams:
                s.wait_stream(cur)
            total = 0
            for w, s in zip(self.workers, streams):
                with torch.cuda.stream(s):
                    total += w.run_steps(k, mini_batch)
            for s in streams:
                cur.wait_stream(s)
            return total
        # CPU: interleave eager steps across workers in grad_accum-sized
        # chunks (so accumulation rounds stay intact per worker)
        total = 0
        chunk = max(1, getattr(self.workers[0].config, "grad_accum", 1))
        done = 0
        while done < k:
            a = min(chunk, k - done)
            for w in self.workers:
                total += w.run_steps(a, mini_batch, use_graph=False)
            done += a
        return total

    def stage_all(self, x, y) -> None:
        for w in self.workers:
            w.stage(x, y)

    def check_failure(self):
        """Surface a PS-service failure ("Too many failures during
        training") to the rank-0 caller — reference parity
        (HogwildSparkModel.py:231-237); the service itself keeps
        draining the wire so no worker rank ever blocks."""
        if self.service is not None:
            self.service.check()

    def close(self):
        if self.rank != 0:
            for w in self.workers:
                w.transport.close()
        failure = None
        if self.service is not None:
            try:
                self.service.join()  # raises the first service failure
            except Exception as e:
                failure = e
        if self.world > 1 and self.device.type == "cuda":
            from sparkflow_amd.parallel.transport import destroy_native

            pairs = getattr(self.workers[0].transport, "_pairs", None)
            if pairs is None and self.service is not None:
                pairs = self.service.pairs
            if pairs is not None:
                torch.cuda.synchronize(self.device)
                destroy_native(pairs)
        if failure is not None:
            raise failure


def build_engine(graph_json: str, optimizer: OptimizerSpec,
                 config: TrainConfig, device=None, backend=None,
                 loss_callback=None, workers_per_gpu: int = 1,
                 remote_workers_per_gpu: Optional[int] = None) -> EngineHandle:
    """Construct PS + worker(s) for this rank (call on every rank).

    ``workers_per_gpu`` > 1 runs that many concurrent Hogwild workers on
    this rank's GPU (the reference's multiple-partitions-per-host model,
    e.g. the canonical README config's 4 partitions on local[4]); on the
    PS rank each gets its own LocalTransport, on worker ranks they share
    the rank's RCCL pair link (driven sequentially by one host thread).

    ``remote_workers_per_gpu`` (default: same as ``workers_per_gpu``)
    sets the worker count on NON-PS ranks independently — at world=8 the
    PS GPU serves 7 ranks x W sessions, so fewer remote workers is the
    first star-contention remedy.  Must be passed identically on every
    rank (the pair process-groups are created collectively)."""
    rank, world, device = init_from_env(backend=backend, device=device)
    wpg = max(1, workers_per_gpu)
    rwpg = wpg if remote_workers_per_gpu is None \
        else max(1, remote_workers_per_gpu)

    if device.type == "cuda":
        from sparkflow_amd import ops as _ops

        if _ops.available():
            # create the replay streams FIRST so they land on distinct
            # HW queues (see dist_session.hip replay_streams_init)
            _ops.hip.replay_streams_init(wpg if rank == 0 else rwpg)

    ps = service = None
    if world > 1:
        # pair communicators exist only for REMOTE workers (rank 0's
        # local workers ride LocalTransport), so their per-rank count is
        # the remote one
        pairs = DistPairs(workers_per_gpu=rwpg)
        pairs.init_native(device)  # raw RCCL channel (GIL-free sessions)
        _warmup_pairs(pairs, device)
        if rank == 0:
            total = (world - 1) * rwpg
            print(f"[sparkflow_amd] engine world={world} device={device} "
                  f"local_workers={wpg} remote_workers/rank={rwpg} "
                  f"raw_rccl_pairs={len(pairs.native)}/{total}"
                  + ("" if len(pairs.native) == total else
                     " (torch-pair fallback on the rest)"),
                  flush=True)
    if rank == 0:
        ps = PSCore(graph_json, optimizer, device=device,
                    acquire_lock=config.acquire_lock,
                    max_errors=config.iters)
        if world > 1:
            service = DistPSService(ps, pairs).start()
        transports = [LocalTransport(ps, device) for _ in range(wpg)]
    else:
        numel = flat_size(module_from_json(graph_json))
        transports = [DistTransport(pairs, numel, device, widx=i)
                      for i in range(rwpg)]

    workers = []
    for i, tr in enumerate(transports):
        cfg_i = config
        if config.seed is not None and i > 0:
            from dataclasses import replace

            cfg_i = replace(config, seed=config.seed + i)
        workers.append(Worker(graph_json, tr, cfg_i, device=device,
                              loss_callback=loss_callback))
    return EngineHandle(rank=rank, world=world, device=device,
                        worker=workers[0], ps=ps, service=service,
                        workers=workers)
