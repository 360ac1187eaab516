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
    one deterministic ping per pair avoids it."""
    import torch.distributed as dist

    t = torch.ones(1, device=device)
    c = torch.ones(1)
    if pairs.rank == 0:
        for w in range(1, pairs.world):
            dist.send(t, dst=w, group=pairs.payload[w])
            dist.send(c, dst=w, group=pairs.ctrl[w])
    else:
        dist.recv(t, src=0, group=pairs.payload[pairs.rank])
        dist.recv(c, src=0, group=pairs.ctrl[pairs.rank])
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

    def barrier(self):
        import torch.distributed as dist

        if self.world > 1:
            dist.barrier()

    def close(self):
        if self.rank != 0:
            self.worker.transport.close()
        if self.service is not None:
            self.service.join()


def build_engine(graph_json: str, optimizer: OptimizerSpec,
                 config: TrainConfig, device=None,
                 loss_callback=None) -> EngineHandle:
    """Construct PS + worker for this rank (call on every rank)."""
    rank, world, device = init_from_env(device=device)

    ps = service = None
    if world > 1:
        pairs = DistPairs()
        _warmup_pairs(pairs, device)
    if rank == 0:
        ps = PSCore(graph_json, optimizer, device=device,
                    acquire_lock=config.acquire_lock,
                    max_errors=config.iters)
        if world > 1:
            service = DistPSService(ps, pairs).start()
        transport = LocalTransport(ps, device)
    else:
        numel = flat_size(module_from_json(graph_json))
        transport = DistTransport(pairs, numel, device)

    worker = Worker(graph_json, transport, config, device=device,
                    loss_callback=loss_callback)
    return EngineHandle(rank=rank, world=world, device=device, worker=worker,
                        ps=ps, service=service)
