"""Training orchestrator — parity with reference ``HogwildSparkModel``
(HogwildSparkModel.py:103-272), re-architected for MI355X.

The PS runs in a spawned daemon process (spawn, never fork — reference
HogwildSparkModel.py:156-166 keeps the same rule for GPU state) serving the
raw-tensor TCP protocol; a readiness handshake replaces the reference's
fixed ``sleep(8)``.  ``train`` accepts either a PySpark RDD (one executor
per GPU; the worker closure connects back to the PS) or a plain list of
partitions (local mode used by tests — the analog of the reference's
Spark ``local[2]`` harness, each partition trained by a concurrent thread).

For torchrun-launched multi-GPU jobs (bench.py / the driver's scaling
runs) use ``sparkflow_amd.parallel.dist_engine`` instead, which replaces
this TCP path with RCCL p2p pair communicators over xGMI.
"""
from __future__ import annotations

import multiprocessing
import socket
import threading
from typing import List, Optional

import numpy as np
import torch

from sparkflow_amd.parallel.ps import OptimizerSpec, PSCore, TrainConfig
from sparkflow_amd.parallel.transport import (
    TcpPSServer,
    TcpTransport,
    wait_for_server,
)
from sparkflow_amd.parallel.worker import run_partition
from sparkflow_amd.utils.serialization import flat_to_list, module_from_json


def _ps_process_main(graph_json, optimizer, port, acquire_lock, max_errors,
                     device):
    if device == "auto":
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    core = PSCore(graph_json, optimizer, device=device,
                  acquire_lock=acquire_lock, max_errors=max_errors)
    server = TcpPSServer(core, port=port).start()
    # Serve until the parent terminates this daemon process (reference
    # stop_server semantics, HogwildSparkModel.py:168-173).
    threading.Event().wait()


class HogwildSparkModel:
    def __init__(self,
                 tensorflowGraph: str = None,
                 iters: int = 1000,
                 tfInput=None,
                 tfLabel=None,
                 optimizer: OptimizerSpec = None,
                 master_url: Optional[str] = None,
                 serverStartup: int = 60,
                 acquire_lock: bool = False,
                 mini_batch: int = -1,
                 mini_stochastic_iters: int = -1,
                 shuffle: bool = True,
                 verbose: int = 0,
                 partition_shuffles: int = 1,
                 loss_callback=None,
                 port: int = 5000,
                 ps_device: str = "auto",
                 worker_device: str = "auto"):
        self.graph_json = tensorflowGraph
        self.iters = iters
        self.tfInput = tfInput
        self.tfLabel = tfLabel
        self.is_supervised = tfLabel is not None
        self.optimizer = optimizer or OptimizerSpec()
        self.acquire_lock = acquire_lock
        self.config = TrainConfig(iters=iters, mini_batch=mini_batch,
                                  mini_stochastic_iters=mini_stochastic_iters,
                                  shuffle=shuffle, verbose=verbose,
                                  acquire_lock=acquire_lock,
                                  partition_shuffles=partition_shuffles,
                                  port=port)
        self.partition_shuffles = partition_shuffles
        self.loss_callback = loss_callback
        self.port = port
        self.ps_device = ps_device
        self.worker_device = worker_device
        self.server = None

        self.start_server(port)
        self.master_url = master_url if master_url is not None \
            else HogwildSparkModel.determine_master(port)
        # readiness handshake instead of the reference's fixed sleep(8)
        wait_for_server("127.0.0.1:%d" % port, timeout=max(serverStartup, 30))

    @staticmethod
    def determine_master(port: int) -> str:
        try:
            return socket.gethostbyname(socket.gethostname()) + ":" + str(port)
        except OSError:
            return "localhost:" + str(port)

    # -- PS lifecycle --------------------------------------------------------

    def start_server(self, port: int):
        ctx = multiprocessing.get_context("spawn")
        self.server = ctx.Process(
            target=_ps_process_main,
            args=(self.graph_json, self.optimizer, port, self.acquire_lock,
                  self.iters, self.ps_device),
            daemon=True,
        )
        self.server.start()

    def stop_server(self):
        if self.server is not None:
            self.server.terminate()
            self.server.join()
            self.server = None

    # -- training ------------------------------------------------------------

    def _worker_device(self) -> str:
        if self.worker_device != "auto":
            return self.worker_device
        return "cuda:0" if torch.cuda.is_available() else "cpu"

    def _train_rdd(self, rdd):
        """Spark path: one worker closure per partition (reference
        HogwildSparkModel.py:258-266)."""
        graph_json = self.graph_json
        cfg = self.config
        supervised = self.is_supervised
        master = self.master_url
        cb = self.loss_callback

        def _closure(part_iter):
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
            transport = TcpTransport(master, device="cpu")
            run_partition(part_iter, graph_json, transport, cfg, supervised,
                          device=device, loss_callback=cb)

        for i in range(self.partition_shuffles):
            rdd.foreachPartition(_closure)
            if self.partition_shuffles - i > 1:
                rdd = rdd.repartition(rdd.getNumPartitions())

    def _train_local(self, partitions: List[list]):
        """Local mode: each partition trains on its own thread against the
        PS process — the no-Spark analog of the reference's local[2] tests."""
        device = self._worker_device()
        master = "127.0.0.1:%d" % self.port

        def _one(part):
            transport = TcpTransport(master, device="cpu")
            run_partition(part, self.graph_json, transport, self.config,
                          self.is_supervised, device=device,
                          loss_callback=self.loss_callback)

        parts = [list(p) for p in partitions]
        for i in range(self.partition_shuffles):
            errs = []

            def _run(p):
                try:
                    _one(p)
                except Exception as e:  # surface worker failures to caller
                    errs.append(e)

            threads = [threading.Thread(target=_run, args=(p,)) for p in parts]
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            if errs:
                raise errs[0]
            if self.partition_shuffles - i > 1:
                rows = [r for p in parts for r in p]
                order = np.random.permutation(len(rows))
                k = len(parts)
                parts = [[rows[j] for j in order[s::k]] for s in range(k)]

    def train(self, data):
        """Train and return the final weights as a per-variable ndarray list
        (reference train contract, HogwildSparkModel.py:246-272)."""
        try:
            if hasattr(data, "foreachPartition"):
                self._train_rdd(data)
            else:
                self._train_local(data)
            transport = TcpTransport("127.0.0.1:%d" % self.port, device="cpu")
            flat = transport.pull()
            transport.close()
            module = module_from_json(self.graph_json)
            weights = flat_to_list(module, flat)
            self.stop_server()
            return weights
        except Exception:
            self.stop_server()
            raise


# idiomatic alias
HogwildTrainer = HogwildSparkModel
