"""Pull/push transports between workers and the parameter server.

The reference moves weights/grads as HTTP + pickle (reference
HogwildSparkModel.py:22-35).  Here every message is ONE
contiguous fp32 buffer (all trainable tensors flattened), in three
transports behind one interface:

* ``LocalTransport``   — PS in the same process (1-GPU jobs, tests): direct
  calls into the PS core; Hogwild raciness comes from PS service threads
  sharing the weight buffer.
* ``TcpTransport``     — raw-socket length-prefixed fp32 frames against a
  threaded PS server.  This is the Spark-executor / CPU-plumbing path
  (BASELINE.json config 1) — same pull/push semantics as the reference's
  GET /parameters / POST /update, minus Flask and pickle.
* ``DistTransport``    — torch.distributed point-to-point: per-worker
  {PS, worker} pair process-groups, payload over the ``nccl`` backend
  (= RCCL over xGMI on an MI355X node) or ``gloo`` on CPU, control
  messages over a per-worker gloo pair group.  The PS star over p2p xGMI
  links gives each worker<->PS pair its own ~153 GB/s link.
"""
from __future__ import annotations

import os
import socket
import struct
import threading
import time
from typing import Optional

import numpy as np
import torch

# control ops (DistTransport control channel + TCP op bytes)
OP_PULL, OP_PUSH, OP_DONE, OP_PING, OP_RUN = 1, 2, 3, 4, 5


class WorkerTransport:
    def pull(self) -> torch.Tensor:
        raise NotImplementedError

    def push(self, flat_grads: torch.Tensor) -> None:
        raise NotImplementedError

    def close(self) -> None:
        pass


class LocalTransport(WorkerTransport):
    """Direct in-process handle to the PS core."""

    def __init__(self, ps_core, device="cpu"):
        self.ps = ps_core
        self.device = torch.device(device)

    def pull(self) -> torch.Tensor:
        if not self.ps.acquire_lock and self.ps.device == self.device:
            # Hogwild same-device fast path: hand out the live HBM buffer;
            # the worker copies it into its own flat buffer (the racy
            # read IS the reference's lock-free /parameters semantics).
            self.ps.pull_count += 1
            return self.ps.weights
        return self.ps.read_weights().to(self.device, non_blocking=True)

    def push(self, flat_grads: torch.Tensor) -> None:
        self.ps.apply_gradients(flat_grads.to(self.ps.device, non_blocking=True))


# ---------------------------------------------------------------------------
# TCP transport
# ---------------------------------------------------------------------------

_HDR = struct.Struct("<Q")


def _recv_exact(sock: socket.socket, n: int) -> bytes:
    chunks = []
    while n:
        b = sock.recv(min(n, 1 << 20))
        if not b:
            raise ConnectionError("peer closed")
        chunks.append(b)
        n -= len(b)
    return b"".join(chunks)


class TcpTransport(WorkerTransport):
    def __init__(self, master_url: str, device="cpu", timeout: float = 120.0):
        host, port = master_url.rsplit(":", 1)
        self.device = torch.device(device)
        self.sock = socket.create_connection((host, int(port)), timeout=timeout)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)

    def pull(self) -> torch.Tensor:
        self.sock.sendall(bytes([OP_PULL]))
        (nbytes,) = _HDR.unpack(_recv_exact(self.sock, _HDR.size))
        buf = _recv_exact(self.sock, nbytes)
        arr = np.frombuffer(buf, dtype=np.float32)
        return torch.from_numpy(arr.copy()).to(self.device)

    def push(self, flat_grads: torch.Tensor) -> None:
        data = flat_grads.detach().to("cpu", torch.float32).numpy().tobytes()
        self.sock.sendall(bytes([OP_PUSH]) + _HDR.pack(len(data)) + data)
        status = _recv_exact(self.sock, 1)[0]
        if status != 0:
            raise RuntimeError("parameter server rejected update (error budget)")

    def ping(self) -> bool:
        self.sock.sendall(bytes([OP_PING]))
        return _recv_exact(self.sock, 1)[0] == 0

    def close(self) -> None:
        try:
            self.sock.sendall(bytes([OP_DONE]))
            self.sock.close()
        except OSError:
            pass


class TcpPSServer:
    """Threaded socket front-end over a PS core — one service thread per
    worker connection (the Hogwild concurrency surface, reference
    HogwildSparkModel.py:244 ``threaded=True``)."""

    def __init__(self, ps_core, host: str = "0.0.0.0", port: int = 5000):
        self.ps = ps_core
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind((host, port))
        self.port = self.sock.getsockname()[1]
        self.sock.listen(128)
        self._stop = threading.Event()
        self._threads = []
        self._accept_thread = threading.Thread(target=self._accept_loop,
                                               daemon=True)

    def start(self):
        self._accept_thread.start()
        return self

    def _accept_loop(self):
        self.sock.settimeout(0.2)
        while not self._stop.is_set():
            try:
                conn, _ = self.sock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            t = threading.Thread(target=self._serve, args=(conn,), daemon=True)
            t.start()
            self._threads.append(t)

    def _serve(self, conn: socket.socket):
        conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        try:
            while not self._stop.is_set():
                op = conn.recv(1)
                if not op:
                    return
                op = op[0]
                if op == OP_PULL:
                    data = self.ps.read_weights().to("cpu", torch.float32) \
                        .numpy().tobytes()
                    conn.sendall(_HDR.pack(len(data)) + data)
                elif op == OP_PUSH:
                    (nbytes,) = _HDR.unpack(_recv_exact(conn, _HDR.size))
                    buf = _recv_exact(conn, nbytes)
                    grads = torch.from_numpy(
                        np.frombuffer(buf, dtype=np.float32).copy())
                    try:
                        self.ps.apply_gradients(grads.to(self.ps.device))
                        conn.sendall(b"\x00")
                    except Exception:
                        conn.sendall(b"\x01")
                        raise
                elif op == OP_PING:
                    conn.sendall(b"\x00")
                elif op == OP_DONE:
                    return
        except (ConnectionError, OSError):
            pass
        finally:
            conn.close()

    def stop(self):
        self._stop.set()
        try:
            self.sock.close()
        except OSError:
            pass


def wait_for_server(master_url: str, timeout: float = 60.0) -> None:
    """Readiness handshake replacing the reference's fixed ``sleep(8)``
    (HogwildSparkModel.py:117,135)."""
    deadline = time.monotonic() + timeout
    last_err: Optional[Exception] = None
    while time.monotonic() < deadline:
        try:
            t = TcpTransport(master_url, timeout=5.0)
            ok = t.ping()
            t.close()
            if ok:
                return
        except (OSError, ConnectionError) as e:
            last_err = e
        time.sleep(0.1)
    raise TimeoutError(f"parameter server at {master_url} not ready: {last_err}")


# ---------------------------------------------------------------------------
# torch.distributed transport (RCCL p2p over xGMI on GPU, gloo on CPU)
# ---------------------------------------------------------------------------


def find_free_port(host: str = "127.0.0.1") -> int:
    """OS-assigned free TCP port (rendezvous bootstrap)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind((host, 0))
    port = s.getsockname()[1]
    s.close()
    return port


class DistPairs:
    """Pair process-groups between rank 0 (PS) and each worker rank.

    Every rank must construct this collectively (``new_group`` is a
    collective call).  Payload groups use the default backend of the world
    (nccl on GPU = one RCCL communicator per worker<->PS xGMI link); control
    groups are always gloo (small CPU int tensors carry PULL/PUSH/DONE).

    With ``workers_per_gpu`` > 1 every LOGICAL worker on a rank gets its
    own pair communicator (keyed ``(rank, widx)``), so concurrent worker
    streams never share an RCCL communicator."""

    def __init__(self, payload_backend: Optional[str] = None,
                 workers_per_gpu: int = 1):
        import torch.distributed as dist

        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        self.workers_per_gpu = max(1, workers_per_gpu)
        self.payload = {}
        self.ctrl = {}
        self.native = {}  # (w, i) -> raw RCCL comm handle (both sides ok)
        for w in range(1, self.world):
            for i in range(self.workers_per_gpu):
                self.payload[(w, i)] = dist.new_group(
                    [0, w], backend=payload_backend)
                self.ctrl[(w, i)] = dist.new_group([0, w], backend="gloo")

    def init_native(self, device) -> None:
        """Bootstrap one raw RCCL communicator per pair for the GIL-free
        C++ session loops (ops/hip/dist_session.hip).  The ncclUniqueId
        travels over the gloo control channel; a success flag is
        exchanged afterwards so both sides agree whether the pair may use
        the raw channel (a one-sided failure would otherwise deadlock
        the wire).  No-op (empty ``native``) on CPU or without the
        extension."""
        import torch.distributed as dist

        if torch.device(device).type != "cuda":
            return
        from sparkflow_amd import ops

        have_ext = ops.available()
        if os.environ.get("SPARKFLOW_NATIVE_SESSION", "1") == "0":
            have_ext = False
        my_can = torch.tensor([1 if have_ext else 0], dtype=torch.int64)
        if self.rank == 0:
            for w in range(1, self.world):
                for i in range(self.workers_per_gpu):
                    cg = self.ctrl[(w, i)]
                    peer_can = torch.zeros(1, dtype=torch.int64)
                    dist.send(my_can, dst=w, group=cg)
                    dist.recv(peer_can, src=w, group=cg)
                    if not (have_ext and int(peer_can.item())):
                        continue
                    raw = ops.hip.nccl_unique_id()
                    idbuf = torch.frombuffer(bytearray(raw),
                                             dtype=torch.uint8).clone()
                    dist.send(idbuf, dst=w, group=cg)
                    handle = None
                    try:
                        handle = ops.hip.nccl_comm_init(
                            bytes(idbuf.numpy().tobytes()), 2, 0)
                    except Exception as e:  # peer fails symmetrically
                        print(f"[sparkflow_amd] raw comm init (0,{w},{i}) "
                              f"failed ({e}); torch pair fallback")
                    # confirm: both sides must agree the pair is usable
                    ok = torch.tensor([1 if handle is not None else 0],
                                      dtype=torch.int64)
                    peer_ok = torch.zeros(1, dtype=torch.int64)
                    dist.send(ok, dst=w, group=cg)
                    dist.recv(peer_ok, src=w, group=cg)
                    if handle is not None and int(peer_ok.item()):
                        self.native[(w, i)] = handle
        else:
            for i in range(self.workers_per_gpu):
                cg = self.ctrl[(self.rank, i)]
                peer_can = torch.zeros(1, dtype=torch.int64)
                dist.recv(peer_can, src=0, group=cg)
                dist.send(my_can, dst=0, group=cg)
                if not (have_ext and int(peer_can.item())):
                    continue
                idbuf = torch.zeros(128, dtype=torch.uint8)
                dist.recv(idbuf, src=0, group=cg)
                handle = None
                try:
                    handle = ops.hip.nccl_comm_init(
                        bytes(idbuf.numpy().tobytes()), 2, 1)
                except Exception as e:
                    print(f"[sparkflow_amd] raw comm init "
                          f"({self.rank},{i}) failed ({e}); torch pair "
                          "fallback")
                ok = torch.tensor([1 if handle is not None else 0],
                                  dtype=torch.int64)
                peer_ok = torch.zeros(1, dtype=torch.int64)
                dist.recv(peer_ok, src=0, group=cg)
                dist.send(ok, dst=0, group=cg)
                if handle is not None and int(peer_ok.item()):
                    self.native[(self.rank, i)] = handle


def destroy_native(pairs: DistPairs) -> None:
    """Tear down this rank's raw RCCL communicators (call after every
    session is closed / service thread joined)."""
    from sparkflow_amd import ops

    for key, h in list(pairs.native.items()):
        try:
            ops.hip.nccl_comm_destroy(h)
        except Exception:
            pass
    pairs.native.clear()


class DistTransport(WorkerTransport):
    """Worker side of the RCCL star.

    Two protocols share the pair link:

    * per-op: every pull/push is announced on the gloo control channel
      (the eager ``Worker.run`` path, where the op pattern varies);
    * session (``run_begin(k)``): ONE control message announces k
      steps, then the link runs k x {recv weights; send grads} with no
      per-step control traffic or host syncs — RCCL enqueues are
      stream-ordered, so the whole session pipelines on the worker's
      HIP stream (this is what the driver's scaling bench exercises).
    """

    def __init__(self, pairs: DistPairs, numel: int, device, widx: int = 0):
        import torch.distributed as dist

        self.dist = dist
        self.rank = pairs.rank
        self._pairs = pairs
        self.pg = pairs.payload[(self.rank, widx)]
        self.cg = pairs.ctrl[(self.rank, widx)]
        self.native_comm = pairs.native.get((self.rank, widx))
        self.device = torch.device(device)
        self.buf = torch.empty(numel, dtype=torch.float32, device=self.device)
        self._ctrl_buf = torch.zeros(3, dtype=torch.int64)
        self._session = 0  # pulls+pushes prepaid by run_begin
        self._session_native = False

    def _ctrl(self, op: int, arg: int = 0, chan: int = 0):
        self._ctrl_buf[0] = op
        self._ctrl_buf[1] = arg
        self._ctrl_buf[2] = chan
        self.dist.send(self._ctrl_buf, dst=0, group=self.cg)

    def run_begin(self, k: int, native: Optional[bool] = None) -> None:
        """Announce a k-step session: the PS will k times {send weights,
        recv grads, apply} on this pair with no further control.  With
        ``native`` (default: whenever the raw comm exists) the payload
        moves on the raw RCCL channel so BOTH sides can run their loop
        in C++ with the GIL released."""
        if native is None:
            native = self.native_comm is not None
        self._ctrl(OP_RUN, k, 1 if native else 0)
        self._session = k
        self._session_native = bool(native)

    def pull(self) -> torch.Tensor:
        self.pull_into(self.buf)
        return self.buf

    def _use_raw(self) -> bool:
        # Whenever the raw comm exists it carries ALL payload (sessions
        # AND per-op) so the torch payload pair group is never touched —
        # build_engine then skips creating those communicators entirely.
        return self.native_comm is not None

    def pull_into(self, out: torch.Tensor) -> None:
        # inside a session, the channel is whatever run_begin declared
        raw = self._session_native if self._session > 0 else self._use_raw()
        if self._session <= 0:
            self._ctrl(OP_PULL, chan=1 if raw else 0)
        if raw:
            from sparkflow_amd import ops

            ops.hip.raw_recv(self.native_comm, 0, out)
        else:
            self.dist.recv(out, src=0, group=self.pg)

    def push(self, flat_grads: torch.Tensor) -> None:
        raw = self._session_native if self._session > 0 else self._use_raw()
        if self._session <= 0:
            self._ctrl(OP_PUSH, chan=1 if raw else 0)
        else:
            self._session -= 1
        if raw:
            from sparkflow_amd import ops

            ops.hip.raw_send(self.native_comm, 0, flat_grads.contiguous())
        else:
            self.dist.send(flat_grads.contiguous(), dst=0, group=self.pg)

    def run_session(self, k: int, flat_params: torch.Tensor,
                    flat_grads: torch.Tensor, graph_exec: int) -> bool:
        """Whole k-step session in ONE C++ call (GIL released): k x
        {recv weights into flat_params, launch the captured step graph,
        send flat_grads}.  Returns False when the raw channel is
        unavailable (caller falls back to the per-step loop)."""
        if self.native_comm is None:
            return False
        from sparkflow_amd import ops

        self.run_begin(k, native=True)
        ops.hip.worker_session(self.native_comm, 0, flat_params, flat_grads,
                               graph_exec, k)
        self._session = 0
        return True

    def close(self) -> None:
        self._ctrl(OP_DONE)


class DistPSService:
    """PS side of the RCCL star: one host thread per worker, each driving
    its own pair communicator and its own HIP stream, so one slow worker
    never blocks another (SURVEY.md §7 hard-part 1).  Hogwild mode lets the
    per-thread optimizer applies race on the weight buffer; lock mode gates
    them through the PS core's RWLock."""

    def __init__(self, ps_core, pairs: DistPairs):
        import torch.distributed as dist

        self.dist = dist
        self.ps = ps_core
        self.pairs = pairs
        self._threads = []
        # First fatal error from any service thread (e.g. "Too many
        # failures during training" past the PS error budget).  The
        # thread that hits it records it here and switches to DRAIN mode
        # — it keeps answering the wire protocol (pulls get the current
        # weights, pushed grads are received and dropped) so no peer is
        # ever left blocked in a recv — and the engine re-raises it to
        # the train caller (reference HogwildSparkModel.py:231-237
        # surfaces the same message).
        self.failure: Optional[Exception] = None
        self._fail_lock = threading.Lock()

    def _record_failure(self, e: Exception) -> None:
        with self._fail_lock:
            if self.failure is None:
                self.failure = e
        print(f"[sparkflow_amd] PS service error (serving continues in "
              f"drain mode): {e}")

    def check(self) -> None:
        """Raise the first recorded service failure (call from the train
        driver on rank 0)."""
        if self.failure is not None:
            raise self.failure

    def start(self):
        for w in range(1, self.pairs.world):
            for i in range(self.pairs.workers_per_gpu):
                t = threading.Thread(target=self._serve_guarded, args=(w, i),
                                     daemon=True)
                t.start()
                self._threads.append(t)
        return self

    def _serve(self, w: int, widx: int = 0):
        dist = self.dist
        pg = self.pairs.payload[(w, widx)]
        cg = self.pairs.ctrl[(w, widx)]
        native_comm = self.pairs.native.get((w, widx))
        dev = self.ps.device
        use_cuda = dev.type == "cuda"
        stream = torch.cuda.Stream(device=dev) if use_cuda else None
        grad_buf = torch.empty(self.ps.numel, dtype=torch.float32, device=dev)
        ctrl = torch.zeros(3, dtype=torch.int64)

        def apply(buf):
            # Past the error budget apply_gradients raises; record the
            # failure ONCE and drop further gradients (drain mode) so
            # the protocol position with the peer never shifts and no
            # worker blocks in a recv.
            if self.failure is not None:
                return
            try:
                self.ps.apply_gradients(buf)
            except Exception as e:
                self._record_failure(e)

        def handle(op: int, native: bool = False):
            native = native and native_comm is not None
            if native:
                from sparkflow_amd import ops

                if op == OP_PULL:
                    self.ps.send_weights(
                        lambda buf: ops.hip.raw_send(native_comm, 1, buf))
                elif op == OP_PUSH:
                    ops.hip.raw_recv(native_comm, 1, grad_buf)
                    apply(grad_buf)
                return
            if op == OP_PULL:
                self.ps.send_weights(
                    lambda buf: dist.send(buf, dst=w, group=pg))
            elif op == OP_PUSH:
                dist.recv(grad_buf, src=w, group=pg)
                apply(grad_buf)

        def run_session(k: int, native: bool):
            # k-step session: {send weights, recv grads, apply} x k with
            # no further control traffic.  On the raw channel with a
            # native-fused optimizer and no lock, the WHOLE loop runs in
            # C++ with the GIL released (ops/hip/dist_session.hip);
            # otherwise per-step ops from Python, still session-paced.
            spec = self.ps.optimizer.native_spec() if use_cuda else None
            if native and spec is not None and not self.ps.acquire_lock \
                    and self.failure is None:
                from sparkflow_amd import ops

                opt_id, state, lr, h1, h2, h3 = spec
                try:
                    ops.hip.ps_session(native_comm, 1, self.ps.weights,
                                       grad_buf, k, opt_id, state, lr, h1,
                                       h2, h3)
                except Exception as e:
                    # ps_session validates args BEFORE any wire traffic,
                    # so a raise here means zero steps ran: record and
                    # serve the whole session per-step (drain applies).
                    self._record_failure(e)
                    for _ in range(k):
                        handle(OP_PULL, native)
                        handle(OP_PUSH, native)
                    return
                self.ps.pull_count += k
                self.ps.update_count += k
                return
            for _ in range(k):
                handle(OP_PULL, native)
                handle(OP_PUSH, native)

        while True:
            dist.recv(ctrl, src=w, group=cg)
            op = int(ctrl[0].item())
            if op == OP_DONE:
                return
            if op == OP_RUN:
                k = int(ctrl[1].item())
                native = bool(int(ctrl[2].item())) and native_comm is not None
                if use_cuda:
                    with torch.cuda.stream(stream):
                        run_session(k, native)
                else:
                    run_session(k, native)
                continue
            per_op_native = bool(int(ctrl[2].item()))
            if use_cuda:
                with torch.cuda.stream(stream):
                    handle(op, per_op_native)
            else:
                handle(op, per_op_native)

    def _serve_guarded(self, w: int, widx: int = 0):
        try:
            self._serve(w, widx)
        except Exception as e:
            # Transport-level death (peer gone, comm torn down): nothing
            # left to serve, but the caller must still see the error.
            self._record_failure(e)

    def join(self):
        for t in self._threads:
            t.join()
        self.check()
