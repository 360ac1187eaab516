// Native (C++/HIP) session loops for the async-PS RCCL star.
//
// The Python-thread-per-worker PS service is correct but GIL-bound: an
// 8-GPU node with 4 workers/GPU makes ~2 torch.distributed calls per
// worker-step on rank 0 — tens of thousands of GIL-serialized Python
// calls per bench.  These entry points run a whole k-step session in
// C++ with the GIL RELEASED:
//
//   ps_session:     k x { ncclSend(weights) ; ncclRecv(grads) ; fused
//                   optimizer apply } on this service thread's stream —
//                   concurrent sessions race their applies on the
//                   weight buffer (Hogwild, reference
//                   HogwildSparkModel.py:219-242 semantics).
//   worker_session: k x { ncclRecv(weights) ; hipGraphLaunch(step) ;
//                   ncclSend(grads) } — the captured compute graph
//                   comes from torch.cuda.CUDAGraph.raw_cuda_graph_exec.
//   multi_replay:   k round-robin launches of W captured full-step
//                   graphs on W streams (the single-GPU multi-worker
//                   Hogwild loop) in one call.
//
// Pair communicators are bootstrapped with ncclGetUniqueId /
// ncclCommInitRank (the unique id travels over the existing gloo
// control channel), independent of torch's process-group internals.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>
#include <rccl/rccl.h>

#include <chrono>
#include <mutex>
#include <thread>
#include <vector>

#include "common.h"

// fused optimizer entry points (optim.hip)
void fused_sgd(torch::Tensor p, torch::Tensor g, double lr);
void fused_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor acc,
                    double lr, double mom, bool nesterov);
void fused_adam_dev(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                    torch::Tensor v, torch::Tensor state, double lr,
                    double b1, double b2, double eps);
void fused_rmsprop(torch::Tensor p, torch::Tensor g, torch::Tensor ms,
                   torch::Tensor mom, double lr, double decay,
                   double momentum, double eps);

#define NCCL_OK(expr)                                                       \
  do {                                                                      \
    ncclResult_t _r = (expr);                                               \
    TORCH_CHECK(_r == ncclSuccess, "RCCL error: ",                          \
                ncclGetErrorString(_r), " at ", __FILE__, ":", __LINE__);   \
  } while (0)

static std::vector<ncclComm_t> g_comms;
static std::mutex g_comms_mu;

// SPARKFLOW_SESSION_CHUNK_ELEMS splits every session/raw payload
// transfer into fixed-size chunks (star-contention knob: with 7 ranks x
// W sessions landing on the PS GPU, chunking lets RCCL interleave the
// concurrent pair transfers instead of serializing whole messages,
// smoothing tail latency).  0 (default) = one transfer per message.
// MUST be set identically on every rank: NCCL matches send/recv counts
// per operation, so both ends of a pair have to agree on the split.
static size_t session_chunk_elems() {
  static size_t cached = [] {
    const char* e = getenv("SPARKFLOW_SESSION_CHUNK_ELEMS");
    long v = e ? atol(e) : 0;
    return v > 0 ? (size_t)v : (size_t)0;
  }();
  return cached;
}

static void chunked_xfer(bool send, float* p, size_t n, int peer,
                         ncclComm_t comm, hipStream_t stream) {
  const size_t c = session_chunk_elems();
  if (c == 0 || c >= n) {
    if (send)
      NCCL_OK(ncclSend(p, n, ncclFloat32, peer, comm, stream));
    else
      NCCL_OK(ncclRecv(p, n, ncclFloat32, peer, comm, stream));
    return;
  }
  for (size_t off = 0; off < n; off += c) {
    const size_t m = std::min(c, n - off);
    if (send)
      NCCL_OK(ncclSend(p + off, m, ncclFloat32, peer, comm, stream));
    else
      NCCL_OK(ncclRecv(p + off, m, ncclFloat32, peer, comm, stream));
  }
}

py::bytes nccl_unique_id() {
  ncclUniqueId id;
  NCCL_OK(ncclGetUniqueId(&id));
  return py::bytes(reinterpret_cast<const char*>(&id), sizeof(id));
}

int64_t nccl_comm_init(py::bytes id_bytes, int64_t nranks, int64_t rank,
                       int64_t timeout_ms) {
  std::string s = id_bytes;
  TORCH_CHECK(s.size() == sizeof(ncclUniqueId), "bad unique id size");
  ncclUniqueId id;
  memcpy(&id, s.data(), sizeof(id));
  ncclComm_t comm = nullptr;
  ncclResult_t st = ncclSuccess;
  {
    // Non-blocking init + bounded poll: ncclCommInitRank is a
    // collective rendezvous, so if the PEER dies before (or while)
    // calling it, a blocking init would hang this thread forever and
    // the ok/peer_ok handshake in DistPairs.init_native would never be
    // reached (the gloo recv after it would deadlock the engine
    // build).  With blocking=0 we poll ncclCommGetAsyncError and abort
    // the half-open comm on timeout; the caller then takes the torch
    // pair fallback and the handshake still completes on both sides.
    py::gil_scoped_release rel;
    ncclConfig_t cfg = NCCL_CONFIG_INITIALIZER;
    cfg.blocking = 0;
    st = ncclCommInitRankConfig(&comm, (int)nranks, id, (int)rank, &cfg);
    if (st == ncclSuccess || st == ncclInProgress) {
      auto dl = std::chrono::steady_clock::now() +
                std::chrono::milliseconds(timeout_ms > 0 ? timeout_ms
                                                         : 30000);
      do {
        ncclResult_t as = ncclSuccess;
        st = ncclCommGetAsyncError(comm, &as) == ncclSuccess
                 ? as
                 : ncclUnhandledCudaError;
        if (st != ncclInProgress) break;
        std::this_thread::sleep_for(std::chrono::milliseconds(2));
      } while (std::chrono::steady_clock::now() < dl);
    }
    if (st != ncclSuccess && comm) ncclCommAbort(comm);
  }
  TORCH_CHECK(st == ncclSuccess, "raw RCCL comm init failed/timed out: ",
              st == ncclInProgress ? "timeout waiting for peer"
                                   : ncclGetErrorString(st));
  std::lock_guard<std::mutex> lk(g_comms_mu);
  g_comms.push_back(comm);
  return (int64_t)g_comms.size() - 1;
}

static ncclComm_t get_comm(int64_t h) {
  std::lock_guard<std::mutex> lk(g_comms_mu);
  TORCH_CHECK(h >= 0 && h < (int64_t)g_comms.size() && g_comms[h],
              "bad comm handle");
  return g_comms[h];
}

void nccl_comm_destroy(int64_t h) {
  ncclComm_t c = get_comm(h);
  {
    py::gil_scoped_release rel;
    ncclCommDestroy(c);
  }
  std::lock_guard<std::mutex> lk(g_comms_mu);
  g_comms[h] = nullptr;
}

// opt ids shared with parallel/optimizers.py native dispatch
enum { OPT_SGD = 0, OPT_ADAM_DEV = 1, OPT_MOMENTUM = 2, OPT_RMSPROP = 3 };

void ps_session(int64_t comm_h, int64_t peer, torch::Tensor weights,
                torch::Tensor grad_buf, int64_t k, int64_t opt_id,
                std::vector<torch::Tensor> state, double lr, double h1,
                double h2, double h3) {
  ncclComm_t comm = get_comm(comm_h);
  TORCH_CHECK(weights.is_cuda() && grad_buf.is_cuda(), "GPU buffers only");
  const size_t n = weights.numel();
  TORCH_CHECK((size_t)grad_buf.numel() == n, "size mismatch");
  at::cuda::CUDAStream stream =
      at::cuda::getStreamFromPool(false, weights.get_device());
  at::cuda::CUDAStreamGuard guard(stream);
  float* wp = weights.data_ptr<float>();
  float* gp = grad_buf.data_ptr<float>();

  py::gil_scoped_release rel;
  for (int64_t i = 0; i < k; ++i) {
    chunked_xfer(true, wp, n, (int)peer, comm, stream.stream());
    chunked_xfer(false, gp, n, (int)peer, comm, stream.stream());
    switch (opt_id) {
      case OPT_SGD:
        fused_sgd(weights, grad_buf, lr);
        break;
      case OPT_ADAM_DEV:
        fused_adam_dev(weights, grad_buf, state[0], state[1], state[2], lr,
                       h1, h2, h3);
        break;
      case OPT_MOMENTUM:
        fused_momentum(weights, grad_buf, state[0], lr, h1, h2 != 0.0);
        break;
      case OPT_RMSPROP:
        fused_rmsprop(weights, grad_buf, state[0], state[1], lr, h1, h2,
                      h3);
        break;
      default:
        TORCH_CHECK(false, "unsupported native optimizer id");
    }
  }
  // the service thread owns this stream; make session completion
  // observable to the Python side that joins on update counters
  HIP_OK(hipStreamSynchronize(stream.stream()));
}

void worker_session(int64_t comm_h, int64_t peer, torch::Tensor w_flat,
                    torch::Tensor g_flat, int64_t graph_exec, int64_t k) {
  ncclComm_t comm = get_comm(comm_h);
  const size_t n = w_flat.numel();
  hipGraphExec_t exec = reinterpret_cast<hipGraphExec_t>(graph_exec);
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  float* wp = w_flat.data_ptr<float>();
  float* gp = g_flat.data_ptr<float>();

  py::gil_scoped_release rel;
  for (int64_t i = 0; i < k; ++i) {
    chunked_xfer(false, wp, n, (int)peer, comm, stream);
    HIP_OK(hipGraphLaunch(exec, stream));
    chunked_xfer(true, gp, (size_t)g_flat.numel(), (int)peer, comm, stream);
  }
}

// Per-op raw-comm helpers: the Python PS loop uses these when a session
// arrives on the raw channel but the optimizer has no native kernel.
void raw_send(int64_t comm_h, int64_t peer, torch::Tensor t) {
  ncclComm_t comm = get_comm(comm_h);
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  py::gil_scoped_release rel;
  chunked_xfer(true, t.data_ptr<float>(), t.numel(), (int)peer, comm,
               stream);
}

void raw_recv(int64_t comm_h, int64_t peer, torch::Tensor t) {
  ncclComm_t comm = get_comm(comm_h);
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  py::gil_scoped_release rel;
  NCCL_OK(ncclRecv(t.data_ptr<float>(), t.numel(), ncclFloat32, (int)peer,
                   comm, stream));
}

// Dedicated replay streams, created ONCE and as early as possible.
// HIP maps streams onto GPU_MAX_HW_QUEUES (4) hardware queues; with
// pool streams the mapping the W replay streams land on varies run to
// run, and when two of them share a hardware queue the graph replays
// serialize pairwise — measured as a random-per-process ~0.54x
// throughput mode (full 2.4 GHz clocks, unaffected by perflevel).
// Creating the W streams first, before torch's pool/capture streams,
// pins them to distinct queues deterministically.
static std::vector<hipStream_t> g_replay_streams;
static std::mutex g_rs_mu;

void replay_streams_init(int64_t n) {
  std::lock_guard<std::mutex> lk(g_rs_mu);
  while ((int64_t)g_replay_streams.size() < n) {
    hipStream_t s;
    HIP_OK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    g_replay_streams.push_back(s);
  }
}

void replay_streams_reroll() {
  // Re-roll the replay streams' hardware-queue placement: create the
  // fresh set FIRST (advancing the runtime's queue assignment) and only
  // then destroy the old one.  Caller must have synchronized the device
  // (no replays in flight on the old streams).
  std::lock_guard<std::mutex> lk(g_rs_mu);
  const size_t n = g_replay_streams.size();
  std::vector<hipStream_t> fresh;
  fresh.reserve(n);
  for (size_t i = 0; i < n; ++i) {
    hipStream_t s;
    HIP_OK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    fresh.push_back(s);
  }
  for (hipStream_t s : g_replay_streams) HIP_OK(hipStreamDestroy(s));
  g_replay_streams = std::move(fresh);
}

void multi_replay(std::vector<int64_t> execs, int64_t k) {
  // W full-step graphs round-robin onto W dedicated streams: the whole
  // multi-worker Hogwild loop is enqueued from one GIL-free C++ call.
  const int W = (int)execs.size();
  replay_streams_init(W);
  at::cuda::CUDAStream cur = at::cuda::getCurrentCUDAStream();
  py::gil_scoped_release rel;
  // order the replay streams after whatever is pending on the current one
  hipEvent_t ev;
  HIP_OK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  HIP_OK(hipEventRecord(ev, cur.stream()));
  for (int w = 0; w < W; ++w)
    HIP_OK(hipStreamWaitEvent(g_replay_streams[w], ev, 0));
  for (int64_t i = 0; i < k; ++i) {
    for (int w = 0; w < W; ++w) {
      HIP_OK(hipGraphLaunch(reinterpret_cast<hipGraphExec_t>(execs[w]),
                            g_replay_streams[w]));
    }
  }
  for (int w = 0; w < W; ++w) {
    HIP_OK(hipEventRecord(ev, g_replay_streams[w]));
    HIP_OK(hipStreamWaitEvent(cur.stream(), ev, 0));
  }
  HIP_OK(hipEventDestroy(ev));
}
