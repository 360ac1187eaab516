// Elementwise / reduction helpers: activation backward masks and the
// rowwise-argmax prediction head (reference tf.argmax(out, 1) —
// SURVEY.md §2.2).  Grid-stride float4 sweeps sized for 256 CUs.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

__global__ void act_bwd_kernel(const float* __restrict__ y,
                               const float* __restrict__ dy,
                               float* __restrict__ dpre, long n, int act) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    dpre[i] = act_grad_from_out(y[i], dy[i], act);
  }
}

__global__ void argmax_rows_kernel(const float* __restrict__ x,
                                   float* __restrict__ out, int M, int C) {
  // one wave per row; lanes stride the columns
  const int row = blockIdx.x * (blockDim.x / WAVE) + (threadIdx.x / WAVE);
  const int lane = threadIdx.x & (WAVE - 1);
  if (row >= M) return;
  float best = -INFINITY;
  int best_idx = 0;
  for (int c = lane; c < C; c += WAVE) {
    const float v = x[(long)row * C + c];
    if (v > best) {
      best = v;
      best_idx = c;
    }
  }
  // reduce (value, index) across the wave; ties -> smallest index
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(best_idx, off, WAVE);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  if (lane == 0) out[row] = (float)best_idx;
}

// Fused minibatch staging for the captured training step: ONE launch
// gathers the x rows (and labels) named by the index pool's current row
// and advances the device-side pool cursor — replacing the
// index_select / gather / add_ / remainder_ kernel train the eager
// sequence needs (~5 launches).  The cursor bump uses a done-ticket:
// every block reads the cursor at start, increments `done` at end, and
// the last-finishing block (which can only fire after every block has
// read) bumps the cursor and self-resets the ticket — so the kernel is
// replay-deterministic inside a hipGraph.
__global__ void gather_batch_kernel(const float* __restrict__ x,
                                    float* __restrict__ xb,
                                    const float* __restrict__ y,
                                    float* __restrict__ yb,
                                    const long* __restrict__ idx_pool,
                                    long* __restrict__ ctr,
                                    int* __restrict__ done, int mbs, long dx,
                                    long dy_row, int cap, long n_rows,
                                    const float* __restrict__ wsrc,
                                    float* __restrict__ wdst, long wn) {
  // One ATOMIC cursor read per block (thread 0), broadcast through LDS:
  // a plain per-thread load of this cross-launch-stored location was
  // observed (rarely) returning garbage on gfx950, sending the row
  // gather to wild addresses — and a per-THREAD atomic read serializes
  // the whole grid on one cache line.
  __shared__ long c_sh;
  if (threadIdx.x == 0) {
    long cv = (long)atomicAdd(reinterpret_cast<unsigned long long*>(ctr),
                              0ull);
    if (cv < 0 || cv >= cap) cv = 0;  // defensive: never index off pool
    c_sh = cv;
  }
  __syncthreads();
  const long c = c_sh;
  const long* row_idx = idx_pool + c * mbs;
  const bool vec4 = (dx & 3) == 0;
  if (wsrc != nullptr) {
    // fused weight pull: the step's flat-param copy rides this launch
    // (grid-stride float4 over the whole grid; wn % 4 handled by tail)
    const long w4 = wn >> 2;
    const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    const float4* s4 = reinterpret_cast<const float4*>(wsrc);
    float4* d4 = reinterpret_cast<float4*>(wdst);
    for (long i = tid; i < w4; i += stride) d4[i] = s4[i];
    for (long i = (w4 << 2) + tid; i < wn; i += stride) wdst[i] = wsrc[i];
  }
  for (int r = blockIdx.x; r < mbs; r += gridDim.x) {
    long src = row_idx[r];
    if (src < 0 || src >= n_rows) src = 0;  // defensive
    const float* xs = x + src * dx;
    float* xd = xb + (long)r * dx;
    if (vec4) {
      const float4* s4 = reinterpret_cast<const float4*>(xs);
      float4* d4 = reinterpret_cast<float4*>(xd);
      for (long i = threadIdx.x; i < (dx >> 2); i += blockDim.x)
        d4[i] = s4[i];
    } else {
      for (long i = threadIdx.x; i < dx; i += blockDim.x) xd[i] = xs[i];
    }
    if (y != nullptr && threadIdx.x == 0) {
      for (long j = 0; j < dy_row; ++j)
        yb[(long)r * dy_row + j] = y[src * dy_row + j];
    }
  }
  __syncthreads();  // all of this block's threads have consumed `c`
  if (threadIdx.x == 0) {
    // done-ticket: the last-finishing block advances the cursor (it can
    // only fire after every block has read `c`) and self-resets the
    // ticket, so the kernel replays identically inside a hipGraph.
    // Plain stores suffice: nothing reads ctr/done again until the next
    // launch, and the kernel boundary orders that.
    if (atomicAdd(done, 1) == (int)gridDim.x - 1) {
      atomicExch(reinterpret_cast<unsigned long long*>(ctr),
                 (unsigned long long)((c + 1) % cap));
      atomicExch(done, 0);
    }
  }
}

void gather_batch(torch::Tensor x, torch::Tensor xb, torch::Tensor idx_pool,
                  torch::Tensor ctr, torch::Tensor done,
                  c10::optional<torch::Tensor> y,
                  c10::optional<torch::Tensor> yb,
                  c10::optional<torch::Tensor> wsrc,
                  c10::optional<torch::Tensor> wdst) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && xb.is_contiguous(),
              "gather_batch: x/xb must be contiguous on GPU");
  TORCH_CHECK(idx_pool.scalar_type() == torch::kInt64 &&
                  ctr.scalar_type() == torch::kInt64 &&
                  done.scalar_type() == torch::kInt32,
              "gather_batch: idx_pool/ctr int64, done int32");
  const int mbs = xb.size(0);
  const long dx = xb.numel() / mbs;
  const long n_rows = x.numel() / dx;
  const int cap = idx_pool.size(0);
  TORCH_CHECK(idx_pool.size(1) == mbs, "pool width != mbs");
  const float* yp = nullptr;
  float* ybp = nullptr;
  long dy_row = 0;
  if (y.has_value() && yb.has_value()) {
    TORCH_CHECK(y->scalar_type() == torch::kFloat32 &&
                    yb->scalar_type() == torch::kFloat32,
                "gather_batch: float labels only (int labels are cast by "
                "the worker)");
    yp = y->data_ptr<float>();
    ybp = yb->data_ptr<float>();
    dy_row = yb->numel() / mbs;
  }
  const float* wsp = nullptr;
  float* wdp = nullptr;
  long wn = 0;
  if (wsrc.has_value() && wdst.has_value()) {
    TORCH_CHECK(wsrc->numel() == wdst->numel(), "weight pull size mismatch");
    wsp = wsrc->data_ptr<float>();
    wdp = wdst->data_ptr<float>();
    wn = wsrc->numel();
  }
  // enough blocks to stream the weight copy + rows, few enough that the
  // same-address ticket atomics stay ~1us
  const int blocks = std::min(mbs, 128);
  hipLaunchKernelGGL(gather_batch_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     xb.data_ptr<float>(), yp, ybp,
                     idx_pool.data_ptr<long>(), ctr.data_ptr<long>(),
                     done.data_ptr<int>(), mbs, dx, dy_row, cap, n_rows,
                     wsp, wdp, wn);
}

// Pack up to 16 parameter-grad tensors into the flat push buffer in ONE
// launch.  The segment pointers/offsets travel BY VALUE as kernel
// arguments (baked at hipGraph capture; grad pointers are stable across
// replays), so no descriptor memory or h2d copy is needed — this replaces
// the per-step torch.cat kernel.
#define PACK_MAX_SEGS 16
struct PackArgs {
  const float* src[PACK_MAX_SEGS];
  long off[PACK_MAX_SEGS];   // dst offset of segment s
  long n[PACK_MAX_SEGS];     // numel of segment s
  int first_block[PACK_MAX_SEGS + 1];  // block range owning segment s
};

__global__ void pack_segments_kernel(PackArgs a, int nseg,
                                     float* __restrict__ dst) {
  int s = 0;
  while (s + 1 < nseg && (int)blockIdx.x >= a.first_block[s + 1]) ++s;
  const int b0 = a.first_block[s], b1 = a.first_block[s + 1];
  const float* src = a.src[s];
  float* d = dst + a.off[s];
  const long n = a.n[s];
  const long tid = ((long)(blockIdx.x - b0)) * blockDim.x + threadIdx.x;
  const long stride = (long)(b1 - b0) * blockDim.x;
  const bool vec4 =
      (n & 3) == 0 && (((size_t)src | (size_t)d) & 15) == 0;
  if (vec4) {
    const float4* s4 = reinterpret_cast<const float4*>(src);
    float4* d4 = reinterpret_cast<float4*>(d);
    for (long i = tid; i < (n >> 2); i += stride) d4[i] = s4[i];
  } else {
    for (long i = tid; i < n; i += stride) d[i] = src[i];
  }
}

void pack_segments(std::vector<torch::Tensor> srcs, torch::Tensor flat) {
  const int nseg = (int)srcs.size();
  TORCH_CHECK(nseg >= 1 && nseg <= PACK_MAX_SEGS,
              "pack_segments: 1..16 segments");
  PackArgs a;
  long off = 0;
  int blocks = 0;
  for (int s = 0; s < nseg; ++s) {
    TORCH_CHECK(srcs[s].is_cuda() && srcs[s].is_contiguous() &&
                    srcs[s].scalar_type() == torch::kFloat32,
                "pack_segments: contiguous fp32 on GPU");
    a.src[s] = srcs[s].data_ptr<float>();
    a.off[s] = off;
    a.n[s] = srcs[s].numel();
    a.first_block[s] = blocks;
    blocks += (int)std::min<long>(128, (a.n[s] + 1023) / 1024);
    off += a.n[s];
  }
  a.first_block[nseg] = blocks;
  TORCH_CHECK(off == flat.numel(), "pack_segments: size mismatch");
  hipLaunchKernelGGL(pack_segments_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), a, nseg,
                     flat.data_ptr<float>());
}

static inline void check_f32e(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous fp32 on GPU");
}

torch::Tensor act_bwd(torch::Tensor y, torch::Tensor dy, int64_t act) {
  check_f32e(y, "y");
  check_f32e(dy, "dy");
  auto dpre = torch::empty_like(dy);
  const long n = dy.numel();
  const int blocks = (int)std::min<long>(4096, (n + 255) / 256);
  hipLaunchKernelGGL(act_bwd_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), y.data_ptr<float>(),
                     dy.data_ptr<float>(), dpre.data_ptr<float>(), n,
                     (int)act);
  return dpre;
}

torch::Tensor argmax_rows(torch::Tensor x) {
  check_f32e(x, "x");
  const int M = x.size(0), C = x.size(1);
  auto out = torch::empty({M}, x.options());
  const int waves_per_block = 4;
  hipLaunchKernelGGL(argmax_rows_kernel, dim3(ceil_div(M, waves_per_block)),
                     dim3(waves_per_block * WAVE), 0,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     out.data_ptr<float>(), M, C);
  return out;
}
