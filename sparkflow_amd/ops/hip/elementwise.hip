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
                                    long dy_row, int cap) {
  const long c = *ctr;
  const long* row_idx = idx_pool + c * mbs;
  // block b gathers x row b (grid.x == mbs)
  const int r = blockIdx.x;
  const long src = row_idx[r];
  const float* xs = x + src * dx;
  float* xd = xb + (long)r * dx;
  for (long i = threadIdx.x; i < dx; i += blockDim.x) xd[i] = xs[i];
  if (y != nullptr && threadIdx.x == 0) {
    for (long j = 0; j < dy_row; ++j) yb[(long)r * dy_row + j] =
        y[src * dy_row + j];
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    __threadfence();
    if (atomicAdd(done, 1) == gridDim.x - 1) {
      *ctr = (c + 1) % cap;
      *done = 0;
      __threadfence();
    }
  }
}

void gather_batch(torch::Tensor x, torch::Tensor xb, torch::Tensor idx_pool,
                  torch::Tensor ctr, torch::Tensor done,
                  c10::optional<torch::Tensor> y,
                  c10::optional<torch::Tensor> yb) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && xb.is_contiguous(),
              "gather_batch: x/xb must be contiguous on GPU");
  TORCH_CHECK(idx_pool.scalar_type() == torch::kInt64 &&
                  ctr.scalar_type() == torch::kInt64 &&
                  done.scalar_type() == torch::kInt32,
              "gather_batch: idx_pool/ctr int64, done int32");
  const int mbs = xb.size(0);
  const long dx = xb.numel() / mbs;
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
  hipLaunchKernelGGL(gather_batch_kernel, dim3(mbs), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     xb.data_ptr<float>(), yp, ybp,
                     idx_pool.data_ptr<long>(), ctr.data_ptr<long>(),
                     done.data_ptr<int>(), mbs, dx, dy_row, cap);
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
