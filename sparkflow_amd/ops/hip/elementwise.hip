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
