// Training-mode BatchNorm2d (NCHW, fp32) — MIOpen's spatial BN kernels
// measured 27% of the ResNet-shape step (MIOpenBatchNormBwdSpatial at
// 1.3 ms/call on 64x64x112x112); these are plain bandwidth-shaped
// plane sweeps: no div/mod in the loops, per-(channel, batch-chunk)
// blocks, channel stats combined with atomics.
//
// fwd:  stats kernel (sum, sumsq per channel) -> normalize kernel
//       (y = (x - mean) * invstd * gamma + beta; saves mean/invstd for
//       backward; first block also folds the running-stat update).
// bwd:  reduce kernel (sum_dy, sum_dy*xhat per channel) -> dx kernel
//       dx = gamma*invstd*(dy - sum_dy/cnt - xhat*sum_dy_xhat/cnt).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// ---- forward -------------------------------------------------------------

__global__ void bn_stats_kernel(const float* __restrict__ x,
                                float* __restrict__ stats, int N, int C,
                                long HW) {
  const int c = blockIdx.x;
  float s = 0.0f, s2 = 0.0f;
  for (int n = blockIdx.y; n < N; n += gridDim.y) {
    const float* p = x + ((long)n * C + c) * HW;
    for (long i = threadIdx.x; i < HW; i += blockDim.x) {
      const float v = p[i];
      s += v;
      s2 += v * v;
    }
  }
  s = wave_sum(s);
  s2 = wave_sum(s2);
  __shared__ float ps[4], ps2[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    ps[wid] = s;
    ps2[wid] = s2;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a = 0.0f, b = 0.0f;
    for (int k = 0; k < (int)(blockDim.x / WAVE); ++k) {
      a += ps[k];
      b += ps2[k];
    }
    atomicAdd(&stats[c], a);
    atomicAdd(&stats[C + c], b);
  }
}

__global__ void bn_norm_kernel(const float* __restrict__ x,
                               float* __restrict__ y,
                               const float* __restrict__ stats,
                               const float* __restrict__ gamma,
                               const float* __restrict__ beta,
                               float* __restrict__ save_mean,
                               float* __restrict__ save_invstd,
                               float* __restrict__ run_mean,
                               float* __restrict__ run_var, float momentum,
                               float eps, int N, int C, long HW, int relu) {
  const int c = blockIdx.x;
  const float cnt = (float)N * HW;
  const float mean = stats[c] / cnt;
  const float var = stats[C + c] / cnt - mean * mean;
  const float invstd = __frsqrt_rn(var + eps);
  const float g = gamma ? gamma[c] : 1.0f;
  const float b = beta ? beta[c] : 0.0f;
  const float scale = g * invstd;
  const float shift = b - mean * scale;
  if (blockIdx.y == 0 && threadIdx.x == 0) {
    save_mean[c] = mean;
    save_invstd[c] = invstd;
    if (run_mean) {  // torch semantics: unbiased var into running stats
      run_mean[c] = (1.0f - momentum) * run_mean[c] + momentum * mean;
      const float ub = var * cnt / fmaxf(cnt - 1.0f, 1.0f);
      run_var[c] = (1.0f - momentum) * run_var[c] + momentum * ub;
    }
  }
  for (int n = blockIdx.y; n < N; n += gridDim.y) {
    const float* xp = x + ((long)n * C + c) * HW;
    float* yp = y + ((long)n * C + c) * HW;
    for (long i = threadIdx.x; i < HW; i += blockDim.x) {
      const float v = xp[i] * scale + shift;
      yp[i] = relu ? fmaxf(v, 0.0f) : v;
    }
  }
}

// ---- backward ------------------------------------------------------------

__global__ void bn_bwd_reduce_kernel(const float* __restrict__ x,
                                     const float* __restrict__ dy,
                                     const float* __restrict__ y,
                                     const float* __restrict__ save_mean,
                                     const float* __restrict__ save_invstd,
                                     float* __restrict__ red, int N, int C,
                                     long HW, int relu) {
  const int c = blockIdx.x;
  const float mean = save_mean[c];
  const float invstd = save_invstd[c];
  float sd = 0.0f, sdx = 0.0f;
  for (int n = blockIdx.y; n < N; n += gridDim.y) {
    const long off = ((long)n * C + c) * HW;
    const float* xp = x + off;
    const float* dp = dy + off;
    const float* yp = relu ? y + off : nullptr;
    for (long i = threadIdx.x; i < HW; i += blockDim.x) {
      float d = dp[i];
      if (relu && yp[i] <= 0.0f) d = 0.0f;  // fused ReLU backward
      sd += d;
      sdx += d * (xp[i] - mean) * invstd;
    }
  }
  sd = wave_sum(sd);
  sdx = wave_sum(sdx);
  __shared__ float ps[4], ps2[4];
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    ps[wid] = sd;
    ps2[wid] = sdx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a = 0.0f, b = 0.0f;
    for (int k = 0; k < (int)(blockDim.x / WAVE); ++k) {
      a += ps[k];
      b += ps2[k];
    }
    atomicAdd(&red[c], a);
    atomicAdd(&red[C + c], b);
  }
}

__global__ void bn_bwd_dx_kernel(const float* __restrict__ x,
                                 const float* __restrict__ dy,
                                 const float* __restrict__ y,
                                 float* __restrict__ dx,
                                 const float* __restrict__ save_mean,
                                 const float* __restrict__ save_invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ red, int N, int C,
                                 long HW, int relu) {
  const int c = blockIdx.x;
  const float mean = save_mean[c];
  const float invstd = save_invstd[c];
  const float g = gamma ? gamma[c] : 1.0f;
  const float cnt = (float)N * HW;
  const float mean_dy = red[c] / cnt;
  const float mean_dyx = red[C + c] / cnt;
  const float scale = g * invstd;
  for (int n = blockIdx.y; n < N; n += gridDim.y) {
    const long off = ((long)n * C + c) * HW;
    const float* xp = x + off;
    const float* dp = dy + off;
    const float* yp = relu ? y + off : nullptr;
    float* op = dx + off;
    for (long i = threadIdx.x; i < HW; i += blockDim.x) {
      float d = dp[i];
      if (relu && yp[i] <= 0.0f) d = 0.0f;
      const float xhat = (xp[i] - mean) * invstd;
      op[i] = scale * (d - mean_dy - xhat * mean_dyx);
    }
  }
}

// ---- host ----------------------------------------------------------------

static inline void check_f32n(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous fp32 on GPU");
}

static inline dim3 bn_grid(int N, int C) {
  const int chunks = std::max(1, std::min(N, 1024 / C + 1));
  return dim3(C, chunks);
}

std::vector<torch::Tensor> bn2d_fwd(torch::Tensor x, torch::Tensor gamma,
                                    torch::Tensor beta,
                                    torch::Tensor run_mean,
                                    torch::Tensor run_var, double momentum,
                                    double eps, bool relu) {
  check_f32n(x, "x");
  const int N = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto y = torch::empty_like(x);
  auto stats = torch::zeros({2 * C}, x.options());
  auto save_mean = torch::empty({C}, x.options());
  auto save_invstd = torch::empty({C}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 grid = bn_grid(N, C);
  hipLaunchKernelGGL(bn_stats_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), stats.data_ptr<float>(), N, C, HW);
  hipLaunchKernelGGL(
      bn_norm_kernel, grid, dim3(256), 0, stream, x.data_ptr<float>(),
      y.data_ptr<float>(), stats.data_ptr<float>(),
      gamma.numel() ? gamma.data_ptr<float>() : nullptr,
      beta.numel() ? beta.data_ptr<float>() : nullptr,
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
      run_mean.numel() ? run_mean.data_ptr<float>() : nullptr,
      run_var.numel() ? run_var.data_ptr<float>() : nullptr,
      (float)momentum, (float)eps, N, C, HW, (int)relu);
  return {y, save_mean, save_invstd};
}

std::vector<torch::Tensor> bn2d_bwd(torch::Tensor x, torch::Tensor dy,
                                    torch::Tensor y, torch::Tensor gamma,
                                    torch::Tensor save_mean,
                                    torch::Tensor save_invstd, bool relu) {
  check_f32n(x, "x");
  check_f32n(dy, "dy");
  const int N = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto red = torch::zeros({2 * C}, x.options());
  auto dx = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 grid = bn_grid(N, C);
  const float* yp = relu ? y.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(bn_bwd_reduce_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), dy.data_ptr<float>(), yp,
                     save_mean.data_ptr<float>(),
                     save_invstd.data_ptr<float>(), red.data_ptr<float>(),
                     N, C, HW, (int)relu);
  hipLaunchKernelGGL(
      bn_bwd_dx_kernel, grid, dim3(256), 0, stream, x.data_ptr<float>(),
      dy.data_ptr<float>(), yp, dx.data_ptr<float>(),
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
      gamma.numel() ? gamma.data_ptr<float>() : nullptr,
      red.data_ptr<float>(), N, C, HW, (int)relu);
  // dgamma = sum(dy * xhat) = red[C:], dbeta = sum(dy) = red[:C]
  // (contiguous views into the reduce buffer — no extra copies)
  return {dx, red.narrow(0, C, C), red.narrow(0, 0, C)};
}

// ---- NHWC variants (x: [rows, C], rows = N*H*W) --------------------------
// grid (ceil(C/64), row_chunks); 256-thread blocks = 64 channel lanes x 4
// row groups.  Reads are 64-channel-wide coalesced row segments; the
// stats/reduce kernels combine the 4 row groups in LDS and emit ONE
// atomic per (block, channel).

#define BNC 64  // channel lanes per block

__global__ void bn_nhwc_stats_kernel(const float* __restrict__ x,
                                     float* __restrict__ stats, long rows,
                                     int C) {
  const int cl = threadIdx.x & (BNC - 1);
  const int rg = threadIdx.x >> 6;       // row group 0..3
  const int c = blockIdx.x * BNC + cl;
  const bool ok = c < C;
  float s = 0.0f, s2 = 0.0f;
  if (ok) {
    for (long r = (long)blockIdx.y * 4 + rg; r < rows;
         r += (long)gridDim.y * 4) {
      const float v = x[r * C + c];
      s += v;
      s2 += v * v;
    }
  }
  __shared__ float ls[4][BNC], ls2[4][BNC];
  ls[rg][cl] = s;
  ls2[rg][cl] = s2;
  __syncthreads();
  if (rg == 0 && ok) {
    s = ls[0][cl] + ls[1][cl] + ls[2][cl] + ls[3][cl];
    s2 = ls2[0][cl] + ls2[1][cl] + ls2[2][cl] + ls2[3][cl];
    atomicAdd(&stats[c], s);
    atomicAdd(&stats[C + c], s2);
  }
}

__global__ void bn_nhwc_norm_kernel(const float* __restrict__ x,
                                    float* __restrict__ y,
                                    const float* __restrict__ stats,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    float* __restrict__ save_mean,
                                    float* __restrict__ save_invstd,
                                    float* __restrict__ run_mean,
                                    float* __restrict__ run_var,
                                    float momentum, float eps, long rows,
                                    int C, int relu) {
  const int cl = threadIdx.x & (BNC - 1);
  const int rg = threadIdx.x >> 6;
  const int c = blockIdx.x * BNC + cl;
  if (c >= C) return;
  const float cnt = (float)rows;
  const float mean = stats[c] / cnt;
  const float var = stats[C + c] / cnt - mean * mean;
  const float invstd = __frsqrt_rn(var + eps);
  const float g = gamma ? gamma[c] : 1.0f;
  const float b = beta ? beta[c] : 0.0f;
  const float scale = g * invstd;
  const float shift = b - mean * scale;
  if (blockIdx.y == 0 && rg == 0) {
    save_mean[c] = mean;
    save_invstd[c] = invstd;
    if (run_mean) {  // torch semantics: unbiased var into running stats
      run_mean[c] = (1.0f - momentum) * run_mean[c] + momentum * mean;
      const float ub = var * cnt / fmaxf(cnt - 1.0f, 1.0f);
      run_var[c] = (1.0f - momentum) * run_var[c] + momentum * ub;
    }
  }
  for (long r = (long)blockIdx.y * 4 + rg; r < rows;
       r += (long)gridDim.y * 4) {
    const float v = x[r * C + c] * scale + shift;
    y[r * C + c] = relu ? fmaxf(v, 0.0f) : v;
  }
}

__global__ void bn_nhwc_bwd_reduce_kernel(const float* __restrict__ x,
                                          const float* __restrict__ dy,
                                          const float* __restrict__ y,
                                          const float* __restrict__ sm,
                                          const float* __restrict__ si,
                                          float* __restrict__ red, long rows,
                                          int C, int relu) {
  const int cl = threadIdx.x & (BNC - 1);
  const int rg = threadIdx.x >> 6;
  const int c = blockIdx.x * BNC + cl;
  const bool ok = c < C;
  float sd = 0.0f, sdx = 0.0f;
  if (ok) {
    const float mean = sm[c], invstd = si[c];
    for (long r = (long)blockIdx.y * 4 + rg; r < rows;
         r += (long)gridDim.y * 4) {
      float d = dy[r * C + c];
      if (relu && y[r * C + c] <= 0.0f) d = 0.0f;
      sd += d;
      sdx += d * (x[r * C + c] - mean) * invstd;
    }
  }
  __shared__ float ls[4][BNC], ls2[4][BNC];
  ls[rg][cl] = sd;
  ls2[rg][cl] = sdx;
  __syncthreads();
  if (rg == 0 && ok) {
    sd = ls[0][cl] + ls[1][cl] + ls[2][cl] + ls[3][cl];
    sdx = ls2[0][cl] + ls2[1][cl] + ls2[2][cl] + ls2[3][cl];
    atomicAdd(&red[c], sd);
    atomicAdd(&red[C + c], sdx);
  }
}

__global__ void bn_nhwc_bwd_dx_kernel(const float* __restrict__ x,
                                      const float* __restrict__ dy,
                                      const float* __restrict__ y,
                                      float* __restrict__ dx,
                                      const float* __restrict__ sm,
                                      const float* __restrict__ si,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ red,
                                      long rows, int C, int relu) {
  const int cl = threadIdx.x & (BNC - 1);
  const int rg = threadIdx.x >> 6;
  const int c = blockIdx.x * BNC + cl;
  if (c >= C) return;
  const float mean = sm[c], invstd = si[c];
  const float g = gamma ? gamma[c] : 1.0f;
  const float cnt = (float)rows;
  const float mean_dy = red[c] / cnt;
  const float mean_dyx = red[C + c] / cnt;
  const float scale = g * invstd;
  for (long r = (long)blockIdx.y * 4 + rg; r < rows;
       r += (long)gridDim.y * 4) {
    float d = dy[r * C + c];
    if (relu && y[r * C + c] <= 0.0f) d = 0.0f;
    const float xhat = (x[r * C + c] - mean) * invstd;
    dx[r * C + c] = scale * (d - mean_dy - xhat * mean_dyx);
  }
}

static inline dim3 bn_nhwc_grid(long rows, int C) {
  const int cgrids = (C + BNC - 1) / BNC;
  long chunks = (rows + 3) / 4;
  chunks = std::min(chunks, (long)std::max(1, 2048 / cgrids));
  return dim3(cgrids, (int)chunks);
}

std::vector<torch::Tensor> bn2d_nhwc_fwd(torch::Tensor x,
                                         torch::Tensor gamma,
                                         torch::Tensor beta,
                                         torch::Tensor run_mean,
                                         torch::Tensor run_var,
                                         double momentum, double eps,
                                         bool relu) {
  check_f32n(x, "x");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto y = torch::empty_like(x);
  auto stats = torch::zeros({2 * C}, x.options());
  auto save_mean = torch::empty({C}, x.options());
  auto save_invstd = torch::empty({C}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 grid = bn_nhwc_grid(rows, C);
  hipLaunchKernelGGL(bn_nhwc_stats_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), stats.data_ptr<float>(), rows, C);
  hipLaunchKernelGGL(
      bn_nhwc_norm_kernel, grid, dim3(256), 0, stream, x.data_ptr<float>(),
      y.data_ptr<float>(), stats.data_ptr<float>(),
      gamma.numel() ? gamma.data_ptr<float>() : nullptr,
      beta.numel() ? beta.data_ptr<float>() : nullptr,
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
      run_mean.numel() ? run_mean.data_ptr<float>() : nullptr,
      run_var.numel() ? run_var.data_ptr<float>() : nullptr,
      (float)momentum, (float)eps, rows, C, (int)relu);
  return {y, save_mean, save_invstd};
}

std::vector<torch::Tensor> bn2d_nhwc_bwd(torch::Tensor x, torch::Tensor dy,
                                         torch::Tensor y,
                                         torch::Tensor gamma,
                                         torch::Tensor save_mean,
                                         torch::Tensor save_invstd,
                                         bool relu) {
  check_f32n(x, "x");
  check_f32n(dy, "dy");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto red = torch::zeros({2 * C}, x.options());
  auto dx = torch::empty_like(x);
  auto stream = at::cuda::getCurrentCUDAStream();
  const dim3 grid = bn_nhwc_grid(rows, C);
  const float* yp = relu ? y.data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(bn_nhwc_bwd_reduce_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), dy.data_ptr<float>(), yp,
                     save_mean.data_ptr<float>(),
                     save_invstd.data_ptr<float>(), red.data_ptr<float>(),
                     rows, C, (int)relu);
  hipLaunchKernelGGL(
      bn_nhwc_bwd_dx_kernel, grid, dim3(256), 0, stream,
      x.data_ptr<float>(), dy.data_ptr<float>(), yp, dx.data_ptr<float>(),
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
      gamma.numel() ? gamma.data_ptr<float>() : nullptr,
      red.data_ptr<float>(), rows, C, (int)relu);
  return {dx, red.narrow(0, C, C), red.narrow(0, 0, C)};
}
