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

// float4 variants (C % 4 == 0 — every ResNet/CNN channel count): 16
// channel-quads x 16 row groups per 256-thread block, so a wave's row
// read is 16 lanes x float4 = 64 consecutive channels (full-width
// transactions; the scalar kernels below move one float per lane and
// measured ~2x off the stream floor on the ResNet shapes).  The norm
// kernel optionally FUSES the residual add + trailing ReLU of a ResNet
// block (out = relu(bn(x) + res)), and the bwd-dx kernel then also
// emits d_res (the relu-masked dy) so no separate torch add/relu
// kernels run on the block boundary.

__global__ void bn_nhwc_stats_v4_kernel(const float* __restrict__ x,
                                        float* __restrict__ stats,
                                        long rows, int C) {
  const int cl = threadIdx.x & 15;          // channel quad
  const int rg = threadIdx.x >> 4;          // row group 0..15
  const int c = blockIdx.x * BNC + cl * 4;
  const bool ok = c < C;
  float4 s = {0, 0, 0, 0}, s2 = {0, 0, 0, 0};
  if (ok) {
    for (long r = (long)blockIdx.y * 16 + rg; r < rows;
         r += (long)gridDim.y * 16) {
      const float4 v = *reinterpret_cast<const float4*>(&x[r * C + c]);
      s.x += v.x; s.y += v.y; s.z += v.z; s.w += v.w;
      s2.x += v.x * v.x; s2.y += v.y * v.y;
      s2.z += v.z * v.z; s2.w += v.w * v.w;
    }
  }
  __shared__ float4 ls[16][16], ls2[16][16];
  ls[rg][cl] = s;
  ls2[rg][cl] = s2;
  __syncthreads();
  if (rg == 0 && ok) {
#pragma unroll
    for (int k = 1; k < 16; ++k) {
      const float4 a = ls[k][cl], b = ls2[k][cl];
      s.x += a.x; s.y += a.y; s.z += a.z; s.w += a.w;
      s2.x += b.x; s2.y += b.y; s2.z += b.z; s2.w += b.w;
    }
    atomicAdd(&stats[c + 0], s.x);
    atomicAdd(&stats[c + 1], s.y);
    atomicAdd(&stats[c + 2], s.z);
    atomicAdd(&stats[c + 3], s.w);
    atomicAdd(&stats[C + c + 0], s2.x);
    atomicAdd(&stats[C + c + 1], s2.y);
    atomicAdd(&stats[C + c + 2], s2.z);
    atomicAdd(&stats[C + c + 3], s2.w);
  }
}

__global__ void bn_nhwc_norm_v4_kernel(
    const float* __restrict__ x, float* __restrict__ y,
    const float* __restrict__ stats, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ save_mean,
    float* __restrict__ save_invstd, float* __restrict__ run_mean,
    float* __restrict__ run_var, const float* __restrict__ res,
    float momentum, float eps, long rows, int C, int relu) {
  const int cl = threadIdx.x & 15;
  const int rg = threadIdx.x >> 4;
  const int c = blockIdx.x * BNC + cl * 4;
  if (c >= C) return;
  const float cnt = (float)rows;
  float4 scale, shift;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const float mean = stats[c + j] / cnt;
    const float var = stats[C + c + j] / cnt - mean * mean;
    const float invstd = __frsqrt_rn(var + eps);
    const float g = gamma ? gamma[c + j] : 1.0f;
    const float b = beta ? beta[c + j] : 0.0f;
    reinterpret_cast<float*>(&scale)[j] = g * invstd;
    reinterpret_cast<float*>(&shift)[j] = b - mean * (g * invstd);
    if (blockIdx.y == 0 && rg == 0) {
      save_mean[c + j] = mean;
      save_invstd[c + j] = invstd;
      if (run_mean) {
        run_mean[c + j] =
            (1.0f - momentum) * run_mean[c + j] + momentum * mean;
        const float ub = var * cnt / fmaxf(cnt - 1.0f, 1.0f);
        run_var[c + j] = (1.0f - momentum) * run_var[c + j] + momentum * ub;
      }
    }
  }
  for (long r = (long)blockIdx.y * 16 + rg; r < rows;
       r += (long)gridDim.y * 16) {
    float4 v = *reinterpret_cast<const float4*>(&x[r * C + c]);
    v.x = v.x * scale.x + shift.x;
    v.y = v.y * scale.y + shift.y;
    v.z = v.z * scale.z + shift.z;
    v.w = v.w * scale.w + shift.w;
    if (res) {
      const float4 rv = *reinterpret_cast<const float4*>(&res[r * C + c]);
      v.x += rv.x; v.y += rv.y; v.z += rv.z; v.w += rv.w;
    }
    if (relu) {
      v.x = fmaxf(v.x, 0.0f);
      v.y = fmaxf(v.y, 0.0f);
      v.z = fmaxf(v.z, 0.0f);
      v.w = fmaxf(v.w, 0.0f);
    }
    *reinterpret_cast<float4*>(&y[r * C + c]) = v;
  }
}

__global__ void bn_nhwc_bwd_reduce_v4_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, const float* __restrict__ sm,
    const float* __restrict__ si, float* __restrict__ red, long rows,
    int C, int relu) {
  const int cl = threadIdx.x & 15;
  const int rg = threadIdx.x >> 4;
  const int c = blockIdx.x * BNC + cl * 4;
  const bool ok = c < C;
  float4 sd = {0, 0, 0, 0}, sdx = {0, 0, 0, 0};
  if (ok) {
    const float4 mean = *reinterpret_cast<const float4*>(&sm[c]);
    const float4 invstd = *reinterpret_cast<const float4*>(&si[c]);
    for (long r = (long)blockIdx.y * 16 + rg; r < rows;
         r += (long)gridDim.y * 16) {
      float4 d = *reinterpret_cast<const float4*>(&dy[r * C + c]);
      if (relu) {
        const float4 yv = *reinterpret_cast<const float4*>(&y[r * C + c]);
        if (yv.x <= 0.0f) d.x = 0.0f;
        if (yv.y <= 0.0f) d.y = 0.0f;
        if (yv.z <= 0.0f) d.z = 0.0f;
        if (yv.w <= 0.0f) d.w = 0.0f;
      }
      const float4 xv = *reinterpret_cast<const float4*>(&x[r * C + c]);
      sd.x += d.x; sd.y += d.y; sd.z += d.z; sd.w += d.w;
      sdx.x += d.x * (xv.x - mean.x) * invstd.x;
      sdx.y += d.y * (xv.y - mean.y) * invstd.y;
      sdx.z += d.z * (xv.z - mean.z) * invstd.z;
      sdx.w += d.w * (xv.w - mean.w) * invstd.w;
    }
  }
  __shared__ float4 ls[16][16], ls2[16][16];
  ls[rg][cl] = sd;
  ls2[rg][cl] = sdx;
  __syncthreads();
  if (rg == 0 && ok) {
#pragma unroll
    for (int k = 1; k < 16; ++k) {
      const float4 a = ls[k][cl], b = ls2[k][cl];
      sd.x += a.x; sd.y += a.y; sd.z += a.z; sd.w += a.w;
      sdx.x += b.x; sdx.y += b.y; sdx.z += b.z; sdx.w += b.w;
    }
    atomicAdd(&red[c + 0], sd.x);
    atomicAdd(&red[c + 1], sd.y);
    atomicAdd(&red[c + 2], sd.z);
    atomicAdd(&red[c + 3], sd.w);
    atomicAdd(&red[C + c + 0], sdx.x);
    atomicAdd(&red[C + c + 1], sdx.y);
    atomicAdd(&red[C + c + 2], sdx.z);
    atomicAdd(&red[C + c + 3], sdx.w);
  }
}

__global__ void bn_nhwc_bwd_dx_v4_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    const float* __restrict__ y, float* __restrict__ dx,
    const float* __restrict__ sm, const float* __restrict__ si,
    const float* __restrict__ gamma, const float* __restrict__ red,
    float* __restrict__ dres, long rows, int C, int relu) {
  const int cl = threadIdx.x & 15;
  const int rg = threadIdx.x >> 4;
  const int c = blockIdx.x * BNC + cl * 4;
  if (c >= C) return;
  const float4 mean = *reinterpret_cast<const float4*>(&sm[c]);
  const float4 invstd = *reinterpret_cast<const float4*>(&si[c]);
  const float cnt = (float)rows;
  float4 scale, mdy, mdyx;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const float g = gamma ? gamma[c + j] : 1.0f;
    reinterpret_cast<float*>(&scale)[j] =
        g * reinterpret_cast<const float*>(&invstd)[j];
    reinterpret_cast<float*>(&mdy)[j] = red[c + j] / cnt;
    reinterpret_cast<float*>(&mdyx)[j] = red[C + c + j] / cnt;
  }
  for (long r = (long)blockIdx.y * 16 + rg; r < rows;
       r += (long)gridDim.y * 16) {
    float4 d = *reinterpret_cast<const float4*>(&dy[r * C + c]);
    if (relu) {
      const float4 yv = *reinterpret_cast<const float4*>(&y[r * C + c]);
      if (yv.x <= 0.0f) d.x = 0.0f;
      if (yv.y <= 0.0f) d.y = 0.0f;
      if (yv.z <= 0.0f) d.z = 0.0f;
      if (yv.w <= 0.0f) d.w = 0.0f;
    }
    if (dres) *reinterpret_cast<float4*>(&dres[r * C + c]) = d;
    const float4 xv = *reinterpret_cast<const float4*>(&x[r * C + c]);
    float4 o;
    o.x = scale.x * (d.x - mdy.x - (xv.x - mean.x) * invstd.x * mdyx.x);
    o.y = scale.y * (d.y - mdy.y - (xv.y - mean.y) * invstd.y * mdyx.y);
    o.z = scale.z * (d.z - mdy.z - (xv.z - mean.z) * invstd.z * mdyx.z);
    o.w = scale.w * (d.w - mdy.w - (xv.w - mean.w) * invstd.w * mdyx.w);
    *reinterpret_cast<float4*>(&dx[r * C + c]) = o;
  }
}

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
                                    const float* __restrict__ res,
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
    float v = x[r * C + c] * scale + shift;
    if (res) v += res[r * C + c];
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
                                      float* __restrict__ dres,
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
    if (dres) dres[r * C + c] = d;
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

static inline dim3 bn_nhwc_grid_v4(long rows, int C) {
  const int cgrids = (C + BNC - 1) / BNC;
  long chunks = (rows + 15) / 16;
  chunks = std::min(chunks, (long)std::max(1, 2048 / cgrids));
  return dim3(cgrids, (int)chunks);
}

std::vector<torch::Tensor> bn2d_nhwc_fwd(torch::Tensor x,
                                         torch::Tensor gamma,
                                         torch::Tensor beta,
                                         torch::Tensor run_mean,
                                         torch::Tensor run_var,
                                         double momentum, double eps,
                                         bool relu, torch::Tensor res) {
  check_f32n(x, "x");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  // MEASURED (round 2, benchmarks/bn_micro.py): the scalar per-lane
  // kernels stream at 3.0-3.5 TB/s effective on the ResNet shapes, the
  // float4 v4 variants only ~2.0 — the scalar layout (one wave = 64
  // consecutive channels of one row) wins; v4 kept behind
  // SPARKFLOW_BN_V4=1 for re-evaluation.  The scalar kernels carry the
  // fused residual too.
  static const bool v4_enabled = [] {
    const char* e = getenv("SPARKFLOW_BN_V4");
    return e && e[0] == '1';
  }();
  const bool v4 = (C & 3) == 0 && v4_enabled;
  TORCH_CHECK(res.numel() == 0 || res.numel() == x.numel(),
              "fused residual needs matching shape");
  auto y = torch::empty_like(x);
  auto stats = torch::zeros({2 * C}, x.options());
  auto save_mean = torch::empty({C}, x.options());
  auto save_invstd = torch::empty({C}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* gp = gamma.numel() ? gamma.data_ptr<float>() : nullptr;
  const float* bp = beta.numel() ? beta.data_ptr<float>() : nullptr;
  float* rmp = run_mean.numel() ? run_mean.data_ptr<float>() : nullptr;
  float* rvp = run_var.numel() ? run_var.data_ptr<float>() : nullptr;
  if (v4) {
    const dim3 grid = bn_nhwc_grid_v4(rows, C);
    hipLaunchKernelGGL(bn_nhwc_stats_v4_kernel, grid, dim3(256), 0, stream,
                       x.data_ptr<float>(), stats.data_ptr<float>(), rows,
                       C);
    hipLaunchKernelGGL(
        bn_nhwc_norm_v4_kernel, grid, dim3(256), 0, stream,
        x.data_ptr<float>(), y.data_ptr<float>(), stats.data_ptr<float>(),
        gp, bp, save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(),
        rmp, rvp, res.numel() ? res.data_ptr<float>() : nullptr,
        (float)momentum, (float)eps, rows, C, (int)relu);
    return {y, save_mean, save_invstd};
  }
  const dim3 grid = bn_nhwc_grid(rows, C);
  hipLaunchKernelGGL(bn_nhwc_stats_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), stats.data_ptr<float>(), rows, C);
  hipLaunchKernelGGL(
      bn_nhwc_norm_kernel, grid, dim3(256), 0, stream, x.data_ptr<float>(),
      y.data_ptr<float>(), stats.data_ptr<float>(), gp, bp,
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(), rmp, rvp,
      res.numel() ? res.data_ptr<float>() : nullptr,
      (float)momentum, (float)eps, rows, C, (int)relu);
  return {y, save_mean, save_invstd};
}

std::vector<torch::Tensor> bn2d_nhwc_bwd(torch::Tensor x, torch::Tensor dy,
                                         torch::Tensor y,
                                         torch::Tensor gamma,
                                         torch::Tensor save_mean,
                                         torch::Tensor save_invstd,
                                         bool relu, bool want_dres) {
  check_f32n(x, "x");
  check_f32n(dy, "dy");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  static const bool v4_enabled = [] {
    const char* e = getenv("SPARKFLOW_BN_V4");
    return e && e[0] == '1';
  }();
  const bool v4 = (C & 3) == 0 && v4_enabled;
  auto red = torch::zeros({2 * C}, x.options());
  auto dx = torch::empty_like(x);
  auto dres = want_dres ? torch::empty_like(x) : torch::empty({0}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* yp = relu ? y.data_ptr<float>() : nullptr;
  const float* gp = gamma.numel() ? gamma.data_ptr<float>() : nullptr;
  if (v4) {
    const dim3 grid = bn_nhwc_grid_v4(rows, C);
    hipLaunchKernelGGL(bn_nhwc_bwd_reduce_v4_kernel, grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       yp, save_mean.data_ptr<float>(),
                       save_invstd.data_ptr<float>(), red.data_ptr<float>(),
                       rows, C, (int)relu);
    hipLaunchKernelGGL(
        bn_nhwc_bwd_dx_v4_kernel, grid, dim3(256), 0, stream,
        x.data_ptr<float>(), dy.data_ptr<float>(), yp, dx.data_ptr<float>(),
        save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(), gp,
        red.data_ptr<float>(),
        want_dres ? dres.data_ptr<float>() : nullptr, rows, C, (int)relu);
    return {dx, red.narrow(0, C, C), red.narrow(0, 0, C), dres};
  }
  const dim3 grid = bn_nhwc_grid(rows, C);
  hipLaunchKernelGGL(bn_nhwc_bwd_reduce_kernel, grid, dim3(256), 0, stream,
                     x.data_ptr<float>(), dy.data_ptr<float>(), yp,
                     save_mean.data_ptr<float>(),
                     save_invstd.data_ptr<float>(), red.data_ptr<float>(),
                     rows, C, (int)relu);
  hipLaunchKernelGGL(
      bn_nhwc_bwd_dx_kernel, grid, dim3(256), 0, stream,
      x.data_ptr<float>(), dy.data_ptr<float>(), yp, dx.data_ptr<float>(),
      save_mean.data_ptr<float>(), save_invstd.data_ptr<float>(), gp,
      red.data_ptr<float>(),
      want_dres ? dres.data_ptr<float>() : nullptr, rows, C, (int)relu);
  return {dx, red.narrow(0, C, C), red.narrow(0, 0, C), dres};
}
