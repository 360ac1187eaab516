// conv2d (NCHW, fp32) + maxpool kernels — the reference's implicit TF
// conv surface (reference examples/cnn_example.py:13-21).
//
// v1 strategy: direct convolution with fused bias+activation (one thread
// per output element, grid-stride), weight-element-per-block reductions
// for wgrad, argmax-index maxpool.  Correctness-first; the MFMA
// implicit-GEMM path is the planned optimization for the ResNet-18-shape
// config (BASELINE.json config 5).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// ---- forward: y[n,co,ho,wo] = act(sum_{ci,kh,kw} x * w + b) ----
__global__ void conv2d_fwd_kernel(const float* __restrict__ x,
                                  const float* __restrict__ w,
                                  const float* __restrict__ bias,
                                  float* __restrict__ y, int N, int Cin,
                                  int H, int W, int Cout, int Kh, int Kw,
                                  int Ho, int Wo, int stride, int pad,
                                  int act, int has_bias) {
  const long total = (long)N * Cout * Ho * Wo;
  const long gstride = (long)gridDim.x * blockDim.x;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gstride) {
    const int wo = idx % Wo;
    const int ho = (idx / Wo) % Ho;
    const int co = (idx / ((long)Wo * Ho)) % Cout;
    const int n = idx / ((long)Wo * Ho * Cout);

    float s = has_bias ? bias[co] : 0.0f;
    const int h0 = ho * stride - pad;
    const int w0 = wo * stride - pad;
    for (int ci = 0; ci < Cin; ++ci) {
      const float* xp = x + (((long)n * Cin + ci) * H) * W;
      const float* wp = w + (((long)co * Cin + ci) * Kh) * Kw;
      for (int kh = 0; kh < Kh; ++kh) {
        const int h = h0 + kh;
        if (h < 0 || h >= H) continue;
        for (int kw = 0; kw < Kw; ++kw) {
          const int ww = w0 + kw;
          if (ww < 0 || ww >= W) continue;
          s += xp[(long)h * W + ww] * wp[kh * Kw + kw];
        }
      }
    }
    y[idx] = act_apply(s, act);
  }
}

// ---- dgrad: dx[n,ci,h,w] = sum_{co,kh,kw aligned} dy * w ----
__global__ void conv2d_dgrad_kernel(const float* __restrict__ dy,
                                    const float* __restrict__ w,
                                    float* __restrict__ dx, int N, int Cin,
                                    int H, int W, int Cout, int Kh, int Kw,
                                    int Ho, int Wo, int stride, int pad) {
  const long total = (long)N * Cin * H * W;
  const long gstride = (long)gridDim.x * blockDim.x;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gstride) {
    const int ww = idx % W;
    const int h = (idx / W) % H;
    const int ci = (idx / ((long)W * H)) % Cin;
    const int n = idx / ((long)W * H * Cin);

    float s = 0.0f;
    for (int kh = 0; kh < Kh; ++kh) {
      const int hp = h + pad - kh;
      if (hp < 0 || hp % stride) continue;
      const int ho = hp / stride;
      if (ho >= Ho) continue;
      for (int kw = 0; kw < Kw; ++kw) {
        const int wp = ww + pad - kw;
        if (wp < 0 || wp % stride) continue;
        const int wo = wp / stride;
        if (wo >= Wo) continue;
        for (int co = 0; co < Cout; ++co) {
          s += dy[(((long)n * Cout + co) * Ho + ho) * Wo + wo] *
               w[(((long)co * Cin + ci) * Kh + kh) * Kw + kw];
        }
      }
    }
    dx[idx] = s;
  }
}

// ---- wgrad: one 256-thread block per filter element, reduction over
// (n, ho, wo) staged through wave shuffles + LDS ----
__global__ void conv2d_wgrad_kernel(const float* __restrict__ x,
                                    const float* __restrict__ dy,
                                    float* __restrict__ dw, int N, int Cin,
                                    int H, int W, int Cout, int Kh, int Kw,
                                    int Ho, int Wo, int stride, int pad) {
  const long widx = blockIdx.x;  // (co, ci, kh, kw)
  const int kw = widx % Kw;
  const int kh = (widx / Kw) % Kh;
  const int ci = (widx / (Kw * Kh)) % Cin;
  const int co = widx / ((long)Kw * Kh * Cin);

  const long total = (long)N * Ho * Wo;
  float s = 0.0f;
  for (long i = threadIdx.x; i < total; i += blockDim.x) {
    const int wo = i % Wo;
    const int ho = (i / Wo) % Ho;
    const int n = i / ((long)Wo * Ho);
    const int h = ho * stride - pad + kh;
    const int ww = wo * stride - pad + kw;
    if (h < 0 || h >= H || ww < 0 || ww >= W) continue;
    s += x[(((long)n * Cin + ci) * H + h) * W + ww] *
         dy[(((long)n * Cout + co) * Ho + ho) * Wo + wo];
  }
  s = wave_sum(s);
  __shared__ float partials[4];
  if ((threadIdx.x & (WAVE - 1)) == 0) partials[threadIdx.x / WAVE] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = 0.0f;
    for (int i = 0; i < (int)(blockDim.x / WAVE); ++i) b += partials[i];
    dw[widx] = b;
  }
}

// db[c] = sum over (n, hw) of dy[n][c][hw].  blockIdx.y chunks the batch
// so small-C convs still fill the chip; contiguous hw sweeps per (n, c)
// plane (no div/mod in the loop); partial sums combine via atomicAdd.
__global__ void conv_bias_grad_kernel(const float* __restrict__ dy,
                                      float* __restrict__ db, int N, int C,
                                      int HW) {
  const int c = blockIdx.x;
  float s = 0.0f;
  for (int n = blockIdx.y; n < N; n += gridDim.y) {
    const float* p = dy + ((long)n * C + c) * HW;
    for (int i = threadIdx.x; i < HW; i += blockDim.x) s += p[i];
  }
  s = wave_sum(s);
  __shared__ float partials[4];
  if ((threadIdx.x & (WAVE - 1)) == 0) partials[threadIdx.x / WAVE] = s;
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = 0.0f;
    for (int i = 0; i < (int)(blockDim.x / WAVE); ++i) b += partials[i];
    if (gridDim.y == 1) {
      db[c] = b;
    } else {
      atomicAdd(&db[c], b);
    }
  }
}

// ---- maxpool with argmax indices (bwd scatters through them) ----
// blockIdx.x owns one (n, c) plane, blockIdx.y chunks ho, threads sweep
// wo — every coordinate is loop state, no div/mod anywhere in the sweep
// (the flat-index decode was ~20x off the bandwidth bound on the
// ResNet-shape 112x112 pool).
__global__ void maxpool_fwd_kernel(const float* __restrict__ x,
                                   float* __restrict__ y,
                                   long* __restrict__ idx, int N, int C,
                                   int H, int W, int K, int stride, int pad,
                                   int Ho, int Wo) {
  const long plane = blockIdx.x;  // n * C + c
  const float* xp = x + plane * H * W;
  float* yp = y + plane * Ho * Wo;
  long* ip = idx + plane * Ho * Wo;
  for (int ho = blockIdx.y; ho < Ho; ho += gridDim.y) {
    const int h0 = ho * stride - pad;
    for (int wo = threadIdx.x; wo < Wo; wo += blockDim.x) {
      const int w0 = wo * stride - pad;
      float best = -INFINITY;
      long best_hw = 0;
      for (int kh = 0; kh < K; ++kh) {
        const int h = h0 + kh;
        if (h < 0) continue;
        if (h >= H) break;
        for (int kw = 0; kw < K; ++kw) {
          const int ww = w0 + kw;
          if (ww < 0) continue;
          if (ww >= W) break;
          const float v = xp[(long)h * W + ww];
          if (v > best) {
            best = v;
            best_hw = (long)h * W + ww;
          }
        }
      }
      yp[(long)ho * Wo + wo] = best;
      // torch max_pool2d index convention: within the plane
      ip[(long)ho * Wo + wo] = best_hw;
    }
  }
}

// Flat-index variant for narrow planes (Wo < wavefront): the div/mod per
// element costs less than idling most lanes of a row-sweep block.
__global__ void maxpool_fwd_flat_kernel(const float* __restrict__ x,
                                        float* __restrict__ y,
                                        long* __restrict__ idx, int N, int C,
                                        int H, int W, int K, int stride,
                                        int pad, int Ho, int Wo) {
  const long total = (long)N * C * Ho * Wo;
  const long gstride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += gstride) {
    const int wo = i % Wo;
    const long pc = i / Wo;         // (n*C+c)*Ho + ho
    const int ho = pc % Ho;
    const long plane = pc / Ho;     // n*C + c
    const float* xp = x + plane * H * W;
    float best = -INFINITY;
    long best_hw = 0;
    for (int kh = 0; kh < K; ++kh) {
      const int h = ho * stride - pad + kh;
      if (h < 0) continue;
      if (h >= H) break;
      for (int kw = 0; kw < K; ++kw) {
        const int ww = wo * stride - pad + kw;
        if (ww < 0) continue;
        if (ww >= W) break;
        const float v = xp[(long)h * W + ww];
        if (v > best) {
          best = v;
          best_hw = (long)h * W + ww;
        }
      }
    }
    y[i] = best;
    idx[i] = best_hw;
  }
}

__global__ void maxpool_bwd_kernel(const float* __restrict__ dy,
                                   const long* __restrict__ idx,
                                   float* __restrict__ dx, int N, int C,
                                   int HW, int HoWo) {
  const long plane = blockIdx.x;  // n * C + c — no div/mod in the sweep
  const float* dyp = dy + plane * HoWo;
  const long* ip = idx + plane * HoWo;
  float* dxp = dx + plane * HW;
  const int span = gridDim.y * blockDim.x;
  for (int i = blockIdx.y * blockDim.x + threadIdx.x; i < HoWo; i += span) {
    atomicAdd(&dxp[ip[i]], dyp[i]);
  }
}

// ---------------------------------------------------------------------------

static inline void check_f32c(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous fp32 on GPU");
}

static inline int out_dim(int in, int k, int stride, int pad) {
  return (in + 2 * pad - k) / stride + 1;
}

torch::Tensor conv2d_act(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad, int64_t act) {
  check_f32c(x, "x");
  check_f32c(w, "w");
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w.size(0), Kh = w.size(2), Kw = w.size(3);
  TORCH_CHECK(w.size(1) == Cin, "conv channel mismatch");
  const int Ho = out_dim(H, Kh, stride, pad), Wo = out_dim(W, Kw, stride, pad);
  auto y = torch::empty({N, Cout, Ho, Wo}, x.options());
  const long total = y.numel();
  const int blocks = (int)std::min<long>(8192, (total + 255) / 256);
  hipLaunchKernelGGL(conv2d_fwd_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     w.data_ptr<float>(),
                     bias.numel() ? bias.data_ptr<float>() : nullptr,
                     y.data_ptr<float>(), N, Cin, H, W, Cout, Kh, Kw, Ho, Wo,
                     (int)stride, (int)pad, (int)act, bias.numel() > 0);
  return y;
}

torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w, int64_t H,
                           int64_t W, int64_t stride, int64_t pad) {
  check_f32c(dy, "dy");
  check_f32c(w, "w");
  const int N = dy.size(0), Cout = dy.size(1), Ho = dy.size(2),
            Wo = dy.size(3);
  const int Cin = w.size(1), Kh = w.size(2), Kw = w.size(3);
  auto dx = torch::empty({N, Cin, H, W}, dy.options());
  const long total = dx.numel();
  const int blocks = (int)std::min<long>(8192, (total + 255) / 256);
  hipLaunchKernelGGL(conv2d_dgrad_kernel, dim3(blocks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), dy.data_ptr<float>(),
                     w.data_ptr<float>(), dx.data_ptr<float>(), N, Cin,
                     (int)H, (int)W, Cout, Kh, Kw, Ho, Wo, (int)stride,
                     (int)pad);
  return dx;
}

torch::Tensor conv2d_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                           int64_t Kw, int64_t stride, int64_t pad) {
  check_f32c(x, "x");
  check_f32c(dy, "dy");
  const int N = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
  auto dw = torch::empty({Cout, Cin, Kh, Kw}, x.options());
  hipLaunchKernelGGL(conv2d_wgrad_kernel, dim3((int)dw.numel()), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                     dy.data_ptr<float>(), dw.data_ptr<float>(), N, Cin, H, W,
                     Cout, (int)Kh, (int)Kw, Ho, Wo, (int)stride, (int)pad);
  return dw;
}

// Zero-stuff dy for the strided-dgrad-as-convolution trick: out[...,
// i*s, j*s] = dy[..., i, j] in a zeroed [N, C, Hd, Wd] buffer (extra
// trailing rows/cols cover the stride remainder).  The dilated tensor
// then feeds the fast implicit-GEMM forward with flipped weights — the
// direct strided dgrad kernel was 25% of the ResNet-shape step.
__global__ void dilate_kernel(const float* __restrict__ dy,
                              float* __restrict__ out, int Ho, int Wo,
                              int s, long plane_out, int Wd) {
  const long plane = blockIdx.x;  // n * C + c
  const float* src = dy + plane * (long)Ho * Wo;
  float* dst = out + plane * plane_out;
  for (int ho = blockIdx.y; ho < Ho; ho += gridDim.y) {
    for (int wo = threadIdx.x; wo < Wo; wo += blockDim.x) {
      dst[((long)ho * s) * Wd + (long)wo * s] = src[(long)ho * Wo + wo];
    }
  }
}

torch::Tensor conv_dilate(torch::Tensor dy, int64_t s, int64_t extra_h,
                          int64_t extra_w) {
  check_f32c(dy, "dy");
  const int N = dy.size(0), C = dy.size(1), Ho = dy.size(2),
            Wo = dy.size(3);
  const long Hd = (long)(Ho - 1) * s + 1 + extra_h;
  const long Wd = (long)(Wo - 1) * s + 1 + extra_w;
  auto out = torch::zeros({N, C, Hd, Wd}, dy.options());
  const int chunks = std::max(1, std::min(Ho, 2048 / (N * C) + 1));
  hipLaunchKernelGGL(dilate_kernel, dim3(N * C, chunks),
                     dim3(std::min(256, std::max(64, Wo))), 0,
                     at::cuda::getCurrentCUDAStream(), dy.data_ptr<float>(),
                     out.data_ptr<float>(), Ho, Wo, (int)s, Hd * Wd,
                     (int)Wd);
  return out;
}

torch::Tensor conv_bias_grad(torch::Tensor dy) {
  check_f32c(dy, "dy");
  const int N = dy.size(0), C = dy.size(1), HW = dy.size(2) * dy.size(3);
  const int nchunks = std::min(N, std::max(1, 512 / C));
  auto db = nchunks > 1 ? torch::zeros({C}, dy.options())
                        : torch::empty({C}, dy.options());
  hipLaunchKernelGGL(conv_bias_grad_kernel, dim3(C, nchunks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), dy.data_ptr<float>(),
                     db.data_ptr<float>(), N, C, HW);
  return db;
}

std::vector<torch::Tensor> maxpool2d_fwd(torch::Tensor x, int64_t k,
                                         int64_t stride, int64_t pad) {
  check_f32c(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = (H + 2 * pad - k) / stride + 1;
  const int Wo = (W + 2 * pad - k) / stride + 1;
  auto y = torch::empty({N, C, Ho, Wo}, x.options());
  auto idx = torch::empty({N, C, Ho, Wo},
                          x.options().dtype(torch::kInt64));
  if (Wo >= 64) {  // wide planes: div-free row sweeps
    const int ho_chunks = std::max(1, std::min(Ho, 2048 / (N * C) + 1));
    hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(N * C, ho_chunks),
                       dim3(std::min(256, Wo)), 0,
                       at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                       y.data_ptr<float>(), idx.data_ptr<long>(), N, C, H,
                       W, (int)k, (int)stride, (int)pad, Ho, Wo);
  } else {  // narrow planes: flat sweep keeps all lanes busy
    const long total = y.numel();
    const int blocks = (int)std::min<long>(8192, (total + 255) / 256);
    hipLaunchKernelGGL(maxpool_fwd_flat_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), x.data_ptr<float>(),
                       y.data_ptr<float>(), idx.data_ptr<long>(), N, C, H,
                       W, (int)k, (int)stride, (int)pad, Ho, Wo);
  }
  return {y, idx};
}

torch::Tensor maxpool2d_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                            int64_t W) {
  check_f32c(dy, "dy");
  const int N = dy.size(0), C = dy.size(1);
  const int HoWo = dy.size(2) * dy.size(3);
  auto dx = torch::zeros({N, C, H, W}, dy.options());
  const int chunks = std::max(1, std::min((HoWo + 255) / 256,
                                          4096 / (N * C) + 1));
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(N * C, chunks), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), dy.data_ptr<float>(),
                     idx.data_ptr<long>(), dx.data_ptr<float>(), N, C,
                     (int)(H * W), HoWo);
  return dx;
}
