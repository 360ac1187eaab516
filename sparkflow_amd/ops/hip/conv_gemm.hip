// Implicit-GEMM convolution on the fp32 MFMA pipeline (im2col gathered at
// the LDS staging step, never materialized) — the MI355X-native conv for
// the CNN / ResNet-18-shape configs (SURVEY.md §2.2 "implicit-GEMM conv
// kernel (im2col-in-LDS)").
//
// GEMM views (all fp32, NCHW):
//   fwd:   C[M=N*Ho*Wo, N'=Cout]       = im2col(x) @ W^T          (+bias+act)
//   wgrad: C[M=Cout, N'=Cin*Kh*Kw]     = dyT @ im2col(x)
//   dgrad: runs as a stride-1 convolution of (zero-stuffed) dy with
//          flipped weights on this same fwd pipeline (conv_dilate in
//          conv.hip); conv.hip's direct dgrad kernel remains as the
//          tested reference.
//
// The MFMA inner loop is the same 64x64-tile / 16-deep structure as
// gemm.hip; only the A/B staging differs (indexed gathers).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"
#include "conv_reduce.h"

#define BM 64
#define BN 64
#define BK 16

struct ConvShape {
  int N, Cin, H, W, Cout, Kh, Kw, Ho, Wo, stride, pad;
};

// gather one im2col element: row m = output pixel (n,ho,wo),
// col k = (ci,kh,kw)
__device__ __forceinline__ float im2col_get(const float* __restrict__ x,
                                            const ConvShape s, int m,
                                            int k) {
  const int wo = m % s.Wo;
  const int ho = (m / s.Wo) % s.Ho;
  const int n = m / (s.Wo * s.Ho);
  const int kw = k % s.Kw;
  const int kh = (k / s.Kw) % s.Kh;
  const int ci = k / (s.Kw * s.Kh);
  const int h = ho * s.stride - s.pad + kh;
  const int w = wo * s.stride - s.pad + kw;
  if (h < 0 || h >= s.H || w < 0 || w >= s.W) return 0.0f;
  return x[(((long)n * s.Cin + ci) * s.H + h) * s.W + w];
}

// ---------------- forward ----------------
// A (im2col x): M x K, gathered.  B = W [Cout, K] -> B'[k][n']=W[n'][k].
template <bool HAS_BIAS>
__global__ __launch_bounds__(256) void conv_gemm_fwd_kernel(
    const float* __restrict__ x, const float* __restrict__ wgt,
    const float* __restrict__ bias, float* __restrict__ y, ConvShape s,
    int M, int Nc, int K, int act) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 32;
  const int wn = (wv & 1) * 32;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;

  f32x4 acc[2][2] = {};

  // Register-prefetched staging (next chunk's gathers overlap this
  // chunk's MFMAs — same pattern as gemm.hip).  The output pixel (n, ho,
  // wo) is FIXED per thread, so its decode happens once; the filter
  // coordinate advances incrementally (+BK with carry), leaving no
  // div/mod in the K loop.
  const int st_row = t >> 2, st_c4 = (t & 3) * 4;

  const int gm = m0 + st_row;
  const int m_dec = gm < M ? gm : 0;
  const int a_wo = m_dec % s.Wo;
  const int a_ho = (m_dec / s.Wo) % s.Ho;
  const long a_nbase = (long)(m_dec / (s.Wo * s.Ho)) * s.Cin * s.H * s.W;
  const int h0 = a_ho * s.stride - s.pad;  // + kh
  const int w0 = a_wo * s.stride - s.pad;  // + kw
  int a_kw = st_c4 % s.Kw;
  int a_kh = (st_c4 / s.Kw) % s.Kh;
  int a_ci = st_c4 / (s.Kw * s.Kh);

  auto load_a = [&](int k0) -> float4 {
    float4 r;
    float* rr = reinterpret_cast<float*>(&r);
    int kw = a_kw, kh = a_kh, ci = a_ci;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int h = h0 + kh, w = w0 + kw;
      float v = 0.0f;
      if (gm < M && k0 + st_c4 + i < K && h >= 0 && h < s.H && w >= 0 &&
          w < s.W) {
        v = x[a_nbase + ((long)ci * s.H + h) * s.W + w];
      }
      rr[i] = v;
      if (++kw == s.Kw) { kw = 0; if (++kh == s.Kh) { kh = 0; ++ci; } }
    }
    return r;
  };
  auto advance_a = [&]() {  // filter coordinate += BK, carry, no div
    a_kw += BK;
    while (a_kw >= s.Kw) {
      a_kw -= s.Kw;
      if (++a_kh >= s.Kh) { a_kh -= s.Kh; ++a_ci; }
    }
  };
  auto load_b = [&](int k0) -> float4 {
    float4 r;
    const int gn = n0 + st_row;
    if (gn < Nc && k0 + BK <= K) {
      r = *reinterpret_cast<const float4*>(&wgt[(long)gn * K + k0 + st_c4]);
    } else {
      float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int gk = k0 + st_c4 + i;
        rr[i] = (gn < Nc && gk < K) ? wgt[(long)gn * K + gk] : 0.0f;
      }
    }
    return r;
  };

  const int fr = lane & 15;
  const int fk = lane >> 4;

  float4 ra = load_a(0);
  float4 rb = load_b(0);
  for (int k0 = 0; k0 < K; k0 += BK) {
    {
      const float* pa = reinterpret_cast<const float*>(&ra);
      const float* pb = reinterpret_cast<const float*>(&rb);
#pragma unroll
      for (int i = 0; i < 4; ++i) As[st_row][st_c4 + i] = pa[i];
#pragma unroll
      for (int i = 0; i < 4; ++i) Bs[st_c4 + i][st_row] = pb[i];
    }
    __syncthreads();
    if (k0 + BK < K) {
      advance_a();
      ra = load_a(k0 + BK);
      rb = load_b(k0 + BK);
    }
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      const float a0 = As[wm + fr][kk * 4 + fk];
      const float a1 = As[wm + 16 + fr][kk * 4 + fk];
      const float b0 = Bs[kk * 4 + fk][wn + fr];
      const float b1 = Bs[kk * 4 + fk][wn + 16 + fr];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: y is NCHW, so output pixel m of channel n' scatters to
  // y[n][n'][ho][wo]
  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
  const int HoWo = s.Ho * s.Wo;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int co = n0 + wn + ni * 16 + fcol;
      if (co >= Nc) continue;
      const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = m0 + wm + mi * 16 + frow + i;
        if (m < M) {
          const int hw = m % HoWo;
          const int n = m / HoWo;
          y[((long)n * Nc + co) * HoWo + hw] =
              act_apply(acc[mi][ni][i] + bv, act);
        }
      }
    }
  }
}

// ---------------- wgrad ----------------
// C[M=Cout, N'=K'=Cin*Kh*Kw] = sum_p dy[p, co] * im2col(x)[p, k']
// A'[m=co][k=p]: dy NCHW gather; B'[k=p][n'=k']: im2col gather.
template <bool SPLITK>
__global__ __launch_bounds__(256) void conv_gemm_wgrad_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    float* __restrict__ dw, ConvShape s, int M, int Nc, int K,
    int k_chunk) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 32;
  const int wn = (wv & 1) * 32;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  const int HoWo = s.Ho * s.Wo;

  f32x4 acc[2][2] = {};

  int k_begin = 0, k_end = K;
  if (SPLITK) {
    k_begin = blockIdx.z * k_chunk;
    k_end = min(K, k_begin + k_chunk);
  }

  // The K (=pixel) loop is where a naive im2col gather pays 6+ integer
  // divisions per element; here every div/mod happens ONCE before the
  // loop and pixel coordinates advance incrementally (+BK with carry) —
  // measured 10x+ on the 28x28 MNIST conv wgrad.
  const int a_row = t >> 2, a_c4 = (t & 3) * 4;  // A: row = co idx in tile
  const int b_kk = t >> 4, b_n4 = (t & 15) * 4;  // B: im2col gather

  // ---- A-side pixel cursor (4 consecutive gp starting k_begin+a_c4) ----
  int a_n = (k_begin + a_c4) / HoWo;
  int a_hw = (k_begin + a_c4) % HoWo;
  const int gm = m0 + a_row;
  // ---- B-side pixel cursor (single gp = k_begin + b_kk) ----
  int b_n = (k_begin + b_kk) / HoWo;
  int b_hw = (k_begin + b_kk) % HoWo;
  int b_ho = b_hw / s.Wo;
  int b_wo = b_hw % s.Wo;
  // B's 4 filter positions (gk fixed for the whole loop): decode once.
  int kw_i[4], kh_i[4];
  long coff_i[4];
  bool gk_ok[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int gk = n0 + b_n4 + i;
    gk_ok[i] = gk < Nc;
    const int g = gk_ok[i] ? gk : 0;
    kw_i[i] = g % s.Kw;
    kh_i[i] = (g / s.Kw) % s.Kh;
    coff_i[i] = (long)(g / (s.Kw * s.Kh)) * s.H * s.W;
  }
  const long n_stride = (long)s.Cin * s.H * s.W;

  auto load_a = [&](int gp0) -> float4 {
    float4 r;
    float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int hw = a_hw + i, n = a_n;
      while (hw >= HoWo) { hw -= HoWo; ++n; }  // HoWo may be tiny
      rr[i] = (gm < M && gp0 + i < K)
                  ? dy[((long)n * M + gm) * HoWo + hw] : 0.0f;
    }
    return r;
  };
  auto load_b = [&](int gp0) -> float4 {
    float4 r;
    float* rr = reinterpret_cast<float*>(&r);
    const bool p_ok = gp0 < K;
    const long nbase = (long)b_n * n_stride;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      float v = 0.0f;
      if (p_ok && gk_ok[i]) {
        const int h = b_ho * s.stride - s.pad + kh_i[i];
        const int w = b_wo * s.stride - s.pad + kw_i[i];
        if (h >= 0 && h < s.H && w >= 0 && w < s.W)
          v = x[nbase + coff_i[i] + (long)h * s.W + w];
      }
      rr[i] = v;
    }
    return r;
  };
  auto advance = [&]() {  // both cursors move +BK pixels, carry, no div
    a_hw += BK;
    while (a_hw >= HoWo) { a_hw -= HoWo; ++a_n; }
    b_wo += BK;
    while (b_wo >= s.Wo) { b_wo -= s.Wo; ++b_ho; }
    while (b_ho >= s.Ho) { b_ho -= s.Ho; ++b_n; }
  };

  const int fr = lane & 15;
  const int fk = lane >> 4;

  float4 ra = load_a(k_begin + a_c4);
  float4 rb = load_b(k_begin + b_kk);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    {
      const float* pa = reinterpret_cast<const float*>(&ra);
      const float* pb = reinterpret_cast<const float*>(&rb);
#pragma unroll
      for (int i = 0; i < 4; ++i) As[a_row][a_c4 + i] = pa[i];
#pragma unroll
      for (int i = 0; i < 4; ++i) Bs[b_kk][b_n4 + i] = pb[i];
    }
    __syncthreads();
    if (k0 + BK < k_end) {
      advance();
      ra = load_a(k0 + BK + a_c4);
      rb = load_b(k0 + BK + b_kk);
    }
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      const float a0 = As[wm + fr][kk * 4 + fk];
      const float a1 = As[wm + 16 + fr][kk * 4 + fk];
      const float b0 = Bs[kk * 4 + fk][wn + fr];
      const float b1 = Bs[kk * 4 + fk][wn + 16 + fr];
      acc[0][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int kcol = n0 + wn + ni * 16 + fcol;
      if (kcol >= Nc) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int co = m0 + wm + mi * 16 + frow + i;
        if (co < M) {
          if (SPLITK) {
            dw[(long)blockIdx.z * M * Nc + (long)co * Nc + kcol] =
                acc[mi][ni][i];
          } else {
            dw[(long)co * Nc + kcol] = acc[mi][ni][i];
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------

static inline void check_f32g(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous fp32 on GPU");
}

torch::Tensor conv_gemm_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t act) {
  check_f32g(x, "x");
  check_f32g(w, "w");
  ConvShape s;
  s.N = x.size(0); s.Cin = x.size(1); s.H = x.size(2); s.W = x.size(3);
  s.Cout = w.size(0); s.Kh = w.size(2); s.Kw = w.size(3);
  s.stride = stride; s.pad = pad;
  s.Ho = (s.H + 2 * s.pad - s.Kh) / s.stride + 1;
  s.Wo = (s.W + 2 * s.pad - s.Kw) / s.stride + 1;
  const int M = s.N * s.Ho * s.Wo;
  const int Nc = s.Cout;
  const int K = s.Cin * s.Kh * s.Kw;
  auto y = torch::empty({s.N, s.Cout, s.Ho, s.Wo}, x.options());
  dim3 grid(ceil_div(Nc, BN), ceil_div(M, BM));
  auto stream = at::cuda::getCurrentCUDAStream();
  if (bias.numel() > 0) {
    hipLaunchKernelGGL((conv_gemm_fwd_kernel<true>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), w.data_ptr<float>(),
                       bias.data_ptr<float>(), y.data_ptr<float>(), s, M, Nc,
                       K, (int)act);
  } else {
    hipLaunchKernelGGL((conv_gemm_fwd_kernel<false>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), w.data_ptr<float>(),
                       nullptr, y.data_ptr<float>(), s, M, Nc, K, (int)act);
  }
  return y;
}

torch::Tensor conv_gemm_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                              int64_t Kw, int64_t stride, int64_t pad) {
  check_f32g(x, "x");
  check_f32g(dy, "dy");
  ConvShape s;
  s.N = x.size(0); s.Cin = x.size(1); s.H = x.size(2); s.W = x.size(3);
  s.Cout = dy.size(1); s.Kh = Kh; s.Kw = Kw;
  s.stride = stride; s.pad = pad;
  s.Ho = dy.size(2); s.Wo = dy.size(3);
  const int M = s.Cout;
  const int Nc = s.Cin * s.Kh * s.Kw;
  const int K = s.N * s.Ho * s.Wo;
  auto dw = torch::empty({s.Cout, s.Cin, (long)Kh, (long)Kw}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  // Conv wgrads have tiny M/N tiles and a huge K (every output pixel):
  // push the split hard — the reduce kernel's extra planes are cheap
  // next to an underfilled chip (conv1 of the MNIST CNN is ONE tile).
  const int base = ceil_div(Nc, BN) * ceil_div(M, BM);
  int splitk = 1;
  const int max_by_k = std::max(1, K / 256);
  const int cap = sf_splitk_env_max();
  while (splitk < 256 && splitk < cap && base * splitk <= 1024 &&
         splitk * 2 <= max_by_k) {
    splitk *= 2;
  }
  if (splitk > 1) {
    const int k_chunk = ceil_div(K, splitk * BK) * BK;
    auto ws = torch::empty({splitk, M, Nc}, x.options());
    dim3 grid(ceil_div(Nc, BN), ceil_div(M, BM), splitk);
    hipLaunchKernelGGL((conv_gemm_wgrad_kernel<true>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       ws.data_ptr<float>(), s, M, Nc, K, k_chunk);
    const long n = (long)M * Nc;
    launch_conv_splitk_reduce(ws.data_ptr<float>(), dw.data_ptr<float>(), n,
                              splitk, stream);
  } else {
    dim3 grid(ceil_div(Nc, BN), ceil_div(M, BM));
    hipLaunchKernelGGL((conv_gemm_wgrad_kernel<false>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       dw.data_ptr<float>(), s, M, Nc, K, K);
  }
  return dw;
}
