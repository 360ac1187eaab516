// NHWC implicit-GEMM convolution (fp32, MFMA) — the channel-last layout
// makes every im2col gather a CONTIGUOUS float4 read along Cin and every
// output store contiguous along Cout, fixing the two things that bound
// the NCHW conv kernels on ResNet shapes (scalar gathers in the K loop,
// scattered epilogue).  Used by the layout="nhwc" modules; weights live
// as [Cout, Kh, Kw, Cin] so the k = (kh, kw, ci) GEMM dimension is the
// weight row's native order (no per-step permutes).
//
// GEMM views:
//   fwd:   C[m=(n,ho,wo)][co] = sum_k x[n, ho*s-p+kh, wo*s-p+kw, ci]
//                                     * w[co][k],   k = (kh, kw, ci)
//   wgrad: dw[co][k]          = sum_p dy[p, co] * im2col(x)[p, k]
// Same 64x64-tile / 16-deep MFMA pipeline as gemm.hip / conv_gemm.hip.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"
#include "conv_reduce.h"

#define BM 64
#define BN 64
#define BK 16

struct ConvShapeHW {  // NHWC shapes
  int N, H, W, Cin, Cout, Kh, Kw, Ho, Wo, stride, pad;
};

// ---------------- forward ----------------
template <bool HAS_BIAS>
__global__ __launch_bounds__(256) void conv_nhwc_fwd_kernel(
    const float* __restrict__ x, const float* __restrict__ wgt,
    const float* __restrict__ bias, float* __restrict__ y, ConvShapeHW s,
    int M, int K, int act) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 32;
  const int wn = (wv & 1) * 32;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  const int Nc = s.Cout;

  f32x4 acc[2][2] = {};

  // A staging: thread owns output pixel gm = m0 + (t>>2) and k-slot
  // (t&3)*4.  The pixel decode happens once; the (khkw, ci) cursor
  // advances incrementally (+BK with carry) — when Cin % 4 == 0 the 4
  // consecutive k's sit inside one (kh, kw) cell, so the gather is ONE
  // float4 from x.
  const int st_row = t >> 2, st_c4 = (t & 3) * 4;
  const int gm = m0 + st_row;
  const int m_dec = gm < M ? gm : 0;
  const int a_wo = m_dec % s.Wo;
  const int a_ho = (m_dec / s.Wo) % s.Ho;
  const long a_nbase =
      (long)(m_dec / (s.Wo * s.Ho)) * s.H * s.W * s.Cin;
  const int h0 = a_ho * s.stride - s.pad;
  const int w0 = a_wo * s.stride - s.pad;
  int a_ci = st_c4 % s.Cin;
  int a_khkw = st_c4 / s.Cin;
  const bool vec = (s.Cin & 3) == 0;

  auto load_a = [&](int k0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gm >= M) return r;
    if (vec) {  // 4 k's inside one (kh,kw) cell
      const int kh = a_khkw / s.Kw, kw = a_khkw % s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + st_c4 + 3 < K && h >= 0 && h < s.H && w >= 0 && w < s.W) {
        r = *reinterpret_cast<const float4*>(
            &x[a_nbase + ((long)h * s.W + w) * s.Cin + a_ci]);
      }
      return r;
    }
    float* rr = reinterpret_cast<float*>(&r);
    int ci = a_ci, khkw = a_khkw;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int kh = khkw / s.Kw, kw = khkw % s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + st_c4 + i < K && h >= 0 && h < s.H && w >= 0 && w < s.W) {
        rr[i] = x[a_nbase + ((long)h * s.W + w) * s.Cin + ci];
      }
      if (++ci == s.Cin) { ci = 0; ++khkw; }
    }
    return r;
  };
  auto advance_a = [&]() {
    a_ci += BK;
    while (a_ci >= s.Cin) { a_ci -= s.Cin; ++a_khkw; }
  };
  // B staging: w[co][k] rows are contiguous in k — float4 straight read.
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

  // epilogue: y[m][co] — contiguous along co (coalesced store)
  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
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
          y[(long)m * Nc + co] = act_apply(acc[mi][ni][i] + bv, act);
        }
      }
    }
  }
}

// ---------------- forward, narrow tile (Cout <= 32) ----------------
// The MNIST CNN's convs have Cout=32 (and its dgrad "Cout"=Cin=32): a
// BN=64 tile wastes half its MFMA work there.  64x32 tile, 4 waves
// each covering a 32x16 quadrant (2x1 16x16x4 frags).
template <bool HAS_BIAS>
__global__ __launch_bounds__(256) void conv_nhwc_fwd_n32_kernel(
    const float* __restrict__ x, const float* __restrict__ wgt,
    const float* __restrict__ bias, float* __restrict__ y, ConvShapeHW s,
    int M, int K, int act) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][32 + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv & 1) * 32;   // wave rows quadrant
  const int wn = (wv >> 1) * 16;  // wave cols half
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * 32;
  const int Nc = s.Cout;

  f32x4 acc[2] = {};

  const int st_row = t >> 2, st_c4 = (t & 3) * 4;
  const int gm = m0 + st_row;
  const int m_dec = gm < M ? gm : 0;
  const int a_wo = m_dec % s.Wo;
  const int a_ho = (m_dec / s.Wo) % s.Ho;
  const long a_nbase = (long)(m_dec / (s.Wo * s.Ho)) * s.H * s.W * s.Cin;
  const int h0 = a_ho * s.stride - s.pad;
  const int w0 = a_wo * s.stride - s.pad;
  int a_ci = st_c4 % s.Cin;
  int a_khkw = st_c4 / s.Cin;
  const bool vec = (s.Cin & 3) == 0;

  auto load_a = [&](int k0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gm >= M) return r;
    if (vec) {
      const int kh = a_khkw / s.Kw, kw = a_khkw % s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + st_c4 + 3 < K && h >= 0 && h < s.H && w >= 0 && w < s.W) {
        r = *reinterpret_cast<const float4*>(
            &x[a_nbase + ((long)h * s.W + w) * s.Cin + a_ci]);
      }
      return r;
    }
    float* rr = reinterpret_cast<float*>(&r);
    int ci = a_ci, khkw = a_khkw;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int kh = khkw / s.Kw, kw = khkw % s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + st_c4 + i < K && h >= 0 && h < s.H && w >= 0 && w < s.W) {
        rr[i] = x[a_nbase + ((long)h * s.W + w) * s.Cin + ci];
      }
      if (++ci == s.Cin) { ci = 0; ++khkw; }
    }
    return r;
  };
  auto advance_a = [&]() {
    a_ci += BK;
    while (a_ci >= s.Cin) { a_ci -= s.Cin; ++a_khkw; }
  };
  // B tile is 16k x 32co: threads with st_row < 32 stage it
  auto load_b = [&](int k0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    const int gn = n0 + st_row;
    if (st_row >= 32) return r;
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
#pragma unroll
      for (int i = 0; i < 4; ++i) As[st_row][st_c4 + i] = pa[i];
      if (st_row < 32) {
        const float* pb = reinterpret_cast<const float*>(&rb);
#pragma unroll
        for (int i = 0; i < 4; ++i) Bs[st_c4 + i][st_row] = pb[i];
      }
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
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc[1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
  const int co = n0 + wn + fcol;
  if (co < Nc) {
    const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = m0 + wm + mi * 16 + frow + i;
        if (m < M) {
          y[(long)m * Nc + co] = act_apply(acc[mi][i] + bv, act);
        }
      }
    }
  }
}

// ---------------- forward, large-M tile ----------------
// ResNet-shape forward convs are LARGE-M (M = N*Ho*Wo is 10^5..10^6)
// with modest Cout (64..512): the dense-GEMM 64x64/BK16 tile leaves
// MFMA starved there (1 LDS read per 2048-FLOP mfma_16x16x4).  This
// kernel uses BM=256 x BN=64 x BK=32 on v_mfma_f32_32x32x2_f32 (the
// CDNA4 f32 GEMM recipe, cdna_hip_programming.md §3: 2x2 32x32 frags
// per wave = 1 LDS read per 4096-FLOP instruction, i.e. half the LDS
// traffic per FLOP) with BN=64 so a Cout=64 layer wastes nothing.
// Each of the 4 waves owns a 64x64 quadrant of the 256-row stripe.
// Requires Cin % 4 == 0 (float4 im2col gathers); other shapes take the
// 64x64 kernel above.
#define BM2 256
#define BN2 64
#define BK2 32

template <bool HAS_BIAS>
__global__ __launch_bounds__(256) void conv_nhwc_fwd_big_kernel(
    const float* __restrict__ x, const float* __restrict__ wgt,
    const float* __restrict__ bias, float* __restrict__ y, ConvShapeHW s,
    int M, int K, int act) {
  __shared__ float As[BM2][BK2 + 1];
  __shared__ float Bs[BK2][BN2 + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = wv * 64;  // wave's 64-row quadrant of the 256-row stripe
  const int m0 = blockIdx.y * BM2;
  const int n0 = blockIdx.x * BN2;
  const int Nc = s.Cout;

  f32x16 acc[2][2] = {};

  // A staging: thread t owns output-pixel row (m0 + t) and ALL BK2 k's
  // of that row as 8 float4 groups.  ONE pixel decode per thread; the
  // (khkw, ci) cursor tracks k-group 0 and groups 1..7 derive from it
  // with at most one Cin-carry each (Cin % 4 == 0 and >= 4).
  const int gm = m0 + t;
  const int m_dec = gm < M ? gm : 0;
  const int a_wo = m_dec % s.Wo;
  const int a_ho = (m_dec / s.Wo) % s.Ho;
  const long a_nbase = (long)(m_dec / (s.Wo * s.Ho)) * s.H * s.W * s.Cin;
  const int h0 = a_ho * s.stride - s.pad;
  const int w0 = a_wo * s.stride - s.pad;
  int a_ci = 0, a_khkw = 0;  // cursor at k-group 0 (k = k0)

  auto load_a = [&](int k0, float4 (&r)[8]) {
#pragma unroll
    for (int g = 0; g < 8; ++g) r[g] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gm >= M) return;
    int ci = a_ci, khkw = a_khkw;
#pragma unroll
    for (int g = 0; g < 8; ++g) {
      const int kh = khkw / s.Kw, kw = khkw - (khkw / s.Kw) * s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + g * 4 + 3 < K && h >= 0 && h < s.H && w >= 0 && w < s.W) {
        r[g] = *reinterpret_cast<const float4*>(
            &x[a_nbase + ((long)h * s.W + w) * s.Cin + ci]);
      }
      ci += 4;
      if (ci >= s.Cin) { ci -= s.Cin; ++khkw; }
    }
  };
  auto advance_a = [&]() {
    a_ci += BK2;
    while (a_ci >= s.Cin) { a_ci -= s.Cin; ++a_khkw; }
  };
  // B staging: thread owns co = n0 + (t & 63), k-offsets (t>>6)*8..+7 —
  // 2 float4 straight reads from the contiguous w row.
  const int b_co = n0 + (t & 63);
  const int b_k8 = (t >> 6) * 8;
  auto load_b = [&](int k0, float4 (&r)[2]) {
    if (b_co < Nc && k0 + b_k8 + 7 < K) {
      r[0] = *reinterpret_cast<const float4*>(
          &wgt[(long)b_co * K + k0 + b_k8]);
      r[1] = *reinterpret_cast<const float4*>(
          &wgt[(long)b_co * K + k0 + b_k8 + 4]);
    } else {
      float* rr = reinterpret_cast<float*>(&r[0]);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int gk = k0 + b_k8 + i;
        rr[i] = (b_co < Nc && gk < K) ? wgt[(long)b_co * K + gk] : 0.0f;
      }
    }
  };

  const int fi = lane & 31;   // mfma A row / D col-lane
  const int fk = lane >> 5;   // mfma k slot (0/1)

  float4 ra[8], rb[2];
  load_a(0, ra);
  load_b(0, rb);
  for (int k0 = 0; k0 < K; k0 += BK2) {
    {
#pragma unroll
      for (int g = 0; g < 8; ++g) {
        const float* p = reinterpret_cast<const float*>(&ra[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) As[t][g * 4 + i] = p[i];
      }
      const float* pb = reinterpret_cast<const float*>(&rb[0]);
#pragma unroll
      for (int i = 0; i < 8; ++i) Bs[b_k8 + i][t & 63] = pb[i];
    }
    __syncthreads();
    if (k0 + BK2 < K) {
      advance_a();
      load_a(k0 + BK2, ra);
      load_b(k0 + BK2, rb);
    }
#pragma unroll
    for (int kk = 0; kk < BK2 / 2; ++kk) {
      const float a0 = As[wm + fi][kk * 2 + fk];
      const float a1 = As[wm + 32 + fi][kk * 2 + fk];
      const float b0 = Bs[kk * 2 + fk][fi];
      const float b1 = Bs[kk * 2 + fk][32 + fi];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: D lane mapping col = lane&31, row = (reg&3) + 8*(reg>>2)
  // + 4*(lane>>5); y[m][co] stores are co-contiguous across lanes.
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int co = n0 + ni * 32 + fi;
    if (co >= Nc) continue;
    const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row = (reg & 3) + 8 * (reg >> 2) + 4 * fk;
        const int m = m0 + wm + mi * 32 + row;
        if (m < M) {
          y[(long)m * Nc + co] = act_apply(acc[mi][ni][reg] + bv, act);
        }
      }
    }
  }
}

// ---------------- forward, 128x128 tile (Cout >= 128) ----------------
// The guide's canonical f32 recipe (cdna_hip_programming.md §3:
// 128x128xBK32 block, 2x2 32x32 frags per wave on mfma_f32_32x32x2) —
// applicable where Cout >= 128 so the BN=128 tile wastes nothing
// (ResNet layers 2-4).  Lighter staging than the 256x64 attempt
// (4+4 float4 per thread instead of 8+2), ~110 VGPR -> 4 waves/SIMD.
#define BM3 128
#define BN3 128

template <bool HAS_BIAS>
__global__ __launch_bounds__(256) void conv_nhwc_fwd_sq_kernel(
    const float* __restrict__ x, const float* __restrict__ wgt,
    const float* __restrict__ bias, float* __restrict__ y, ConvShapeHW s,
    int M, int K, int act) {
  __shared__ float As[BM3][BK2 + 1];
  __shared__ float Bs[BK2][BN3 + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 64;
  const int wn = (wv & 1) * 64;
  const int m0 = blockIdx.y * BM3;
  const int n0 = blockIdx.x * BN3;
  const int Nc = s.Cout;

  f32x16 acc[2][2] = {};

  // A: thread owns pixel row m0 + (t>>1) and k-slots (t&1)*16 + 4i
  const int a_row = t >> 1, a_k16 = (t & 1) * 16;
  const int gm = m0 + a_row;
  const int m_dec = gm < M ? gm : 0;
  const int a_wo = m_dec % s.Wo;
  const int a_ho = (m_dec / s.Wo) % s.Ho;
  const long a_nbase = (long)(m_dec / (s.Wo * s.Ho)) * s.H * s.W * s.Cin;
  const int h0 = a_ho * s.stride - s.pad;
  const int w0 = a_wo * s.stride - s.pad;
  int a_ci = a_k16 % s.Cin;          // cursor at k-group (t&1)*16
  int a_khkw = a_k16 / s.Cin;

  auto load_a = [&](int k0, float4 (&r)[4]) {
#pragma unroll
    for (int g = 0; g < 4; ++g) r[g] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gm >= M) return;
    int ci = a_ci, khkw = a_khkw;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      const int kh = khkw / s.Kw, kw = khkw - (khkw / s.Kw) * s.Kw;
      const int h = h0 + kh, w = w0 + kw;
      if (k0 + a_k16 + g * 4 + 3 < K && h >= 0 && h < s.H && w >= 0 &&
          w < s.W) {
        r[g] = *reinterpret_cast<const float4*>(
            &x[a_nbase + ((long)h * s.W + w) * s.Cin + ci]);
      }
      ci += 4;
      if (ci >= s.Cin) { ci -= s.Cin; ++khkw; }
    }
  };
  auto advance_a = [&]() {
    a_ci += BK2;
    while (a_ci >= s.Cin) { a_ci -= s.Cin; ++a_khkw; }
  };
  // B: thread owns co = n0 + (t & 127), k-offsets (t>>7)*16 + 4i
  const int b_co = n0 + (t & 127);
  const int b_k16 = (t >> 7) * 16;
  auto load_b = [&](int k0, float4 (&r)[4]) {
    if (b_co < Nc && k0 + b_k16 + 15 < K) {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        r[g] = *reinterpret_cast<const float4*>(
            &wgt[(long)b_co * K + k0 + b_k16 + g * 4]);
      }
    } else {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        float* rr = reinterpret_cast<float*>(&r[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int gk = k0 + b_k16 + g * 4 + i;
          rr[i] = (b_co < Nc && gk < K) ? wgt[(long)b_co * K + gk] : 0.0f;
        }
      }
    }
  };

  const int fi = lane & 31;
  const int fk = lane >> 5;

  float4 ra[4], rb[4];
  load_a(0, ra);
  load_b(0, rb);
  for (int k0 = 0; k0 < K; k0 += BK2) {
    {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const float* p = reinterpret_cast<const float*>(&ra[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) As[a_row][a_k16 + g * 4 + i] = p[i];
      }
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const float* p = reinterpret_cast<const float*>(&rb[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) Bs[b_k16 + g * 4 + i][t & 127] = p[i];
      }
    }
    __syncthreads();
    if (k0 + BK2 < K) {
      advance_a();
      load_a(k0 + BK2, ra);
      load_b(k0 + BK2, rb);
    }
#pragma unroll
    for (int kk = 0; kk < BK2 / 2; ++kk) {
      const float a0 = As[wm + fi][kk * 2 + fk];
      const float a1 = As[wm + 32 + fi][kk * 2 + fk];
      const float b0 = Bs[kk * 2 + fk][wn + fi];
      const float b1 = Bs[kk * 2 + fk][wn + 32 + fi];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int co = n0 + wn + ni * 32 + fi;
    if (co >= Nc) continue;
    const float bv = HAS_BIAS ? bias[co] : 0.0f;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row = (reg & 3) + 8 * (reg >> 2) + 4 * fk;
        const int m = m0 + wm + mi * 32 + row;
        if (m < M) {
          y[(long)m * Nc + co] = act_apply(acc[mi][ni][reg] + bv, act);
        }
      }
    }
  }
}

// ---------------- wgrad, 128x128 tile (Cout >= 128) ----------------
// Same mfma_f32_32x32x2 recipe as the sq forward, for the wgrad GEMM
// dw[co][k'] = sum_p dy[p, co] * im2col(x)[p, k'] on the Cout >= 128
// ResNet layers (the 64x64 kernel stays for smaller dw).  BK2=32 deep;
// A (dy, transposed at staging) and B (im2col) both float4.
template <bool SPLITK>
__global__ __launch_bounds__(256) void conv_nhwc_wgrad_sq_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    float* __restrict__ dw, ConvShapeHW s, int M, int Nc, int K,
    int k_chunk) {
  __shared__ float As[BM3][BK2 + 1];
  __shared__ float Bs[BK2][BN3 + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 64;
  const int wn = (wv & 1) * 64;
  const int m0 = blockIdx.y * BM3;  // co tile
  const int n0 = blockIdx.x * BN3;  // k' tile

  f32x16 acc[2][2] = {};

  int k_begin = 0, k_end = K;
  if (SPLITK) {
    k_begin = blockIdx.z * k_chunk;
    k_end = min(K, k_begin + k_chunk);
  }

  // A: As[co][p] = dy[p][co] — thread owns pixel-row (t>>3) and co-slots
  // (t&7)*16 + 4g (4 float4 per step; dy rows are co-contiguous)
  const int a_kk = t >> 3, a_m16 = (t & 7) * 16;
  // B: Bs[p][k'] — thread owns pixel-row (t>>3) and k'-slots
  // (t&7)*16 + 4g
  const int b_kk = t >> 3, b_n16 = (t & 7) * 16;
  const int HoWo = s.Ho * s.Wo;

  int pa_n = (k_begin + a_kk) / HoWo;
  int pa_hw = (k_begin + a_kk) % HoWo;
  int pb_n = (k_begin + b_kk) / HoWo;
  int pb_ho = ((k_begin + b_kk) % HoWo) / s.Wo;
  int pb_wo = ((k_begin + b_kk) % HoWo) % s.Wo;
  // decode the 16 k' columns once (fixed for the whole K loop)
  int kw_i[16], kh_i[16], ci_i[16];
  bool gk_ok[16], gvec[4];
  const bool cin4 = (s.Cin & 3) == 0;
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    const int gk = n0 + b_n16 + i;
    gk_ok[i] = gk < Nc;
    const int g = gk_ok[i] ? gk : 0;
    ci_i[i] = g % s.Cin;
    const int khkw = g / s.Cin;
    kw_i[i] = khkw % s.Kw;
    kh_i[i] = khkw / s.Kw;
  }
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    gvec[g] = cin4 && gk_ok[g * 4 + 3] && kh_i[g * 4] == kh_i[g * 4 + 3] &&
              kw_i[g * 4] == kw_i[g * 4 + 3];
  }
  const long n_stride = (long)s.H * s.W * s.Cin;

  auto load_a = [&](int gp0, float4 (&r)[4]) {
#pragma unroll
    for (int g = 0; g < 4; ++g) r[g] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 >= k_end) return;
    const long rowbase = ((long)pa_n * HoWo + pa_hw) * s.Cout;
    if (m0 + a_m16 + 15 < s.Cout && (s.Cout & 3) == 0) {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        r[g] = *reinterpret_cast<const float4*>(
            &dy[rowbase + m0 + a_m16 + g * 4]);
      }
    } else {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        float* rr = reinterpret_cast<float*>(&r[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int co = m0 + a_m16 + g * 4 + i;
          rr[i] = co < s.Cout ? dy[rowbase + co] : 0.0f;
        }
      }
    }
  };
  auto load_b = [&](int gp0, float4 (&r)[4]) {
#pragma unroll
    for (int g = 0; g < 4; ++g) r[g] = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 >= k_end) return;
    const long nbase = (long)pb_n * n_stride;
#pragma unroll
    for (int g = 0; g < 4; ++g) {
      if (gvec[g]) {
        const int h = pb_ho * s.stride - s.pad + kh_i[g * 4];
        const int w = pb_wo * s.stride - s.pad + kw_i[g * 4];
        if (h >= 0 && h < s.H && w >= 0 && w < s.W) {
          r[g] = *reinterpret_cast<const float4*>(
              &x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[g * 4]]);
        }
      } else {
        float* rr = reinterpret_cast<float*>(&r[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int j = g * 4 + i;
          if (gk_ok[j]) {
            const int h = pb_ho * s.stride - s.pad + kh_i[j];
            const int w = pb_wo * s.stride - s.pad + kw_i[j];
            if (h >= 0 && h < s.H && w >= 0 && w < s.W)
              rr[i] = x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[j]];
          }
        }
      }
    }
  };
  auto advance = [&]() {  // both pixel cursors += BK2
    pa_hw += BK2;
    while (pa_hw >= HoWo) { pa_hw -= HoWo; ++pa_n; }
    pb_wo += BK2;
    while (pb_wo >= s.Wo) { pb_wo -= s.Wo; ++pb_ho; }
    while (pb_ho >= s.Ho) { pb_ho -= s.Ho; ++pb_n; }
  };

  const int fi = lane & 31;
  const int fk = lane >> 5;

  float4 ra[4], rb[4];
  load_a(k_begin + a_kk, ra);
  load_b(k_begin + b_kk, rb);
  for (int k0 = k_begin; k0 < k_end; k0 += BK2) {
    {
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const float* pa = reinterpret_cast<const float*>(&ra[g]);
        const float* pb = reinterpret_cast<const float*>(&rb[g]);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          As[a_m16 + g * 4 + i][a_kk] = pa[i];
          Bs[b_kk][b_n16 + g * 4 + i] = pb[i];
        }
      }
    }
    __syncthreads();
    if (k0 + BK2 < k_end) {
      advance();
      load_a(k0 + BK2 + a_kk, ra);
      load_b(k0 + BK2 + b_kk, rb);
    }
#pragma unroll
    for (int kk = 0; kk < BK2 / 2; ++kk) {
      const float a0 = As[wm + fi][kk * 2 + fk];
      const float a1 = As[wm + 32 + fi][kk * 2 + fk];
      const float b0 = Bs[kk * 2 + fk][wn + fi];
      const float b1 = Bs[kk * 2 + fk][wn + 32 + fi];
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc[1][1], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int kcol = n0 + wn + ni * 32 + fi;
    if (kcol >= Nc) continue;
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row = (reg & 3) + 8 * (reg >> 2) + 4 * fk;
        const int co = m0 + wm + mi * 32 + row;
        if (co < M) {
          if (SPLITK) {
            dw[(long)blockIdx.z * M * Nc + (long)co * Nc + kcol] =
                acc[mi][ni][reg];
          } else {
            dw[(long)co * Nc + kcol] = acc[mi][ni][reg];
          }
        }
      }
    }
  }
}

// ---------------- wgrad, narrow tile (Cout <= 32) ----------------
// dw for the MNIST CNN's 32-channel convs: BM=64 wastes half the MFMA
// work when M = Cout <= 32.  32x64 tile; 4 waves each 16 co x 32 k'
// (1x2 16x16x4 frags).
template <bool SPLITK>
__global__ __launch_bounds__(256) void conv_nhwc_wgrad_m32_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    float* __restrict__ dw, ConvShapeHW s, int M, int Nc, int K,
    int k_chunk) {
  __shared__ float As[32][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv & 1) * 16;   // wave's co half
  const int wn = (wv >> 1) * 32;  // wave's k' quadrant
  const int m0 = blockIdx.y * 32;  // co tile
  const int n0 = blockIdx.x * BN;  // k' tile

  f32x4 acc[2] = {};

  int k_begin = 0, k_end = K;
  if (SPLITK) {
    k_begin = blockIdx.z * k_chunk;
    k_end = min(K, k_begin + k_chunk);
  }

  // A: As[co][p] = dy[p][co] — threads with (t&15)*4 < 32 stage it
  const int a_kk = t >> 4, a_m4 = (t & 15) * 4;
  const int b_kk = t >> 4, b_n4 = (t & 15) * 4;
  const int HoWo = s.Ho * s.Wo;

  int pa_n = (k_begin + a_kk) / HoWo;
  int pa_hw = (k_begin + a_kk) % HoWo;
  int pb_n = (k_begin + b_kk) / HoWo;
  int pb_ho = ((k_begin + b_kk) % HoWo) / s.Wo;
  int pb_wo = ((k_begin + b_kk) % HoWo) % s.Wo;
  int kw_i[4], kh_i[4], ci_i[4];
  bool gk_ok[4];
  bool bvec = (s.Cin & 3) == 0;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int gk = n0 + b_n4 + i;
    gk_ok[i] = gk < Nc;
    const int g = gk_ok[i] ? gk : 0;
    ci_i[i] = g % s.Cin;
    const int khkw = g / s.Cin;
    kw_i[i] = khkw % s.Kw;
    kh_i[i] = khkw / s.Kw;
  }
  bvec = bvec && gk_ok[3] && kh_i[0] == kh_i[3] && kw_i[0] == kw_i[3];
  const long n_stride = (long)s.H * s.W * s.Cin;

  auto load_a = [&](int gp0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 >= k_end || a_m4 >= 32) return r;
    int n = pa_n, hw = pa_hw;
    if ((s.Cout & 3) == 0 && m0 + a_m4 + 3 < s.Cout) {
      r = *reinterpret_cast<const float4*>(
          &dy[((long)n * HoWo + hw) * s.Cout + m0 + a_m4]);
    } else {
      float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int co = m0 + a_m4 + i;
        rr[i] = co < s.Cout
                    ? dy[((long)n * HoWo + hw) * s.Cout + co] : 0.0f;
      }
    }
    return r;
  };
  auto load_b = [&](int gp0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 >= k_end) return r;
    const long nbase = (long)pb_n * n_stride;
    if (bvec) {
      const int h = pb_ho * s.stride - s.pad + kh_i[0];
      const int w = pb_wo * s.stride - s.pad + kw_i[0];
      if (h >= 0 && h < s.H && w >= 0 && w < s.W) {
        r = *reinterpret_cast<const float4*>(
            &x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[0]]);
      }
      return r;
    }
    float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (gk_ok[i]) {
        const int h = pb_ho * s.stride - s.pad + kh_i[i];
        const int w = pb_wo * s.stride - s.pad + kw_i[i];
        if (h >= 0 && h < s.H && w >= 0 && w < s.W)
          rr[i] = x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[i]];
      }
    }
    return r;
  };
  auto advance = [&]() {
    pa_hw += BK;
    while (pa_hw >= HoWo) { pa_hw -= HoWo; ++pa_n; }
    pb_wo += BK;
    while (pb_wo >= s.Wo) { pb_wo -= s.Wo; ++pb_ho; }
    while (pb_ho >= s.Ho) { pb_ho -= s.Ho; ++pb_n; }
  };

  const int fr = lane & 15;
  const int fk = lane >> 4;

  float4 ra = load_a(k_begin + a_kk);
  float4 rb = load_b(k_begin + b_kk);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    {
      if (a_m4 < 32) {
        const float* pa = reinterpret_cast<const float*>(&ra);
#pragma unroll
        for (int i = 0; i < 4; ++i) As[a_m4 + i][a_kk] = pa[i];
      }
      const float* pb = reinterpret_cast<const float*>(&rb);
#pragma unroll
      for (int i = 0; i < 4; ++i) Bs[b_kk][b_n4 + i] = pb[i];
    }
    __syncthreads();
    if (k0 + BK < k_end) {
      advance();
      ra = load_a(k0 + BK + a_kk);
      rb = load_b(k0 + BK + b_kk);
    }
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      const float a0 = As[wm + fr][kk * 4 + fk];
      const float b0 = Bs[kk * 4 + fk][wn + fr];
      const float b1 = Bs[kk * 4 + fk][wn + 16 + fr];
      acc[0] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc[0], 0, 0, 0);
      acc[1] = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc[1], 0, 0, 0);
    }
    __syncthreads();
  }

  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
#pragma unroll
  for (int ni = 0; ni < 2; ++ni) {
    const int kcol = n0 + wn + ni * 16 + fcol;
    if (kcol >= Nc) continue;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int co = m0 + wm + frow + i;
      if (co < M) {
        if (SPLITK) {
          dw[(long)blockIdx.z * M * Nc + (long)co * Nc + kcol] =
              acc[ni][i];
        } else {
          dw[(long)co * Nc + kcol] = acc[ni][i];
        }
      }
    }
  }
}

// ---------------- wgrad ----------------
// dw[co][k'] = sum_p dy[p, co] * im2col(x)[p, k'] — A staged transposed
// (dy contiguous along co), B's im2col contiguous along ci.
template <bool SPLITK>
__global__ __launch_bounds__(256) void conv_nhwc_wgrad_kernel(
    const float* __restrict__ x, const float* __restrict__ dy,
    float* __restrict__ dw, ConvShapeHW s, int M, int Nc, int K,
    int k_chunk) {
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 32;
  const int wn = (wv & 1) * 32;
  const int m0 = blockIdx.y * BM;  // co tile
  const int n0 = blockIdx.x * BN;  // k' tile

  f32x4 acc[2][2] = {};

  int k_begin = 0, k_end = K;
  if (SPLITK) {
    k_begin = blockIdx.z * k_chunk;
    k_end = min(K, k_begin + k_chunk);
  }

  // A: As[co][p] = dy[p][co] — thread owns pixel-row kk = t>>4 and 4
  // consecutive co's, ONE float4 per chunk (dy NHWC = [p][co]).
  const int a_kk = t >> 4, a_m4 = (t & 15) * 4;
  // B: Bs[p][k'] — thread owns pixel-row b_kk = t>>4 and 4 consecutive
  // k' = (kh,kw,ci): float4 from x when Cin % 4 == 0.
  const int b_kk = t >> 4, b_n4 = (t & 15) * 4;
  const int HoWo = s.Ho * s.Wo;

  int pa_n = (k_begin + a_kk) / HoWo;
  int pa_hw = (k_begin + a_kk) % HoWo;
  int pb_n = (k_begin + b_kk) / HoWo;
  int pb_ho = ((k_begin + b_kk) % HoWo) / s.Wo;
  int pb_wo = ((k_begin + b_kk) % HoWo) % s.Wo;
  // decode the 4 k' columns once (fixed for the whole K loop)
  int kw_i[4], kh_i[4], ci_i[4];
  bool gk_ok[4];
  bool bvec = (s.Cin & 3) == 0;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int gk = n0 + b_n4 + i;
    gk_ok[i] = gk < Nc;
    const int g = gk_ok[i] ? gk : 0;
    ci_i[i] = g % s.Cin;
    const int khkw = g / s.Cin;
    kw_i[i] = khkw % s.Kw;
    kh_i[i] = khkw / s.Kw;
  }
  // the float4 fast path additionally needs the 4 k's in one (kh,kw)
  bvec = bvec && gk_ok[3] && kh_i[0] == kh_i[3] && kw_i[0] == kw_i[3];
  const long n_stride = (long)s.H * s.W * s.Cin;

  auto load_a = [&](int gp0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 < K) {
      int n = pa_n, hw = pa_hw;
      if ((s.Cout & 3) == 0 && m0 + a_m4 + 3 < s.Cout) {
        r = *reinterpret_cast<const float4*>(
            &dy[((long)n * HoWo + hw) * s.Cout + m0 + a_m4]);
      } else {
        float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int co = m0 + a_m4 + i;
          rr[i] = co < s.Cout
                      ? dy[((long)n * HoWo + hw) * s.Cout + co] : 0.0f;
        }
      }
    }
    return r;
  };
  auto load_b = [&](int gp0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gp0 >= K) return r;
    const long nbase = (long)pb_n * n_stride;
    if (bvec) {
      const int h = pb_ho * s.stride - s.pad + kh_i[0];
      const int w = pb_wo * s.stride - s.pad + kw_i[0];
      if (h >= 0 && h < s.H && w >= 0 && w < s.W) {
        r = *reinterpret_cast<const float4*>(
            &x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[0]]);
      }
      return r;
    }
    float* rr = reinterpret_cast<float*>(&r);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (gk_ok[i]) {
        const int h = pb_ho * s.stride - s.pad + kh_i[i];
        const int w = pb_wo * s.stride - s.pad + kw_i[i];
        if (h >= 0 && h < s.H && w >= 0 && w < s.W)
          rr[i] = x[nbase + ((long)h * s.W + w) * s.Cin + ci_i[i]];
      }
    }
    return r;
  };
  auto advance = [&]() {  // both pixel cursors += BK
    pa_hw += BK;
    while (pa_hw >= HoWo) { pa_hw -= HoWo; ++pa_n; }
    pb_wo += BK;
    while (pb_wo >= s.Wo) { pb_wo -= s.Wo; ++pb_ho; }
    while (pb_ho >= s.Ho) { pb_ho -= s.Ho; ++pb_n; }
  };

  const int fr = lane & 15;
  const int fk = lane >> 4;

  float4 ra = load_a(k_begin + a_kk);
  float4 rb = load_b(k_begin + b_kk);
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    {
      const float* pa = reinterpret_cast<const float*>(&ra);
      const float* pb = reinterpret_cast<const float*>(&rb);
#pragma unroll
      for (int i = 0; i < 4; ++i) As[a_m4 + i][a_kk] = pa[i];
#pragma unroll
      for (int i = 0; i < 4; ++i) Bs[b_kk][b_n4 + i] = pb[i];
    }
    __syncthreads();
    if (k0 + BK < k_end) {
      advance();
      ra = load_a(k0 + BK + a_kk);
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

// ---------------- direct strided dgrad ----------------
// dgrad for stride s > 1 WITHOUT zero-stuffing: the dilated-dy
// formulation (dilate + stride-1 fwd conv) runs s^2 = 4x the useful
// MFMA work on zeros (plus a dilate kernel and a big fill).  Here dx is
// computed directly as s^2 sub-convolutions: output pixels of class
// (h%s, w%s) see only the filter taps kh ≡ (p'-h) (mod s) — the K
// dimension per class is nkh*nkw*Cout ≈ K/s^2, and dy is read
// unstrided.  blockIdx.z picks the class; same 64x64/BK16
// mfma_16x16x4 scheme as the fwd kernel (measured best, see
// profiles/conv_micro_r02.md).  Requires Cout % 4 == 0.
//
//   dx[n, h, w, ci] = sum_{khi,kwi,co} dy[n, hh+khi+oh_off,
//                        ww+kwi+ow_off, co] * wt[ci, kh0+khi*s,
//                        kw0+kwi*s, co]
// with h = a + s*hh, kh0 = (p'-a) mod s, oh_off = (a+kh0-p')/s and the
// same for w/b/kw0/ow_off; wt = w.flip(1,2).permute(3,1,2,0) exactly as
// the dilated path's operand (ops/functional.py Conv2dActNHWC.backward).
__global__ __launch_bounds__(256) void conv_nhwc_dgrad_s_kernel(
    const float* __restrict__ dy, const float* __restrict__ wt,
    float* __restrict__ dx, ConvShapeHW s_, int pprime) {
  // geometry for this class --------------------------------------------
  const int S = s_.stride;
  const int a = blockIdx.z / S, b = blockIdx.z % S;
  const int kh0 = ((pprime - a) % S + S) % S;
  const int kw0 = ((pprime - b) % S + S) % S;
  const int nkh = s_.Kh > kh0 ? (s_.Kh - kh0 + S - 1) / S : 0;
  const int nkw = s_.Kw > kw0 ? (s_.Kw - kw0 + S - 1) / S : 0;
  const int Ha = s_.H > a ? (s_.H - a + S - 1) / S : 0;
  const int Wb = s_.W > b ? (s_.W - b + S - 1) / S : 0;
  const int oh_off = (a + kh0 - pprime) / S;
  const int ow_off = (b + kw0 - pprime) / S;
  const int M = s_.N * Ha * Wb;
  const int K = nkh * nkw * s_.Cout;  // may be 0 -> nothing to write
  const int Nc = s_.Cin;              // GEMM columns = input channels

  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BK][BN + 4];

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wv = t >> 6;
  const int wm = (wv >> 1) * 32;
  const int wn = (wv & 1) * 32;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  if (m0 >= M || K == 0) return;  // classes have different M grids

  f32x4 acc[2][2] = {};

  const int st_row = t >> 2, st_c4 = (t & 3) * 4;
  // A: thread owns dx pixel gm = m0 + st_row, k-slot st_c4 (4
  // consecutive co inside one (khi, kwi) — Cout % 4 == 0)
  const int gm = m0 + st_row;
  const int m_dec = gm < M ? gm : 0;
  const int a_ww = m_dec % Wb;
  const int a_hh = (m_dec / Wb) % Ha;
  const int a_n = m_dec / (Wb * Ha);
  const long dy_nbase = (long)a_n * s_.Ho * s_.Wo * s_.Cout;
  // (khi, kwi, co) cursor for k-slot st_c4, advanced by BK per step
  int a_co = st_c4 % s_.Cout;
  int a_kk = st_c4 / s_.Cout;  // khi*nkw + kwi

  auto load_a = [&](int k0) -> float4 {
    // slots never straddle K: slot starts and K are both %4 == 0
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gm >= M || k0 + st_c4 >= K) return r;
    const int khi = a_kk / nkw, kwi = a_kk - (a_kk / nkw) * nkw;
    const int oh = a_hh + khi + oh_off;
    const int ow = a_ww + kwi + ow_off;
    if (oh >= 0 && oh < s_.Ho && ow >= 0 && ow < s_.Wo) {
      r = *reinterpret_cast<const float4*>(
          &dy[dy_nbase + ((long)oh * s_.Wo + ow) * s_.Cout + a_co]);
    }
    return r;
  };
  auto advance_a = [&]() {
    a_co += BK;
    while (a_co >= s_.Cout) { a_co -= s_.Cout; ++a_kk; }
  };
  // B: thread owns wt row ci = n0 + st_row, k-slot st_c4; same cursor
  int b_co = st_c4 % s_.Cout;
  int b_kk = st_c4 / s_.Cout;
  const int gci = n0 + st_row;
  const long wt_row = (long)(gci < Nc ? gci : 0) * s_.Kh * s_.Kw * s_.Cout;

  auto load_b = [&](int k0) -> float4 {
    float4 r = {0.0f, 0.0f, 0.0f, 0.0f};
    if (gci >= Nc || k0 + st_c4 >= K) return r;
    const int khi = b_kk / nkw, kwi = b_kk - (b_kk / nkw) * nkw;
    const int kh = kh0 + khi * S, kw = kw0 + kwi * S;
    r = *reinterpret_cast<const float4*>(
        &wt[wt_row + ((long)kh * s_.Kw + kw) * s_.Cout + b_co]);
    return r;
  };
  auto advance_b = [&]() {
    b_co += BK;
    while (b_co >= s_.Cout) { b_co -= s_.Cout; ++b_kk; }
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
      advance_b();
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

  // epilogue: dx[(n, a + s*hh, b + s*ww), ci] — ci-contiguous stores
  const int fcol = lane & 15;
  const int frow = (lane >> 4) * 4;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int ci = n0 + wn + ni * 16 + fcol;
      if (ci >= Nc) continue;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int m = m0 + wm + mi * 16 + frow + i;
        if (m < M) {
          const int ww = m % Wb;
          const int hh = (m / Wb) % Ha;
          const int n = m / (Wb * Ha);
          dx[(((long)n * s_.H + a + (long)S * hh) * s_.W + b +
              (long)S * ww) * s_.Cin + ci] = acc[mi][ni][i];
        }
      }
    }
  }
}

// Zero-stuff dy (NHWC) for strided dgrad-as-convolution: rows stay
// channel-contiguous (a row copy is C contiguous floats).
__global__ void dilate_nhwc_kernel(const float* __restrict__ dy,
                                   float* __restrict__ out, int Ho, int Wo,
                                   int C, int s, long plane_out, int Wd) {
  const long n = blockIdx.x;
  const float* src = dy + n * (long)Ho * Wo * C;
  float* dst = out + n * plane_out;
  for (int ho = blockIdx.y; ho < Ho; ho += gridDim.y) {
    for (int wo = 0; wo < Wo; ++wo) {
      const float* srow = src + ((long)ho * Wo + wo) * C;
      float* drow = dst + (((long)ho * s) * Wd + (long)wo * s) * C;
      for (int c = threadIdx.x; c < C; c += blockDim.x) drow[c] = srow[c];
    }
  }
}

// ---------------------------------------------------------------------------

static inline void check_f32w(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous(),
              name, " must be contiguous fp32 on GPU");
}

// x: [N, H, W, Cin], w: [Cout, Kh, Kw, Cin] -> y: [N, Ho, Wo, Cout]
torch::Tensor conv_nhwc_fwd(torch::Tensor x, torch::Tensor w,
                            torch::Tensor bias, int64_t stride, int64_t pad,
                            int64_t act) {
  check_f32w(x, "x");
  check_f32w(w, "w");
  ConvShapeHW s;
  s.N = x.size(0); s.H = x.size(1); s.W = x.size(2); s.Cin = x.size(3);
  s.Cout = w.size(0); s.Kh = w.size(1); s.Kw = w.size(2);
  TORCH_CHECK(w.size(3) == s.Cin, "Cin mismatch");
  s.stride = stride; s.pad = pad;
  s.Ho = (s.H + 2 * s.pad - s.Kh) / s.stride + 1;
  s.Wo = (s.W + 2 * s.pad - s.Kw) / s.stride + 1;
  const int M = s.N * s.Ho * s.Wo;
  const int K = s.Kh * s.Kw * s.Cin;
  auto y = torch::empty({s.N, s.Ho, s.Wo, s.Cout}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  const float* bp = bias.numel() > 0 ? bias.data_ptr<float>() : nullptr;
  // MEASURED NEGATIVE (round 2, profiles/conv_micro_r02.md): the
  // 256x64/BK32 mfma_32x32x2 kernel (140 VGPR, 3 waves/SIMD) is ~15%
  // slower than the 64x64/BK16 kernel on every ResNet-18 layer shape at
  // mbs=300 — the 64x64 kernel already runs at 87-100 TF (56-64% of the
  // 155 TF f32 MFMA peak), so halving LDS traffic buys nothing and the
  // lost occupancy (56 VGPR -> 8 waves/SIMD vs 3) costs latency hiding
  // on the im2col gathers.  Kept for re-evaluation: SPARKFLOW_CONV_BIG=1
  // opts in; default is the 64x64 kernel.
  static const bool big_enabled = [] {
    const char* e = getenv("SPARKFLOW_CONV_BIG");
    return e && e[0] == '1';
  }();
  // 128x128/BK32 mfma_32x32x2 tile for Cout >= 128 shapes (the guide's
  // canonical f32 recipe; SPARKFLOW_CONV_SQ=0 disables)
  static const bool sq_enabled = [] {
    const char* e = getenv("SPARKFLOW_CONV_SQ");
    return !(e && e[0] == '0');
  }();
  const long sq_tiles = (long)ceil_div(M, BM3) * ceil_div(s.Cout, BN3);
  if (sq_enabled && (s.Cin & 3) == 0 && s.Cout >= BN3 && K >= BK2 &&
      sq_tiles >= 128) {
    dim3 grid(ceil_div(s.Cout, BN3), ceil_div(M, BM3));
    if (bp) {
      hipLaunchKernelGGL((conv_nhwc_fwd_sq_kernel<true>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         bp, y.data_ptr<float>(), s, M, K, (int)act);
    } else {
      hipLaunchKernelGGL((conv_nhwc_fwd_sq_kernel<false>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         nullptr, y.data_ptr<float>(), s, M, K, (int)act);
    }
    return y;
  }
  const long big_tiles = (long)ceil_div(M, BM2) * ceil_div(s.Cout, BN2);
  if (big_enabled && (s.Cin & 3) == 0 && K >= BK2 && big_tiles >= 128) {
    dim3 grid(ceil_div(s.Cout, BN2), ceil_div(M, BM2));
    if (bp) {
      hipLaunchKernelGGL((conv_nhwc_fwd_big_kernel<true>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         bp, y.data_ptr<float>(), s, M, K, (int)act);
    } else {
      hipLaunchKernelGGL((conv_nhwc_fwd_big_kernel<false>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         nullptr, y.data_ptr<float>(), s, M, K, (int)act);
    }
    return y;
  }
  if (s.Cout <= 32) {  // narrow tile: BN=64 would waste half its MFMAs
    dim3 grid(ceil_div(s.Cout, 32), ceil_div(M, BM));
    if (bp) {
      hipLaunchKernelGGL((conv_nhwc_fwd_n32_kernel<true>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         bp, y.data_ptr<float>(), s, M, K, (int)act);
    } else {
      hipLaunchKernelGGL((conv_nhwc_fwd_n32_kernel<false>), grid, dim3(256),
                         0, stream, x.data_ptr<float>(), w.data_ptr<float>(),
                         nullptr, y.data_ptr<float>(), s, M, K, (int)act);
    }
    return y;
  }
  dim3 grid(ceil_div(s.Cout, BN), ceil_div(M, BM));
  if (bp) {
    hipLaunchKernelGGL((conv_nhwc_fwd_kernel<true>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), w.data_ptr<float>(),
                       bp, y.data_ptr<float>(), s, M, K,
                       (int)act);
  } else {
    hipLaunchKernelGGL((conv_nhwc_fwd_kernel<false>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), w.data_ptr<float>(),
                       nullptr, y.data_ptr<float>(), s, M, K, (int)act);
  }
  return y;
}

// x: [N, H, W, Cin], dy: [N, Ho, Wo, Cout] -> dw: [Cout, Kh, Kw, Cin]
torch::Tensor conv_nhwc_wgrad(torch::Tensor x, torch::Tensor dy, int64_t Kh,
                              int64_t Kw, int64_t stride, int64_t pad) {
  check_f32w(x, "x");
  check_f32w(dy, "dy");
  ConvShapeHW s;
  s.N = x.size(0); s.H = x.size(1); s.W = x.size(2); s.Cin = x.size(3);
  s.Cout = dy.size(3); s.Kh = Kh; s.Kw = Kw;
  s.stride = stride; s.pad = pad;
  s.Ho = dy.size(1); s.Wo = dy.size(2);
  const int M = s.Cout;
  const int Nc = s.Kh * s.Kw * s.Cin;
  const int K = s.N * s.Ho * s.Wo;
  auto dw = torch::empty({s.Cout, (long)Kh, (long)Kw, s.Cin}, x.options());
  auto stream = at::cuda::getCurrentCUDAStream();
  // 128x128 mfma_32x32x2 wgrad tile for Cout >= 128 (SPARKFLOW_CONV_SQ=0
  // disables, same flag as the sq forward)
  static const bool sq_enabled = [] {
    const char* e = getenv("SPARKFLOW_CONV_SQ");
    return !(e && e[0] == '0');
  }();
  // measured (benchmarks/wgrad_micro.py): the sq tile wins at
  // Cout >= 256 (l3 +10%, l4 +2%) and loses slightly at 128 — dispatch
  // at M >= 2*BM3
  if (sq_enabled && M >= 2 * BM3 && (s.Cout & 3) == 0) {
    const int base_sq = ceil_div(Nc, BN3) * ceil_div(M, BM3);
    int splitk = 1;
    const int max_by_k = std::max(1, K / 256);
    const int cap = sf_splitk_env_max();
    while (splitk < 256 && splitk < cap && base_sq * splitk <= 1024 &&
           splitk * 2 <= max_by_k) {
      splitk *= 2;
    }
    while (base_sq * splitk < 512 && splitk < cap &&
           splitk * 2 <= max_by_k && splitk < 1024) {
      splitk *= 2;
    }
    if (splitk > 1) {
      const int k_chunk = ceil_div(K, splitk * BK2) * BK2;
      auto ws = torch::empty({splitk, M, Nc}, x.options());
      dim3 grid(ceil_div(Nc, BN3), ceil_div(M, BM3), splitk);
      hipLaunchKernelGGL((conv_nhwc_wgrad_sq_kernel<true>), grid,
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         dy.data_ptr<float>(), ws.data_ptr<float>(), s, M,
                         Nc, K, k_chunk);
      const long n = (long)M * Nc;
      launch_conv_splitk_reduce(ws.data_ptr<float>(), dw.data_ptr<float>(),
                                n, splitk, stream);
    } else {
      dim3 grid(ceil_div(Nc, BN3), ceil_div(M, BM3));
      hipLaunchKernelGGL((conv_nhwc_wgrad_sq_kernel<false>), grid,
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         dy.data_ptr<float>(), dw.data_ptr<float>(), s, M,
                         Nc, K, K);
    }
    return dw;
  }
  // narrow-M tile for Cout <= 32 (same rationale as the fwd n32 tile)
  if (M <= 32) {
    const int base32 = ceil_div(Nc, BN) * ceil_div(M, 32);
    int splitk = 1;
    const int max_by_k = std::max(1, K / 256);
    const int cap = sf_splitk_env_max();
    while (splitk < 256 && splitk < cap && base32 * splitk <= 1024 &&
           splitk * 2 <= max_by_k) {
      splitk *= 2;
    }
    while (base32 * splitk < 512 && splitk < cap &&
           splitk * 2 <= max_by_k && splitk < 1024) {
      splitk *= 2;
    }
    if (splitk > 1) {
      const int k_chunk = ceil_div(K, splitk * BK) * BK;
      auto ws = torch::empty({splitk, M, Nc}, x.options());
      dim3 grid(ceil_div(Nc, BN), ceil_div(M, 32), splitk);
      hipLaunchKernelGGL((conv_nhwc_wgrad_m32_kernel<true>), grid,
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         dy.data_ptr<float>(), ws.data_ptr<float>(), s, M,
                         Nc, K, k_chunk);
      const long n = (long)M * Nc;
      launch_conv_splitk_reduce(ws.data_ptr<float>(), dw.data_ptr<float>(),
                                n, splitk, stream);
    } else {
      dim3 grid(ceil_div(Nc, BN), ceil_div(M, 32));
      hipLaunchKernelGGL((conv_nhwc_wgrad_m32_kernel<false>), grid,
                         dim3(256), 0, stream, x.data_ptr<float>(),
                         dy.data_ptr<float>(), dw.data_ptr<float>(), s, M,
                         Nc, K, K);
    }
    return dw;
  }
  const int base = ceil_div(Nc, BN) * ceil_div(M, BM);
  int splitk = 1;
  const int max_by_k = std::max(1, K / 256);
  const int cap = sf_splitk_env_max();
  while (splitk < 256 && splitk < cap && base * splitk <= 1024 &&
         splitk * 2 <= max_by_k) {
    splitk *= 2;
  }
  // tiny-dw wgrads (base = 1 tile, e.g. MNIST conv1 dw[32][25]) stall at
  // splitk=256 = 256 workgroups = 1 per CU with poor latency hiding;
  // keep doubling toward the 512-workgroup fill target
  while (base * splitk < 512 && splitk < cap && splitk * 2 <= max_by_k &&
         splitk < 1024) {
    splitk *= 2;
  }
  if (splitk > 1) {
    const int k_chunk = ceil_div(K, splitk * BK) * BK;
    auto ws = torch::empty({splitk, M, Nc}, x.options());
    dim3 grid(ceil_div(Nc, BN), ceil_div(M, BM), splitk);
    hipLaunchKernelGGL((conv_nhwc_wgrad_kernel<true>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       ws.data_ptr<float>(), s, M, Nc, K, k_chunk);
    const long n = (long)M * Nc;
    launch_conv_splitk_reduce(ws.data_ptr<float>(), dw.data_ptr<float>(), n,
                              splitk, stream);
  } else {
    dim3 grid(ceil_div(Nc, BN), ceil_div(M, BM));
    hipLaunchKernelGGL((conv_nhwc_wgrad_kernel<false>), grid, dim3(256), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       dw.data_ptr<float>(), s, M, Nc, K, K);
  }
  return dw;
}

// dy: [N, Ho, Wo, Cout] (act-bwd'ed), wt: [Cin, Kh, Kw, Cout]
// (= w.flip(1,2).permute(3,1,2,0)) -> dx: [N, H, W, Cin].
torch::Tensor conv_nhwc_dgrad_strided(torch::Tensor dy, torch::Tensor wt,
                                      int64_t H, int64_t W, int64_t stride,
                                      int64_t pad) {
  check_f32w(dy, "dy");
  check_f32w(wt, "wt");
  ConvShapeHW s;
  s.N = dy.size(0); s.Ho = dy.size(1); s.Wo = dy.size(2);
  s.Cout = dy.size(3);
  s.Cin = wt.size(0); s.Kh = wt.size(1); s.Kw = wt.size(2);
  TORCH_CHECK(wt.size(3) == s.Cout, "Cout mismatch");
  TORCH_CHECK((s.Cout & 3) == 0, "strided dgrad needs Cout % 4 == 0");
  s.H = H; s.W = W; s.stride = stride; s.pad = pad;
  const int S = stride;
  const int pprime = s.Kh - 1 - pad;
  // the NHWC conv family takes one kernel size / one pad (square);
  // the class decomposition below relies on it
  TORCH_CHECK(s.Kh == s.Kw, "strided dgrad expects square kernels");
  // any class with zero taps leaves its dx pixels unwritten -> zeros
  bool any_empty = false;
  for (int a = 0; a < S && !any_empty; ++a) {
    const int kh0 = ((pprime - a) % S + S) % S;
    if (s.Kh <= kh0) any_empty = true;
  }
  auto dx = any_empty
                ? torch::zeros({s.N, (long)H, (long)W, s.Cin},
                               dy.options())
                : torch::empty({s.N, (long)H, (long)W, s.Cin},
                               dy.options());
  const long maxM = (long)s.N * ((H + S - 1) / S) * ((W + S - 1) / S);
  dim3 grid(ceil_div(s.Cin, BN), ceil_div((int)maxM, BM), S * S);
  hipLaunchKernelGGL(conv_nhwc_dgrad_s_kernel, grid, dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(),
                     dy.data_ptr<float>(), wt.data_ptr<float>(),
                     dx.data_ptr<float>(), s, pprime);
  return dx;
}

torch::Tensor conv_nhwc_dilate(torch::Tensor dy, int64_t s, int64_t extra_h,
                               int64_t extra_w) {
  check_f32w(dy, "dy");
  const int N = dy.size(0), Ho = dy.size(1), Wo = dy.size(2),
            C = dy.size(3);
  const long Hd = (long)(Ho - 1) * s + 1 + extra_h;
  const long Wd = (long)(Wo - 1) * s + 1 + extra_w;
  auto out = torch::zeros({N, Hd, Wd, C}, dy.options());
  const int chunks = std::max(1, std::min(Ho, 2048 / std::max(1, N)));
  hipLaunchKernelGGL(dilate_nhwc_kernel, dim3(N, chunks),
                     dim3(std::min(256, std::max(64, C))), 0,
                     at::cuda::getCurrentCUDAStream(), dy.data_ptr<float>(),
                     out.data_ptr<float>(), Ho, Wo, C, (int)s, Hd * Wd * C,
                     (int)Wd);
  return out;
}

// db[co] = sum over (n, ho, wo) of dy[..., co] — rows are co-contiguous.
// Same access shape as the BN stats kernel (64 channel lanes x 4 row
// groups per 256-thread block, LDS combine, one atomic per channel per
// block): the round-1 version strided one thread per channel down the
// rows and measured 26% of the MNIST-CNN step at 0.43 TB/s.
__global__ void conv_nhwc_bias_grad_kernel(const float* __restrict__ dy,
                                           float* __restrict__ db, long rows,
                                           int C, int lg_lanes) {
  // lane width adapts to C (e.g. C=32 -> 32 lanes x 8 row groups, so a
  // wave reads 2 FULL consecutive rows — no idle lanes, contiguous)
  const int lanes = 1 << lg_lanes;
  const int rgs = 256 >> lg_lanes;
  const int cl = threadIdx.x & (lanes - 1);
  const int rg = threadIdx.x >> lg_lanes;
  const int c = blockIdx.x * lanes + cl;
  float s = 0.0f;
  if (c < C) {
    for (long r = (long)blockIdx.y * rgs + rg; r < rows;
         r += (long)gridDim.y * rgs) {
      s += dy[r * C + c];
    }
  }
  __shared__ float ls[256];
  ls[threadIdx.x] = s;
  __syncthreads();
  if (rg == 0 && c < C) {
    for (int k = 1; k < rgs; ++k) s += ls[k * lanes + cl];
    atomicAdd(&db[c], s);
  }
}

torch::Tensor conv_nhwc_bias_grad(torch::Tensor dy) {
  check_f32w(dy, "dy");
  const int C = dy.size(3);
  const long rows = dy.numel() / C;
  auto db = torch::zeros({C}, dy.options());
  int lg = 4;  // >= 16 lanes
  while ((1 << (lg + 1)) <= C && lg < 6) ++lg;
  const int lanes = 1 << lg;
  const int cgrids = (C + lanes - 1) / lanes;
  const int rgs = 256 >> lg;
  long chunks = (rows + rgs - 1) / rgs;
  // cap the per-address atomic chain (chunks atomics per channel):
  // CNN-sized tensors are small enough that 2048 blocks made the
  // same-address atomics the bound, not the streaming reads
  chunks = std::min(chunks, (long)std::max(1, 512 / cgrids));
  hipLaunchKernelGGL(conv_nhwc_bias_grad_kernel, dim3(cgrids, (int)chunks),
                     dim3(256), 0, at::cuda::getCurrentCUDAStream(),
                     dy.data_ptr<float>(), db.data_ptr<float>(), rows, C,
                     lg);
  return db;
}

// ---- NHWC maxpool --------------------------------------------------------
// Thread-per-output-float4 along C (NHWC rows are channel-contiguous, so
// the K*K window reads and the y/idx stores are all full-width float4
// transactions); grid-stride over every (n, ho, wo, c4).  idx stores
// h*W+w (per-plane convention, layout-independent).  The round-1
// kernel swept wo serially per thread with scalar loads and ran ~10x
// off the bandwidth floor (2.35 ms on the 300x112x112x64 ResNet pool,
// profiles/); this one is a plain streaming kernel.
__global__ void maxpool_nhwc_fwd_kernel(const float* __restrict__ x,
                                        float* __restrict__ y,
                                        long* __restrict__ idx, int N, int C,
                                        int H, int W, int K, int stride,
                                        int pad, int Ho, int Wo) {
  const int C4 = C >> 2;  // caller guarantees C % 4 == 0
  const long total = (long)N * Ho * Wo * C4;
  const long span = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += span) {
    const int c4 = i % C4;
    long rest = i / C4;
    const int wo = rest % Wo;
    rest /= Wo;
    const int ho = rest % Ho;
    const int n = rest / Ho;
    const float* xp = x + (long)n * H * W * C + c4 * 4;
    const int h0 = ho * stride - pad;
    const int w0 = wo * stride - pad;
    float4 best = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
    long bhw[4] = {0, 0, 0, 0};
    for (int kh = 0; kh < K; ++kh) {
      const int h = h0 + kh;
      if (h < 0) continue;
      if (h >= H) break;
      for (int kw = 0; kw < K; ++kw) {
        const int ww = w0 + kw;
        if (ww < 0) continue;
        if (ww >= W) break;
        const float4 v = *reinterpret_cast<const float4*>(
            &xp[((long)h * W + ww) * C]);
        const long hw = (long)h * W + ww;
        if (v.x > best.x) { best.x = v.x; bhw[0] = hw; }
        if (v.y > best.y) { best.y = v.y; bhw[1] = hw; }
        if (v.z > best.z) { best.z = v.z; bhw[2] = hw; }
        if (v.w > best.w) { best.w = v.w; bhw[3] = hw; }
      }
    }
    const long obase = i * 4;  // (((n*Ho+ho)*Wo+wo)*C + c4*4)
    *reinterpret_cast<float4*>(&y[obase]) = best;
    idx[obase + 0] = bhw[0];
    idx[obase + 1] = bhw[1];
    idx[obase + 2] = bhw[2];
    idx[obase + 3] = bhw[3];
  }
}

// backward as a GATHER over the <= ceil(K/stride)^2 windows that can
// contain each input pixel (no atomics, no dx pre-zeroing, coalesced
// float4 stores) — each window's winner index is compared against this
// pixel's h*W+w.
__global__ void maxpool_nhwc_bwd_kernel(const float* __restrict__ dy,
                                        const long* __restrict__ idx,
                                        float* __restrict__ dx, int N, int C,
                                        int H, int W, int K, int stride,
                                        int pad, int Ho, int Wo) {
  const int C4 = C >> 2;
  const long total = (long)N * H * W * C4;
  const long span = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += span) {
    const int c4 = i % C4;
    long rest = i / C4;
    const int w = rest % W;
    rest /= W;
    const int h = rest % H;
    const int n = rest / H;
    const long me = (long)h * W + w;
    float4 acc = {0.0f, 0.0f, 0.0f, 0.0f};
    // windows (ho, wo) with ho*stride - pad <= h < ho*stride - pad + K
    const int ho_hi = min(Ho - 1, (h + pad) / stride);
    const int ho_lo = max(0, (h + pad - K + stride) / stride);
    const int wo_hi = min(Wo - 1, (w + pad) / stride);
    const int wo_lo = max(0, (w + pad - K + stride) / stride);
    const float* dyp = dy + ((long)n * Ho * Wo) * C + c4 * 4;
    const long* ip = idx + ((long)n * Ho * Wo) * C + c4 * 4;
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        const long o = ((long)ho * Wo + wo) * C;
        const float4 d = *reinterpret_cast<const float4*>(&dyp[o]);
        if (ip[o + 0] == me) acc.x += d.x;
        if (ip[o + 1] == me) acc.y += d.y;
        if (ip[o + 2] == me) acc.z += d.z;
        if (ip[o + 3] == me) acc.w += d.w;
      }
    }
    *reinterpret_cast<float4*>(
        &dx[(((long)n * H + h) * W + w) * C + c4 * 4]) = acc;
  }
}

// scalar fallbacks for C % 4 != 0 (round-1 kernels, unchanged shape)
__global__ void maxpool_nhwc_fwd_scalar_kernel(
    const float* __restrict__ x, float* __restrict__ y,
    long* __restrict__ idx, int N, int C, int H, int W, int K, int stride,
    int pad, int Ho, int Wo) {
  const int n = blockIdx.x;
  const float* xp = x + (long)n * H * W * C;
  for (int ho = blockIdx.y; ho < Ho; ho += gridDim.y) {
    const int h0 = ho * stride - pad;
    for (int wo = 0; wo < Wo; ++wo) {
      const int w0 = wo * stride - pad;
      const long obase = (((long)n * Ho + ho) * Wo + wo) * C;
      for (int c = threadIdx.x; c < C; c += blockDim.x) {
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
            const float v = xp[((long)h * W + ww) * C + c];
            if (v > best) {
              best = v;
              best_hw = (long)h * W + ww;
            }
          }
        }
        y[obase + c] = best;
        idx[obase + c] = best_hw;
      }
    }
  }
}

__global__ void maxpool_nhwc_bwd_scalar_kernel(const float* __restrict__ dy,
                                               const long* __restrict__ idx,
                                               float* __restrict__ dx, int N,
                                               int C, int HW, int HoWo) {
  const int n = blockIdx.x;
  const float* dyp = dy + (long)n * HoWo * C;
  const long* ip = idx + (long)n * HoWo * C;
  float* dxp = dx + (long)n * HW * C;
  const int span = gridDim.y * blockDim.x;
  for (long i = (long)blockIdx.y * blockDim.x + threadIdx.x;
       i < (long)HoWo * C; i += span) {
    const int c = i % C;
    atomicAdd(&dxp[ip[i] * C + c], dyp[i]);
  }
}

std::vector<torch::Tensor> maxpool_nhwc_fwd(torch::Tensor x, int64_t k,
                                            int64_t stride, int64_t pad) {
  check_f32w(x, "x");
  const int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  const int Ho = (H + 2 * pad - k) / stride + 1;
  const int Wo = (W + 2 * pad - k) / stride + 1;
  auto y = torch::empty({N, Ho, Wo, C}, x.options());
  auto idx = torch::empty({N, Ho, Wo, C}, x.options().dtype(torch::kInt64));
  auto stream = at::cuda::getCurrentCUDAStream();
  if ((C & 3) == 0) {
    const long total = (long)N * Ho * Wo * (C / 4);
    const int blocks =
        (int)std::min<long>(8192, (total + 255) / 256);
    hipLaunchKernelGGL(maxpool_nhwc_fwd_kernel, dim3(blocks), dim3(256), 0,
                       stream, x.data_ptr<float>(), y.data_ptr<float>(),
                       idx.data_ptr<long>(), N, C, H, W, (int)k,
                       (int)stride, (int)pad, Ho, Wo);
  } else {
    const int chunks = std::max(1, std::min(Ho, 2048 / std::max(1, N)));
    hipLaunchKernelGGL(maxpool_nhwc_fwd_scalar_kernel, dim3(N, chunks),
                       dim3(std::min(256, std::max(64, C))), 0, stream,
                       x.data_ptr<float>(), y.data_ptr<float>(),
                       idx.data_ptr<long>(), N, C, H, W, (int)k,
                       (int)stride, (int)pad, Ho, Wo);
  }
  return {y, idx};
}

torch::Tensor maxpool_nhwc_bwd(torch::Tensor dy, torch::Tensor idx,
                               int64_t H, int64_t W, int64_t k,
                               int64_t stride, int64_t pad) {
  check_f32w(dy, "dy");
  const int N = dy.size(0), C = dy.size(3);
  const int Ho = dy.size(1), Wo = dy.size(2);
  auto stream = at::cuda::getCurrentCUDAStream();
  if ((C & 3) == 0) {
    auto dx = torch::empty({N, (long)H, (long)W, C}, dy.options());
    const long total = (long)N * H * W * (C / 4);
    const int blocks =
        (int)std::min<long>(8192, (total + 255) / 256);
    hipLaunchKernelGGL(maxpool_nhwc_bwd_kernel, dim3(blocks), dim3(256), 0,
                       stream, dy.data_ptr<float>(), idx.data_ptr<long>(),
                       dx.data_ptr<float>(), N, C, (int)H, (int)W, (int)k,
                       (int)stride, (int)pad, Ho, Wo);
    return dx;
  }
  auto dx = torch::zeros({N, (long)H, (long)W, C}, dy.options());
  const int HoWo = Ho * Wo;
  const int chunks =
      std::max(1, std::min((HoWo * C + 255) / 256,
                           4096 / std::max(1, N)));
  hipLaunchKernelGGL(maxpool_nhwc_bwd_scalar_kernel, dim3(N, chunks),
                     dim3(256), 0, stream, dy.data_ptr<float>(),
                     idx.data_ptr<long>(), dx.data_ptr<float>(), N, C,
                     (int)(H * W), HoWo);
  return dx;
}
