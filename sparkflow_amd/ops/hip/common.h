// Common helpers for the sparkflow_amd gfx950 (CDNA4) kernels.
//
// All kernels are fp32 (the reference's TF-1.x numerics) and are written
// MI355X-first: 64-wide wavefronts, MFMA matrix cores
// (v_mfma_f32_16x16x4_f32 — exact fp32 at the 157 TF matrix rate), LDS
// tiling, grid-stride elementwise sweeps sized for 256 CUs.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64

#define HIP_OK(expr)                                                          \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,      \
             __LINE__);                                                       \
    }                                                                         \
  } while (0)

// fp32 MFMA accumulator fragment (v_mfma_f32_16x16x4_f32: 4 f32 per lane)
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// Activation codes shared with sparkflow_amd/ops/functional.py
#define ACT_NONE 0
#define ACT_RELU 1
#define ACT_SIGMOID 2

__device__ __forceinline__ float act_apply(float x, int act) {
  if (act == ACT_RELU) return x > 0.0f ? x : 0.0f;
  if (act == ACT_SIGMOID) return 1.0f / (1.0f + __expf(-x));
  return x;
}

// d(pre)/d(out) recomputed from the activation OUTPUT y.
__device__ __forceinline__ float act_grad_from_out(float y, float dy, int act) {
  if (act == ACT_RELU) return y > 0.0f ? dy : 0.0f;
  if (act == ACT_SIGMOID) return dy * y * (1.0f - y);
  return dy;
}

// Wave-level sum over all 64 lanes (result in every lane).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

static inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

// SPARKFLOW_SPLITK_MAX env override for split-K experiments (unset = no
// cap; structural limits at the call sites govern).  Defined in gemm.hip.
int sf_splitk_env_max();
