// Conv split-K reduce kernels, shared by the NCHW and NHWC conv TUs.
// Defined `static` in this header because the extension builds with
// -fno-gpu-rdc (no cross-TU device linking).
#pragma once

#include "common.h"

// One block per output element, threads parallel over the z planes —
// used when splitk >= 64 (a per-thread z loop would be a serial latency
// chain); shallow splits use the flat variants below.
static __global__ void conv_splitk_reduce_kernel(
    const float* __restrict__ W, float* __restrict__ C, long n, int splitk) {
  for (long i = blockIdx.x; i < n; i += gridDim.x) {
    float sum = 0.0f;
    for (int z = threadIdx.x; z < splitk; z += blockDim.x)
      sum += W[(long)z * n + i];
    sum = wave_sum(sum);
    __shared__ float partials[4];
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) partials[wid] = sum;
    __syncthreads();
    if (threadIdx.x == 0) {
      float b = 0.0f;
      for (int k = 0; k < (int)(blockDim.x / WAVE); ++k) b += partials[k];
      C[i] = b;
    }
    __syncthreads();
  }
}

static __global__ void conv_splitk_reduce_flat_kernel(
    const float4* __restrict__ W, float4* __restrict__ C, long n4,
    int splitk) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 s = {0.0f, 0.0f, 0.0f, 0.0f};
    for (int z = 0; z < splitk; ++z) {
      const float4 w = W[(long)z * n4 + i];
      s.x += w.x; s.y += w.y; s.z += w.z; s.w += w.w;
    }
    C[i] = s;
  }
}

static __global__ void conv_splitk_reduce_flat1_kernel(
    const float* __restrict__ W, float* __restrict__ C, long n, int splitk) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float s = 0.0f;
    for (int z = 0; z < splitk; ++z) s += W[(long)z * n + i];
    C[i] = s;
  }
}

// Shared launch helper: reduce ws[splitk, n] into out[n].
static inline void launch_conv_splitk_reduce(const float* ws, float* out,
                                             long n, int splitk,
                                             hipStream_t stream) {
  if (splitk >= 64) {
    const int blocks = (int)std::min<long>(8192, n);
    hipLaunchKernelGGL(conv_splitk_reduce_kernel, dim3(blocks), dim3(256),
                       0, stream, ws, out, n, splitk);
  } else if ((n & 3) == 0) {
    const long n4 = n >> 2;
    const int blocks = (int)std::min<long>(4096, (n4 + 255) / 256);
    hipLaunchKernelGGL(conv_splitk_reduce_flat_kernel, dim3(blocks),
                       dim3(256), 0, stream,
                       reinterpret_cast<const float4*>(ws),
                       reinterpret_cast<float4*>(out), n4, splitk);
  } else {
    const int blocks = (int)std::min<long>(4096, (n + 255) / 256);
    hipLaunchKernelGGL(conv_splitk_reduce_flat1_kernel, dim3(blocks),
                       dim3(256), 0, stream, ws, out, n, splitk);
  }
}
