// Fused single-launch optimizer applies over the PS's flat fp32 weight
// buffer (the north-star "fused multi-tensor SGD and Adam HIP kernels on
// the rank-0 PS GPU" — BASELINE.json; reference applies TF optimizers per
// incoming POST, HogwildSparkModel.py:219-242).
//
// Because the canonical weights are ONE contiguous tensor, a whole-model
// apply is a single grid-stride float4 sweep: one kernel launch per
// gradient push regardless of how many variables the model has (vs ~5
// launches for a torch optimizer even on a single tensor).  These kernels
// are what makes concurrent Hogwild applies cheap: each PS service thread
// enqueues one launch on its own HIP stream.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

#define GS_LOOP(i, n)                                                         \
  const long _stride = (long)gridDim.x * blockDim.x;                          \
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;           \
       i += _stride)

__global__ void sgd_kernel(float* __restrict__ p, const float* __restrict__ g,
                           float lr, long n) {
  GS_LOOP(i, n) { p[i] -= lr * g[i]; }
}

__global__ void momentum_kernel(float* __restrict__ p,
                                const float* __restrict__ g,
                                float* __restrict__ acc, float lr, float mom,
                                int nesterov, long n) {
  GS_LOOP(i, n) {
    const float a = mom * acc[i] + g[i];
    acc[i] = a;
    p[i] -= nesterov ? lr * (g[i] + mom * a) : lr * a;
  }
}

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            float lr_t, float b1, float b2, float eps,
                            long n) {
  GS_LOOP(i, n) {
    const float gi = g[i];
    const float mi = b1 * m[i] + (1.0f - b1) * gi;
    const float vi = b2 * v[i] + (1.0f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr_t * mi / (__fsqrt_rn(vi) + eps);
  }
}

// Device-state Adam in ONE launch: the step count and the NEXT step's
// bias-corrected lr live in a small device buffer
// (state[0]=t, state[1]=lr_t for step t+1, state[2]=block ticket), so a
// PS apply is hipGraph-capturable with the bias correction still
// advancing per replay.  Every thread reads this step's lr_t before the
// block's __syncthreads(); the LAST block to finish (ticket) advances t
// and precomputes lr_t for the next step — look-ahead scheduling that
// replaces the former 1-thread prep launch.  Concurrent Hogwild applies
// racing the ticket is the same benign lr_t race the prep kernel had.
__global__ void adam_dev_kernel(float* __restrict__ p,
                                const float* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                float* __restrict__ state, float lr,
                                float b1, float b2, float eps, long n) {
  const float lr_t = state[1];
  GS_LOOP(i, n) {
    const float gi = g[i];
    const float mi = b1 * m[i] + (1.0f - b1) * gi;
    const float vi = b2 * v[i] + (1.0f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    p[i] -= lr_t * mi / (__fsqrt_rn(vi) + eps);
  }
  __syncthreads();  // every thread of this block has consumed lr_t
  if (threadIdx.x == 0) {
    int* tick = reinterpret_cast<int*>(state + 2);
    if (atomicAdd(tick, 1) == (int)gridDim.x - 1) {
      // atomicSub (not a plain store) so increments from a CONCURRENT
      // Hogwild apply sharing this dev_state are never wiped: each +1
      // is preserved and every crossing of gridDim.x-1 fires exactly
      // one ticket, so t counts every apply.  atomicAdd on t for the
      // same reason; the lr_t look-ahead write stays a benign
      // last-writer race (one step of staleness at most).
      atomicSub(tick, (int)gridDim.x);
      const float tdone = atomicAdd(state, 1.0f) + 1.0f;  // this step
      const float tn = tdone + 1.0f;
      state[1] = lr * __fsqrt_rn(1.0f - __powf(b2, tn)) /
                 (1.0f - __powf(b1, tn));
    }
  }
}

__global__ void rmsprop_kernel(float* __restrict__ p,
                               const float* __restrict__ g,
                               float* __restrict__ ms,
                               float* __restrict__ mom, float lr, float decay,
                               float momentum, float eps, long n) {
  GS_LOOP(i, n) {
    const float gi = g[i];
    const float m2 = decay * ms[i] + (1.0f - decay) * gi * gi;
    ms[i] = m2;
    // TF rmsprop: epsilon inside the sqrt
    const float mo = momentum * mom[i] + lr * gi / __fsqrt_rn(m2 + eps);
    mom[i] = mo;
    p[i] -= mo;
  }
}

static inline void check_flat(const torch::Tensor& t, const char* name,
                              long n) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                  t.is_contiguous() && t.numel() == n,
              name, " must be contiguous fp32 on GPU with matching numel");
}

static inline int opt_blocks(long n) {
  return (int)std::min<long>(2048, (n + 255) / 256);
}

// The ticketed (look-ahead lr_t) adam kernel pays one same-address
// atomic per block — cap its grid so the ticket line stays short; the
// grid-stride loop amortizes the extra elements per block.
static inline int adam_blocks(long n) {
  return (int)std::min<long>(256, (n + 1023) / 1024);
}

void fused_sgd(torch::Tensor p, torch::Tensor g, double lr) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(g, "g", n);
  hipLaunchKernelGGL(sgd_kernel, dim3(opt_blocks(n)), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), (float)lr, n);
}

void fused_momentum(torch::Tensor p, torch::Tensor g, torch::Tensor acc,
                    double lr, double mom, bool nesterov) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(g, "g", n);
  check_flat(acc, "acc", n);
  hipLaunchKernelGGL(momentum_kernel, dim3(opt_blocks(n)), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), acc.data_ptr<float>(), (float)lr,
                     (float)mom, (int)nesterov, n);
}

void fused_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr_t, double b1, double b2,
                double eps) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(g, "g", n);
  check_flat(m, "m", n);
  check_flat(v, "v", n);
  hipLaunchKernelGGL(adam_kernel, dim3(opt_blocks(n)), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), (float)lr_t, (float)b1, (float)b2,
                     (float)eps, n);
}

void fused_adam_dev(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                    torch::Tensor v, torch::Tensor state, double lr,
                    double b1, double b2, double eps) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(g, "g", n);
  check_flat(m, "m", n);
  check_flat(v, "v", n);
  TORCH_CHECK(state.numel() >= 3 && state.is_cuda(), "bad adam state");
  auto stream = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(adam_dev_kernel, dim3(adam_blocks(n)), dim3(256), 0,
                     stream, p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(),
                     state.data_ptr<float>(), (float)lr, (float)b1,
                     (float)b2, (float)eps, n);
}

void fused_rmsprop(torch::Tensor p, torch::Tensor g, torch::Tensor ms,
                   torch::Tensor mom, double lr, double decay,
                   double momentum, double eps) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(g, "g", n);
  check_flat(ms, "ms", n);
  check_flat(mom, "mom", n);
  hipLaunchKernelGGL(rmsprop_kernel, dim3(opt_blocks(n)), dim3(256), 0,
                     at::cuda::getCurrentCUDAStream(), p.data_ptr<float>(),
                     g.data_ptr<float>(), ms.data_ptr<float>(),
                     mom.data_ptr<float>(), (float)lr, (float)decay,
                     (float)momentum, (float)eps, n);
}

// ---------------------------------------------------------------------------
// Fused split-K-reduce + Adam apply.
//
// The MLP step's tail was 6 bandwidth-trivial launches — three wgrad
// split-K reduces, a scalar reduce, pack_segments, adam — each paying
// the ~4-7 us launch floor (profiles/mlp_w4_kernels.md).  Because every
// wgrad workspace is [splits, Kin+1, Nout] whose plane is EXACTLY the
// layer's contiguous (weight, bias) slice of the flat buffer, one kernel
// can sum the partials of every layer and feed them straight into the
// device-state Adam update on the PS weights — the per-tensor grads, the
// packed flat-grad buffer, and five launches disappear.  The z-loop add
// order and the Adam math are identical to splitk_reduce_kernel +
// adam_dev_kernel, so the fused path is numerically identical too.
#define RA_MAX_SEG 16
struct RASeg {
  const float* ws;  // [splits, plane] workspace, plane == end - base
  long plane;       // elements per split slice
  long end;         // exclusive flat-buffer end of this segment
  int splits;
};
struct RAArgs {
  RASeg seg[RA_MAX_SEG];
  int nseg;
};

__global__ void reduce_apply_adam_kernel(RAArgs a, float* __restrict__ p,
                                         float* __restrict__ m,
                                         float* __restrict__ v,
                                         float* __restrict__ state, float lr,
                                         float b1, float b2, float eps) {
  const float lr_t = state[1];
  long base = 0;
  for (int k = 0; k < a.nseg; ++k) {  // uniform: no divergent seg lookup
    const float* __restrict__ ws = a.seg[k].ws;
    const long plane = a.seg[k].plane;
    const int splits = a.seg[k].splits;
    if (((plane | base) & 3) == 0) {
      // float4 path: 4x fewer load/RMW chains (the kernel is latency-
      // bound, not bandwidth-bound, at ~270k elements on 65k threads)
      const float4* __restrict__ ws4 = reinterpret_cast<const float4*>(ws);
      float4* __restrict__ p4 = reinterpret_cast<float4*>(p + base);
      float4* __restrict__ m4 = reinterpret_cast<float4*>(m + base);
      float4* __restrict__ v4 = reinterpret_cast<float4*>(v + base);
      const long plane4 = plane >> 2;
      GS_LOOP(off, plane4) {
        float4 g = {0.0f, 0.0f, 0.0f, 0.0f};
        for (int z = 0; z < splits; ++z) {
          const float4 w = ws4[(long)z * plane4 + off];
          g.x += w.x; g.y += w.y; g.z += w.z; g.w += w.w;
        }
        float4 mi = m4[off], vi = v4[off], pi = p4[off];
#define RA_ADAM1(c)                                                           \
  mi.c = b1 * mi.c + (1.0f - b1) * g.c;                                       \
  vi.c = b2 * vi.c + (1.0f - b2) * g.c * g.c;                                 \
  pi.c -= lr_t * mi.c / (__fsqrt_rn(vi.c) + eps);
        RA_ADAM1(x) RA_ADAM1(y) RA_ADAM1(z) RA_ADAM1(w)
#undef RA_ADAM1
        m4[off] = mi;
        v4[off] = vi;
        p4[off] = pi;
      }
    } else {
      GS_LOOP(off, plane) {
        float gi = 0.0f;
        for (int z = 0; z < splits; ++z) gi += ws[(long)z * plane + off];
        const long i = base + off;
        const float mi = b1 * m[i] + (1.0f - b1) * gi;
        const float vi = b2 * v[i] + (1.0f - b2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        p[i] -= lr_t * mi / (__fsqrt_rn(vi) + eps);
      }
    }
    base = a.seg[k].end;
  }
  __syncthreads();  // every thread of this block has consumed lr_t
  if (threadIdx.x == 0) {
    int* tick = reinterpret_cast<int*>(state + 2);
    if (atomicAdd(tick, 1) == (int)gridDim.x - 1) {
      // see adam_dev_kernel: atomicSub/atomicAdd so concurrent Hogwild
      // applies sharing one dev_state never lose a t increment
      atomicSub(tick, (int)gridDim.x);
      const float tdone = atomicAdd(state, 1.0f) + 1.0f;
      const float tn = tdone + 1.0f;
      state[1] = lr * __fsqrt_rn(1.0f - __powf(b2, tn)) /
                 (1.0f - __powf(b1, tn));
    }
  }
}

// ws[k]: [splits_k, plane_k] fp32 workspaces ordered by flat offset;
// offsets[k] is segment k's start and segments must tile [0, p.numel())
// contiguously (the caller derives offsets from the params' positions in
// the flat buffer and falls back to the unfused path when they don't).
void reduce_apply_adam(std::vector<torch::Tensor> ws,
                       std::vector<int64_t> offsets, torch::Tensor p,
                       torch::Tensor m, torch::Tensor v, torch::Tensor state,
                       double lr, double b1, double b2, double eps) {
  const long n = p.numel();
  check_flat(p, "p", n);
  check_flat(m, "m", n);
  check_flat(v, "v", n);
  TORCH_CHECK(state.numel() >= 3 && state.is_cuda(), "bad adam state");
  TORCH_CHECK(ws.size() == offsets.size() && !ws.empty() &&
                  ws.size() <= RA_MAX_SEG,
              "reduce_apply_adam: 1..16 segments");
  RAArgs a;
  a.nseg = (int)ws.size();
  long expect = 0;
  for (int k = 0; k < a.nseg; ++k) {
    const auto& t = ws[k];
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                    t.is_contiguous() && t.dim() >= 2,
                "bad workspace ", k);
    const int splits = (int)t.size(0);
    const long plane = t.numel() / splits;
    TORCH_CHECK(offsets[k] == expect,
                "workspaces must tile the flat buffer contiguously");
    a.seg[k].ws = t.data_ptr<float>();
    a.seg[k].plane = plane;
    a.seg[k].splits = splits;
    expect += plane;
    a.seg[k].end = expect;
  }
  TORCH_CHECK(expect == n, "workspaces must cover the whole flat buffer");
  hipLaunchKernelGGL(reduce_apply_adam_kernel, dim3(adam_blocks(n)),
                     dim3(256), 0, at::cuda::getCurrentCUDAStream(), a,
                     p.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), state.data_ptr<float>(),
                     (float)lr, (float)b1, (float)b2, (float)eps);
}
