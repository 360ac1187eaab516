"""MI355X-native compute ops.

The hot ops of the reference's implicit TF-C++ surface are
hand-written CDNA4 HIP kernels under ``sparkflow_amd/ops/hip/`` compiled
for gfx950 into an in-tree extension ``_build/sparkflow_hip.so``:

* fused fp32 MFMA GEMM + bias + activation (fwd / dgrad / wgrad)
* fused softmax-cross-entropy and MSE loss (+grad)
* conv2d implicit-GEMM (fwd / dgrad / wgrad), maxpool2d
* fused single-launch optimizer applies (SGD / momentum / Adam / RMSProp)
  used by the rank-0 GPU parameter server

Dispatch contract: on a GPU tensor these kernels are THE execution path —
if the extension is missing on a machine with a GPU, ops raise instead of
silently falling back to eager PyTorch.  On CPU tensors (no GPU in the
container) plain fp32 torch ops implement the same math for tests.
"""
from __future__ import annotations

import torch

from sparkflow_amd.ops import loader

hip = loader.LazyExt()


def available() -> bool:
    return loader.try_load() is not None


def use_hip(t: torch.Tensor) -> bool:
    """True iff ``t`` lives on a GPU.  Raises (loudly) when a GPU tensor is
    seen but the gfx950 extension is not importable — a silent eager
    fallback on the GPU box would invalidate every benchmark number."""
    if not t.is_cuda:
        return False
    if loader.try_load() is None:
        raise RuntimeError(
            "sparkflow_amd HIP extension is not built but a GPU tensor was "
            "used. Build it with: python -c 'import __graft_entry__ as g; g.build()' "
            "(or python setup.py build_ext --inplace). Original error: %s"
            % (loader.LAST_ERROR,)
        )
    return True
