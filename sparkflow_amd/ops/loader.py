"""Build/load the in-tree gfx950 HIP extension.

The built ``.so`` lives at ``sparkflow_amd/ops/_build/sparkflow_hip.so`` —
in-tree so the gpurun snapshot carries it to the GPU box (a JIT cache under
``~/.cache`` would not travel).
"""
from __future__ import annotations

import importlib.util
import os
import sys
from pathlib import Path

HERE = Path(__file__).resolve().parent
BUILD_DIR = HERE / "_build"
EXT_NAME = "sparkflow_hip"
SOURCES = [
    HERE / "hip" / "bindings.cpp",
    HERE / "hip" / "gemm.hip",
    HERE / "hip" / "elementwise.hip",
    HERE / "hip" / "losses.hip",
    HERE / "hip" / "optim.hip",
    HERE / "hip" / "conv.hip",
    HERE / "hip" / "conv_gemm.hip",
    HERE / "hip" / "conv_nhwc.hip",
    HERE / "hip" / "norm.hip",
    HERE / "hip" / "dist_session.hip",
]

_ext = None
_tried = False
LAST_ERROR: Exception | None = None


def _so_path() -> Path | None:
    if not BUILD_DIR.is_dir():
        return None
    for cand in sorted(BUILD_DIR.glob(EXT_NAME + "*.so")):
        return cand
    return None


def try_load():
    """Import the prebuilt extension if present; never builds. Returns the
    module or None (error kept in LAST_ERROR)."""
    global _ext, _tried, LAST_ERROR
    if _ext is not None or _tried:
        return _ext
    _tried = True
    so = _so_path()
    if so is None:
        LAST_ERROR = FileNotFoundError(f"no {EXT_NAME}*.so under {BUILD_DIR}")
        return None
    try:
        import torch  # noqa: F401  (the extension links against torch libs)

        spec = importlib.util.spec_from_file_location(EXT_NAME, so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        sys.modules[EXT_NAME] = mod
        _ext = mod
    except Exception as e:  # keep the cause for use_hip()'s loud error
        LAST_ERROR = e
        _ext = None
    return _ext


def build(verbose: bool = True):
    """Compile every HIP source for gfx950 into the in-tree build dir.

    Uses torch.utils.cpp_extension (drives hipcc on ROCm); works with no GPU
    present (cross-compile) as long as PYTORCH_ROCM_ARCH=gfx950 is set.
    """
    global _ext, _tried
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(min(16, os.cpu_count() or 8)))
    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    from torch.utils import cpp_extension

    mod = cpp_extension.load(
        name=EXT_NAME,
        sources=[str(s) for s in SOURCES],
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        extra_ldflags=["-L/opt/rocm/lib", "-lrccl"],
        verbose=verbose,
        is_python_module=True,
    )
    _ext = mod
    _tried = True
    sys.modules[EXT_NAME] = mod
    return mod


class LazyExt:
    """Attribute proxy: ``ops.hip.fused_adam(...)`` resolves at call time."""

    def __getattr__(self, item):
        ext = try_load()
        if ext is None:
            raise RuntimeError(
                f"sparkflow_hip extension not available: {LAST_ERROR}"
            )
        return getattr(ext, item)
