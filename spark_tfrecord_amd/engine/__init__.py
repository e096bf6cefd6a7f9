"""Engine dispatch: CPU host codec vs gfx950 GPU pipeline.

The native extension is mandatory — there is no pure-Python fallback, and on
a GPU machine the GPU kernels must be present (a silent eager fallback would
invalidate benchmarks)."""

from __future__ import annotations

import os


def native():
    try:
        from .. import _native
    except ImportError as e:  # pragma: no cover
        raise ImportError(
            "spark_tfrecord_amd._native is not built; run `python build_native.py` "
            "(hipcc, offload-arch gfx950)") from e
    return _native


def gpu_available() -> bool:
    if os.environ.get("TFREC_FORCE_CPU"):
        return False
    try:
        import torch
    except ImportError:
        return False
    if not torch.cuda.is_available():
        return False
    n = native()
    if not getattr(n, "HAS_GPU_KERNELS", False):
        raise RuntimeError(
            "A GPU is visible but spark_tfrecord_amd._native was built without "
            "HIP kernels — rebuild with build_native.py (refusing to fall back "
            "to the CPU path silently)")
    return True


def resolve_engine(engine: str) -> str:
    engine = (engine or "auto").lower()
    if engine == "auto":
        return "gpu" if gpu_available() else "cpu"
    if engine == "gpu":
        if not gpu_available():
            raise RuntimeError("engine='gpu' requested but no HIP GPU is available")
        return "gpu"
    if engine == "cpu":
        return "cpu"
    raise ValueError(f"unknown engine {engine!r}")
