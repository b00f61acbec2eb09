"""Loader for the in-tree gfx950 HIP extension (``byzpy_amd._hip_ops``).

The extension is built in-tree by ``setup.py build_ext --inplace`` (or
``__graft_entry__.build()``) so the .so travels with the repo snapshot.
Policy: on CUDA(ROCm) tensors the HIP kernels are REQUIRED — ops raise
instead of silently falling back to eager torch (the CPU eager path is the
parity oracle, not a production path).
"""
from __future__ import annotations

import importlib
from typing import Any, Optional

_ext: Optional[Any] = None
_tried = False
_err: Optional[BaseException] = None


def extension() -> Optional[Any]:
    global _ext, _tried, _err
    if not _tried:
        _tried = True
        try:
            import torch  # noqa: F401  (libtorch must be loaded first)

            _ext = importlib.import_module("byzpy_amd._hip_ops")
        except Exception as e:  # noqa: BLE001
            _err = e
            _ext = None
    return _ext


def available() -> bool:
    return extension() is not None


def require() -> Any:
    ext = extension()
    if ext is None:
        raise RuntimeError(
            "byzpy_amd._hip_ops HIP extension is not built/loadable but a CUDA "
            "tensor reached a hot op. Build it with `python setup.py "
            f"build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). Cause: {_err!r}"
        )
    return ext
