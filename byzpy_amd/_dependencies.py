"""Dependency inventory and environment probing.

Reference parity: byzpy/_dependencies.py:9-88 — the reference sniffed
nvidia-smi/CUDA to pick cupy/ucxx extras. The MI355X build has one target
stack (torch-ROCm + RCCL + the in-tree gfx950 extension), so this module
reports what is required and what the current environment provides
(``byzpy-amd doctor`` renders it).
"""
from __future__ import annotations

from typing import Dict, List

CPU_DEPS: List[str] = ["torch", "numpy", "cloudpickle"]
GPU_STACK: List[str] = [
    "torch-rocm (hip runtime)",
    "rccl (torch.distributed backend 'nccl')",
    "byzpy_amd._hip_ops (in-tree gfx950 extension)",
]

# env overrides mirroring the reference's BYZPY_FORCE_GPU / BYZPY_FORCE_CPU
FORCE_GPU_ENV = "BYZPY_AMD_FORCE_GPU"
FORCE_CPU_ENV = "BYZPY_AMD_FORCE_CPU"


def gpu_forced() -> bool:
    import os

    return bool(os.environ.get(FORCE_GPU_ENV))


def cpu_forced() -> bool:
    import os

    return bool(os.environ.get(FORCE_CPU_ENV))


def probe() -> Dict[str, object]:
    """Probe the runtime stack; never raises."""
    info: Dict[str, object] = {"cpu_deps": {}, "gpu_stack": {}}
    for mod in ("torch", "numpy", "cloudpickle"):
        try:
            m = __import__(mod)
            info["cpu_deps"][mod] = getattr(m, "__version__", "ok")
        except Exception as e:  # noqa: BLE001
            info["cpu_deps"][mod] = f"MISSING ({e!r})"
    try:
        import torch

        info["gpu_stack"]["hip"] = getattr(torch.version, "hip", None)
        use_gpu = torch.cuda.is_available() and not cpu_forced()
        info["gpu_stack"]["device_visible"] = use_gpu
        info["gpu_stack"]["rccl"] = torch.distributed.is_nccl_available()
    except Exception as e:  # noqa: BLE001
        info["gpu_stack"]["torch"] = f"MISSING ({e!r})"
    from byzpy_amd import hip

    info["gpu_stack"]["hip_extension"] = hip.available()
    return info
