"""byzpy_amd — MI355X-native Byzantine-robust distributed learning framework.

A from-scratch rebuild of the capabilities of ByzPy (reference:
/root/reference) designed for AMD Instinct MI355X (gfx950):

- single array backend: PyTorch-ROCm tensors (CPU eager path doubles as the
  numerical reference for kernel parity tests),
- hand-written HIP/CDNA4 kernels for every robust-aggregation hot path
  (byzpy_amd/hip/csrc/*.hip), loaded from the in-tree extension ``_hip_ops``,
- RCCL over xGMI (torch.distributed, backend "nccl") for multi-GPU
  d-sharded aggregation (byzpy_amd/parallel/).

Public API mirrors the reference surface (reference: python/byzpy/__init__.py:1-4):
``run_operator`` / ``OperatorExecutor`` plus the aggregator / attack /
pre-aggregator operator families.
"""

from byzpy_amd._version import __version__
from byzpy_amd.graph.executor import OperatorExecutor, run_operator

__all__ = ["OperatorExecutor", "run_operator", "__version__"]
