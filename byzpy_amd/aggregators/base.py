"""Aggregator operator base.

Reference parity: aggregators/base.py:38-103. ``aggregate()`` is the
public one-shot API; as an Operator, ``compute`` routes graph input key
"gradients" into it. Subtask machinery (CPU pools): the (n, d) matrix is
registered once (shm for process pools, zero-copy view for thread pools)
and chunk subtasks reopen it.
"""
from __future__ import annotations

from typing import Any, List, Sequence, Tuple

import torch

from byzpy_amd.ops.base import Operator, OpContext
from byzpy_amd.storage.shared_store import (
    SharedTensorHandle,
    cleanup_tensor,
    register_tensor,
)
from byzpy_amd.utils.flatten import LikeTemplate, stack_gradients, to_like


def build_matrix_ref(
    ctx: OpContext, gradients: Any
) -> Tuple[Any, torch.Tensor, LikeTemplate, List[SharedTensorHandle]]:
    """Build the (n, d) matrix and a reference shippable to pool workers:
    a shm handle for process pools, the tensor itself (zero-copy) otherwise.
    Returns (ref, X, like, handles_to_cleanup)."""
    X, like = stack_gradients(gradients)
    handles: List[SharedTensorHandle] = []
    use_shm = bool(
        ctx.pool is not None and getattr(ctx.pool, "prefers_shared_memory", False)
    ) and not X.is_cuda
    if use_shm:
        h = register_tensor(X)
        handles.append(h)
        return h, X, like, handles
    return X, X, like, handles


def cleanup_handles(handles: Sequence[SharedTensorHandle]) -> None:
    for h in handles:
        cleanup_tensor(h)


class Aggregator(Operator):
    name = "aggregator"
    input_key = "gradients"

    # -- public API --------------------------------------------------------
    def aggregate(self, gradients: Any) -> Any:
        X, like = stack_gradients(gradients)
        out = self._aggregate(X)
        return to_like(out.reshape(-1), like)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    # -- Operator ----------------------------------------------------------
    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        return self.aggregate(inputs[self.input_key])

    # -- subtask plumbing shared by subclasses ------------------------------
    def _matrix_ref(self, ctx: OpContext, gradients: Any):
        return build_matrix_ref(ctx, gradients)

    @staticmethod
    def _cleanup(handles: Sequence[SharedTensorHandle]) -> None:
        cleanup_handles(handles)
