"""Pre-aggregators: Clipping, Bucketing, NearestNeighborMixing, ARC.

Reference parity: pre_aggregators/{clipping,bucketing,nnm,arc}.py. On GPU
these are single kernel launches (the reference's process-pool slowdowns on
these cheap ops — BASELINE.md note — cannot happen by construction).
"""
from __future__ import annotations

import random
from typing import Any, Iterable, List, Optional, Sequence

import torch

from byzpy_amd.aggregators import _subtask_fns as SF
from byzpy_amd.aggregators._chunking import chunk_ranges, select_adaptive_chunk_size
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops.base import OpContext
from byzpy_amd.pre_aggregators.base import PreAggregator
from byzpy_amd.utils.flatten import to_like


class Clipping(PreAggregator):
    """Scale each row to L2 norm at most ``threshold``; row-chunked in-place
    clipping on CPU pools (reference clipping.py:92-127)."""

    name = "pre-agg/clipping"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, threshold: float = 2.0, *, chunk_size: int = 32) -> None:
        if threshold < 0:
            raise ValueError("threshold must be >= 0")
        self.threshold = float(threshold)
        self.chunk_size = int(chunk_size)

    def _pre_aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.clip_rows(X, self.threshold)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        vectors = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, vectors)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (like, handles)
        n = X.shape[0]
        chunk = max(1, min(self.chunk_size, n))
        return [
            SubTask(fn=SF.clip_rows_chunk, args=(ref, lo, hi, self.threshold))
            for lo, hi in chunk_ranges(n, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        like, handles = ctx.metadata.pop("_op_pending")
        try:
            out = torch.cat(results, dim=0)
            return [to_like(row, like) for row in out]
        finally:
            self._cleanup(handles)

class Bucketing(PreAggregator):
    """Random permutation -> buckets of ``bucket_size`` -> per-bucket mean;
    ``perm``/``rng`` injectable for determinism (bucketing.py:93-94)."""

    name = "pre-agg/bucketing"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(
        self,
        bucket_size: int,
        *,
        feature_chunk_size: int = 8192,
        perm: Optional[Iterable[int]] = None,
        rng: Optional[random.Random] = None,
    ) -> None:
        if bucket_size < 1:
            raise ValueError("bucket_size must be >= 1")
        self.bucket_size = int(bucket_size)
        self.feature_chunk_size = int(feature_chunk_size)
        self.perm = None if perm is None else [int(i) for i in perm]
        self.rng = rng or random.Random()

    def _draw_perm(self, n: int) -> List[int]:
        if self.perm is not None:
            return list(self.perm)
        perm = list(range(n))
        self.rng.shuffle(perm)
        return perm

    def _pre_aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.bucketing(X, self.bucket_size, self._draw_perm(X.shape[0]))

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        vectors = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, vectors)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (like, handles)
        perm = self._draw_perm(X.shape[0])
        n = X.shape[0]
        buckets = [
            perm[lo:hi] for lo, hi in chunk_ranges(n, self.bucket_size)
        ]
        return [SubTask(fn=SF.bucket_mean_chunk, args=(ref, b)) for b in buckets]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        like, handles = ctx.metadata.pop("_op_pending")
        try:
            return [to_like(r, like) for r in results]
        finally:
            self._cleanup(handles)



class NearestNeighborMixing(PreAggregator):
    """Replace x_i with the mean of its n-f nearest neighbours (incl. self);
    feature-chunked partial Gram on CPU pools (nnm.py:182-187)."""

    name = "pre-agg/nnm"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, *, feature_chunk_size: int = 8192) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.feature_chunk_size = int(feature_chunk_size)

    def _pre_aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.nnm(X, self.f)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        vectors = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, vectors)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (ref, X, like, handles)
        d = X.shape[1]
        chunk = select_adaptive_chunk_size(d, ctx.pool_size, self.feature_chunk_size)
        return [
            SubTask(fn=SF.gram_feature_chunk, args=(ref, lo, hi))
            for lo, hi in chunk_ranges(d, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        ref, X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            G = sum(results)
            norms = torch.diagonal(G)
            D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
            k = X.shape[0] - self.f
            idx = torch.topk(D2, k=k, dim=1, largest=False).indices
            out = X.float()[idx].mean(dim=1).to(X.dtype)
            return [to_like(row, like) for row in out]
        finally:
            self._cleanup(handles)



class ARC(PreAggregator):
    """Adaptive robust clipping: clip the floor(2f/n * (n-f)) largest-norm
    rows to the next-largest remaining norm (arc.py:36-61)."""

    name = "pre-agg/arc"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int = 0, *, chunk_size: int = 32) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.chunk_size = int(chunk_size)

    def _pre_aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.arc_clip(X, self.f)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        vectors = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, vectors)
        if X.is_cuda:
            return []
        n = X.shape[0]
        k = int(2 * self.f / n * (n - self.f))
        norms = X.float().norm(dim=1)
        if k <= 0:
            scale = torch.ones(n)
        else:
            order = torch.argsort(norms, descending=True)
            threshold = norms[order[k]]
            scale = torch.clamp(threshold / norms.clamp_min(1e-20), max=1.0)
        ctx.metadata["_op_pending"] = (like, handles)
        chunk = max(1, min(self.chunk_size, n))
        return [
            SubTask(
                fn=SF.scale_rows_chunk, args=(ref, lo, hi, scale[lo:hi].tolist())
            )
            for lo, hi in chunk_ranges(n, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        like, handles = ctx.metadata.pop("_op_pending")
        try:
            out = torch.cat(results, dim=0)
            return [to_like(row, like) for row in out]
        finally:
            self._cleanup(handles)

