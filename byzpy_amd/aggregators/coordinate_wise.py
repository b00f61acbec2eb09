"""Coordinate-wise robust aggregators (K1-K3 on the GPU path).

Reference parity: aggregators/coordinate_wise/{median,trimmed_mean,
mean_of_medians}.py. Deviation (documented, SURVEY.md §7 hard part 5):
both the direct and the chunked path compute the TRUE median (even n:
mean of the two middles); the reference's chunked path returned the lower
middle.
"""
from __future__ import annotations

from typing import Any, List, Sequence

import torch

from byzpy_amd.aggregators import _subtask_fns as SF
from byzpy_amd.aggregators._chunking import chunk_ranges, select_adaptive_chunk_size
from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops.base import OpContext
from byzpy_amd.utils.flatten import to_like


class _FeatureChunkedAggregator(Aggregator):
    """Shared machinery: chunk the feature dim, run a per-chunk fn, and
    concatenate the (d_chunk,) partials back in submission order."""

    supports_subtasks = True
    max_subtasks_inflight = 0
    default_chunk_size = 8192

    def __init__(self, *, chunk_size: int | None = None) -> None:
        self.chunk_size = int(chunk_size or self.default_chunk_size)

    def _chunk_fn(self):
        raise NotImplementedError

    def _chunk_args(self) -> tuple:
        return ()

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, gradients)
        if X.is_cuda:
            return []  # GPU: single-kernel compute path
        ctx.metadata["_op_pending"] = (like, handles)
        d = X.shape[1]
        chunk = select_adaptive_chunk_size(d, ctx.pool_size, self.chunk_size)
        fn = self._chunk_fn()
        extra = self._chunk_args()
        return [
            SubTask(fn=fn, args=(ref, lo, hi, *extra), name=f"{self.name}[{lo}:{hi}]")
            for lo, hi in chunk_ranges(d, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        like, handles = ctx.metadata.pop("_op_pending")
        try:
            out = torch.cat([torch.as_tensor(r) for r in results])
            return to_like(out, like)
        finally:
            self._cleanup(handles)


class CoordinateWiseMedian(_FeatureChunkedAggregator):
    """Per-coordinate median over the worker axis; robust to
    floor((n-1)/2) byzantine workers (reference median.py:28). Both the
    direct and pooled paths compute the TRUE median (even n: mean of the
    two middle values)."""

    name = "coordinate-wise-median"
    default_chunk_size = 8192

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.median(X)

    def _chunk_fn(self):
        return SF.median_chunk


class CoordinateWiseTrimmedMean(_FeatureChunkedAggregator):
    """Per-coordinate sort, drop f from each end, mean the middle n-2f
    (Yin et al. 2018; reference trimmed_mean.py:27)."""

    name = "coordinate-wise-trimmed-mean"
    default_chunk_size = 4096

    def __init__(self, f: int, *, chunk_size: int = 4096) -> None:
        super().__init__(chunk_size=chunk_size)
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.trimmed_mean(X, self.f)

    def _chunk_fn(self):
        return SF.trimmed_mean_chunk

    def _chunk_args(self) -> tuple:
        return (self.f,)


class MeanOfMedians(_FeatureChunkedAggregator):
    """MeaMed: per coordinate, mean of the n-f values closest to the median."""

    name = "mean-of-medians"
    default_chunk_size = 8192

    def __init__(self, f: int, *, chunk_size: int = 8192) -> None:
        super().__init__(chunk_size=chunk_size)
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.mean_of_medians(X, self.f)

    def _chunk_fn(self):
        return SF.meamed_chunk

    def _chunk_args(self) -> tuple:
        return (self.f,)
