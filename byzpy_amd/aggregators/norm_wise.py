"""Norm-based robust aggregators (CenteredClipping, CGE, CAF).

Reference parity: aggregators/norm_wise/*.py. GPU path: fused per-iteration
clip kernels (K7), row-norm reduction + gather-mean (K8), CAF via rocBLAS
matvecs (library GEMV is the sanctioned path for plain shapes).
"""
from __future__ import annotations

from typing import Any, List, Sequence

import torch

from byzpy_amd.aggregators import _subtask_fns as SF
from byzpy_amd.aggregators._chunking import chunk_ranges, select_adaptive_chunk_size
from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops import functional as F
from byzpy_amd.ops.base import OpContext
from byzpy_amd.storage.shared_store import register_tensor, write_handle
from byzpy_amd.utils.flatten import to_like


class CenteredClipping(Aggregator):
    """v <- v + (1/n) sum clip(x_i - v, c_tau), M iterations (Karimireddy
    et al. 2021); barriered row-chunk fan-out per iteration on CPU pools."""

    name = "centered-clipping"
    supports_barriered_subtasks = True
    max_subtasks_inflight = 0

    def __init__(
        self,
        *,
        c_tau: float,
        M: int = 10,
        eps: float = 1e-12,
        init: str = "mean",
        chunk_size: int = 32,
    ) -> None:
        if c_tau < 0:
            raise ValueError("c_tau must be >= 0")
        if M <= 0 or eps <= 0:
            raise ValueError("M and eps must be > 0")
        if init not in {"mean", "median", "zero"}:
            raise ValueError("init must be one of {'mean','median','zero'}")
        self.c_tau, self.M, self.eps = float(c_tau), int(M), float(eps)
        self.init = init
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.centered_clipping(
            X, c_tau=self.c_tau, M=self.M, eps=self.eps, init=self.init
        )

    async def run_barriered_subtasks(self, ctx: OpContext, **inputs: Any) -> Any:
        gradients = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, gradients)
        if X.is_cuda:
            self._cleanup(handles)
            return to_like(self._aggregate(X), like)
        use_shm = handles != []
        center_handle = None
        try:
            n = X.shape[0]
            chunk = max(1, min(self.chunk_size, n))
            if self.init == "mean":
                v = X.float().mean(dim=0)
            elif self.init == "median":
                v = F.median(X).float()
            else:
                v = torch.zeros(X.shape[1], dtype=torch.float32)
            if use_shm:
                center_handle = register_tensor(v)
            for _ in range(self.M):
                if use_shm:
                    write_handle(center_handle, v)
                    center_ref = center_handle
                else:
                    center_ref = v
                tasks = [
                    SubTask(
                        fn=SF.cc_chunk,
                        args=(ref, lo, hi, center_ref, self.c_tau, self.eps),
                    )
                    for lo, hi in chunk_ranges(n, chunk)
                ]
                partials = await self._run_subtasks(ctx, tasks)
                v = v + sum(partials) / n
            return to_like(v.to(X.dtype), like)
        finally:
            if center_handle is not None:
                self._cleanup([center_handle])
            self._cleanup(handles)


class ComparativeGradientElimination(Aggregator):
    """CGE: mean of the n-f smallest-L2-norm rows; feature-chunked partial
    norms on CPU pools."""

    name = "comparative-gradient-elimination"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, *, chunk_size: int = 8192) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.cge(X, self.f)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, gradients)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (X, like, handles)
        d = X.shape[1]
        chunk = select_adaptive_chunk_size(d, ctx.pool_size, self.chunk_size)
        return [
            SubTask(fn=SF.row_sqnorm_chunk, args=(ref, lo, hi))
            for lo, hi in chunk_ranges(d, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            norms = sum(results)
            k = X.shape[0] - self.f
            idx = torch.argsort(norms, stable=True)[:k]
            out = X.float()[idx].mean(dim=0).to(X.dtype)
            return to_like(out, like)
        finally:
            self._cleanup(handles)


class CAF(Aggregator):
    """Covariance-agnostic filter (power-iteration downweighting)."""

    name = "caf"
    supports_subtasks = False

    def __init__(self, f: int, *, chunk_size: int = 256, power_iters: int = 3) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.chunk_size = int(chunk_size)
        self.power_iters = int(power_iters)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.caf(X, self.f, power_iters=self.power_iters)
