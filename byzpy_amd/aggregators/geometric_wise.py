"""Geometry-based robust aggregators (Krum family, GeoMedian, MDA, MoNNA,
SMEA).

Reference parity: aggregators/geometric_wise/*.py. GPU path: the pairwise
Gram runs on the MFMA split-K kernel (K4); selection/scoring on small
(n, n) tensors; Weiszfeld as fused per-iteration kernels (K6).
"""
from __future__ import annotations

import itertools
from typing import Any, List, Sequence

import torch

from byzpy_amd.aggregators import _subtask_fns as SF
from byzpy_amd.aggregators._chunking import chunk_ranges
from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops import functional as F
from byzpy_amd.ops.base import OpContext
from byzpy_amd.storage.shared_store import register_tensor, write_handle
from byzpy_amd.utils.flatten import stack_gradients, to_like


class MultiKrum(Aggregator):
    """Pairwise sq-dists -> score_i = sum of n-f-1 nearest -> mean of the q
    best (Blanchard et al. 2017). Row-chunked Gram on CPU pools."""

    name = "multi-krum"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, q: int, *, chunk_size: int = 32) -> None:
        if f < 0 or q < 1:
            raise ValueError("need f >= 0 and q >= 1")
        self.f, self.q = int(f), int(q)
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.multi_krum(X, self.f, self.q)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, gradients)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (X, like, handles)
        n = X.shape[0]
        chunk = max(1, min(self.chunk_size, n))
        return [
            SubTask(fn=SF.gram_row_chunk, args=(ref, lo, hi), name=f"gram[{lo}:{hi}]")
            for lo, hi in chunk_ranges(n, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            G = torch.cat(results, dim=0)
            norms = torch.diagonal(G)
            D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
            n = X.shape[0]
            D2 = D2 + torch.diag(torch.full((n,), float("inf")))
            k = n - self.f - 1
            scores = torch.topk(D2, k=k, dim=1, largest=False).values.sum(dim=1)
            winners = torch.topk(scores, k=self.q, largest=False).indices
            out = X.float()[winners].mean(dim=0).to(X.dtype)
            return to_like(out, like)
        finally:
            self._cleanup(handles)


class Krum(MultiKrum):
    """MultiKrum(q=1) returning the single winner itself."""

    name = "krum"

    def __init__(self, f: int, *, chunk_size: int = 32) -> None:
        super().__init__(f, 1, chunk_size=chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.krum(X, self.f)

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            G = torch.cat(results, dim=0)
            norms = torch.diagonal(G)
            D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
            n = X.shape[0]
            D2 = D2 + torch.diag(torch.full((n,), float("inf")))
            k = n - self.f - 1
            scores = torch.topk(D2, k=k, dim=1, largest=False).values.sum(dim=1)
            return to_like(X[int(torch.argmin(scores))].clone(), like)
        finally:
            self._cleanup(handles)


class GeometricMedian(Aggregator):
    """Weiszfeld fixed point; barriered per-iteration fan-out on CPU pools
    (reference geometric_median.py:106-158), fused kernels on GPU."""

    name = "geometric-median"
    supports_barriered_subtasks = True
    max_subtasks_inflight = 0

    def __init__(
        self,
        *,
        tol: float = 1e-6,
        max_iter: int = 256,
        eps: float = 1e-12,
        init: str = "median",
        chunk_size: int = 32,
        fixed_iters: "int | None" = None,
    ) -> None:
        if tol <= 0 or max_iter <= 0 or eps <= 0:
            raise ValueError("tol, max_iter, eps must be > 0")
        if init not in {"median", "mean"}:
            raise ValueError("init must be 'median' or 'mean'")
        if fixed_iters is not None and fixed_iters <= 0:
            raise ValueError("fixed_iters must be > 0")
        self.tol, self.max_iter, self.eps = float(tol), int(max_iter), float(eps)
        self.init = init
        self.chunk_size = int(chunk_size)
        # fixed_iters: exactly that many Weiszfeld steps, NO convergence
        # polls — fully async on GPU (per-node streams overlap; hipGraph
        # capture-safe). The poll-free path is what unblocks config-4
        # gossip (round-1 weakness 5).
        self.fixed_iters = None if fixed_iters is None else int(fixed_iters)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.geometric_median(
            X, tol=self.tol, max_iter=self.max_iter, eps=self.eps,
            init=self.init, fixed_iters=self.fixed_iters
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
            iters = self.fixed_iters if self.fixed_iters is not None else self.max_iter
            z = (F.median(X) if self.init == "median" else X.float().mean(dim=0)).float()
            if use_shm:
                # ONE segment, rewritten per iteration (reference
                # geometric_median.py:126 _write_handle semantics)
                center_handle = register_tensor(z)
            for _ in range(iters):
                if use_shm:
                    write_handle(center_handle, z)
                    center_ref = center_handle
                else:
                    center_ref = z
                tasks = [
                    SubTask(
                        fn=SF.weiszfeld_chunk,
                        args=(ref, lo, hi, center_ref, self.eps),
                    )
                    for lo, hi in chunk_ranges(n, chunk)
                ]
                partials = await self._run_subtasks(ctx, tasks)
                num = sum(p[0] for p in partials)
                den = sum(p[1] for p in partials)
                z_new = num / den
                shift = float((z_new - z).norm())
                z = z_new
                if self.fixed_iters is None and shift <= self.tol:
                    break
            return to_like(z.to(X.dtype), like)
        finally:
            if center_handle is not None:
                self._cleanup([center_handle])
            self._cleanup(handles)


class MinimumDiameterAveraging(Aggregator):
    """Exact min-diameter (n-f)-subset mean; combo-batched subtasks on CPU
    pools, host branch-and-bound over the device-computed D2 on GPU."""

    name = "minimum-diameter-averaging"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, *, chunk_size: int = 256) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.minimum_diameter_averaging(X, self.f)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        X, like = stack_gradients(gradients)
        if X.is_cuda:
            return []
        n = X.shape[0]
        m = n - self.f
        D2 = F.pairwise_sq_dists(X)
        use_shm = bool(getattr(ctx.pool, "prefers_shared_memory", False))
        handles = []
        if use_shm:
            ref = register_tensor(D2)
            handles.append(ref)
        else:
            ref = D2
        ctx.metadata["_op_pending"] = (X, like, handles)
        combos = list(itertools.combinations(range(n), m))
        batch = max(1, self.chunk_size)
        return [
            SubTask(fn=SF.mda_combo_chunk, args=(ref, combos[i : i + batch], m))
            for i in range(0, len(combos), batch)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            # min diameter, then lexicographic subset (canonical tie-break)
            best_diam, best = min(results, key=lambda r: (r[0], r[1]))
            idx = torch.tensor(best, dtype=torch.long)
            out = X.float()[idx].mean(dim=0).to(X.dtype)
            return to_like(out, like)
        finally:
            self._cleanup(handles)


class MoNNA(Aggregator):
    """Mean of the n-f nearest neighbours of a trusted reference row."""

    name = "monna"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, *, reference_index: int = 0, chunk_size: int = 32) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.reference_index = int(reference_index)
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.monna(X, self.f, self.reference_index)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        ref, X, like, handles = self._matrix_ref(ctx, gradients)
        if X.is_cuda:
            return []
        ctx.metadata["_op_pending"] = (X, like, handles)
        n = X.shape[0]
        chunk = max(1, min(self.chunk_size, n))
        return [
            SubTask(
                fn=SF.ref_dist_chunk, args=(ref, lo, hi, self.reference_index)
            )
            for lo, hi in chunk_ranges(n, chunk)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            d2 = torch.cat(results)
            d2[self.reference_index] = -1.0  # reference-first tiebreak
            k = X.shape[0] - self.f
            idx = torch.argsort(d2, stable=True)[:k]
            out = X.float()[idx].mean(dim=0).to(X.dtype)
            return to_like(out, like)
        finally:
            self._cleanup(handles)


class SMEA(Aggregator):
    """Smallest max-eigenvalue (n-f)-subset mean; Gram once, combo-batched
    eigensolves."""

    name = "smea"
    supports_subtasks = True
    max_subtasks_inflight = 0

    def __init__(self, f: int, *, chunk_size: int = 256) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.chunk_size = int(chunk_size)

    def _aggregate(self, X: torch.Tensor) -> torch.Tensor:
        return D.smea(X, self.f)

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        gradients = inputs[self.input_key]
        X, like = stack_gradients(gradients)
        if X.is_cuda:
            return []
        n = X.shape[0]
        m = n - self.f
        Xf = X.float()
        G = Xf @ Xf.T
        use_shm = bool(getattr(ctx.pool, "prefers_shared_memory", False))
        handles = []
        if use_shm:
            ref = register_tensor(G)
            handles.append(ref)
        else:
            ref = G
        ctx.metadata["_op_pending"] = (X, like, handles)
        combos = list(itertools.combinations(range(n), m))
        batch = max(1, self.chunk_size)
        return [
            SubTask(fn=SF.smea_combo_chunk, args=(ref, combos[i : i + batch], m))
            for i in range(0, len(combos), batch)
        ]

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        X, like, handles = ctx.metadata.pop("_op_pending")
        try:
            best_ev, best = min(results, key=lambda r: r[0])
            idx = torch.tensor(best, dtype=torch.long)
            out = X.float()[idx].mean(dim=0).to(X.dtype)
            return to_like(out, like)
        finally:
            self._cleanup(handles)
