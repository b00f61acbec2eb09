"""Operator base — the unit every aggregator / attack / pre-aggregator is.

Reference parity: engine/graph/operator.py (Operator, OpContext,
MessageTriggerOp; run() dispatch order barriered -> subtasks -> compute;
windowed in-flight subtask scheduling operator.py:96-179; affinity
round-robin 182-196).

MI355X design note: on CUDA(ROCm) tensors an operator's ``compute`` runs as
one-or-few HIP kernel launches, so the subtask fan-out path is a CPU-pool
facility; the windowed scheduler maps to stream depth for stream workers.
"""
from __future__ import annotations

import asyncio
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence

from byzpy_amd.graph.subtask import SubTask


@dataclass
class OpContext:
    """Execution context handed to Operator.run by a scheduler."""

    pool: Any = None
    metadata: Dict[str, Any] = field(default_factory=dict)

    @property
    def pool_size(self) -> int:
        if "pool_size" in self.metadata:
            return int(self.metadata["pool_size"])
        if self.pool is not None:
            return int(getattr(self.pool, "size", 1))
        return 1

    @property
    def worker_affinities(self) -> Optional[List[str]]:
        return self.metadata.get("worker_affinities")


class Operator:
    """Base class for graph operators.

    Dispatch order in :meth:`run` (reference operator.py:55-70):
    barriered subtasks -> windowed subtask fan-out -> plain compute.
    """

    name: str = "operator"
    input_key: str = "value"
    supports_subtasks: bool = False
    supports_barriered_subtasks: bool = False
    #: 0 => auto (pool.size * 8); negative => pool.size * |limit|
    max_subtasks_inflight: int = 0

    # -- public entry ------------------------------------------------------
    async def run(self, ctx: OpContext, **inputs: Any) -> Any:
        if ctx.pool is not None and self.supports_barriered_subtasks:
            return await self.run_barriered_subtasks(ctx, **inputs)
        if ctx.pool is not None and self.supports_subtasks:
            subtasks = list(self.create_subtasks(ctx, **inputs))
            if subtasks:
                self._assign_worker_affinities(ctx, subtasks)
                try:
                    results = await self._run_subtasks_windowed(ctx, subtasks)
                except BaseException:
                    # a failed fan-out never reaches reduce_subtasks (whose
                    # finally releases shared-memory handles): release any
                    # pending op state here so POSIX shm never leaks
                    self._release_pending(ctx)
                    raise
                return self.reduce_subtasks(ctx, results, **inputs)
        return self.compute(ctx, **inputs)

    def _release_pending(self, ctx: OpContext) -> None:
        """Best-effort release of create_subtasks state when the fan-out
        fails before reduce. Operators stash (like, handles) tuples under
        the ``_op_pending`` metadata key (aggregators/_chunking pattern)."""
        pending = ctx.metadata.pop("_op_pending", None)
        if not pending:
            return
        try:
            from byzpy_amd.aggregators.base import cleanup_handles

            handles = pending[-1]
            cleanup_handles(handles or ())
        except Exception:
            pass

    # -- overridables ------------------------------------------------------
    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        raise NotImplementedError

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        return []

    def reduce_subtasks(self, ctx: OpContext, results: List[Any], **inputs: Any) -> Any:
        raise NotImplementedError

    async def run_barriered_subtasks(self, ctx: OpContext, **inputs: Any) -> Any:
        raise NotImplementedError

    # -- machinery ---------------------------------------------------------
    def _inflight_limit(self, ctx: OpContext) -> int:
        pool_size = max(1, ctx.pool_size)
        limit = self.max_subtasks_inflight
        if limit == 0:
            return pool_size * 8
        if limit < 0:
            return pool_size * abs(limit)
        return limit

    def _assign_worker_affinities(self, ctx: OpContext, subtasks: List[SubTask]) -> None:
        """Round-robin affinity hints over the pool's per-worker capabilities."""
        affinities = ctx.worker_affinities
        if not affinities:
            return
        for i, st in enumerate(subtasks):
            if st.affinity is None:
                st.affinity = affinities[i % len(affinities)]

    async def _run_subtasks_windowed(
        self, ctx: OpContext, subtasks: List[SubTask]
    ) -> List[Any]:
        """Sliding-window scheduler: keep <= limit subtasks in flight; results
        are returned in submission order. An optional shared semaphore in
        metadata["subtask_semaphore"] provides cross-operator back-pressure
        (reference parallel_scheduler.py:80-86)."""
        limit = max(1, self._inflight_limit(ctx))
        sem: Optional[asyncio.Semaphore] = ctx.metadata.get("subtask_semaphore")
        results: List[Any] = [None] * len(subtasks)
        window = asyncio.Semaphore(limit)

        async def _one(idx: int, st: SubTask) -> None:
            async with window:
                if sem is not None:
                    async with sem:
                        results[idx] = await ctx.pool.run_subtask(st)
                else:
                    results[idx] = await ctx.pool.run_subtask(st)

        await asyncio.gather(*(_one(i, st) for i, st in enumerate(subtasks)))
        return results

    async def _run_subtasks(self, ctx: OpContext, subtasks: List[SubTask]) -> List[Any]:
        """Barrier helper for iterative operators: run a batch to completion."""
        self._assign_worker_affinities(ctx, list(subtasks))
        return await self._run_subtasks_windowed(ctx, list(subtasks))


class MessageTriggerOp(Operator):
    """Blocks a graph node until a message arrives (reference operator.py:199-217)."""

    name = "message-trigger"

    def __init__(self, message_type: str, *, timeout: Optional[float] = None) -> None:
        self.message_type = message_type
        self.timeout = timeout

    async def run(self, ctx: OpContext, **inputs: Any) -> Any:
        scheduler = ctx.metadata.get("scheduler")
        if scheduler is None or not hasattr(scheduler, "wait_for_message"):
            raise RuntimeError("MessageTriggerOp requires a message-aware scheduler")
        return await scheduler.wait_for_message(self.message_type, timeout=self.timeout)
