"""OperatorExecutor / run_operator — the one-liner API.

Reference parity: engine/graph/executor.py (key auto-detection 33-68; lazy
pool creation + graph/scheduler caching 124-258; _InputMappingOperator
195-234; run_operator 266-291).
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Sequence

from byzpy_amd.graph.ops import make_single_operator_graph
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.graph.scheduler import NodeScheduler
from byzpy_amd.ops.base import Operator


def _detect_input_key(op: Operator, inputs: Dict[str, Any]) -> str:
    from byzpy_amd.aggregators.base import Aggregator
    from byzpy_amd.attacks.base import Attack
    from byzpy_amd.pre_aggregators.base import PreAggregator

    if isinstance(op, (Aggregator, PreAggregator)):
        return op.input_key
    if isinstance(op, Attack):
        if len(inputs) == 1:
            return next(iter(inputs))
        raise ValueError(
            "Attack operators need explicit input keys (pass exactly one input "
            "or use a ComputationGraph)"
        )
    return op.input_key


class OperatorExecutor:
    """Runs a single operator, optionally over a worker pool. Use as an
    async context manager, or via the module-level ``run_operator``."""

    def __init__(
        self,
        op: Operator,
        *,
        pool: Optional[ActorPool] = None,
        pool_config: Optional[Sequence[ActorPoolConfig] | ActorPoolConfig] = None,
    ) -> None:
        self.op = op
        self._pool = pool
        self._pool_config = pool_config
        self._owns_pool = pool is None and pool_config is not None
        self._scheduler: Optional[NodeScheduler] = None
        self._graph_key: Optional[str] = None

    async def __aenter__(self) -> "OperatorExecutor":
        await self._ensure_pool()
        return self

    async def __aexit__(self, *exc: Any) -> None:
        await self.aclose()

    async def _ensure_pool(self) -> None:
        if self._pool is None and self._pool_config is not None:
            self._pool = ActorPool(self._pool_config)
        if self._pool is not None:
            await self._pool.start()

    async def aclose(self) -> None:
        if self._owns_pool and self._pool is not None:
            await self._pool.close()
            self._pool = None

    async def run(self, inputs: Dict[str, Any]) -> Any:
        await self._ensure_pool()
        user_key = next(iter(inputs)) if len(inputs) == 1 else None
        op_key = _detect_input_key(self.op, inputs)
        if user_key is not None and user_key != op_key:
            # map the user's key onto the operator's expected key
            inputs = {op_key: inputs[user_key]}
        graph_key = op_key
        if self._scheduler is None or self._graph_key != graph_key:
            graph = make_single_operator_graph(self.op, {op_key: op_key})
            self._scheduler = NodeScheduler(graph, pool=self._pool)
            self._graph_key = graph_key
        return await self._scheduler.run(inputs)


async def run_operator(
    op: Operator,
    inputs: Dict[str, Any],
    *,
    pool: Optional[ActorPool] = None,
    pool_config: Optional[Sequence[ActorPoolConfig] | ActorPoolConfig] = None,
) -> Any:
    """One-shot convenience: build an executor, run, tear down."""
    executor = OperatorExecutor(op, pool=pool, pool_config=pool_config)
    try:
        await executor._ensure_pool()
        return await executor.run(inputs)
    finally:
        await executor.aclose()
