"""ParameterServer — robust-aggregation training round over node actors.

Reference parity: engine/parameter_server/ps.py:18-158 — honest gradients
stream in as they complete (asyncio.as_completed), Byzantine gradients are
conditioned on them, optional pre-aggregation, aggregation via a
NodeScheduler (when pooled) or the aggregator directly, then the update
fans out via apply_server_gradient.

MI355X note: this is the actor-orchestration engine (nodes may be
thread / process / stream actors on one box). The one-process-per-GPU
RCCL engine for 8x MI355X is RcclParameterServer (rccl.py).
"""
from __future__ import annotations

import asyncio
from typing import Any, List, Optional, Sequence

from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.pool import ActorPool
from byzpy_amd.graph.scheduler import NodeScheduler
from byzpy_amd.pre_aggregators.base import PreAggregator


class ParameterServer:
    def __init__(
        self,
        honest_nodes: Sequence[Any],
        byzantine_nodes: Sequence[Any],
        aggregator: Aggregator,
        *,
        pre_aggregator: Optional[PreAggregator] = None,
        pool: Optional[ActorPool] = None,
    ) -> None:
        self.honest_nodes = list(honest_nodes)
        self.byzantine_nodes = list(byzantine_nodes)
        self.aggregator = aggregator
        self.pre_aggregator = pre_aggregator
        self.pool = pool
        self._scheduler: Optional[NodeScheduler] = None
        if pool is not None:
            graph = ComputationGraph(
                [GraphNode("aggregate", aggregator, {"gradients": GraphInput("gradients")})]
            )
            self._scheduler = NodeScheduler(graph, pool=pool)

    async def _stream_honest(self) -> List[Any]:
        tasks = [
            asyncio.ensure_future(h.honest_gradient_for_next_batch())
            for h in self.honest_nodes
        ]
        grads: List[Any] = []
        for fut in asyncio.as_completed(tasks):
            grads.append(await fut)
        return grads

    async def _stream_byzantine(self, honest_grads: List[Any]) -> List[Any]:
        tasks = [
            asyncio.ensure_future(
                b.byzantine_gradient_for_next_batch(honest_grads=honest_grads)
            )
            for b in self.byzantine_nodes
        ]
        return list(await asyncio.gather(*tasks)) if tasks else []

    async def round(self) -> Any:
        honest = await self._stream_honest()
        byz = await self._stream_byzantine(honest)
        gradients = honest + byz
        if self.pre_aggregator is not None:
            gradients = self.pre_aggregator.pre_aggregate(gradients)
        if self._scheduler is not None:
            update = await self._scheduler.run({"gradients": gradients})
        else:
            update = self.aggregator.aggregate(gradients)
        await asyncio.gather(
            *(n.apply_server_gradient(update) for n in self.honest_nodes + self.byzantine_nodes)
        )
        return update
