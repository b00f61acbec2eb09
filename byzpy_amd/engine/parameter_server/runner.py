"""Prototype parameter server over the NodeRunner/NodeCluster layer.

Reference parity: engine/parameter_server/runner.py:49-90 (per-node
step/on_msg closures over the prototype runner) and
engine/parameter_server/decentralized.py (DecentralizedParameterServer
back-compat wrapper). Kept for API completeness; production paths are
ParameterServer (actors) and RcclParameterServer (one rank per GPU).
"""
from __future__ import annotations

from typing import Callable, Optional, Sequence

import torch

from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.engine.node_runner import NodeCluster, NodeRunner


class ParameterServerRunner:
    """Each worker is a NodeRunner whose step produces a gradient; the
    server aggregates every round and pushes the update to all workers."""

    def __init__(
        self,
        gradient_fns: Sequence[Callable[[int], torch.Tensor]],
        aggregator: Aggregator,
        *,
        apply_fns: Optional[Sequence[Callable[[torch.Tensor], None]]] = None,
    ) -> None:
        self.aggregator = aggregator
        self.apply_fns = list(apply_fns or [])
        self.cluster = NodeCluster()
        for i, fn in enumerate(gradient_fns):
            self.cluster.add(NodeRunner(f"worker-{i}", step_fn=fn))
        self.rounds_done = 0

    async def start(self) -> None:
        await self.cluster.start_all()

    async def round(self) -> torch.Tensor:
        grads = list(self.cluster.step_all().values())
        update = self.aggregator.aggregate(grads)
        for fn in self.apply_fns:
            fn(update)
        self.rounds_done += 1
        return update

    async def stop(self) -> None:
        await self.cluster.stop_all()


class DecentralizedParameterServer(ParameterServerRunner):
    """Back-compat alias (reference decentralized.py:14-38)."""
