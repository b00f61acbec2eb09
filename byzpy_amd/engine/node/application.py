"""NodeApplication — named computation-graph pipelines over a worker pool.

Reference parity: engine/node/application.py (NodePipeline; register/run
with merged metadata 91-112; sync wrapper refusing inside a running loop
114-138; Honest/Byzantine reserved pipeline names 144-261).
"""
from __future__ import annotations

import asyncio
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from byzpy_amd.graph.graph import ComputationGraph
from byzpy_amd.graph.pool import ActorPool
from byzpy_amd.graph.scheduler import NodeScheduler


@dataclass
class NodePipeline:
    graph: ComputationGraph
    metadata: Dict[str, Any] = field(default_factory=dict)


class NodeApplication:
    reserved_pipelines: tuple = ()

    def __init__(self, pool: Optional[ActorPool] = None) -> None:
        self.pool = pool
        self._pipelines: Dict[str, NodePipeline] = {}

    def register_pipeline(
        self,
        name: str,
        graph: ComputationGraph,
        metadata: Optional[dict] = None,
        *,
        _internal: bool = False,
    ) -> None:
        """Register a named pipeline. Names in ``reserved_pipelines`` are
        wired by the node itself (reference application.py:144-261) and
        cannot be overridden by application code."""
        if not _internal and name in self.reserved_pipelines:
            raise ValueError(
                f"pipeline name {name!r} is reserved by {type(self).__name__}"
            )
        self._pipelines[name] = NodePipeline(graph, dict(metadata or {}))

    def has_pipeline(self, name: str) -> bool:
        return name in self._pipelines

    @property
    def pipeline_names(self):
        return list(self._pipelines)

    async def start(self) -> None:
        if self.pool is not None:
            await self.pool.start()

    async def close(self) -> None:
        if self.pool is not None:
            await self.pool.close()

    async def run_pipeline(
        self,
        name: str,
        inputs: Optional[Dict[str, Any]] = None,
        metadata: Optional[dict] = None,
    ) -> Any:
        pipe = self._pipelines.get(name)
        if pipe is None:
            raise KeyError(f"unknown pipeline {name!r}")
        md = dict(pipe.metadata)
        md.update(metadata or {})
        scheduler = NodeScheduler(pipe.graph, pool=self.pool, metadata=md)
        return await scheduler.run(inputs or {})

    def run_pipeline_sync(
        self,
        name: str,
        inputs: Optional[Dict[str, Any]] = None,
        metadata: Optional[dict] = None,
    ) -> Any:
        try:
            asyncio.get_running_loop()
        except RuntimeError:
            return asyncio.run(self.run_pipeline(name, inputs, metadata))
        raise RuntimeError(
            "run_pipeline_sync cannot be called from inside a running event "
            "loop; await run_pipeline instead"
        )


class HonestNodeApplication(NodeApplication):
    reserved_pipelines = ("aggregate", "honest_gradient")


class ByzantineNodeApplication(NodeApplication):
    reserved_pipelines = ("attack",)
