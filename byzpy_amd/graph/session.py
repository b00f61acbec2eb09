"""ExecutionSession — incremental graph re-execution with result caching.

Reference parity: engine/graph/session.py (ExecutionFuture 27-160; node
result cache + graph pruning of cached deps into GraphInputs 306-370;
all-nodes-as-outputs so intermediates get cached 250-267; execute_async
280-304).
"""
from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional

from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.parallel_scheduler import ParallelScheduler


class ExecutionFuture:
    def __init__(self, task: asyncio.Task) -> None:
        self._task = task

    def done(self) -> bool:
        return self._task.done()

    def cancel(self) -> bool:
        return self._task.cancel()

    def __await__(self):
        return self._task.__await__()

    async def result(self) -> Any:
        return await self._task


class ExecutionSession:
    def __init__(self, graph: ComputationGraph, pool: Any = None) -> None:
        self.graph = graph
        self.pool = pool
        self._cache: Dict[str, Any] = {}

    @property
    def cached_nodes(self) -> List[str]:
        return list(self._cache)

    def invalidate(self, *names: str) -> None:
        if not names:
            self._cache.clear()
            return
        # invalidating a node invalidates its dependents transitively
        dirty = set(names)
        changed = True
        while changed:
            changed = False
            for node in self.graph.nodes.values():
                if node.name in dirty:
                    continue
                for spec in node.inputs.values():
                    if isinstance(spec, str) and spec in dirty:
                        dirty.add(node.name)
                        changed = True
                        break
        for name in dirty:
            self._cache.pop(name, None)

    def _pruned_graph(self) -> ComputationGraph:
        """Rewrite the graph so cached dependencies become GraphInputs."""
        nodes: List[GraphNode] = []
        for name in self.graph.topo_order:
            if name in self._cache:
                continue
            node = self.graph.nodes[name]
            new_inputs: Dict[str, Any] = {}
            for arg, spec in node.inputs.items():
                if isinstance(spec, str) and spec in self._cache:
                    new_inputs[arg] = GraphInput(f"__cached__/{spec}")
                else:
                    new_inputs[arg] = spec
            nodes.append(GraphNode(name=name, op=node.op, inputs=new_inputs))
        # all remaining nodes become outputs so intermediates get cached
        return ComputationGraph(nodes, outputs=[n.name for n in nodes] or None)

    async def execute(self, inputs: Optional[Dict[str, Any]] = None) -> Any:
        inputs = dict(inputs or {})
        pruned = self._pruned_graph()
        if pruned.nodes:
            for name, value in self._cache.items():
                inputs[f"__cached__/{name}"] = value
            scheduler = ParallelScheduler(pruned, pool=self.pool)
            results = await scheduler.run(inputs)
            if len(pruned.outputs) == 1:
                results = {pruned.outputs[0]: results}
            self._cache.update(results)
        if len(self.graph.outputs) == 1:
            return self._cache[self.graph.outputs[0]]
        return {o: self._cache[o] for o in self.graph.outputs}

    def execute_async(self, inputs: Optional[Dict[str, Any]] = None) -> ExecutionFuture:
        task = asyncio.get_running_loop().create_task(self.execute(inputs))
        return ExecutionFuture(task)
