"""NodeScheduler — sequential topo-order execution; MessageAwareNodeScheduler
adds message-triggered inputs.

Reference parity: engine/graph/scheduler.py (NodeScheduler.run 51-68;
MessageAwareNodeScheduler message cache + futures 124-269).
"""
from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional

from byzpy_amd.graph.graph import ComputationGraph, GraphInput, MessageSource
from byzpy_amd.ops.base import OpContext


class NodeScheduler:
    def __init__(self, graph: ComputationGraph, pool: Any = None, metadata: Optional[dict] = None) -> None:
        self.graph = graph
        self.pool = pool
        self.metadata = dict(metadata or {})

    def _base_metadata(self) -> Dict[str, Any]:
        md = dict(self.metadata)
        if self.pool is not None:
            md.setdefault("pool_size", getattr(self.pool, "size", 1))
            affinities = getattr(self.pool, "worker_affinities", None)
            if affinities:
                md.setdefault("worker_affinities", affinities)
        md.setdefault("scheduler", self)
        return md

    async def _resolve_input(self, spec: Any, results: Dict[str, Any], inputs: Dict[str, Any]) -> Any:
        if isinstance(spec, str):
            return results[spec]
        if isinstance(spec, GraphInput):
            if spec.name not in inputs:
                raise KeyError(f"missing graph input {spec.name!r}")
            return inputs[spec.name]
        if isinstance(spec, MessageSource):
            msg = await self.wait_for_message(spec.message_type, timeout=spec.timeout)
            return msg.get(spec.field) if spec.field else msg
        raise TypeError(f"bad input spec {spec!r}")

    async def run(self, inputs: Optional[Dict[str, Any]] = None) -> Any:
        inputs = dict(inputs or {})
        required = self.graph.required_inputs()
        missing = [r for r in required if r not in inputs]
        if missing:
            raise KeyError(f"missing graph inputs: {missing}")
        results: Dict[str, Any] = {}
        md = self._base_metadata()
        for name in self.graph.topo_order:
            node = self.graph.nodes[name]
            kwargs = {
                arg: await self._resolve_input(spec, results, inputs)
                for arg, spec in node.inputs.items()
            }
            ctx = OpContext(pool=self.pool, metadata=dict(md))
            results[name] = await node.op.run(ctx, **kwargs)
        if len(self.graph.outputs) == 1:
            return results[self.graph.outputs[0]]
        return {o: results[o] for o in self.graph.outputs}

    # message API (overridden by MessageAwareNodeScheduler)
    async def wait_for_message(self, message_type: str, timeout: Optional[float] = None) -> Any:
        raise RuntimeError("this scheduler is not message-aware")


class MessageAwareNodeScheduler(NodeScheduler):
    """Adds deliver_message/wait_for_message with a cache so messages that
    arrive before the consumer are not lost.

    The per-type cache is a BOUNDED FIFO (``cache_limit``, default 1024):
    every incoming node message is mirrored here whether or not any graph
    ever consumes that type, so an unbounded list is a leak — in the P2P
    gossip engine it retained one broadcast payload per sender per round
    forever (caught by the device long-run soak). When the cap is hit the
    OLDEST unconsumed message is dropped, which matches the consumer's
    pop-oldest order: a reader that far behind has already lost ordering
    guarantees."""

    def __init__(self, graph: ComputationGraph, pool: Any = None, metadata: Optional[dict] = None,
                 cache_limit: int = 1024) -> None:
        super().__init__(graph, pool, metadata)
        self._cache: Dict[str, List[Any]] = {}
        self._waiters: Dict[str, List[asyncio.Future]] = {}
        self.cache_limit = int(cache_limit)

    def deliver_message(self, message_type: str, payload: Any) -> None:
        # skip and prune futures that are already done (e.g. a wait_for
        # timed out): deliver to the first still-live waiter
        waiters = self._waiters.get(message_type)
        while waiters:
            fut = waiters.pop(0)
            if not fut.done():
                fut.set_result(payload)
                return
        cache = self._cache.setdefault(message_type, [])
        cache.append(payload)
        if len(cache) > self.cache_limit:
            del cache[0]

    async def wait_for_message(self, message_type: str, timeout: Optional[float] = None) -> Any:
        cached = self._cache.get(message_type)
        if cached:
            return cached.pop(0)
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._waiters.setdefault(message_type, []).append(fut)
        if timeout is not None:
            return await asyncio.wait_for(fut, timeout)
        return await fut

    def set_graph(self, graph: ComputationGraph) -> None:
        """Swap the graph (DecentralizedNode.execute_pipeline semantics)."""
        self.graph = graph
