"""DecentralizedNode — message-driven node with swappable pipelines.

Reference parity: engine/node/decentralized.py (message loop feeding
scheduler.deliver_message + registered handlers 95-123; send/broadcast/
multicast with topology enforcement 125-175; autonomous background tasks
223-253; per-call graph swap 185-208).
"""
from __future__ import annotations

import asyncio
from typing import Any, Awaitable, Callable, Dict, List, Optional

from byzpy_amd.engine.node.application import NodeApplication
from byzpy_amd.engine.node.context import InProcessContext, NodeContext
from byzpy_amd.engine.node.router import MessageRouter
from byzpy_amd.graph.graph import ComputationGraph
from byzpy_amd.graph.scheduler import MessageAwareNodeScheduler

Handler = Callable[[dict], Awaitable[None]]


class DecentralizedNode:
    def __init__(
        self,
        node_id: str,
        *,
        context: Optional[NodeContext] = None,
        topology: Any = None,
        application: Optional[NodeApplication] = None,
    ) -> None:
        self.node_id = node_id
        self.context = context or InProcessContext(node_id)
        self.app = application or NodeApplication()
        self.router = MessageRouter(node_id, self.context, topology)
        self.scheduler = MessageAwareNodeScheduler(ComputationGraph([]))
        self._handlers: Dict[str, List[Handler]] = {}
        self._loop_task: Optional[asyncio.Task] = None
        self._autonomous: List[asyncio.Task] = []
        self._running = False

    # -- lifecycle ---------------------------------------------------------
    async def attach_context(self, context: NodeContext) -> None:
        self.context = context
        self.router.context = context

    async def start(self) -> None:
        await self.context.start(self)
        await self.start_local()

    async def start_local(self) -> None:
        self._running = True
        if isinstance(self.context, InProcessContext):
            self._loop_task = asyncio.get_running_loop().create_task(
                self._message_processing_loop()
            )

    async def stop_local(self) -> None:
        self._running = False
        for t in self._autonomous:
            t.cancel()
        if self._loop_task is not None:
            self._loop_task.cancel()
            try:
                await self._loop_task
            except asyncio.CancelledError:
                pass
            self._loop_task = None

    async def stop(self) -> None:
        await self.stop_local()
        await self.context.shutdown()

    # -- messaging ---------------------------------------------------------
    async def _message_processing_loop(self) -> None:
        while self._running:
            msg = await self.context.receive(timeout=0.1)
            if msg is not None:
                await self.handle_incoming_message(msg)

    async def handle_incoming_message(self, message: dict) -> None:
        mtype = message.get("type", "")
        self.scheduler.deliver_message(mtype, message)
        for handler in self._handlers.get(mtype, []):
            result = handler(message)
            if asyncio.iscoroutine(result):
                await result

    def register_handler(self, message_type: str, handler: Handler) -> None:
        self._handlers.setdefault(message_type, []).append(handler)

    async def send_message(self, target_id: str, message_type: str, payload: dict) -> None:
        await self.router.route_direct(
            target_id, {"type": message_type, "sender": self.node_id, **payload}
        )

    async def broadcast_message(self, message_type: str, payload: dict) -> int:
        return await self.router.route_broadcast(
            {"type": message_type, "sender": self.node_id, **payload}
        )

    async def multicast_message(
        self, target_ids: List[str], message_type: str, payload: dict
    ) -> None:
        await self.router.route_multicast(
            target_ids, {"type": message_type, "sender": self.node_id, **payload}
        )

    # -- pipelines ---------------------------------------------------------
    def register_pipeline(self, name: str, graph: ComputationGraph, metadata=None) -> None:
        self.app.register_pipeline(name, graph, metadata)

    async def execute_pipeline(self, name: str, inputs: Optional[dict] = None) -> Any:
        pipe = self.app._pipelines[name]
        self.scheduler.set_graph(pipe.graph)
        self.scheduler.pool = self.app.pool
        md = dict(pipe.metadata)
        self.scheduler.metadata = md
        return await self.scheduler.run(inputs or {})

    # -- autonomous behaviors ----------------------------------------------
    def start_autonomous_task(
        self, fn: Callable[["DecentralizedNode"], Awaitable[None]], interval: float
    ) -> None:
        async def loop() -> None:
            while self._running:
                await fn(self)
                await asyncio.sleep(interval)

        self._autonomous.append(asyncio.get_running_loop().create_task(loop()))
