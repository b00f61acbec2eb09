"""Prototype per-node runner + cluster (legacy surface).

Reference parity: engine/node_runner.py:33-174 (cmd/inbox/result queues,
auto-step timer) and engine/node_cluster.py:16-60. Kept for API
completeness; the production paths are the actor pools + RCCL engines.
"""
from __future__ import annotations

import asyncio
from typing import Any, Callable, Dict, List, Optional

from byzpy_amd.engine.transport import LocalTransport, Transport


class NodeRunner:
    """Drives one node object: periodic step() plus message dispatch."""

    def __init__(
        self,
        name: str,
        step_fn: Callable[[int], Any],
        *,
        transport: Optional[Transport] = None,
        on_msg: Optional[Callable[[Any], None]] = None,
        auto_step_interval: Optional[float] = None,
    ) -> None:
        self.name = name
        self.step_fn = step_fn
        self.transport = transport or LocalTransport(name)
        self.on_msg = on_msg
        self.auto_step_interval = auto_step_interval
        self.steps_done = 0
        self._auto_task: Optional[asyncio.Task] = None
        self.results: List[Any] = []

    async def start(self) -> None:
        if self.on_msg is not None:
            self.transport.on_message(self.on_msg)
        await self.transport.start()
        if self.auto_step_interval is not None:
            self._auto_task = asyncio.get_running_loop().create_task(self._auto_loop())

    async def _auto_loop(self) -> None:
        while True:
            await asyncio.sleep(self.auto_step_interval)
            self.step()

    def step(self) -> Any:
        result = self.step_fn(self.steps_done)
        self.steps_done += 1
        self.results.append(result)
        return result

    async def send(self, target: str, payload: Any) -> None:
        await self.transport.send(target, payload)

    async def stop(self) -> None:
        if self._auto_task is not None:
            self._auto_task.cancel()
            try:
                await self._auto_task
            except asyncio.CancelledError:
                pass
        await self.transport.stop()


class NodeCluster:
    def __init__(self) -> None:
        self.runners: Dict[str, NodeRunner] = {}

    def add(self, runner: NodeRunner) -> None:
        self.runners[runner.name] = runner

    async def start_all(self) -> None:
        for r in self.runners.values():
            await r.start()

    def step_all(self) -> Dict[str, Any]:
        return {name: r.step() for name, r in self.runners.items()}

    async def stop_all(self) -> None:
        for r in self.runners.values():
            await r.stop()
