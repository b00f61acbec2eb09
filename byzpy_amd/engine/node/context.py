"""Node execution contexts — where a DecentralizedNode's body runs.

Reference parity: engine/node/context.py (NodeContext ABC 11-53;
InProcessContext 56-123; ProcessContext with a child event loop and
parent-relayed child->child routing 136-490). The MI355X build keeps
InProcess (pure asyncio) and Process (spawn + queues) contexts for the
orchestration layer; bulk tensor traffic between GPUs never rides these —
it goes over RCCL (byzpy_amd/parallel/).

RemoteContext / mesh-TCP contexts: byzpy_amd/engine/node/remote.py.
"""
from __future__ import annotations

import asyncio
import multiprocessing as mp
import threading
from abc import ABC, abstractmethod
from typing import Any, Callable, Dict, Optional

import cloudpickle


class NodeContext(ABC):
    @abstractmethod
    async def start(self, node: Any) -> None: ...

    @abstractmethod
    async def send_message(self, target_id: str, message: dict) -> None: ...

    @abstractmethod
    async def shutdown(self) -> None: ...


class InProcessContext(NodeContext):
    """All nodes share one event loop; a class-level registry routes
    messages directly into each node's inbox."""

    _registry: Dict[str, "InProcessContext"] = {}

    def __init__(self, node_id: str) -> None:
        self.node_id = node_id
        self.node: Any = None
        self.inbox: asyncio.Queue = asyncio.Queue()

    async def start(self, node: Any) -> None:
        self.node = node
        InProcessContext._registry[self.node_id] = self

    async def send_message(self, target_id: str, message: dict) -> None:
        target = InProcessContext._registry.get(target_id)
        if target is None:
            raise RuntimeError(f"unknown node {target_id!r}")
        target.inbox.put_nowait(message)

    async def receive(self, timeout: float = 0.1) -> Optional[dict]:
        try:
            return await asyncio.wait_for(self.inbox.get(), timeout)
        except asyncio.TimeoutError:
            return None

    async def shutdown(self) -> None:
        InProcessContext._registry.pop(self.node_id, None)


def _process_node_main(node_blob: bytes, node_id: str, cmd_q, out_q) -> None:
    """Child process body: build the node (cloudpickled factory), run its
    event loop, drain the command queue. Outgoing messages are relayed to
    the parent which routes them (reference context.py:319-490)."""
    import asyncio as aio

    factory = cloudpickle.loads(node_blob)

    async def main() -> None:
        node = factory()
        bridge = _SubprocessBridgeContext(node_id, out_q)
        await node.attach_context(bridge)
        loop = aio.get_running_loop()

        def drain_cmd() -> None:
            while True:
                try:
                    cmd = cmd_q.get()
                except (EOFError, OSError):
                    break
                if cmd is None:
                    loop.call_soon_threadsafe(stop_evt.set)
                    break
                kind, payload = cloudpickle.loads(cmd)
                if kind == "message":
                    aio.run_coroutine_threadsafe(
                        node.handle_incoming_message(payload), loop
                    )
                elif kind == "pipeline":
                    name, inputs, req_id = payload

                    async def run(name=name, inputs=inputs, req_id=req_id):
                        try:
                            result = await node.execute_pipeline(name, inputs)
                            out_q.put(cloudpickle.dumps(("result", (req_id, result))))
                        except asyncio.CancelledError:
                            raise  # loop shutdown: never mask cancellation
                        except BaseException as e:  # noqa: BLE001
                            out_q.put(cloudpickle.dumps(("error", (req_id, repr(e)))))

                    aio.run_coroutine_threadsafe(run(), loop)

        stop_evt = aio.Event()
        t = threading.Thread(target=drain_cmd, daemon=True)
        t.start()
        await node.start_local()
        await stop_evt.wait()
        await node.stop_local()

    aio.run(main())


class _SubprocessBridgeContext(NodeContext):
    """Inside the child: outgoing sends go up to the parent for routing."""

    def __init__(self, node_id: str, out_q) -> None:
        self.node_id = node_id
        self.out_q = out_q

    async def start(self, node: Any) -> None:
        pass

    async def send_message(self, target_id: str, message: dict) -> None:
        self.out_q.put(cloudpickle.dumps(("route", (target_id, message))))

    async def shutdown(self) -> None:
        pass


class ProcessContext(NodeContext):
    """Runs the node in a spawned child process. The parent relays
    child->child messages via the cluster's routing callback."""

    def __init__(self, node_id: str, node_factory: Callable[[], Any]) -> None:
        self.node_id = node_id
        self._factory = node_factory
        self._proc: Optional[mp.Process] = None
        self._cmd_q = None
        self._out_q = None
        self._route_cb: Optional[Callable] = None
        self._results: Dict[str, asyncio.Future] = {}
        self._drain_task: Optional[asyncio.Task] = None
        self._req_counter = 0

    def set_route_callback(self, cb: Callable) -> None:
        self._route_cb = cb

    async def start(self, node: Any = None) -> None:
        ctx = mp.get_context("spawn")
        self._cmd_q = ctx.Queue()
        self._out_q = ctx.Queue()
        blob = cloudpickle.dumps(self._factory)
        self._proc = ctx.Process(
            target=_process_node_main,
            args=(blob, self.node_id, self._cmd_q, self._out_q),
            daemon=True,
        )
        self._proc.start()
        self._drain_task = asyncio.get_running_loop().create_task(self._drain())

    async def _drain(self) -> None:
        loop = asyncio.get_running_loop()
        while True:
            blob = await loop.run_in_executor(None, self._out_q.get)
            if blob is None:
                break
            kind, payload = cloudpickle.loads(blob)
            if kind == "route" and self._route_cb is not None:
                target_id, message = payload
                await self._route_cb(target_id, message)
            elif kind in ("result", "error"):
                req_id, value = payload
                fut = self._results.pop(req_id, None)
                if fut is not None and not fut.done():
                    if kind == "result":
                        fut.set_result(value)
                    else:
                        fut.set_exception(RuntimeError(value))

    async def send_message(self, target_id: str, message: dict) -> None:
        # parent -> child delivery (target is THIS context's node)
        self._cmd_q.put(cloudpickle.dumps(("message", message)))

    async def execute_pipeline(self, name: str, inputs: Optional[dict] = None) -> Any:
        self._req_counter += 1
        req_id = f"req-{self._req_counter}"
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._results[req_id] = fut
        self._cmd_q.put(cloudpickle.dumps(("pipeline", (name, inputs or {}, req_id))))
        return await fut

    async def shutdown(self) -> None:
        if self._proc is None:
            return
        try:
            self._cmd_q.put(None)
        except Exception:
            pass
        self._out_q.put(None)
        if self._drain_task is not None:
            await self._drain_task
        self._proc.join(timeout=5)
        if self._proc.is_alive():
            self._proc.terminate()
            self._proc.join(timeout=5)
        self._proc = None
