"""Remote TCP actor backend + server (control plane).

Reference parity: engine/actor/backends/remote.py (RemoteActorBackend
19-253; RemoteActorServer 302-428). MI355X note: this is the CONTROL
plane for multi-node bootstrap; bulk device data moves over RCCL
(SURVEY.md C7), so payloads here are pickled host objects.
"""
from __future__ import annotations

import asyncio
import itertools
import uuid
from typing import Any, Dict, Optional, Tuple

import cloudpickle

from byzpy_amd.actor._wire import recv_obj, send_obj
from byzpy_amd.actor.channels import Endpoint

_ids = itertools.count()


class RemoteActorBackend:
    """Client of a RemoteActorServer; one persistent connection per actor."""

    scheme = "tcp"

    def __init__(self, host: str, port: int) -> None:
        self.host, self.port = host, int(port)
        self.actor_id = f"tcp-{next(_ids)}-{uuid.uuid4().hex[:8]}"
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._lock = asyncio.Lock()

    async def start(self) -> None:
        self._reader, self._writer = await asyncio.open_connection(self.host, self.port)
        await self._request(("hello", self.actor_id))

    async def _request(self, msg: Any) -> Any:
        async with self._lock:
            await send_obj(self._writer, msg)
            status, payload = await recv_obj(self._reader)
        if status != "ok":
            raise RuntimeError(f"remote actor error: {payload}")
        return payload

    async def construct(self, factory: Any, /, *args: Any, **kwargs: Any) -> None:
        blob = cloudpickle.dumps((factory, args, kwargs))
        await self._request(("construct", self.actor_id, blob))

    async def call(self, method: str, /, *args: Any, **kwargs: Any) -> Any:
        return await self._request(("call", self.actor_id, method, args, kwargs))

    async def close(self) -> None:
        if self._writer is None:
            return
        try:
            await self._request(("close", self.actor_id))
        except Exception:
            pass
        self._writer.close()
        self._writer = None
        self._reader = None

    def get_endpoint(self) -> Endpoint:
        return Endpoint(
            scheme="tcp", address=f"{self.host}:{self.port}", actor_id=self.actor_id
        )

    async def chan_open(self, name: str) -> None:
        await self._request(("chan_open", self.actor_id, name))

    async def chan_put(self, endpoint: Endpoint, name: str, payload: Any) -> None:
        from byzpy_amd.actor.router import channel_router

        target = channel_router.lookup(endpoint.scheme, endpoint.actor_id)
        if target is not None and endpoint.scheme != "tcp":
            await target._deliver(name, payload)
            return
        if endpoint.scheme == "tcp" and endpoint.address == f"{self.host}:{self.port}":
            # same-server fast path (reference remote.py:149-166)
            await self._request(("chan_deliver", endpoint.actor_id, name, payload))
            return
        from byzpy_amd.actor.transports import tcp

        await tcp.chan_put(endpoint, name, payload)

    async def chan_get(self, name: str) -> Any:
        return await self._request(("chan_get", self.actor_id, name))


class RemoteActorServer:
    """Hosts actors for TCP clients: construct / call / channel ops."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0) -> None:
        self.host, self.port = host, int(port)
        self._server: Optional[asyncio.AbstractServer] = None
        self._actors: Dict[str, Any] = {}
        self._mailboxes: Dict[Tuple[str, str], asyncio.Queue] = {}

    async def start(self) -> None:
        self._server = await asyncio.start_server(self._serve, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    async def _serve(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    msg = await recv_obj(reader)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                try:
                    result = await self._handle(msg)
                    await send_obj(writer, ("ok", result))
                except asyncio.CancelledError:
                    raise  # server shutdown: never mask cancellation
                except BaseException as e:  # noqa: BLE001
                    await send_obj(writer, ("err", repr(e)))
        finally:
            writer.close()

    async def _handle(self, msg: Any) -> Any:
        op = msg[0]
        if op == "hello":
            return None
        if op == "construct":
            _, actor_id, blob = msg
            factory, args, kwargs = cloudpickle.loads(blob)
            self._actors[actor_id] = factory(*args, **kwargs)
            return None
        if op == "call":
            _, actor_id, method, args, kwargs = msg
            fn = getattr(self._actors[actor_id], method)
            result = fn(*args, **kwargs)
            if asyncio.iscoroutine(result):
                result = await result
            return result
        if op == "chan_open":
            _, actor_id, name = msg
            self._mailboxes.setdefault((actor_id, name), asyncio.Queue())
            return None
        if op == "chan_deliver":
            _, actor_id, name, payload = msg
            self._mailboxes.setdefault((actor_id, name), asyncio.Queue()).put_nowait(payload)
            return None
        if op == "chan_get":
            _, actor_id, name = msg
            q = self._mailboxes.setdefault((actor_id, name), asyncio.Queue())
            return await q.get()
        if op == "close":
            _, actor_id = msg
            self._actors.pop(actor_id, None)
            return None
        raise ValueError(f"unknown op {op!r}")
