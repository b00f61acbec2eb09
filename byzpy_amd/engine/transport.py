"""Pluggable transport for the prototype runner layer.

Reference parity: engine/transport/{base,local,tcp_simple}.py — a minimal
Transport protocol with an in-process handler-dispatch LocalTransport and
a socket mailbox TcpTransport (length-prefixed pickle frames).
"""
from __future__ import annotations

import asyncio
import pickle
import struct
from typing import Any, Callable, Dict, Optional


class Transport:
    async def start(self) -> None: ...

    async def send(self, target: str, payload: Any) -> None: ...

    def on_message(self, handler: Callable[[Any], None]) -> None: ...

    async def stop(self) -> None: ...


class LocalTransport(Transport):
    _registry: Dict[str, "LocalTransport"] = {}

    def __init__(self, name: str) -> None:
        self.name = name
        self._handler: Optional[Callable[[Any], None]] = None

    async def start(self) -> None:
        LocalTransport._registry[self.name] = self

    def on_message(self, handler: Callable[[Any], None]) -> None:
        self._handler = handler

    async def send(self, target: str, payload: Any) -> None:
        peer = LocalTransport._registry.get(target)
        if peer is None or peer._handler is None:
            raise RuntimeError(f"no local transport {target!r}")
        peer._handler(payload)

    async def stop(self) -> None:
        LocalTransport._registry.pop(self.name, None)


class TcpTransport(Transport):
    """One asyncio TCP mailbox per endpoint; peers are (host, port) pairs."""

    def __init__(self, name: str, host: str = "127.0.0.1", port: int = 0) -> None:
        self.name = name
        self.host, self.port = host, int(port)
        self.peers: Dict[str, tuple] = {}
        self._handler: Optional[Callable[[Any], None]] = None
        self._server: Optional[asyncio.AbstractServer] = None

    def add_peer(self, name: str, host: str, port: int) -> None:
        self.peers[name] = (host, port)

    async def start(self) -> None:
        self._server = await asyncio.start_server(self._serve, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]

    def on_message(self, handler: Callable[[Any], None]) -> None:
        self._handler = handler

    async def _serve(self, reader, writer) -> None:
        try:
            while True:
                try:
                    header = await reader.readexactly(8)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                (length,) = struct.unpack("!Q", header)
                payload = pickle.loads(await reader.readexactly(length))
                if self._handler is not None:
                    self._handler(payload)
        finally:
            writer.close()

    async def send(self, target: str, payload: Any) -> None:
        host, port = self.peers[target]
        _, writer = await asyncio.open_connection(host, port)
        blob = pickle.dumps(payload)
        writer.write(struct.pack("!Q", len(blob)) + blob)
        await writer.drain()
        writer.close()

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
