"""Remote node fabric: server-hosted and serverless-mesh TCP contexts.

Reference parity: engine/node/remote_server.py (RemoteNodeServer hosting
DecentralizedNodes, routing client<->local and client<->client 125-224),
remote_client.py (RemoteNodeClient with background receive loop), and
context.py's RemoteContext (565-705) / MeshRemoteContext (708-1055 — every
node runs its own asyncio TCP server, dials peers, reconnect monitor).

MI355X note: this is the CONTROL plane for multi-node topologies; bulk
gradient traffic between GPUs rides RCCL. Frames are
4-byte-length cloudpickle.
"""
from __future__ import annotations

import asyncio
import struct
from typing import Any, Dict, Optional

import cloudpickle

from byzpy_amd.engine.node.context import NodeContext


async def _send_frame(writer: asyncio.StreamWriter, obj: Any) -> None:
    blob = cloudpickle.dumps(obj)
    writer.write(struct.pack("!I", len(blob)) + blob)
    await writer.drain()


async def _recv_frame(reader: asyncio.StreamReader) -> Any:
    header = await reader.readexactly(4)
    (length,) = struct.unpack("!I", header)
    return cloudpickle.loads(await reader.readexactly(length))


class RemoteNodeServer:
    """Hosts node registrations and routes messages between clients (and
    optionally locally-hosted DecentralizedNodes)."""

    def __init__(self, host: str = "127.0.0.1", port: int = 0) -> None:
        self.host, self.port = host, int(port)
        self._server: Optional[asyncio.AbstractServer] = None
        self._clients: Dict[str, asyncio.StreamWriter] = {}
        self._local_nodes: Dict[str, Any] = {}

    def host_node(self, node: Any) -> None:
        self._local_nodes[node.node_id] = node

    async def start(self) -> None:
        self._server = await asyncio.start_server(self._serve, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        for node in self._local_nodes.values():
            await node.attach_context(ServerNodeContext(node.node_id, self))
            await node.start_local()

    async def stop(self) -> None:
        for node in self._local_nodes.values():
            await node.stop_local()
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    async def route(self, target_id: str, message: dict) -> None:
        if target_id in self._local_nodes:
            await self._local_nodes[target_id].handle_incoming_message(message)
            return
        writer = self._clients.get(target_id)
        if writer is None:
            raise RuntimeError(f"no route to node {target_id!r}")
        await _send_frame(writer, ("message", message))

    async def _serve(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        registered: Optional[str] = None
        try:
            while True:
                try:
                    kind, payload = await _recv_frame(reader)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                if kind == "register":
                    registered = payload
                    self._clients[payload] = writer
                    await _send_frame(writer, ("ok", payload))
                elif kind == "send":
                    target_id, message = payload
                    try:
                        await self.route(target_id, message)
                        await _send_frame(writer, ("ok", None))
                    except Exception as e:  # noqa: BLE001
                        await _send_frame(writer, ("err", repr(e)))
        finally:
            if registered is not None:
                self._clients.pop(registered, None)
            writer.close()


class RemoteNodeClient:
    def __init__(self, node_id: str, host: str, port: int) -> None:
        self.node_id = node_id
        self.host, self.port = host, int(port)
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._inbox: asyncio.Queue = asyncio.Queue()
        self._recv_task: Optional[asyncio.Task] = None
        self._pending_acks: asyncio.Queue = asyncio.Queue()

    async def connect(self) -> None:
        self._reader, self._writer = await asyncio.open_connection(self.host, self.port)
        await _send_frame(self._writer, ("register", self.node_id))
        kind, _ = await _recv_frame(self._reader)
        if kind != "ok":
            raise RuntimeError("node registration failed")
        self._recv_task = asyncio.get_running_loop().create_task(self._recv_loop())

    async def _recv_loop(self) -> None:
        while True:
            try:
                kind, payload = await _recv_frame(self._reader)
            except (asyncio.IncompleteReadError, ConnectionResetError):
                break
            if kind == "message":
                self._inbox.put_nowait(payload)
            else:  # ok / err ack for a send
                self._pending_acks.put_nowait((kind, payload))

    async def send(self, target_id: str, message: dict) -> None:
        await _send_frame(self._writer, ("send", (target_id, message)))
        kind, payload = await self._pending_acks.get()
        if kind != "ok":
            raise RuntimeError(f"remote send failed: {payload}")

    async def receive(self, timeout: float = 0.1) -> Optional[dict]:
        try:
            return await asyncio.wait_for(self._inbox.get(), timeout)
        except asyncio.TimeoutError:
            return None

    async def close(self) -> None:
        if self._recv_task is not None:
            self._recv_task.cancel()
            try:
                await self._recv_task
            except asyncio.CancelledError:
                pass
        if self._writer is not None:
            self._writer.close()


class ServerNodeContext(NodeContext):
    """Context for a node HOSTED on a RemoteNodeServer: outgoing messages
    go through the server's routing table."""

    def __init__(self, node_id: str, server: RemoteNodeServer) -> None:
        self.node_id = node_id
        self.server = server

    async def start(self, node: Any) -> None:
        pass

    async def send_message(self, target_id: str, message: dict) -> None:
        await self.server.route(target_id, message)

    async def shutdown(self) -> None:
        pass


class RemoteContext(NodeContext):
    """Context for a node living in THIS process but registered with a
    RemoteNodeServer; a background loop pumps received messages into the
    node (reference context.py:565-705)."""

    def __init__(self, node_id: str, host: str, port: int) -> None:
        self.node_id = node_id
        self.client = RemoteNodeClient(node_id, host, port)
        self.node: Any = None
        self._pump: Optional[asyncio.Task] = None

    async def start(self, node: Any) -> None:
        self.node = node
        await self.client.connect()
        self._pump = asyncio.get_running_loop().create_task(self._pump_loop())

    async def _pump_loop(self) -> None:
        while True:
            msg = await self.client.receive(timeout=0.2)
            if msg is not None and self.node is not None:
                await self.node.handle_incoming_message(msg)

    async def receive(self, timeout: float = 0.1) -> Optional[dict]:
        return None  # messages are pumped directly into the node

    async def send_message(self, target_id: str, message: dict) -> None:
        await self.client.send(target_id, message)

    async def shutdown(self) -> None:
        if self._pump is not None:
            self._pump.cancel()
            try:
                await self._pump
            except asyncio.CancelledError:
                pass
        await self.client.close()


class MeshRemoteContext(NodeContext):
    """Serverless mesh: every node runs its own TCP server and dials its
    peers directly; dead peers are re-dialed by a reconnect monitor
    (reference context.py:708-1055)."""

    def __init__(
        self,
        node_id: str,
        host: str = "127.0.0.1",
        port: int = 0,
        *,
        peers: Optional[Dict[str, tuple]] = None,
        reconnect_interval: float = 2.0,
    ) -> None:
        self.node_id = node_id
        self.host, self.port = host, int(port)
        self.peers: Dict[str, tuple] = dict(peers or {})
        self.reconnect_interval = reconnect_interval
        self.node: Any = None
        self._server: Optional[asyncio.AbstractServer] = None
        self._out: Dict[str, asyncio.StreamWriter] = {}
        self._monitor: Optional[asyncio.Task] = None

    def add_peer(self, node_id: str, host: str, port: int) -> None:
        self.peers[node_id] = (host, port)

    async def start(self, node: Any) -> None:
        self.node = node
        self._server = await asyncio.start_server(self._serve, self.host, self.port)
        self.port = self._server.sockets[0].getsockname()[1]
        self._monitor = asyncio.get_running_loop().create_task(self._reconnect_loop())

    async def _serve(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    kind, payload = await _recv_frame(reader)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                if kind == "message" and self.node is not None:
                    await self.node.handle_incoming_message(payload)
        finally:
            writer.close()

    async def _dial(self, node_id: str) -> Optional[asyncio.StreamWriter]:
        host, port = self.peers[node_id]
        try:
            _, writer = await asyncio.open_connection(host, port)
            self._out[node_id] = writer
            return writer
        except OSError:
            return None

    async def _reconnect_loop(self) -> None:
        while True:
            for nid in list(self.peers):
                w = self._out.get(nid)
                if w is None or w.is_closing():
                    await self._dial(nid)
            await asyncio.sleep(self.reconnect_interval)

    async def send_message(self, target_id: str, message: dict) -> None:
        writer = self._out.get(target_id)
        if writer is None or writer.is_closing():
            writer = await self._dial(target_id)
        if writer is None:
            raise RuntimeError(f"peer {target_id!r} unreachable")
        await _send_frame(writer, ("message", message))

    async def receive(self, timeout: float = 0.1) -> Optional[dict]:
        return None  # inbound messages are delivered by _serve directly

    async def shutdown(self) -> None:
        if self._monitor is not None:
            self._monitor.cancel()
            try:
                await self._monitor
            except asyncio.CancelledError:
                pass
        for w in self._out.values():
            w.close()
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
