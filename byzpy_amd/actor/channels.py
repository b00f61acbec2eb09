"""Endpoints and channel references.

Reference parity: engine/actor/channels.py:13-65.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any

from byzpy_amd.actor.ipc import unwrap_payload


@dataclass(frozen=True)
class Endpoint:
    scheme: str  # "thread" | "process" | "stream" | "tcp"
    address: str
    actor_id: str


class ChannelRef:
    """Mailbox handle bound to (backend, channel name); recv applies the
    universal IPC unwrap."""

    def __init__(self, backend: Any, name: str) -> None:
        self.backend = backend
        self.name = name

    async def send(self, endpoint: Endpoint, payload: Any) -> None:
        await self.backend.chan_put(endpoint, self.name, payload)

    async def recv(self) -> Any:
        return unwrap_payload(await self.backend.chan_get(self.name))


async def open_channel(backend: Any, name: str) -> ChannelRef:
    await backend.chan_open(name)
    return ChannelRef(backend, name)
