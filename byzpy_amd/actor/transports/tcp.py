"""Stateless TCP channel helper: one fresh connection per chan op.

Reference parity: engine/actor/transports/tcp.py:16-67.
"""
from __future__ import annotations

from typing import Any

from byzpy_amd.actor._wire import recv_obj, send_obj
from byzpy_amd.actor.channels import Endpoint

import asyncio


async def chan_put(endpoint: Endpoint, name: str, payload: Any) -> None:
    host, port = endpoint.address.rsplit(":", 1)
    reader, writer = await asyncio.open_connection(host, int(port))
    try:
        await send_obj(writer, ("chan_deliver", endpoint.actor_id, name, payload))
        status, detail = await recv_obj(reader)
        if status != "ok":
            raise RuntimeError(f"tcp chan_put failed: {detail}")
    finally:
        writer.close()


async def chan_get(endpoint: Endpoint, name: str) -> Any:
    host, port = endpoint.address.rsplit(":", 1)
    reader, writer = await asyncio.open_connection(host, int(port))
    try:
        await send_obj(writer, ("chan_get", endpoint.actor_id, name))
        status, payload = await recv_obj(reader)
        if status != "ok":
            raise RuntimeError(f"tcp chan_get failed: {payload}")
        return payload
    finally:
        writer.close()
