"""TCP framing: 4-byte big-endian length prefix + pickle.

Reference parity: engine/actor/_wire.py:5-18.
"""
from __future__ import annotations

import asyncio
import pickle
import struct
from typing import Any


async def send_obj(writer: asyncio.StreamWriter, obj: Any) -> None:
    blob = pickle.dumps(obj)
    writer.write(struct.pack("!I", len(blob)) + blob)
    await writer.drain()


async def recv_obj(reader: asyncio.StreamReader) -> Any:
    header = await reader.readexactly(4)
    (length,) = struct.unpack("!I", header)
    blob = await reader.readexactly(length)
    return pickle.loads(blob)
