"""ThreadActorBackend — one worker thread per actor, asyncio mailboxes.

Reference parity: engine/actor/backends/thread.py:14-171. chan_put routes
locally via the channel_router for thread/process/stream schemes.
"""
from __future__ import annotations

import asyncio
import itertools
import uuid
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict

from byzpy_amd.actor.channels import Endpoint
from byzpy_amd.actor.router import channel_router

_ids = itertools.count()


class ThreadActorBackend:
    scheme = "thread"

    def __init__(self) -> None:
        self.actor_id = f"thread-{next(_ids)}-{uuid.uuid4().hex[:8]}"
        self._executor: ThreadPoolExecutor | None = None
        self._obj: Any = None
        self._mailboxes: Dict[str, asyncio.Queue] = {}

    async def start(self) -> None:
        if self._executor is None:
            self._executor = ThreadPoolExecutor(
                max_workers=1, thread_name_prefix=self.actor_id
            )
            channel_router.register(self.scheme, self.actor_id, self)

    async def construct(self, factory: Any, /, *args: Any, **kwargs: Any) -> None:
        loop = asyncio.get_running_loop()
        self._obj = await loop.run_in_executor(
            self._executor, lambda: factory(*args, **kwargs)
        )

    async def call(self, method: str, /, *args: Any, **kwargs: Any) -> Any:
        fn = getattr(self._obj, method)
        if asyncio.iscoroutinefunction(fn):
            return await fn(*args, **kwargs)
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(self._executor, lambda: fn(*args, **kwargs))

    async def close(self) -> None:
        if self._executor is not None:
            self._executor.shutdown(wait=True)
            self._executor = None
        channel_router.unregister(self.scheme, self.actor_id)
        self._obj = None

    def get_endpoint(self) -> Endpoint:
        return Endpoint(scheme=self.scheme, address="local", actor_id=self.actor_id)

    # -- channels ----------------------------------------------------------
    async def chan_open(self, name: str) -> None:
        self._mailboxes.setdefault(name, asyncio.Queue())

    async def _deliver(self, name: str, payload: Any) -> None:
        self._mailboxes.setdefault(name, asyncio.Queue()).put_nowait(payload)

    async def chan_put(self, endpoint: Endpoint, name: str, payload: Any) -> None:
        target = channel_router.lookup(endpoint.scheme, endpoint.actor_id)
        if target is not None:
            await target._deliver(name, payload)
            return
        if endpoint.scheme == "tcp":
            from byzpy_amd.actor.transports import tcp

            await tcp.chan_put(endpoint, name, payload)
            return
        raise RuntimeError(f"no route to endpoint {endpoint!r}")

    async def chan_get(self, name: str) -> Any:
        q = self._mailboxes.setdefault(name, asyncio.Queue())
        return await q.get()
