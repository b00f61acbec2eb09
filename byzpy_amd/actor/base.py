"""ActorBackend protocol + ActorRef async RPC proxy.

Reference parity: engine/actor/base.py:8-60. The MI355X fleet collapses the
reference's thread/process/gpu/remote zoo into: thread (CPU), process
(CPU, shm tensors), stream (one HIP stream per worker, tensors resident in
HBM) and remote-TCP (control plane; device data moves over RCCL).
"""
from __future__ import annotations

from typing import Any, Protocol, runtime_checkable


@runtime_checkable
class ActorBackend(Protocol):
    async def start(self) -> None: ...

    async def construct(self, factory: Any, /, *args: Any, **kwargs: Any) -> None: ...

    async def call(self, method: str, /, *args: Any, **kwargs: Any) -> Any: ...

    async def close(self) -> None: ...

    def get_endpoint(self) -> Any: ...

    async def chan_open(self, name: str) -> None: ...

    async def chan_put(self, endpoint: Any, name: str, payload: Any) -> None: ...

    async def chan_get(self, name: str) -> Any: ...


class ActorRef:
    """``ref.method(*a, **kw)`` -> awaitable backend.call("method", ...)."""

    def __init__(self, backend: ActorBackend) -> None:
        self._backend = backend

    @property
    def backend(self) -> ActorBackend:
        return self._backend

    def __getattr__(self, name: str):
        if name.startswith("_"):
            raise AttributeError(name)

        async def _call(*args: Any, **kwargs: Any) -> Any:
            return await self._backend.call(name, *args, **kwargs)

        _call.__name__ = name
        return _call
