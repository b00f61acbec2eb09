"""Node actors — construct a Node subclass by value on any actor backend.

Reference parity: engine/node/actors.py:41-91. ``NodeActor.__getattr__``
delegates to the ActorRef so every node method becomes an async RPC.
MI355X note: ``spawn(..., backend="stream:K")`` puts the node's tensors on
GPU K with a dedicated HIP stream.
"""
from __future__ import annotations

from typing import Any

from byzpy_amd.actor.base import ActorRef
from byzpy_amd.actor.factory import resolve_backend


class NodeActor:
    def __init__(self, backend: Any, ref: ActorRef) -> None:
        self._backend = backend
        self._ref = ref

    @classmethod
    async def spawn(cls, node_cls: type, *args: Any, backend: Any = "thread", **kwargs: Any):
        be = resolve_backend(backend)
        await be.start()
        await be.construct(node_cls, *args, **kwargs)
        return cls(be, ActorRef(be))

    @property
    def backend(self) -> Any:
        return self._backend

    async def close(self) -> None:
        await self._backend.close()

    def __getattr__(self, name: str):
        if name.startswith("_"):
            raise AttributeError(name)
        return getattr(self._ref, name)


class HonestNodeActor(NodeActor):
    pass


class ByzantineNodeActor(NodeActor):
    pass
