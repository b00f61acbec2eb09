"""StreamActorBackend — the MI355X-native actor: one HIP stream per worker.

This replaces the reference's in-process GPUActorBackend
(engine/actor/backends/gpu.py:23-200) and its whole UCX satellite: a worker
is a CDNA4 HIP stream (+ the calling host thread); gradients live in HBM
and mailbox sends between stream actors in one process are zero-copy
device references (stream-ordered). Cross-process device movement is
RCCL's job (byzpy_amd/parallel/), not this backend's.

Without a GPU the backend still runs (streams become no-ops), so CPU CI
exercises the same code path.
"""
from __future__ import annotations

import asyncio
import contextlib
import itertools
import uuid
from typing import Any, Dict, Optional

import torch

from byzpy_amd.actor.channels import Endpoint
from byzpy_amd.actor.router import channel_router

_ids = itertools.count()


class StreamActorBackend:
    scheme = "stream"

    def __init__(self, device: Optional[int] = None) -> None:
        self.actor_id = f"stream-{next(_ids)}-{uuid.uuid4().hex[:8]}"
        self.device_index = device
        self._obj: Any = None
        self._stream: Optional[torch.cuda.Stream] = None
        self._mailboxes: Dict[str, asyncio.Queue] = {}

    @property
    def stream(self) -> Optional[torch.cuda.Stream]:
        return self._stream

    async def start(self) -> None:
        if torch.cuda.is_available():
            idx = self.device_index if self.device_index is not None else 0
            with torch.cuda.device(idx):
                self._stream = torch.cuda.Stream(device=idx)
        channel_router.register(self.scheme, self.actor_id, self)

    @contextlib.contextmanager
    def _on_stream(self):
        if self._stream is not None:
            with torch.cuda.device(self._stream.device):
                with torch.cuda.stream(self._stream):
                    yield
        else:
            yield

    async def construct(self, factory: Any, /, *args: Any, **kwargs: Any) -> None:
        with self._on_stream():
            self._obj = factory(*args, **kwargs)

    async def call(self, method: str, /, *args: Any, **kwargs: Any) -> Any:
        fn = getattr(self._obj, method)
        with self._on_stream():
            result = fn(*args, **kwargs)
        if asyncio.iscoroutine(result):
            return await result
        return result

    async def close(self) -> None:
        if self._stream is not None:
            self._stream.synchronize()
            self._stream = None
        channel_router.unregister(self.scheme, self.actor_id)
        self._obj = None

    def get_endpoint(self) -> Endpoint:
        addr = f"cuda:{self.device_index or 0}" if self._stream is not None else "cpu"
        return Endpoint(scheme=self.scheme, address=addr, actor_id=self.actor_id)

    # -- channels: zero-copy device references, stream-ordered -------------
    async def chan_open(self, name: str) -> None:
        self._mailboxes.setdefault(name, asyncio.Queue())

    async def _deliver(self, name: str, payload: Any) -> None:
        self._mailboxes.setdefault(name, asyncio.Queue()).put_nowait(payload)

    async def chan_put(self, endpoint: Endpoint, name: str, payload: Any) -> None:
        target = channel_router.lookup(endpoint.scheme, endpoint.actor_id)
        if target is not None:
            if self._stream is not None and isinstance(payload, torch.Tensor) and payload.is_cuda:
                # make the payload visible to the consumer stream before use
                evt = torch.cuda.Event()
                evt.record(self._stream)
                payload = _EventedTensor(payload, evt)
            await target._deliver(name, payload)
            return
        raise RuntimeError(f"no route to endpoint {endpoint!r}")

    async def chan_get(self, name: str) -> Any:
        q = self._mailboxes.setdefault(name, asyncio.Queue())
        item = await q.get()
        if isinstance(item, _EventedTensor):
            if self._stream is not None:
                self._stream.wait_event(item.event)
            return item.tensor
        return item


class _EventedTensor:
    """Device tensor + the producer-stream event that orders its validity."""

    __slots__ = ("tensor", "event")

    def __init__(self, tensor: torch.Tensor, event: "torch.cuda.Event") -> None:
        self.tensor = tensor
        self.event = event
