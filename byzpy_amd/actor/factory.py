"""resolve_backend(spec) — string spec to actor backend instance.

Reference parity: engine/actor/factory.py:14-67. Specs:
"thread" | "process" | "stream" | "stream:N" (HIP stream on device N) |
"gpu" (alias of stream) | "tcp://host:port".
"""
from __future__ import annotations

from typing import Any


def resolve_backend(spec: Any):
    if not isinstance(spec, str):
        return spec  # already a backend instance
    if spec == "thread":
        from byzpy_amd.actor.backends.thread import ThreadActorBackend

        return ThreadActorBackend()
    if spec == "process":
        from byzpy_amd.actor.backends.process import ProcessActorBackend

        return ProcessActorBackend()
    if spec in ("stream", "gpu"):
        from byzpy_amd.actor.backends.stream import StreamActorBackend

        return StreamActorBackend()
    if spec.startswith("stream:") or spec.startswith("gpu:"):
        from byzpy_amd.actor.backends.stream import StreamActorBackend

        return StreamActorBackend(device=int(spec.split(":", 1)[1]))
    if spec.startswith("tcp://"):
        from byzpy_amd.actor.backends.remote import RemoteActorBackend

        hostport = spec[len("tcp://") :]
        host, port = hostport.rsplit(":", 1)
        return RemoteActorBackend(host, int(port))
    raise ValueError(f"unknown actor backend spec {spec!r}")
