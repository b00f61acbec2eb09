"""Process-local channel router: {scheme -> {actor_id -> backend}}.

Reference parity: engine/actor/router.py:24-55 (global singleton breaks
import cycles between backends).
"""
from __future__ import annotations

from typing import Any, Dict, Optional


class ChannelRouter:
    def __init__(self) -> None:
        self._registry: Dict[str, Dict[str, Any]] = {}

    def register(self, scheme: str, actor_id: str, backend: Any) -> None:
        self._registry.setdefault(scheme, {})[actor_id] = backend

    def unregister(self, scheme: str, actor_id: str) -> None:
        self._registry.get(scheme, {}).pop(actor_id, None)

    def lookup(self, scheme: str, actor_id: str) -> Optional[Any]:
        return self._registry.get(scheme, {}).get(actor_id)


channel_router = ChannelRouter()
