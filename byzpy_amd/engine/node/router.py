"""MessageRouter — topology-constrained direct/broadcast/multicast.

Reference parity: engine/node/router.py:10-257 (int<->string node-id
mapping; broadcast swallows per-neighbor failures 181-186; multicast
validates all targets first 211-214).
"""
from __future__ import annotations

import logging
from typing import Any, Dict, List, Sequence

logger = logging.getLogger(__name__)


class MessageRouter:
    def __init__(self, node_id: str, context: Any, topology: Any = None) -> None:
        self.node_id = node_id
        self.context = context
        self.topology = topology
        # topology index <-> node id
        self._index_to_id: Dict[int, str] = {}
        self._id_to_index: Dict[str, int] = {}

    def set_node_mapping(self, index_to_id: Dict[int, str]) -> None:
        self._index_to_id = dict(index_to_id)
        self._id_to_index = {v: k for k, v in self._index_to_id.items()}

    def _out_neighbor_ids(self) -> List[str]:
        if self.topology is None or self.node_id not in self._id_to_index:
            return [i for i in self._index_to_id.values() if i != self.node_id]
        idx = self._id_to_index[self.node_id]
        return [self._index_to_id[j] for j in self.topology.out_neighbors(idx)]

    def _check_allowed(self, target_id: str) -> None:
        if self.topology is None:
            return
        if target_id not in self._id_to_index or self.node_id not in self._id_to_index:
            raise ValueError(f"unknown node {target_id!r}")
        src = self._id_to_index[self.node_id]
        dst = self._id_to_index[target_id]
        if dst not in self.topology.out_neighbors(src):
            raise ValueError(
                f"topology forbids {self.node_id!r} -> {target_id!r}"
            )

    async def route_direct(self, target_id: str, message: dict) -> None:
        self._check_allowed(target_id)
        await self.context.send_message(target_id, message)

    async def route_broadcast(self, message: dict) -> int:
        sent = 0
        for nid in self._out_neighbor_ids():
            try:
                await self.context.send_message(nid, message)
                sent += 1
            except Exception as e:  # noqa: BLE001 — per-neighbor tolerance
                logger.warning("broadcast to %s failed: %r", nid, e)
        return sent

    async def route_multicast(self, target_ids: Sequence[str], message: dict) -> None:
        for t in target_ids:
            self._check_allowed(t)
        for t in target_ids:
            await self.context.send_message(t, message)

    async def route_reply(self, original: dict, message: dict) -> None:
        sender = original.get("sender")
        if sender is None:
            raise ValueError("original message has no sender")
        message.setdefault("reply_to", original.get("type"))
        await self.context.send_message(sender, message)
