"""PeerToPeer facade.

Reference parity: engine/peer_to_peer/train.py:17-86 (bootstrap / round /
shutdown over the decentralized runner).
"""
from __future__ import annotations

from typing import Any, Optional, Sequence

from byzpy_amd.engine.peer_to_peer.runner import DecentralizedPeerToPeer
from byzpy_amd.engine.peer_to_peer.topology import Topology


class PeerToPeer:
    def __init__(
        self,
        honest_nodes: Sequence[Any],
        byzantine_nodes: Sequence[Any],
        aggregator: Any,
        *,
        topology: Optional[Topology] = None,
        pre_aggregator: Any = None,
        lr: float = 0.1,
    ) -> None:
        self._runner = DecentralizedPeerToPeer(
            honest_nodes,
            byzantine_nodes,
            aggregator,
            topology=topology,
            pre_aggregator=pre_aggregator,
            lr=lr,
        )

    async def bootstrap(self) -> None:
        await self._runner.bootstrap()

    async def round(self) -> None:
        await self._runner.bootstrap()
        await self._runner.run_round_async()

    async def shutdown(self) -> None:
        await self._runner.shutdown()
