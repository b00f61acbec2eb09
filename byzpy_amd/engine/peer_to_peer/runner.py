"""DecentralizedPeerToPeer — gossip training rounds over DecentralizedNodes.

Reference parity: engine/peer_to_peer/runner.py:284-389 (round = half-step
pipelines -> broadcast "gradient" messages -> byz nodes attack cached
neighbor vectors -> honest nodes aggregate [self]+received). Improvement
over the reference's fixed asyncio.sleep(0.1) propagation barriers
(runner.py:322/371): rounds use count-based completion — each node awaits
exactly one vector per in-neighbor per round.
"""
from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional, Sequence

import torch

from byzpy_amd.engine.node.decentralized import DecentralizedNode
from byzpy_amd.engine.node.cluster import DecentralizedCluster
from byzpy_amd.engine.peer_to_peer.topology import Topology


class _P2PNodeShell(DecentralizedNode):
    """DecentralizedNode wrapper that collects per-round neighbor vectors."""

    def __init__(self, node_id: str, obj: Any, is_byzantine: bool, **kw: Any) -> None:
        super().__init__(node_id, **kw)
        self.obj = obj
        self.is_byzantine = is_byzantine
        self._round_vectors: Dict[int, List[torch.Tensor]] = {}
        self._round_conds: Dict[int, asyncio.Condition] = {}
        self._expected = 0
        self.register_handler("gradient", self._on_gradient)
        # gossip shells consume every message through the handler above;
        # the scheduler's mailbox mirror would otherwise retain one
        # broadcast payload per sender per round (caught by the device
        # long-run soak — see MessageAwareNodeScheduler.cache_limit)
        self.scheduler.cache_limit = 0

    def set_expected(self, count: int) -> None:
        self._expected = count

    def _cond(self, rnd: int) -> asyncio.Condition:
        return self._round_conds.setdefault(rnd, asyncio.Condition())

    async def _on_gradient(self, msg: dict) -> None:
        rnd = int(msg["round"])
        cond = self._cond(rnd)
        async with cond:
            self._round_vectors.setdefault(rnd, []).append(msg["vector"])
            cond.notify_all()

    async def wait_round(
        self, rnd: int, count: Optional[int] = None, timeout: float = 30.0
    ) -> List[torch.Tensor]:
        """Await `count` vectors for round `rnd` (default: all in-neighbors).
        Byzantine turns pass the HONEST in-neighbor count: waiting for the
        full neighborhood would deadlock two byzantine neighbors, each
        waiting for the other's not-yet-broadcast vector."""
        want = self._expected if count is None else count
        if want <= 0:
            return []
        cond = self._cond(rnd)
        async with cond:
            await asyncio.wait_for(
                cond.wait_for(
                    lambda: len(self._round_vectors.get(rnd, ())) >= want
                ),
                timeout,
            )
            return list(self._round_vectors.get(rnd, ()))[:want] if count is not None else self._round_vectors.get(rnd, [])

    def drop_round(self, rnd: int) -> None:
        self._round_vectors.pop(rnd, None)
        self._round_conds.pop(rnd, None)


class DecentralizedPeerToPeer:
    def __init__(
        self,
        honest_nodes: Sequence[Any],  # P2PHonestMixin objects
        byzantine_nodes: Sequence[Any],  # P2PByzantineMixin objects
        aggregator: Any,
        *,
        topology: Optional[Topology] = None,
        pre_aggregator: Any = None,
        lr: float = 0.1,
    ) -> None:
        self.aggregator = aggregator
        self.pre_aggregator = pre_aggregator
        self.lr = lr
        n = len(honest_nodes) + len(byzantine_nodes)
        self.topology = topology or Topology.complete(n)
        self.cluster = DecentralizedCluster(self.topology)
        self.shells: List[_P2PNodeShell] = []
        for i, obj in enumerate(list(honest_nodes) + list(byzantine_nodes)):
            shell = _P2PNodeShell(
                f"node-{i}", obj, is_byzantine=i >= len(honest_nodes)
            )
            self.shells.append(shell)
            self.cluster.add_node(shell)
        self.n_honest = len(honest_nodes)
        self._round = 0
        self._started = False

    async def bootstrap(self) -> None:
        if self._started:
            return
        await self.cluster.start_all()
        for i, shell in enumerate(self.shells):
            shell.set_expected(len(self.topology.in_neighbors(i)))
        self._started = True

    async def run_round_async(self) -> None:
        rnd = self._round
        self._round += 1
        honest = [s for s in self.shells if not s.is_byzantine]
        byz = [s for s in self.shells if s.is_byzantine]

        # 1. local half-steps (possibly heavy: run concurrently)
        halves: Dict[str, torch.Tensor] = {}
        for s in honest:
            halves[s.node_id] = s.obj.p2p_half_step(self.lr)

        # 2. honest nodes broadcast theta-half to out-neighbors
        await asyncio.gather(
            *(
                s.broadcast_message("gradient", {"vector": halves[s.node_id], "round": rnd})
                for s in honest
            )
        )

        # 3. byzantine nodes attack the honest vectors they can see, then
        # broadcast the malicious vector. They wait only for HONEST
        # in-neighbors (honest indices are < n_honest): waiting for the
        # full neighborhood would deadlock two byzantine neighbors.
        async def byz_turn(s: _P2PNodeShell) -> None:
            idx = self.shells.index(s)
            h_in = sum(
                1 for j in self.topology.in_neighbors(idx) if j < self.n_honest
            )
            received = await s.wait_round(rnd, count=h_in) if h_in else []
            honest_vecs = received or list(halves.values())
            vec = s.obj.p2p_broadcast_vector(honest_vecs)
            await s.broadcast_message("gradient", {"vector": vec, "round": rnd})

        await asyncio.gather(*(byz_turn(s) for s in byz))

        # 4. honest nodes aggregate [self] + everything received this round
        async def honest_aggregate(s: _P2PNodeShell) -> None:
            received = await s.wait_round(rnd)
            vectors = [halves[s.node_id]] + received
            s.obj.p2p_aggregate_and_set(vectors, self.aggregator, self.pre_aggregator)

        await asyncio.gather(*(honest_aggregate(s) for s in honest))
        for s in self.shells:
            s.drop_round(rnd)

    async def shutdown(self) -> None:
        await self.cluster.shutdown_all()
        self._started = False
