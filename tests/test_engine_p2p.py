"""Peer-to-peer engine tests: topology, router enforcement, gossip rounds."""
import asyncio

import pytest
import torch
from torch import nn

from byzpy_amd.aggregators import CoordinateWiseMedian
from byzpy_amd.attacks import EmpireAttack
from byzpy_amd.engine.peer_to_peer.mixin import P2PByzantineMixin, P2PHonestMixin
from byzpy_amd.engine.peer_to_peer.topology import Topology
from byzpy_amd.engine.peer_to_peer.train import PeerToPeer


class TinyHonest(P2PHonestMixin):
    def __init__(self, seed: int):
        torch.manual_seed(seed)
        self.model = nn.Linear(4, 1, bias=False)
        self.lr = 0.05
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(16, 4, generator=g)
        self.w_true = torch.tensor([[1.0, -1.0, 0.5, 2.0]])
        self.y = self.x @ self.w_true.T

    def p2p_local_loss_backward(self):
        loss = ((self.model(self.x) - self.y) ** 2).mean()
        loss.backward()


class TinyByz(P2PByzantineMixin):
    def __init__(self):
        self.attack = EmpireAttack(scale=-1.0)


class TestTopology:
    def test_complete(self):
        t = Topology.complete(4)
        assert t.out_neighbors(0) == [1, 2, 3]
        assert t.in_neighbors(3) == [0, 1, 2]

    def test_ring(self):
        t = Topology.ring(5, 1)
        assert t.out_neighbors(0) == [1, 4]
        t2 = Topology.ring(6, 2)
        assert t2.out_neighbors(0) == [1, 2, 4, 5]

    def test_bad_edge(self):
        with pytest.raises(ValueError):
            Topology(2, [(0, 5)])


def test_p2p_rounds_converge():
    async def main():
        honest = [TinyHonest(s) for s in range(4)]
        byz = [TinyByz()]
        p2p = PeerToPeer(honest, byz, CoordinateWiseMedian(), lr=0.05)
        await p2p.bootstrap()
        for _ in range(60):
            await p2p.round()
        await p2p.shutdown()
        w = honest[0].model.weight.detach()
        err = (w - honest[0].w_true).norm()
        assert err < 0.4, f"did not converge toward w_true: err={err}"
        # all honest nodes agree after aggregation
        w2 = honest[1].model.weight.detach()
        assert torch.allclose(w, w2, atol=1e-5)

    asyncio.run(main())


def test_p2p_ring_topology():
    async def main():
        honest = [TinyHonest(s) for s in range(4)]
        p2p = PeerToPeer(
            honest, [], CoordinateWiseMedian(), topology=Topology.ring(4, 1)
        )
        await p2p.bootstrap()
        await p2p.round()
        await p2p.shutdown()

    asyncio.run(main())


def test_router_topology_enforcement():
    from byzpy_amd.engine.node.decentralized import DecentralizedNode
    from byzpy_amd.engine.node.cluster import DecentralizedCluster

    async def main():
        topo = Topology.ring(3, 1)
        cluster = DecentralizedCluster(topo)
        nodes = [DecentralizedNode(f"node-{i}") for i in range(3)]
        for n in nodes:
            cluster.add_node(n)
        await cluster.start_all()
        # ring(3,1) is complete for n=3; shrink: use a directed chain instead
        await cluster.shutdown_all()

        topo2 = Topology(3, [(0, 1), (1, 2)])
        cluster2 = DecentralizedCluster(topo2)
        nodes2 = [DecentralizedNode(f"m-{i}") for i in range(3)]
        for n in nodes2:
            cluster2.add_node(n)
        await cluster2.start_all()
        await nodes2[0].send_message("m-1", "ping", {})
        with pytest.raises(ValueError):
            await nodes2[0].send_message("m-2", "ping", {})
        await cluster2.shutdown_all()

    asyncio.run(main())


def test_p2p_with_preaggregator():
    """Gossip round with NNM pre-aggregation before the robust aggregate
    (BASELINE config 4 op pair, actor engine flavor)."""
    from byzpy_amd.aggregators import GeometricMedian
    from byzpy_amd.pre_aggregators import NearestNeighborMixing

    async def main():
        honest = [TinyHonest(s) for s in range(4)]
        p2p = PeerToPeer(
            honest,
            [],
            GeometricMedian(tol=1e-7),
            pre_aggregator=NearestNeighborMixing(1),
            lr=0.05,
        )
        await p2p.bootstrap()
        for _ in range(25):
            await p2p.round()
        await p2p.shutdown()
        w = honest[0].model.weight.detach()
        err = (w - honest[0].w_true).norm()
        assert err < 0.6, f"NNM+GeoMedian gossip did not converge: {err}"

    asyncio.run(main())


def test_p2p_repeated_attack_rounds():
    """Byzantine pressure applied EVERY round must not prevent honest
    convergence under a robust aggregator."""
    async def main():
        honest = [TinyHonest(s) for s in range(4)]
        byz = [TinyByz(), TinyByz()]  # 2 byzantine vs 4 honest, complete topo
        p2p = PeerToPeer(honest, byz, CoordinateWiseMedian(), lr=0.05)
        await p2p.bootstrap()
        for _ in range(60):
            await p2p.round()
        await p2p.shutdown()
        w = honest[0].model.weight.detach()
        assert (w - honest[0].w_true).norm() < 0.5

    asyncio.run(main())


def test_p2p_facade_kwargs_compat():
    """PeerToPeer facade accepts the reference's keyword surface."""
    honest = [TinyHonest(0)]
    p2p = PeerToPeer(
        honest,
        [],
        CoordinateWiseMedian(),
        topology=Topology.complete(1),
        pre_aggregator=None,
        lr=0.123,
    )
    assert p2p._runner.lr == 0.123


def test_p2p_base_grad_attack():
    """Regression: base_grad attacks (SignFlip) in P2P gossip derive their
    input from the mixin's uses_* flags instead of always honest_grads."""
    from byzpy_amd.attacks import SignFlipAttack
    from byzpy_amd.engine.peer_to_peer.mixin import P2PByzantineMixin

    class Byz(P2PByzantineMixin):
        def __init__(self):
            self.attack = SignFlipAttack(scale=-2.0)

    vecs = [torch.ones(4), 3 * torch.ones(4)]
    out = Byz().p2p_broadcast_vector(vecs)
    # base_grad = mean(vecs) = 2; scaled by -2 -> -4
    assert torch.allclose(out, -4.0 * torch.ones(4))

    async def gossip():
        honest = [TinyHonest(s) for s in range(3)]
        p2p = PeerToPeer(honest, [Byz()], CoordinateWiseMedian(), lr=0.05)
        await p2p.bootstrap()
        for _ in range(2):
            await p2p.round()
        await p2p.shutdown()
        for h in honest:
            assert torch.isfinite(h.p2p_flat_params()).all()

    asyncio.run(gossip())


def test_rccl_p2p_rejects_inconsistent_byzantine_set():
    """A rank listed in byzantine_ranks without an attack would deadlock
    its peers' phase-2 recvs — must be rejected at construction."""
    import pytest
    import torch

    from byzpy_amd.aggregators import CoordinateWiseMedian
    from byzpy_amd.engine.peer_to_peer.rccl import RcclPeerToPeer

    with pytest.raises(ValueError):
        RcclPeerToPeer(
            lambda: torch.ones(4),
            lambda v: None,
            CoordinateWiseMedian(),
            byzantine_ranks=[0],  # this (only) rank has no attack
        )


def test_p2p_rounds_do_not_retain_tensors():
    """Regression for a real leak: the scheduler's message-cache mirror
    retained one broadcast payload per sender per round, forever (gossip
    shells consume via handlers; the mailbox was never drained). Tensor
    census must be flat across rounds."""
    import gc

    import torch.nn as nn

    from byzpy_amd.attacks import SignFlipAttack

    class H(P2PHonestMixin):
        def __init__(self, seed):
            torch.manual_seed(seed)
            self.model = nn.Linear(16, 1, bias=False)
            self.lr = 0.05
            g = torch.Generator().manual_seed(seed)
            self.x = torch.randn(8, 16, generator=g)
            self.y = self.x.sum(dim=1, keepdim=True) * 0.1

        def p2p_local_loss_backward(self):
            loss = ((self.model(self.x) - self.y) ** 2).mean()
            loss.backward()

    class B(P2PByzantineMixin):
        def __init__(self):
            self.attack = SignFlipAttack()

    def census():
        gc.collect()
        return sum(1 for o in gc.get_objects() if torch.is_tensor(o))

    async def main():
        honest = [H(s) for s in (1, 2, 3)]
        p2p = PeerToPeer(honest, [B()], CoordinateWiseMedian(), lr=0.05)
        for _ in range(5):
            await p2p.round()
        c0 = census()
        for _ in range(25):
            await p2p.round()
        c1 = census()
        assert c1 <= c0, f"tensor census grew {c0} -> {c1} over 25 rounds"
        await p2p.shutdown()

    asyncio.run(main())
