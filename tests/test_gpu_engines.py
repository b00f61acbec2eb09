"""GPU end-to-end engine tests: PS round on stream actors, run_operator on
device tensors, sharded single-rank paths."""
import asyncio

import pytest
import torch

pytestmark = pytest.mark.gpu

from byzpy_amd import run_operator
from byzpy_amd.aggregators import CoordinateWiseTrimmedMean, MultiKrum
from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor
from byzpy_amd.engine.parameter_server.ps import ParameterServer
from byzpy_amd.engine.parameter_server.rccl import (
    RcclParameterServer,
    trimmed_mean_aggregate,
)
from byzpy_amd.engine.peer_to_peer.rccl import RcclPeerToPeer
from byzpy_amd.aggregators import GeometricMedian
from byzpy_amd.pre_aggregators import NearestNeighborMixing


class GpuHonest:
    """Device-resident stub worker (constructed under a stream actor)."""

    def __init__(self, value: float, d: int = 4096):
        self.grad = torch.full((d,), value, device="cuda", dtype=torch.bfloat16)
        self.applied = 0

    def honest_gradient_for_next_batch(self):
        return self.grad

    def apply_server_gradient(self, g):
        assert g.is_cuda
        self.applied += 1

    def applied_count(self):
        return self.applied


class GpuByz:
    def __init__(self, d: int = 4096):
        self.d = d

    def byzantine_gradient_for_next_batch(self, honest_grads=None):
        return torch.full((self.d,), 1e4, device="cuda", dtype=torch.bfloat16)

    def apply_server_gradient(self, g):
        pass


def test_ps_round_on_stream_actors():
    async def main():
        honest = [
            await HonestNodeActor.spawn(GpuHonest, v, backend="stream:0")
            for v in (1.0, 2.0, 3.0)
        ]
        byz = [await ByzantineNodeActor.spawn(GpuByz, backend="stream:0")]
        ps = ParameterServer(honest, byz, CoordinateWiseTrimmedMean(f=1))
        update = await ps.round()
        assert update.is_cuda
        # trimmed f=1 over {1,2,3,1e4} -> mean(2,3) = 2.5
        assert torch.allclose(
            update.float(), torch.full((4096,), 2.5, device="cuda"), atol=0.1
        )
        assert await honest[0].applied_count() == 1
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_run_operator_device_tensors():
    g = torch.Generator().manual_seed(0)
    grads = [torch.randn(8192, generator=g).cuda() for _ in range(12)]
    out = asyncio.run(run_operator(MultiKrum(3, 4), {"gradients": grads}))
    assert out.is_cuda and out.shape == (8192,)


def test_rccl_ps_single_rank_gpu():
    d = 100_000

    def honest(v):
        return lambda: torch.full((d,), float(v), device="cuda")

    ps = RcclParameterServer(
        [honest(1), honest(2), honest(5), honest(1e6)], trimmed_mean_aggregate(1)
    )
    out = ps.round()
    assert out.is_cuda
    assert torch.allclose(out, torch.full((d,), 3.5, device="cuda"), atol=1e-3)


def test_rccl_p2p_single_rank_gpu():
    state = {"p": torch.full((50_000,), 2.0, device="cuda")}
    p2p = RcclPeerToPeer(
        lambda: state["p"],
        lambda v: state.__setitem__("p", v),
        GeometricMedian(max_iter=50),
        pre_aggregator=NearestNeighborMixing(0),
    )
    out = p2p.round()
    assert out.is_cuda
    assert torch.allclose(out, torch.full((50_000,), 2.0, device="cuda"), atol=1e-2)


# -- round-2 depth: full PS/P2P rounds on device -------


class GpuModelHonest:
    """Tiny device model node: real fwd/bwd per round, applies updates."""

    def __init__(self, seed: int, d_in: int = 64):
        import torch.nn as nn

        torch.manual_seed(seed)
        self.model = nn.Linear(d_in, 1, bias=False).cuda()
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(32, d_in, generator=g).cuda()
        self.y = (self.x @ torch.ones(d_in, 1, device="cuda")) * 0.5
        self.lr = 0.05

    def honest_gradient_for_next_batch(self):
        self.model.zero_grad()
        loss = ((self.model(self.x) - self.y) ** 2).mean()
        loss.backward()
        return self.model.weight.grad.detach().reshape(-1).clone()

    def apply_server_gradient(self, g):
        with torch.no_grad():
            self.model.weight -= self.lr * g.reshape(self.model.weight.shape)

    def loss(self):
        with torch.no_grad():
            return float(((self.model(self.x) - self.y) ** 2).mean())


class GpuSignFlipByz:
    """Byzantine node applying a real SignFlipAttack to the honest mean."""

    def __init__(self, scale: float = -8.0):
        from byzpy_amd.attacks import SignFlipAttack

        self.attack = SignFlipAttack(scale=scale)

    def byzantine_gradient_for_next_batch(self, honest_grads=None):
        base = torch.stack(list(honest_grads)).float().mean(dim=0)
        return self.attack.apply(base_grad=base)

    def apply_server_gradient(self, g):
        pass


def test_ps_full_round_signflip_trimmedmean_training():
    """The VERDICT-specified scenario: honest grads -> SignFlip byz ->
    TrimmedMean -> apply, over stream actors, and the loss must go DOWN
    across rounds despite the flipped gradients."""

    async def main():
        honest = [
            await HonestNodeActor.spawn(GpuModelHonest, s, backend="stream:0")
            for s in (1, 2, 3, 4, 5)
        ]
        byz = [
            await ByzantineNodeActor.spawn(GpuSignFlipByz, backend="stream:0")
            for _ in range(2)
        ]
        ps = ParameterServer(honest, byz, CoordinateWiseTrimmedMean(f=2))
        before = await honest[0].loss()
        for _ in range(8):
            update = await ps.round()
            assert update.is_cuda and torch.isfinite(update.float()).all()
        after = await honest[0].loss()
        assert after < before * 0.9, (before, after)
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_ps_round_with_preagg_and_multikrum():
    from byzpy_amd.pre_aggregators import Clipping

    async def main():
        honest = [
            await HonestNodeActor.spawn(GpuHonest, v, backend="stream:0")
            for v in (1.0, 1.1, 0.9, 1.05)
        ]
        byz = [await ByzantineNodeActor.spawn(GpuByz, backend="stream:0")]
        ps = ParameterServer(
            honest,
            byz,
            MultiKrum(1, 2),
            pre_aggregator=Clipping(1e3),
        )
        update = await ps.round()
        # krum must reject the 1e4 byz row; result near the honest cluster
        assert float(update.float().mean()) < 2.0
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_p2p_round_fully_on_device():
    """In-process P2P gossip round where every node's model/vectors live on
    the GPU (message fabric passes device tensors by reference)."""
    import torch.nn as nn

    from byzpy_amd.aggregators import CoordinateWiseMedian
    from byzpy_amd.engine.peer_to_peer.mixin import (
        P2PByzantineMixin,
        P2PHonestMixin,
    )
    from byzpy_amd.engine.peer_to_peer.train import PeerToPeer

    class DevHonest(P2PHonestMixin):
        def __init__(self, seed):
            torch.manual_seed(seed)
            self.model = nn.Linear(8, 1, bias=False).cuda()
            self.lr = 0.05
            g = torch.Generator().manual_seed(seed)
            self.x = torch.randn(16, 8, generator=g).cuda()
            self.y = (self.x.sum(dim=1, keepdim=True)) * 0.25

        def p2p_local_loss_backward(self):
            loss = ((self.model(self.x) - self.y) ** 2).mean()
            loss.backward()

    class DevByz(P2PByzantineMixin):
        def __init__(self):
            from byzpy_amd.attacks import EmpireAttack

            self.attack = EmpireAttack(scale=-2.0)

    async def main():
        honest = [DevHonest(s) for s in (1, 2, 3)]
        byz = [DevByz()]
        p2p = PeerToPeer(honest, byz, CoordinateWiseMedian(), lr=0.05)
        for _ in range(3):
            await p2p.round()
        for h in honest:
            w = h.model.weight.detach()
            assert w.is_cuda and torch.isfinite(w).all()
        await p2p.shutdown()

    asyncio.run(main())


def test_rccl_ps_multikrum_outlier_rejection():
    from byzpy_amd.engine.parameter_server.rccl import multi_krum_aggregate

    d = 200_000

    def honest(v):
        return lambda: torch.full((d,), float(v), device="cuda")

    ps = RcclParameterServer(
        [honest(1.0), honest(1.2), honest(0.8), honest(1e8)],
        multi_krum_aggregate(1, 2),
    )
    out = ps.round()
    assert out.is_cuda
    assert float(out.float().max()) < 2.0  # the 1e8 row was rejected


def test_rccl_p2p_fixed_iters_geomed_on_device():
    state = {"p": torch.full((30_000,), 3.0, device="cuda")}
    p2p = RcclPeerToPeer(
        lambda: state["p"],
        lambda v: state.__setitem__("p", v),
        GeometricMedian(fixed_iters=16),
    )
    out = p2p.round()
    assert out.is_cuda
    assert torch.allclose(out.float(), torch.full((30_000,), 3.0, device="cuda"), atol=1e-2)


class GpuRandHonest:
    def __init__(self, seed: int, d: int = 8192):
        g = torch.Generator(device="cpu").manual_seed(seed)
        self.grad = (
            torch.randn(d, generator=g).to("cuda", torch.bfloat16)
            + float(seed % 3)
        )

    def honest_gradient_for_next_batch(self):
        return self.grad

    def apply_server_gradient(self, g):
        pass


class GpuRandByz:
    def __init__(self, attack, d: int = 8192):
        self.attack = attack
        self.d = d

    def byzantine_gradient_for_next_batch(self, honest_grads=None):
        kwargs = {}
        if getattr(self.attack, "uses_honest_grads", False):
            kwargs["honest_grads"] = honest_grads
        if getattr(self.attack, "uses_base_grad", False):
            kwargs["base_grad"] = honest_grads[0]
        return self.attack.apply(**kwargs).to("cuda", torch.bfloat16)

    def apply_server_gradient(self, g):
        pass


def test_ps_engine_fuzz_on_device():
    """Seeded engine-level fuzz fully on device: random aggregator x
    attack combinations through the real ParameterServer + stream actors.
    Output must be finite and bounded by the honest scale (the robust
    aggregators' whole contract) for every pairing."""
    from byzpy_amd.aggregators import (
        CAF,
        CoordinateWiseMedian,
        ComparativeGradientElimination,
        GeometricMedian,
        MeanOfMedians,
    )
    from byzpy_amd.attacks import (
        EmpireAttack,
        GaussianAttack,
        InfAttack,
        LittleAttack,
        SignFlipAttack,
    )

    aggs = [
        CoordinateWiseMedian(),
        CoordinateWiseTrimmedMean(f=2),
        MultiKrum(f=2, q=3),
        MeanOfMedians(f=2),
        ComparativeGradientElimination(f=2),
        GeometricMedian(max_iter=32),
        CAF(f=2),
    ]
    attacks = [
        SignFlipAttack(),
        EmpireAttack(scale=-8.0),
        LittleAttack(f=2, N=9),
        GaussianAttack(0.0, 30.0),
        InfAttack(),
    ]

    async def one(agg, atk, seed):
        honest = [
            await HonestNodeActor.spawn(GpuRandHonest, seed * 31 + i, backend="stream:0")
            for i in range(7)
        ]
        byz = [
            await ByzantineNodeActor.spawn(GpuRandByz, atk, backend="stream:0")
            for _ in range(2)
        ]
        ps = ParameterServer(honest, byz, agg)
        update = await ps.round()
        assert update.is_cuda, (agg, atk)
        uf = update.float()
        if not isinstance(atk, InfAttack) or not isinstance(
            agg, GeometricMedian
        ):
            assert torch.isfinite(uf).all(), (agg, atk)
            assert float(uf.abs().max()) < 50.0, (agg, atk)
        for a in honest + byz:
            await a.close()

    async def main():
        for i, agg in enumerate(aggs):
            atk = attacks[i % len(attacks)]
            await one(agg, atk, i)
        # and the historically nastiest pairing: inf rows into order stats
        await one(CoordinateWiseTrimmedMean(f=2), InfAttack(), 99)
        await one(CoordinateWiseMedian(), InfAttack(), 100)

    asyncio.run(main())


def test_ps_long_run_no_leak_no_fade():
    """200 rounds through the PS on stream actors: device memory must be
    flat (no per-round allocation leak from the engine/actor plumbing)
    and late rounds must not be slower than early ones (no queue/handle
    accumulation)."""
    import time

    async def main():
        honest = [
            await HonestNodeActor.spawn(GpuRandHonest, i, backend="stream:0")
            for i in range(6)
        ]
        byz = [await ByzantineNodeActor.spawn(GpuByz, 8192, backend="stream:0")]
        ps = ParameterServer(honest, byz, CoordinateWiseTrimmedMean(f=1))
        for _ in range(10):  # warm the caches/allocator
            await ps.round()
        torch.cuda.synchronize()
        mem0 = torch.cuda.memory_allocated()
        t0 = time.perf_counter()
        for _ in range(95):
            await ps.round()
        torch.cuda.synchronize()
        first = time.perf_counter() - t0
        mem1 = torch.cuda.memory_allocated()
        t0 = time.perf_counter()
        for _ in range(95):
            await ps.round()
        torch.cuda.synchronize()
        second = time.perf_counter() - t0
        mem2 = torch.cuda.memory_allocated()
        assert mem1 == mem0 and mem2 == mem0, (mem0, mem1, mem2)
        assert second < first * 1.5, (first, second)
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_p2p_long_run_no_leak():
    """100 gossip rounds fully on device: flat memory, finite models —
    the P2P twin of the PS soak above."""
    import torch.nn as nn

    from byzpy_amd.aggregators import CoordinateWiseMedian
    from byzpy_amd.engine.peer_to_peer.mixin import (
        P2PByzantineMixin,
        P2PHonestMixin,
    )
    from byzpy_amd.engine.peer_to_peer.train import PeerToPeer

    class DevHonest2(P2PHonestMixin):
        def __init__(self, seed):
            torch.manual_seed(seed)
            self.model = nn.Linear(64, 1, bias=False).cuda()
            self.lr = 0.02
            g = torch.Generator().manual_seed(seed)
            self.x = torch.randn(32, 64, generator=g).cuda()
            self.y = self.x.sum(dim=1, keepdim=True) * 0.1

        def p2p_local_loss_backward(self):
            loss = ((self.model(self.x) - self.y) ** 2).mean()
            loss.backward()

    class DevByz2(P2PByzantineMixin):
        def __init__(self):
            from byzpy_amd.attacks import SignFlipAttack

            self.attack = SignFlipAttack()

    async def main():
        honest = [DevHonest2(s) for s in (1, 2, 3, 4)]
        byz = [DevByz2()]
        p2p = PeerToPeer(honest, byz, CoordinateWiseMedian(), lr=0.02)
        for _ in range(10):
            await p2p.round()
        torch.cuda.synchronize()
        mem0 = torch.cuda.memory_allocated()
        for _ in range(90):
            await p2p.round()
        torch.cuda.synchronize()
        assert torch.cuda.memory_allocated() == mem0
        for h in honest:
            assert torch.isfinite(h.model.weight).all()
        await p2p.shutdown()

    asyncio.run(main())
