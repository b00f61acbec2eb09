"""End-to-end Byzantine-resilience: robust aggregation protects training
where plain averaging is destroyed (the framework's core claim)."""
import asyncio

import pytest
import torch
from torch import nn

from byzpy_amd.aggregators import (
    CoordinateWiseMedian,
    CoordinateWiseTrimmedMean,
    GeometricMedian,
    MultiKrum,
)
from byzpy_amd.engine.parameter_server.ps import ParameterServer
from byzpy_amd.ops.base import Operator, OpContext


class MeanAggregator(Operator):
    """Non-robust baseline (what the attack defeats)."""

    name = "mean"
    input_key = "gradients"

    def aggregate(self, gradients):
        return torch.stack(list(gradients)).mean(dim=0)

    def compute(self, ctx, **inputs):
        return self.aggregate(inputs[self.input_key])


class LinearWorker:
    def __init__(self, seed: int):
        torch.manual_seed(0)
        self.model = nn.Linear(6, 1, bias=False)
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(32, 6, generator=g)
        self.w_true = torch.ones(1, 6)
        self.y = self.x @ self.w_true.T
        self.lr = 0.1

    async def honest_gradient_for_next_batch(self):
        self.model.zero_grad(set_to_none=True)
        loss = ((self.model(self.x) - self.y) ** 2).mean()
        loss.backward()
        return self.model.weight.grad.reshape(-1).clone()

    async def apply_server_gradient(self, g):
        with torch.no_grad():
            self.model.weight.add_(g.reshape(self.model.weight.shape), alpha=-self.lr)

    def error(self):
        return float((self.model.weight.detach() - self.w_true).norm())


class Saboteur:
    """Sends a huge anti-gradient."""

    async def byzantine_gradient_for_next_batch(self, honest_grads=None):
        base = torch.stack([g for g in honest_grads]).mean(dim=0)
        return -50.0 * base + 100.0

    async def apply_server_gradient(self, g):
        pass


def _train(aggregator, rounds=30, n_honest=6, n_byz=2):
    async def main():
        honest = [LinearWorker(s) for s in range(n_honest)]
        byz = [Saboteur() for _ in range(n_byz)]
        ps = ParameterServer(honest, byz, aggregator)
        for _ in range(rounds):
            await ps.round()
        return honest[0].error()

    return asyncio.run(main())


@pytest.mark.parametrize(
    "agg",
    [
        CoordinateWiseMedian(),
        CoordinateWiseTrimmedMean(2),
        MultiKrum(2, 3),
        GeometricMedian(max_iter=100),
    ],
    ids=lambda a: a.name,
)
def test_robust_aggregator_survives_attack(agg):
    err = _train(agg)
    assert err < 0.3, f"{agg.name} failed to converge under attack: err={err}"


def test_plain_mean_is_destroyed():
    err = _train(MeanAggregator(), rounds=10)
    assert err > 1.0, f"attack unexpectedly harmless to plain mean: err={err}"


def test_ps_random_config_sweep():
    """Seeded mini-sweep of PS configurations (aggregator x pre-agg x
    attack): every combination must produce a finite, correctly-shaped
    aggregate. (A 25-config version of this sweep runs ad hoc; this seeded
    5-config subset guards regressions.)"""
    import asyncio
    import random

    import torch

    from byzpy_amd.aggregators import (
        CenteredClipping,
        CoordinateWiseMedian,
        GeometricMedian,
        MeanOfMedians,
        MultiKrum,
    )
    from byzpy_amd.attacks import EmpireAttack, GaussianAttack, SignFlipAttack
    from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor
    from byzpy_amd.engine.node.base import ByzantineNode, HonestNode
    from byzpy_amd.engine.parameter_server.ps import ParameterServer
    from byzpy_amd.pre_aggregators import ARC, Clipping

    class H(HonestNode):
        def __init__(self, seed, d):
            g = torch.Generator().manual_seed(seed)
            self.g0 = torch.randn(d, generator=g)

        def next_batch(self):
            return None, None

        def honest_gradient(self, x, y):
            return self.g0

        def apply_server_gradient(self, grad):
            self.last = grad

    class B(ByzantineNode):
        def __init__(self, atk):
            self.atk = atk

        def next_batch(self):
            return None, None

        def byzantine_gradient(self, x, y, honest_grads=None):
            if self.atk.uses_honest_grads:
                return self.atk.apply(honest_grads=honest_grads)
            return self.atk.apply(base_grad=honest_grads[0])

        def apply_server_gradient(self, grad):
            pass

    rng = random.Random(3)

    async def run_one(i):
        d = rng.randint(8, 64)
        agg = rng.choice(
            [
                CoordinateWiseMedian(),
                MeanOfMedians(1),
                MultiKrum(1, 2),
                GeometricMedian(),
                CenteredClipping(c_tau=1.0),
            ]
        )
        pre = rng.choice([None, Clipping(5.0), ARC(1)])
        atk = rng.choice([EmpireAttack(scale=-5.0), SignFlipAttack(), GaussianAttack(seed=1)])
        honest = [
            await HonestNodeActor.spawn(H, 10 + j, d, backend="thread")
            for j in range(4)
        ]
        byz = [await ByzantineNodeActor.spawn(B, atk, backend="thread")]
        ps = ParameterServer(honest, byz, agg, pre_aggregator=pre)
        try:
            out = await ps.round()
            assert out.shape == (d,)
            assert torch.isfinite(out).all()
        finally:
            for nd in honest + byz:
                await nd.close()

    async def main():
        for i in range(5):
            await run_one(i)

    asyncio.run(main())
