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
