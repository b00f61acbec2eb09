"""ParameterServer engine tests with stub nodes (SURVEY.md §4 pattern 6c)."""
import asyncio

import pytest
import torch

from byzpy_amd.aggregators import CoordinateWiseMedian, CoordinateWiseTrimmedMean
from byzpy_amd.attacks import SignFlipAttack
from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor
from byzpy_amd.engine.parameter_server.ps import ParameterServer
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.utils.training import train_with_progress


class StubHonest:
    """Canned-gradient honest node (picklable for actor construct)."""

    def __init__(self, value: float, d: int = 8):
        self.value = value
        self.d = d
        self.applied = []

    def honest_gradient_for_next_batch(self):
        return torch.full((self.d,), self.value)

    def apply_server_gradient(self, g):
        self.applied.append(g)
        return None

    def applied_count(self):
        return len(self.applied)


class StubByz:
    def __init__(self, d: int = 8):
        self.d = d

    def byzantine_gradient_for_next_batch(self, honest_grads=None):
        return torch.full((self.d,), 1e6)

    def apply_server_gradient(self, g):
        return None


def test_ps_round_median_robust():
    async def main():
        honest = [await HonestNodeActor.spawn(StubHonest, v) for v in (1.0, 2.0, 3.0)]
        byz = [await ByzantineNodeActor.spawn(StubByz)]
        ps = ParameterServer(honest, byz, CoordinateWiseMedian())
        update = await ps.round()
        # median of {1,2,3,1e6} = 2.5; outlier rejected
        assert torch.allclose(update, torch.full((8,), 2.5))
        # fan-out reached every honest node
        assert await honest[0].applied_count() == 1
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_ps_round_with_pool_and_preagg():
    from byzpy_amd.pre_aggregators import Clipping

    async def main():
        honest = [await HonestNodeActor.spawn(StubHonest, v) for v in (0.1, 0.2, 0.3)]
        byz = [await ByzantineNodeActor.spawn(StubByz)]
        pool = ActorPool(ActorPoolConfig(backend="thread", count=2))
        await pool.start()
        ps = ParameterServer(
            honest,
            byz,
            CoordinateWiseTrimmedMean(1, chunk_size=4),
            pre_aggregator=Clipping(1.0),
            pool=pool,
        )
        update = await ps.round()
        assert update.shape == (8,)
        assert update.abs().max() < 1.0  # clipped + trimmed
        await pool.close()
        for a in honest + byz:
            await a.close()

    asyncio.run(main())


def test_train_with_progress():
    async def main():
        honest = [await HonestNodeActor.spawn(StubHonest, 1.0)]
        ps = ParameterServer(honest, [], CoordinateWiseMedian())
        await train_with_progress(ps, 3, progress=False)
        assert await honest[0].applied_count() == 3
        await honest[0].close()

    asyncio.run(main())
