"""Chained pre-aggregator -> aggregator graphs (every combination) must
match the manual composition — the reference's core usage pattern."""
import asyncio

import pytest
import torch

from byzpy_amd.aggregators import (
    CoordinateWiseMedian,
    CoordinateWiseTrimmedMean,
    GeometricMedian,
    MultiKrum,
)
from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.lazy import GraphBuilder
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.graph.scheduler import NodeScheduler
from byzpy_amd.pre_aggregators import ARC, Bucketing, Clipping, NearestNeighborMixing


def _grads(n=12, d=65, seed=4):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(d, generator=g) for _ in range(n)]


PRES = [
    Clipping(1.5),
    Bucketing(3, perm=list(range(12))),
    NearestNeighborMixing(2),
    ARC(2),
]
AGGS = [
    CoordinateWiseMedian(),
    CoordinateWiseTrimmedMean(1),
    MultiKrum(1, 2),
]


@pytest.mark.parametrize("pre", PRES, ids=[p.name for p in PRES])
@pytest.mark.parametrize("agg", AGGS, ids=[a.name for a in AGGS])
def test_chained_graph_matches_manual(pre, agg):
    grads = _grads()
    manual = agg.aggregate(pre.pre_aggregate(grads))

    g = ComputationGraph(
        [
            GraphNode("pre", pre, {"vectors": GraphInput("g")}),
            GraphNode("agg", agg, {"gradients": "pre"}),
        ]
    )

    async def run():
        return await NodeScheduler(g).run({"g": grads})

    out = asyncio.run(run())
    assert torch.allclose(out, manual, atol=1e-4), f"{pre.name}->{agg.name}"


@pytest.mark.parametrize("pre", PRES[:2], ids=[p.name for p in PRES[:2]])
def test_chained_graph_pooled(pre):
    grads = _grads()
    agg = CoordinateWiseMedian(chunk_size=16)
    manual = agg.aggregate(pre.pre_aggregate(grads))

    async def run():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=3))
        await pool.start()
        g = ComputationGraph(
            [
                GraphNode("pre", pre, {"vectors": GraphInput("g")}),
                GraphNode("agg", agg, {"gradients": "pre"}),
            ]
        )
        out = await NodeScheduler(g, pool=pool).run({"g": grads})
        await pool.close()
        return out

    out = asyncio.run(run())
    assert torch.allclose(out, manual, atol=1e-4)


def test_lazy_builder_chain():
    grads = _grads()
    pre, agg = Clipping(1.0), CoordinateWiseMedian()
    b = GraphBuilder()
    node = b.input("g").apply(pre).apply(agg, input_key="gradients")
    graph = b.build([node.name])

    async def run():
        return await NodeScheduler(graph).run({"g": grads})

    out = asyncio.run(run())
    manual = agg.aggregate(pre.pre_aggregate(grads))
    assert torch.allclose(out, manual, atol=1e-5)


def test_diamond_graph_two_aggregates():
    """One pre-agg feeding two different aggregators (multi-output)."""
    grads = _grads()
    pre = Clipping(1.0)
    med, tm = CoordinateWiseMedian(), CoordinateWiseTrimmedMean(1)
    g = ComputationGraph(
        [
            GraphNode("pre", pre, {"vectors": GraphInput("g")}),
            GraphNode("median", med, {"gradients": "pre"}),
            GraphNode("trimmed", tm, {"gradients": "pre"}),
        ],
        outputs=["median", "trimmed"],
    )

    async def run():
        return await NodeScheduler(g).run({"g": grads})

    out = asyncio.run(run())
    clipped = pre.pre_aggregate(grads)
    assert torch.allclose(out["median"], med.aggregate(clipped), atol=1e-5)
    assert torch.allclose(out["trimmed"], tm.aggregate(clipped), atol=1e-5)
