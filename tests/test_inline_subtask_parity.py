"""SURVEY §4 pattern 2, exact form: call create_subtasks with a synthetic
OpContext, run every subtask FN INLINE (no pool at all), reduce, and
compare against the direct aggregate. Isolates the chunk decomposition
from pool scheduling."""
import pytest
import torch

from byzpy_amd.aggregators import (
    CenteredClipping,
    ComparativeGradientElimination,
    CoordinateWiseMedian,
    CoordinateWiseTrimmedMean,
    GeometricMedian,
    MeanOfMedians,
    MinimumDiameterAveraging,
    MoNNA,
    MultiKrum,
    SMEA,
)
from byzpy_amd.ops.base import OpContext
from byzpy_amd.pre_aggregators import ARC, Clipping, NearestNeighborMixing


class _InlinePool:
    """Just enough pool surface for create_subtasks sizing decisions."""

    size = 3
    prefers_shared_memory = False
    worker_affinities = ["worker::a", "worker::b", "worker::c"]


def _grads(n=11, d=201, seed=8):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(d, generator=g) for _ in range(n)]


AGGS = [
    CoordinateWiseMedian(chunk_size=32),
    CoordinateWiseTrimmedMean(2, chunk_size=32),
    MeanOfMedians(2, chunk_size=32),
    MultiKrum(2, 3, chunk_size=4),
    MoNNA(2, chunk_size=4),
    SMEA(7, chunk_size=64),
    MinimumDiameterAveraging(2, chunk_size=64),
    ComparativeGradientElimination(2, chunk_size=32),
]


@pytest.mark.parametrize("agg", AGGS, ids=[a.name for a in AGGS])
def test_inline_chunk_decomposition_matches_direct(agg):
    grads = _grads()
    direct = agg.aggregate(grads)
    ctx = OpContext(pool=_InlinePool())
    subtasks = list(agg.create_subtasks(ctx, gradients=grads))
    # MDA's seeded search can collapse to one anchor batch at tiny n
    min_tasks = 1 if agg.name == "minimum-diameter-averaging" else 2
    assert len(subtasks) >= min_tasks, f"{agg.name}: expected a fan-out"
    results = [st.fn(*st.args, **(st.kwargs or {})) for st in subtasks]
    out = agg.reduce_subtasks(ctx, results, gradients=grads)
    assert torch.allclose(out, direct, atol=1e-4), agg.name


@pytest.mark.parametrize(
    "pre",
    [Clipping(1.0, chunk_size=4), NearestNeighborMixing(2, feature_chunk_size=64),
     ARC(2, chunk_size=4)],
    ids=["clipping", "nnm", "arc"],
)
def test_inline_preagg_decomposition_matches_direct(pre):
    vecs = _grads()
    direct = pre.pre_aggregate(vecs)
    ctx = OpContext(pool=_InlinePool())
    subtasks = list(pre.create_subtasks(ctx, vectors=vecs))
    assert subtasks
    results = [st.fn(*st.args, **(st.kwargs or {})) for st in subtasks]
    out = pre.reduce_subtasks(ctx, results, vectors=vecs)
    assert len(out) == len(direct)
    for a, b in zip(out, direct):
        assert torch.allclose(a, b, atol=1e-4), pre.name


def test_inline_barriered_geomed():
    """Barriered operators (GeometricMedian) drive their own loop; the
    inline equivalent is run_barriered_subtasks with a stub pool that
    executes subtasks synchronously."""
    import asyncio

    class _SyncPool(_InlinePool):
        async def run_subtask(self, st):
            return st.fn(*st.args, **(st.kwargs or {}))

    agg = GeometricMedian(chunk_size=4)
    grads = _grads()
    direct = agg.aggregate(grads)
    ctx = OpContext(pool=_SyncPool())
    out = asyncio.run(agg.run_barriered_subtasks(ctx, gradients=grads))
    assert torch.allclose(out, direct, atol=1e-3)


def test_inline_barriered_cc():
    import asyncio

    class _SyncPool(_InlinePool):
        async def run_subtask(self, st):
            return st.fn(*st.args, **(st.kwargs or {}))

    agg = CenteredClipping(c_tau=0.6, chunk_size=4)
    grads = _grads()
    direct = agg.aggregate(grads)
    ctx = OpContext(pool=_SyncPool())
    out = asyncio.run(agg.run_barriered_subtasks(ctx, gradients=grads))
    assert torch.allclose(out, direct, atol=1e-3)
