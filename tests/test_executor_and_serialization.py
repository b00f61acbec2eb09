"""Executor key auto-detection, attacks/pre-aggs via run_operator,
operator serialization round-trips (SURVEY.md §4 patterns 1/9)."""
import asyncio

import cloudpickle
import pytest
import torch

from byzpy_amd import OperatorExecutor, run_operator
from byzpy_amd.aggregators import (
    CenteredClipping,
    CoordinateWiseMedian,
    GeometricMedian,
    MultiKrum,
)
from byzpy_amd.attacks import EmpireAttack, LittleAttack
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.ops import functional as F
from byzpy_amd.pre_aggregators import Clipping, NearestNeighborMixing


def _grads(n=8, d=33, seed=0):
    g = torch.Generator().manual_seed(seed)
    return list(torch.randn(n, d, generator=g))


def test_run_operator_no_pool():
    out = asyncio.run(run_operator(CoordinateWiseMedian(), {"gradients": _grads()}))
    assert out.shape == (33,)


def test_run_operator_key_remap():
    # user passes a custom key; executor maps it onto the operator's
    grads = _grads()
    out = asyncio.run(run_operator(CoordinateWiseMedian(), {"my_grads": grads}))
    assert torch.allclose(out, F.median(torch.stack(grads)), atol=1e-6)


def test_run_operator_preagg():
    vecs = _grads()
    out = asyncio.run(run_operator(Clipping(1.0), {"vectors": vecs}))
    assert len(out) == len(vecs)
    assert all(v.norm() <= 1.0 + 1e-4 for v in out)


def test_run_operator_attack_single_key():
    grads = _grads()
    out = asyncio.run(run_operator(EmpireAttack(scale=-1.0), {"honest_grads": grads}))
    assert torch.allclose(out, -torch.stack(grads).mean(dim=0), atol=1e-5)


def test_executor_reuses_pool_and_graph():
    async def main():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=2))
        await pool.start()
        ex = OperatorExecutor(CoordinateWiseMedian(chunk_size=8), pool=pool)
        grads = _grads()
        a = await ex.run({"gradients": grads})
        b = await ex.run({"gradients": grads})
        assert torch.allclose(a, b)
        # owned pool stays open (executor doesn't own it)
        out = await pool.run_subtask(
            __import__("byzpy_amd.graph.subtask", fromlist=["SubTask"]).SubTask(
                fn=len, args=([1, 2],)
            )
        )
        assert out == 2
        await pool.close()

    asyncio.run(main())


@pytest.mark.parametrize(
    "op",
    [
        CoordinateWiseMedian(chunk_size=16),
        MultiKrum(2, 3),
        GeometricMedian(max_iter=50),
        CenteredClipping(c_tau=0.5),
        Clipping(1.0),
        NearestNeighborMixing(2),
        EmpireAttack(),
        LittleAttack(2),
    ],
    ids=lambda o: o.name,
)
def test_operator_cloudpickle_roundtrip(op):
    blob = cloudpickle.dumps(op)
    clone = cloudpickle.loads(blob)
    grads = _grads(10, 17)
    if hasattr(op, "aggregate"):
        a, b = op.aggregate(grads), clone.aggregate(grads)
        assert torch.allclose(a, b, atol=1e-5)
    elif hasattr(op, "pre_aggregate"):
        a, b = op.pre_aggregate(grads), clone.pre_aggregate(grads)
        for x, y in zip(a, b):
            assert torch.allclose(x, y, atol=1e-5)
    else:
        a = op.apply(honest_grads=grads)
        b = clone.apply(honest_grads=grads)
        assert torch.allclose(a, b, atol=1e-5)


def test_numpy_gradients_roundtrip():
    import numpy as np

    grads = [np.random.RandomState(i).randn(12).astype(np.float32) for i in range(6)]
    out = CoordinateWiseMedian().aggregate(grads)
    assert isinstance(out, np.ndarray) and out.shape == (12,)


def test_param_list_gradients_roundtrip():
    gen = torch.Generator().manual_seed(0)
    grads = [
        [torch.randn(3, 4, generator=gen), torch.randn(5, generator=gen)]
        for _ in range(5)
    ]
    out = CoordinateWiseMedian().aggregate(grads)
    assert isinstance(out, list) and out[0].shape == (3, 4) and out[1].shape == (5,)


def test_shared_handle_gradients():
    from byzpy_amd.storage.shared_store import cleanup_tensor, register_tensor

    g = torch.Generator().manual_seed(3)
    tensors = [torch.randn(9, generator=g) for _ in range(5)]
    handles = [register_tensor(t) for t in tensors]
    try:
        out = CoordinateWiseMedian().aggregate(handles)
        ref = F.median(torch.stack(tensors))
        assert torch.allclose(torch.as_tensor(out).reshape(-1), ref, atol=1e-6)
    finally:
        for h in handles:
            cleanup_tensor(h)


class TestAttackExecutorSweep:
    """Every attack routed through run_operator with explicit input keys."""

    @staticmethod
    def _honest(n=6, d=9, seed=2):
        g = torch.Generator().manual_seed(seed)
        return [torch.randn(d, generator=g) for _ in range(n)]

    @pytest.mark.parametrize(
        "atk_name",
        ["empire", "little", "gaussian", "inf", "mimic"],
    )
    def test_honest_grads_attacks(self, atk_name):
        from byzpy_amd.attacks import (
            EmpireAttack,
            GaussianAttack,
            InfAttack,
            LittleAttack,
            MimicAttack,
        )

        atk = {
            "empire": EmpireAttack(scale=-2.0),
            "little": LittleAttack(f=1),
            "gaussian": GaussianAttack(seed=5),
            "inf": InfAttack(),
            "mimic": MimicAttack(epsilon=2),
        }[atk_name]
        honest = self._honest()
        direct = atk.apply(honest_grads=honest)

        async def run():
            return await run_operator(atk, {"honest_grads": honest})

        out = asyncio.run(run())
        assert out.shape == direct.shape
        if atk_name != "gaussian":  # gaussian reseeds per call? seeded: equal
            assert torch.allclose(out, direct, equal_nan=True, atol=1e-5)

    def test_sign_flip_via_executor(self):
        from byzpy_amd.attacks import SignFlipAttack

        base = torch.arange(5.0)

        async def run():
            return await run_operator(
                SignFlipAttack(scale=-1.5), {"base_grad": base}
            )

        out = asyncio.run(run())
        assert torch.allclose(out, -1.5 * base)


def test_run_operator_repeated_no_retention():
    """200 run_operator calls through the cached executor/pool path:
    tensor census flat (the graph/scheduler caches must be bounded)."""
    import gc

    import torch

    from byzpy_amd import run_operator
    from byzpy_amd.aggregators import CoordinateWiseMedian

    agg = CoordinateWiseMedian()
    X = [torch.randn(4, 64) for _ in range(3)]

    def once():
        async def batch():
            for i in range(10):
                out = await run_operator(agg, {"gradients": list(X[i % 3])})
                assert out.shape == (64,)

        asyncio.run(batch())

    once()
    gc.collect()
    c0 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    for _ in range(19):
        once()
    gc.collect()
    c1 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    assert c1 <= c0 + 8, (c0, c1)
