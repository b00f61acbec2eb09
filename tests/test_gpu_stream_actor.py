"""Stream actor backend on a real GPU: HIP-stream workers, device-resident
state, event-ordered zero-copy mailboxes."""
import asyncio

import pytest
import torch

pytestmark = pytest.mark.gpu

from byzpy_amd.actor.backends.stream import StreamActorBackend
from byzpy_amd.actor.base import ActorRef
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.graph.subtask import SubTask


class DeviceWorker:
    """Holds device state; methods run under the worker's HIP stream."""

    def __init__(self, d: int):
        self.state = torch.zeros(d, device="cuda")

    def accumulate(self, t):
        self.state += t
        return None

    def norm(self):
        return float(self.state.norm())

    def make(self, scale: float):
        return torch.full((4,), scale, device="cuda")


def test_stream_actor_device_state():
    async def main():
        b = StreamActorBackend(device=0)
        await b.start()
        assert b.stream is not None
        await b.construct(DeviceWorker, 1024)
        ref = ActorRef(b)
        t = torch.ones(1024, device="cuda")
        await ref.accumulate(t)
        await ref.accumulate(t)
        assert abs(await ref.norm() - (4 * 1024) ** 0.5) < 1e-3
        await b.close()

    asyncio.run(main())


def test_stream_channels_event_ordering():
    async def main():
        a, b = StreamActorBackend(device=0), StreamActorBackend(device=0)
        await a.start()
        await b.start()
        await a.construct(DeviceWorker, 8)
        await b.construct(DeviceWorker, 8)
        await b.chan_open("inbox")
        payload = torch.full((1024,), 3.0, device="cuda")
        await a.chan_put(b.get_endpoint(), "inbox", payload)
        got = await b.chan_get("inbox")
        assert got.is_cuda and torch.allclose(got, payload)
        await a.close()
        await b.close()

    asyncio.run(main())


def test_stream_pool_runs_device_subtasks():
    def work(scale):
        x = torch.full((4096,), scale, device="cuda")
        return float(x.sum())

    async def main():
        pool = ActorPool(ActorPoolConfig(backend="stream:0", count=2, name="gpu"))
        await pool.start()
        outs = await asyncio.gather(
            *(pool.run_subtask(SubTask(fn=work, args=(float(i),))) for i in range(8))
        )
        assert outs == [4096.0 * i for i in range(8)]
        # capability routing: these workers advertise the gpu capability
        out = await pool.run_subtask(SubTask(fn=work, args=(2.0,), affinity="gpu"))
        assert out == 8192.0
        await pool.close()

    asyncio.run(main())


# -- round-2 depth: windowed in-flight limit as stream depth ---------------

_INFLIGHT = {"cur": 0, "max": 0}


def _probe_fn(i):
    """Tracks concurrent executions (stream workers are in-process, so the
    module global is shared)."""
    import threading
    import time as _time

    lock = getattr(_probe_fn, "_lock", None)
    if lock is None:
        lock = threading.Lock()
        _probe_fn._lock = lock
    with lock:
        _INFLIGHT["cur"] += 1
        _INFLIGHT["max"] = max(_INFLIGHT["max"], _INFLIGHT["cur"])
    x = torch.full((2048,), float(i), device="cuda")
    s = float(x.sum())
    _time.sleep(0.02)
    with lock:
        _INFLIGHT["cur"] -= 1
    return s


def test_windowed_inflight_limit_is_stream_depth():
    """graph/pool.py promises the operator's windowed in-flight limit acts
    as stream depth for stream workers: with max_subtasks_inflight=-1 on a
    2-worker stream pool, at most 2 subtasks may ever run concurrently."""
    from byzpy_amd.ops.base import Operator, OpContext

    class ProbeOp(Operator):
        name = "probe"
        input_key = "xs"
        supports_subtasks = True
        max_subtasks_inflight = -1  # pool.size * 1

        def create_subtasks(self, ctx, **inputs):
            return [SubTask(fn=_probe_fn, args=(i,)) for i in range(10)]

        def reduce_subtasks(self, ctx, results, **inputs):
            return results

    async def main():
        _INFLIGHT["cur"] = 0
        _INFLIGHT["max"] = 0
        pool = ActorPool(ActorPoolConfig(backend="stream:0", count=2, name="g"))
        await pool.start()
        ctx = OpContext(pool=pool, metadata={"pool_size": 2})
        out = await ProbeOp().run(ctx, xs=None)
        assert out == [2048.0 * i for i in range(10)]
        assert _INFLIGHT["max"] <= 2, _INFLIGHT["max"]
        await pool.close()

    asyncio.run(main())


def test_stream_pool_runs_hip_kernel_subtasks():
    """SubTasks that launch the HIP kernels themselves (D.median / D.gram)
    through stream workers — the kernel path under the pool scheduler."""
    from byzpy_amd.hip import dispatch as D

    def kmedian(seed):
        g = torch.Generator().manual_seed(seed)
        X = torch.randn(16, 4096, generator=g).to("cuda", torch.bfloat16)
        out = D.median(X)
        from byzpy_amd.ops import functional as F

        ref = F.median(X.float().cpu())
        assert torch.allclose(out.float().cpu(), ref, atol=5e-2, rtol=5e-2)
        return float(out.float().sum())

    def kgram(seed):
        g = torch.Generator().manual_seed(seed)
        X = torch.randn(8, 8192, generator=g).to("cuda", torch.bfloat16)
        G = D.gram(X)
        Xf = X.float()
        assert (G - Xf @ Xf.T).abs().max().item() < 0.5
        return float(G.trace())

    async def main():
        pool = ActorPool(ActorPoolConfig(backend="stream:0", count=3, name="k"))
        await pool.start()
        outs = await asyncio.gather(
            *(pool.run_subtask(SubTask(fn=kmedian, args=(s,))) for s in range(4)),
            *(pool.run_subtask(SubTask(fn=kgram, args=(s,))) for s in range(4)),
        )
        assert len(outs) == 8 and all(isinstance(o, float) for o in outs)
        await pool.close()

    asyncio.run(main())
