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
