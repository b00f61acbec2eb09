"""ActorPool scheduling with a fake backend (SURVEY.md §4 pattern 3) and
real thread/process backends (pattern 4)."""
import asyncio

import pytest

from byzpy_amd.actor.channels import Endpoint
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig, _infer_capabilities
from byzpy_amd.graph.subtask import SubTask


class _FakeBackend:
    scheme = "fake"

    def __init__(self):
        self.obj = None
        self.calls = []
        self.mailboxes = {}

    async def start(self):
        pass

    async def construct(self, factory, *args, **kwargs):
        self.obj = factory(*args, **kwargs)

    async def call(self, method, *args, **kwargs):
        self.calls.append(method)
        return getattr(self.obj, method)(*args, **kwargs)

    async def close(self):
        pass

    def get_endpoint(self):
        return Endpoint(scheme="fake", address="local", actor_id=str(id(self)))

    async def chan_open(self, name):
        self.mailboxes.setdefault(name, asyncio.Queue())

    async def chan_put(self, endpoint, name, payload):
        self.mailboxes.setdefault(name, asyncio.Queue()).put_nowait(payload)

    async def chan_get(self, name):
        return await self.mailboxes.setdefault(name, asyncio.Queue()).get()


def _sq(x):
    return x * x


def _boom():
    raise RuntimeError("boom")


def test_capability_inference():
    assert _infer_capabilities("stream") == ("gpu",)
    assert _infer_capabilities("gpu:1") == ("gpu",)
    assert _infer_capabilities("thread") == ("cpu",)
    assert _infer_capabilities("process") == ("cpu",)


def test_fake_backend_pool_runs_subtasks():
    async def main():
        pool = ActorPool(ActorPoolConfig(backend=_FakeBackend(), count=1, name="fake"))
        # backend instances can't be replicated; use count=1 per config
        await pool.start()
        out = await pool.run_subtask(SubTask(fn=_sq, args=(7,)))
        assert out == 49
        await pool.close()

    asyncio.run(main())


def test_affinity_routing_and_mismatch():
    async def main():
        pool = ActorPool(
            [
                ActorPoolConfig(backend="thread", count=1, name="cpuw"),
                ActorPoolConfig(
                    backend="thread", count=1, name="special", capabilities=("cpu", "fast")
                ),
            ]
        )
        await pool.start()
        # route to the 'fast' worker by capability
        out = await pool.run_subtask(SubTask(fn=_sq, args=(3,), affinity="fast"))
        assert out == 9
        # per-worker label affinity
        label = pool.worker_affinities[0]
        out = await pool.run_subtask(SubTask(fn=_sq, args=(4,), affinity=label))
        assert out == 16
        with pytest.raises(RuntimeError, match="affinity"):
            await pool.run_subtask(SubTask(fn=_sq, args=(1,), affinity="nope"))
        await pool.close()

    asyncio.run(main())


def test_retry_exhaustion():
    async def main():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=1))
        await pool.start()
        with pytest.raises(RuntimeError, match="boom"):
            await pool.run_subtask(SubTask(fn=_boom, max_retries=2))
        await pool.close()

    asyncio.run(main())


def test_pool_channels_thread():
    async def main():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=2))
        await pool.start()
        chan = await pool.open_channel("c")
        await chan.send(0, 1, {"hello": 1})
        msg = await chan.recv(1)
        assert msg == {"hello": 1}
        await pool.close()

    asyncio.run(main())


def test_many_concurrent_subtasks():
    async def main():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=4))
        await pool.start()
        outs = await asyncio.gather(
            *(pool.run_subtask(SubTask(fn=_sq, args=(i,))) for i in range(50))
        )
        assert outs == [i * i for i in range(50)]
        await pool.close()

    asyncio.run(main())


def _pid():
    import os

    return os.getpid()


def test_process_worker_respawn_after_crash():
    """A killed worker process heals on retry (aux §5.3 fault recovery)."""

    async def main():
        pool = ActorPool(ActorPoolConfig(backend="process", count=1))
        await pool.start()
        pid1 = await pool.run_subtask(SubTask(fn=_pid))
        # kill the worker out from under the pool
        import os
        import signal

        os.kill(pid1, signal.SIGKILL)
        await asyncio.sleep(0.2)
        pid2 = await pool.run_subtask(SubTask(fn=_pid, max_retries=2))
        assert pid2 != pid1
        await pool.close()

    asyncio.run(main())
