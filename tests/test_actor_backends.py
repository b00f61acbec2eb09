"""Real-backend integration tests (SURVEY.md §4 pattern 4): thread, process
and loopback-TCP actors driven end-to-end."""
import asyncio

import numpy as np
import pytest
import torch

from byzpy_amd.actor.base import ActorRef
from byzpy_amd.actor.backends.process import ProcessActorBackend
from byzpy_amd.actor.backends.remote import RemoteActorBackend, RemoteActorServer
from byzpy_amd.actor.backends.stream import StreamActorBackend
from byzpy_amd.actor.backends.thread import ThreadActorBackend
from byzpy_amd.actor.channels import open_channel
from byzpy_amd.actor.ipc import unwrap_payload, wrap_payload


class Counter:
    def __init__(self, start=0):
        self.value = start

    def incr(self, by=1):
        self.value += by
        return self.value

    def tensor_double(self, t):
        return t * 2


def test_thread_actor_construct_call():
    async def main():
        b = ThreadActorBackend()
        await b.start()
        await b.construct(Counter, 10)
        ref = ActorRef(b)
        assert await ref.incr() == 11
        assert await ref.incr(by=5) == 16
        await b.close()

    asyncio.run(main())


def test_process_actor_tensor_roundtrip():
    async def main():
        b = ProcessActorBackend()
        await b.start()
        await b.construct(Counter)
        ref = ActorRef(b)
        t = torch.arange(6, dtype=torch.float32)
        out = await ref.tensor_double(t)
        assert torch.allclose(out, t * 2)
        await b.close()

    asyncio.run(main())


def test_stream_actor_cpu_degraded():
    async def main():
        b = StreamActorBackend()
        await b.start()
        await b.construct(Counter, 1)
        ref = ActorRef(b)
        assert await ref.incr() == 2
        await b.close()

    asyncio.run(main())


def test_thread_channels():
    async def main():
        a, b = ThreadActorBackend(), ThreadActorBackend()
        await a.start()
        await b.start()
        chan_b = await open_channel(b, "inbox")
        chan_a = await open_channel(a, "inbox")
        await chan_a.send(b.get_endpoint(), {"x": 1})
        assert await chan_b.recv() == {"x": 1}
        await a.close()
        await b.close()

    asyncio.run(main())


def test_ipc_wrap_unwrap():
    t = torch.randn(4, 3)
    arr = np.ones((2, 2), dtype=np.float64)
    payload = {"a": [t, arr], "b": 5}
    wrapped = wrap_payload(payload)
    out = unwrap_payload(wrapped)
    assert torch.allclose(out["a"][0], t)
    assert (out["a"][1] == arr).all()
    assert out["b"] == 5


def test_ipc_refuses_cuda_wrap():
    class FakeCuda:
        pass

    # only meaningful with a GPU; on CPU verify the tensor path works
    t = torch.randn(2)
    assert torch.allclose(unwrap_payload(wrap_payload(t)), t)


def test_remote_tcp_actor_loopback():
    async def main():
        server = RemoteActorServer("127.0.0.1", 0)
        await server.start()
        try:
            client = RemoteActorBackend("127.0.0.1", server.port)
            await client.start()
            await client.construct(Counter, 100)
            ref = ActorRef(client)
            assert await ref.incr() == 101
            # channels via the same server
            await client.chan_open("inbox")
            await client.chan_put(client.get_endpoint(), "inbox", {"m": 1})
            assert await client.chan_get("inbox") == {"m": 1}
            await client.close()
        finally:
            await server.stop()

    asyncio.run(main())


def test_process_chan_get_before_put_no_deadlock():
    """Regression: a chan_get on an empty mailbox must not wedge the child's
    request loop — the deliver that fills it arrives over the same pipe."""
    import asyncio

    from byzpy_amd.actor.backends.process import ProcessActorBackend

    class Obj:
        pass

    async def main():
        be = ProcessActorBackend()
        await be.start()
        await be.construct(Obj)
        await be.chan_open("m")

        async def getter():
            return await be.chan_get("m")

        task = asyncio.create_task(getter())
        await asyncio.sleep(0.2)  # getter is polling an empty mailbox
        await be.chan_put(be.get_endpoint(), "m", {"x": 7})
        out = await asyncio.wait_for(task, timeout=10)
        await be.close()
        return out

    out = asyncio.run(main())
    assert out["x"] == 7


def test_actor_spawn_close_churn_no_leak():
    """Spawn and close many actors: thread/process handles, queues and
    tensor references must not accumulate across lifecycles."""
    import gc
    import threading

    import torch

    from byzpy_amd.engine.node.actors import HonestNodeActor

    class W:
        def __init__(self):
            self.g = torch.ones(128)

        def honest_gradient_for_next_batch(self):
            return self.g

        def apply_server_gradient(self, g):
            pass

    async def churn(n):
        for _ in range(n):
            a = await HonestNodeActor.spawn(W, backend="thread")
            out = await a.honest_gradient_for_next_batch()
            assert out.shape == (128,)
            await a.close()

    asyncio.run(churn(5))
    gc.collect()
    threads0 = threading.active_count()
    c0 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    asyncio.run(churn(25))
    gc.collect()
    threads1 = threading.active_count()
    c1 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    assert threads1 <= threads0 + 1, (threads0, threads1)
    assert c1 <= c0 + 4, (c0, c1)


def test_process_actor_churn_no_zombies():
    """Spawn/close process-backend actors repeatedly: child processes
    must actually exit (no zombie/orphan accumulation)."""
    import multiprocessing as mp

    import torch

    from byzpy_amd.engine.node.actors import HonestNodeActor

    class W:
        def __init__(self):
            self.g = torch.ones(64)

        def honest_gradient_for_next_batch(self):
            return self.g

        def apply_server_gradient(self, g):
            pass

    async def churn(n):
        for _ in range(n):
            a = await HonestNodeActor.spawn(W, backend="process")
            out = await a.honest_gradient_for_next_batch()
            assert out.shape == (64,)
            await a.close()

    asyncio.run(churn(2))
    base = len(mp.active_children())
    asyncio.run(churn(6))
    leftover = len(mp.active_children())
    assert leftover <= base, (base, leftover)
