"""Graph/scheduler semantics (SURVEY.md §4 pattern 8)."""
import asyncio

import pytest
import torch

from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.lazy import GraphBuilder
from byzpy_amd.graph.ops import CallableOp
from byzpy_amd.graph.parallel_scheduler import ParallelScheduler
from byzpy_amd.graph.scheduler import MessageAwareNodeScheduler, NodeScheduler
from byzpy_amd.graph.session import ExecutionSession


def op(fn, name="f"):
    return CallableOp(fn, name=name)


def test_topo_order():
    g = ComputationGraph(
        [
            GraphNode("c", op(lambda b: b + 1), {"b": "b"}),
            GraphNode("a", op(lambda x: x * 2), {"x": GraphInput("x")}),
            GraphNode("b", op(lambda a: a + 3), {"a": "a"}),
        ]
    )
    assert g.topo_order.index("a") < g.topo_order.index("b") < g.topo_order.index("c")
    assert g.outputs == ["c"]


def test_cycle_detection():
    with pytest.raises(ValueError, match="cycle"):
        ComputationGraph(
            [
                GraphNode("a", op(lambda b: b), {"b": "b"}),
                GraphNode("b", op(lambda a: a), {"a": "a"}),
            ]
        )


def test_unknown_dep():
    with pytest.raises(ValueError, match="unknown"):
        ComputationGraph([GraphNode("a", op(lambda b: b), {"b": "nope"})])


def test_scheduler_runs_chain():
    g = ComputationGraph(
        [
            GraphNode("a", op(lambda x: x * 2), {"x": GraphInput("x")}),
            GraphNode("b", op(lambda a: a + 3), {"a": "a"}),
        ]
    )
    out = asyncio.run(NodeScheduler(g).run({"x": 5}))
    assert out == 13


def test_scheduler_missing_input():
    g = ComputationGraph([GraphNode("a", op(lambda x: x), {"x": GraphInput("x")})])
    with pytest.raises(KeyError):
        asyncio.run(NodeScheduler(g).run({}))


def test_parallel_scheduler_concurrency():
    order = []

    async def slow(tag, delay):
        order.append(f"start-{tag}")
        await asyncio.sleep(delay)
        order.append(f"end-{tag}")
        return tag

    class AsyncOp(CallableOp):
        async def run(self, ctx, **inputs):
            return await self.fn(**inputs)

    g = ComputationGraph(
        [
            GraphNode("a", AsyncOp(lambda x: slow("a", 0.05)), {"x": GraphInput("x")}),
            GraphNode("b", AsyncOp(lambda x: slow("b", 0.01)), {"x": GraphInput("x")}),
            GraphNode(
                "c", op(lambda a, b: f"{a}{b}"), {"a": "a", "b": "b"}
            ),
        ],
        outputs=["c"],
    )
    out = asyncio.run(ParallelScheduler(g).run({"x": 1}))
    assert out == "ab"
    # a and b overlapped
    assert order.index("start-b") < order.index("end-a")


def test_message_scheduler():
    async def main():
        g = ComputationGraph(
            [
                GraphNode(
                    "a",
                    op(lambda m: m["v"] * 10),
                    {"m": GraphInput.from_message("grad")},
                )
            ]
        )
        sched = MessageAwareNodeScheduler(g)
        task = asyncio.get_running_loop().create_task(sched.run({}))
        await asyncio.sleep(0.01)
        sched.deliver_message("grad", {"v": 4})
        return await task

    assert asyncio.run(main()) == 40


def test_message_cache_before_wait():
    async def main():
        g = ComputationGraph(
            [GraphNode("a", op(lambda m: m), {"m": GraphInput.from_message("t")})]
        )
        sched = MessageAwareNodeScheduler(g)
        sched.deliver_message("t", 123)
        return await sched.run({})

    assert asyncio.run(main()) == 123


def test_lazy_builder():
    b = GraphBuilder()
    node = b.input("x").apply(op(lambda v: v + 1), input_key="v", name="inc")
    node.apply(op(lambda v: v * 2), input_key="v", name="dbl")
    g = b.build(outputs=["dbl"])
    assert asyncio.run(NodeScheduler(g).run({"x": 4})) == 10


def test_session_caching():
    calls = {"a": 0, "b": 0}

    def fa(x):
        calls["a"] += 1
        return x + 1

    def fb(a):
        calls["b"] += 1
        return a * 2

    g = ComputationGraph(
        [
            GraphNode("a", op(fa), {"x": GraphInput("x")}),
            GraphNode("b", op(fb), {"a": "a"}),
        ],
        outputs=["b"],
    )

    async def main():
        s = ExecutionSession(g)
        r1 = await s.execute({"x": 1})
        r2 = await s.execute({"x": 1})  # fully cached
        assert r1 == r2 == 4
        assert calls == {"a": 1, "b": 1}
        s.invalidate("b")
        r3 = await s.execute({"x": 1})
        assert r3 == 4
        assert calls == {"a": 1, "b": 2}  # a stayed cached
        s.invalidate("a")  # invalidates b transitively
        await s.execute({"x": 1})
        assert calls == {"a": 2, "b": 3}

    asyncio.run(main())


def test_session_execute_async():
    g = ComputationGraph(
        [GraphNode("a", op(lambda x: x * 3), {"x": GraphInput("x")})]
    )

    async def main():
        s = ExecutionSession(g)
        fut = s.execute_async({"x": 7})
        return await fut.result()

    assert asyncio.run(main()) == 21
