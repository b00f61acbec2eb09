"""Node runtime: applications, distributed nodes, decentralized messaging,
process contexts (SURVEY.md §4 pattern 6)."""
import asyncio

import pytest
import torch

from byzpy_amd.aggregators import CoordinateWiseMedian
from byzpy_amd.attacks import EmpireAttack
from byzpy_amd.engine.node.application import NodeApplication
from byzpy_amd.engine.node.cluster import DecentralizedCluster
from byzpy_amd.engine.node.decentralized import DecentralizedNode
from byzpy_amd.engine.node.distributed import (
    DistributedByzantineNode,
    DistributedHonestNode,
)
from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.ops import CallableOp


def test_application_pipelines():
    async def main():
        app = NodeApplication()
        app.register_pipeline(
            "double",
            ComputationGraph(
                [GraphNode("d", CallableOp(lambda x: x * 2), {"x": GraphInput("x")})]
            ),
        )
        assert app.has_pipeline("double")
        assert await app.run_pipeline("double", {"x": 21}) == 42

    asyncio.run(main())


def test_application_sync_wrapper_refuses_in_loop():
    app = NodeApplication()
    app.register_pipeline(
        "id",
        ComputationGraph(
            [GraphNode("i", CallableOp(lambda x: x), {"x": GraphInput("x")})]
        ),
    )
    assert app.run_pipeline_sync("id", {"x": 5}) == 5

    async def inside():
        with pytest.raises(RuntimeError, match="running event loop"):
            app.run_pipeline_sync("id", {"x": 5})

    asyncio.run(inside())


class MyHonest(DistributedHonestNode):
    def __init__(self):
        super().__init__(CoordinateWiseMedian())

    def local_honest_gradient(self, x, y):
        return torch.ones(4)

    def next_batch(self):
        return torch.zeros(1), torch.zeros(1)

    def apply_server_gradient(self, gradient):
        self.last = gradient


def test_distributed_honest_node():
    async def main():
        node = MyHonest()
        g = await node.app.run_pipeline(
            "honest_gradient", {"x": torch.zeros(1), "y": torch.zeros(1)}
        )
        assert torch.allclose(g, torch.ones(4))
        agg = await node.aggregate([torch.ones(4), torch.zeros(4), 2 * torch.ones(4)])
        assert torch.allclose(agg, torch.ones(4))

    asyncio.run(main())


class MyByz(DistributedByzantineNode):
    def byzantine_gradient(self, x, y, honest_grads=None):
        return -torch.stack([g for g in honest_grads]).mean(dim=0)

    def next_batch(self):
        return torch.zeros(1), torch.zeros(1)

    def apply_server_gradient(self, gradient):
        pass


def test_distributed_byzantine_override():
    async def main():
        node = MyByz()
        out = await node.run_attack(honest_grads=[torch.ones(3), 3 * torch.ones(3)])
        assert torch.allclose(out, -2 * torch.ones(3))

    asyncio.run(main())


def test_distributed_byzantine_attack_op():
    async def main():
        node = DistributedByzantineNode(EmpireAttack(scale=-1.0))
        out = await node.run_attack(honest_grads=[torch.ones(3), 3 * torch.ones(3)])
        assert torch.allclose(out, -2 * torch.ones(3))

    asyncio.run(main())


def test_decentralized_messaging_and_handlers():
    async def main():
        cluster = DecentralizedCluster()
        a = DecentralizedNode("a")
        b = DecentralizedNode("b")
        got = []
        b.register_handler("hello", lambda m: got.append(m["v"]))
        cluster.add_node(a)
        cluster.add_node(b)
        await cluster.start_all()
        await a.send_message("b", "hello", {"v": 7})
        await asyncio.sleep(0.15)
        assert got == [7]
        # broadcast
        await a.broadcast_message("hello", {"v": 8})
        await asyncio.sleep(0.15)
        assert got == [7, 8]
        await cluster.shutdown_all()

    asyncio.run(main())


def test_decentralized_pipeline_swap():
    async def main():
        node = DecentralizedNode("p")
        node.register_pipeline(
            "a",
            ComputationGraph(
                [GraphNode("n", CallableOp(lambda x: x + 1), {"x": GraphInput("x")})]
            ),
        )
        node.register_pipeline(
            "b",
            ComputationGraph(
                [GraphNode("n", CallableOp(lambda x: x * 10), {"x": GraphInput("x")})]
            ),
        )
        assert await node.execute_pipeline("a", {"x": 1}) == 2
        assert await node.execute_pipeline("b", {"x": 1}) == 10

    asyncio.run(main())


def _make_child_node():
    # module-level factory (cloudpickled by value into the child)
    node = DecentralizedNode("child")
    node.register_pipeline(
        "triple",
        ComputationGraph(
            [GraphNode("t", CallableOp(lambda x: x * 3), {"x": GraphInput("x")})]
        ),
    )
    return node


def test_process_context_pipeline():
    async def main():
        cluster = DecentralizedCluster()
        ctx = cluster.add_process_node("child", _make_child_node)
        await cluster.start_all()
        try:
            out = await asyncio.wait_for(ctx.execute_pipeline("triple", {"x": 14}), 30)
            assert out == 42
        finally:
            await cluster.shutdown_all()

    asyncio.run(main())


def test_autonomous_task():
    async def main():
        node = DecentralizedNode("auto")
        ticks = []

        async def tick(n):
            ticks.append(1)

        await node.start()
        node.start_autonomous_task(tick, 0.02)
        await asyncio.sleep(0.1)
        await node.stop()
        assert len(ticks) >= 2

    asyncio.run(main())
