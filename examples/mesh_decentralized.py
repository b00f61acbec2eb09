"""Serverless mesh of decentralized nodes over TCP (MeshRemoteContext):
every node runs its own asyncio server, dials its peers, and gossips a
robust aggregate — no central server, reconnect monitor included.

  python examples/mesh_decentralized.py
"""
from __future__ import annotations

import asyncio
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.aggregators import CoordinateWiseMedian
from byzpy_amd.engine.node.remote import MeshRemoteContext


class GossipNode:
    def __init__(self, name: str, value: torch.Tensor) -> None:
        self.name = name
        self.value = value
        self.received: dict[str, torch.Tensor] = {}
        self.got_all = asyncio.Event()
        self.expect = 0

    async def handle_incoming_message(self, msg: dict) -> None:
        self.received[msg["from"]] = msg["vector"]
        if len(self.received) >= self.expect:
            self.got_all.set()


async def main() -> None:
    d = 16
    torch.manual_seed(0)
    names = ["alpha", "beta", "gamma"]
    nodes = {nm: GossipNode(nm, torch.randn(d)) for nm in names}
    ctxs: dict[str, MeshRemoteContext] = {}

    # start every node's server, then dial all peers
    for nm in names:
        ctxs[nm] = MeshRemoteContext(nm, host="127.0.0.1", port=0,
                                     reconnect_interval=0.5)
        await ctxs[nm].start(nodes[nm])
    for nm in names:
        for peer in names:
            if peer != nm:
                ctxs[nm].add_peer(peer, "127.0.0.1", ctxs[peer].port)
    await asyncio.sleep(0.3)  # let the reconnect monitor dial everyone

    # one gossip round: broadcast my value, robustly aggregate all
    for nm in names:
        nodes[nm].expect = len(names) - 1
    for nm in names:
        for peer in names:
            if peer != nm:
                await ctxs[nm].send_message(
                    peer, {"from": nm, "vector": nodes[nm].value}
                )
    agg = CoordinateWiseMedian()
    for nm in names:
        await asyncio.wait_for(nodes[nm].got_all.wait(), timeout=10)
        vectors = [nodes[nm].value] + list(nodes[nm].received.values())
        nodes[nm].value = agg.aggregate(vectors)

    v0 = nodes[names[0]].value
    consensus = all(torch.allclose(nodes[nm].value, v0) for nm in names)
    print(f"mesh gossip round complete; consensus={consensus}")
    for nm in names:
        await ctxs[nm].shutdown()


if __name__ == "__main__":
    asyncio.run(main())
