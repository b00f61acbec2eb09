"""DecentralizedCluster — builds and wires a set of DecentralizedNodes.

Reference parity: engine/node/cluster.py:12-108 (topology-index -> node-id
map pushed into every router before start_all).
"""
from __future__ import annotations

from typing import Any, Dict, List

from byzpy_amd.engine.node.context import ProcessContext
from byzpy_amd.engine.node.decentralized import DecentralizedNode


class DecentralizedCluster:
    def __init__(self, topology: Any = None) -> None:
        self.topology = topology
        self.nodes: Dict[str, DecentralizedNode] = {}
        self.process_contexts: Dict[str, ProcessContext] = {}
        self._order: List[str] = []

    def add_node(self, node: DecentralizedNode) -> None:
        node.router.topology = self.topology
        self.nodes[node.node_id] = node
        self._order.append(node.node_id)

    def add_process_node(self, node_id: str, factory) -> ProcessContext:
        """Node runs in a child process; the parent relays its messages."""
        ctx = ProcessContext(node_id, factory)
        ctx.set_route_callback(self._route)
        self.process_contexts[node_id] = ctx
        self._order.append(node_id)
        return ctx

    async def _route(self, target_id: str, message: dict) -> None:
        if target_id in self.nodes:
            await self.nodes[target_id].handle_incoming_message(message)
        elif target_id in self.process_contexts:
            await self.process_contexts[target_id].send_message(target_id, message)
        else:
            raise RuntimeError(f"unknown node {target_id!r}")

    def _index_map(self) -> Dict[int, str]:
        return {i: nid for i, nid in enumerate(self._order)}

    async def start_all(self) -> None:
        idx_map = self._index_map()
        for node in self.nodes.values():
            node.router.set_node_mapping(idx_map)
        for node in self.nodes.values():
            await node.start()
        for ctx in self.process_contexts.values():
            await ctx.start()

    async def shutdown_all(self) -> None:
        for node in self.nodes.values():
            await node.stop()
        for ctx in self.process_contexts.values():
            await ctx.shutdown()
