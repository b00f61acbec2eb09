"""GraphBuilder / LazyNode — fluent graph construction.

Reference parity: engine/graph/lazy.py:24-226 (builder.input(name) ->
LazyNode; LazyNode.apply(op, ...) registers GraphNodes with auto input
keys; builder.build(outputs)).
"""
from __future__ import annotations

import itertools
from typing import Any, Dict, List, Optional, Sequence, Union

from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.ops.base import Operator


class LazyNode:
    def __init__(self, builder: "GraphBuilder", ref: Union[str, GraphInput]) -> None:
        self._builder = builder
        self._ref = ref

    @property
    def name(self) -> str:
        return self._ref.name if isinstance(self._ref, GraphInput) else self._ref

    def apply(
        self,
        op: Operator,
        *,
        input_key: Optional[str] = None,
        extra_inputs: Optional[Dict[str, Union[str, GraphInput, "LazyNode"]]] = None,
        name: Optional[str] = None,
    ) -> "LazyNode":
        key = input_key or getattr(op, "input_key", None) or "vectors"
        node_name = name or self._builder._fresh_name(getattr(op, "name", "op"))
        inputs: Dict[str, Any] = {key: self._ref}
        for arg, src in (extra_inputs or {}).items():
            inputs[arg] = src._ref if isinstance(src, LazyNode) else src
        self._builder._nodes.append(GraphNode(name=node_name, op=op, inputs=inputs))
        return LazyNode(self._builder, node_name)


class GraphBuilder:
    def __init__(self) -> None:
        self._nodes: List[GraphNode] = []
        self._counter = itertools.count()

    def _fresh_name(self, base: str) -> str:
        return f"{base}#{next(self._counter)}"

    def input(self, name: str) -> LazyNode:
        return LazyNode(self, GraphInput(name))

    def node(self, name: str) -> LazyNode:
        return LazyNode(self, name)

    def build(
        self, outputs: Optional[Sequence[Union[str, LazyNode]]] = None
    ) -> ComputationGraph:
        out_names = None
        if outputs is not None:
            out_names = [o.name if isinstance(o, LazyNode) else o for o in outputs]
        return ComputationGraph(list(self._nodes), outputs=out_names)
