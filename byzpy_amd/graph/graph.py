"""ComputationGraph — named operator nodes wired by input mappings.

Reference parity: engine/graph/graph.py:23-131 (GraphInput, MessageSource
via GraphInput.from_message, GraphNode, topo sort, required-input
collection, default output = last topo node).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Mapping, Optional, Sequence, Union

from byzpy_amd.ops.base import Operator


@dataclass(frozen=True)
class GraphInput:
    """Named external input."""

    name: str

    @staticmethod
    def from_message(
        message_type: str, *, field: Optional[str] = None, timeout: Optional[float] = None
    ) -> "MessageSource":
        return MessageSource(message_type=message_type, field=field, timeout=timeout)


@dataclass(frozen=True)
class MessageSource:
    """An input satisfied by an incoming message (message-aware scheduler)."""

    message_type: str
    field: Optional[str] = None
    timeout: Optional[float] = None


def graph_input(name: str) -> GraphInput:
    return GraphInput(name)


InputSpec = Union[str, GraphInput, MessageSource]


@dataclass
class GraphNode:
    name: str
    op: Operator
    inputs: Mapping[str, InputSpec] = field(default_factory=dict)


class ComputationGraph:
    def __init__(
        self, nodes: Sequence[GraphNode], outputs: Optional[Sequence[str]] = None
    ) -> None:
        self.nodes: Dict[str, GraphNode] = {}
        for node in nodes:
            if node.name in self.nodes:
                raise ValueError(f"duplicate node name {node.name!r}")
            self.nodes[node.name] = node
        for node in nodes:
            for arg, spec in node.inputs.items():
                if isinstance(spec, str) and spec not in self.nodes:
                    raise ValueError(
                        f"node {node.name!r} input {arg!r} references unknown "
                        f"node {spec!r}"
                    )
        self.topo_order: List[str] = self._topo_sort()
        if outputs is None:
            self.outputs = [self.topo_order[-1]] if self.topo_order else []
        else:
            for o in outputs:
                if o not in self.nodes:
                    raise ValueError(f"unknown output node {o!r}")
            self.outputs = list(outputs)

    def _topo_sort(self) -> List[str]:
        indeg: Dict[str, int] = {n: 0 for n in self.nodes}
        dependents: Dict[str, List[str]] = {n: [] for n in self.nodes}
        for node in self.nodes.values():
            for spec in node.inputs.values():
                if isinstance(spec, str):
                    indeg[node.name] += 1
                    dependents[spec].append(node.name)
        # deterministic order: seed with declaration order
        ready = [n for n in self.nodes if indeg[n] == 0]
        order: List[str] = []
        while ready:
            cur = ready.pop(0)
            order.append(cur)
            for dep in dependents[cur]:
                indeg[dep] -= 1
                if indeg[dep] == 0:
                    ready.append(dep)
        if len(order) != len(self.nodes):
            raise ValueError("computation graph has a cycle")
        return order

    def required_inputs(self) -> List[str]:
        names = []
        for node in self.nodes.values():
            for spec in node.inputs.values():
                if isinstance(spec, GraphInput) and spec.name not in names:
                    names.append(spec.name)
        return names

    def message_sources(self) -> List[MessageSource]:
        out = []
        for node in self.nodes.values():
            for spec in node.inputs.values():
                if isinstance(spec, MessageSource):
                    out.append(spec)
        return out
