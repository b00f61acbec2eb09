"""Utility operators: CallableOp, RemoteCallableOp, single-op graphs.

Reference parity: engine/graph/ops.py:10-92.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Optional, Sequence

from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.ops.base import Operator, OpContext


class CallableOp(Operator):
    """Wraps a local callable; graph inputs map to keyword arguments."""

    def __init__(self, fn: Callable[..., Any], *, name: str = "callable") -> None:
        self.fn = fn
        self.name = name

    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        return self.fn(**inputs)


class RemoteCallableOp(Operator):
    """Ships the callable to the pool as a single subtask."""

    supports_subtasks = True

    def __init__(self, fn: Callable[..., Any], *, name: str = "remote-callable") -> None:
        self.fn = fn
        self.name = name

    def create_subtasks(self, ctx: OpContext, **inputs: Any) -> Sequence[SubTask]:
        return [SubTask(fn=_kwargs_call, args=(self.fn,), kwargs=dict(inputs))]

    def reduce_subtasks(self, ctx: OpContext, results, **inputs: Any) -> Any:
        return results[0]

    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        return self.fn(**inputs)


def _kwargs_call(fn: Callable[..., Any], **kwargs: Any) -> Any:
    return fn(**kwargs)


def make_single_operator_graph(
    op: Operator,
    input_keys: Optional[Dict[str, str]] = None,
    *,
    node_name: str = "op",
) -> ComputationGraph:
    """Graph with one node whose operator inputs come from graph inputs.
    ``input_keys`` maps operator arg -> external input name (defaults to the
    operator's own input_key for both)."""
    if input_keys is None:
        input_keys = {op.input_key: op.input_key}
    inputs = {arg: GraphInput(ext) for arg, ext in input_keys.items()}
    return ComputationGraph([GraphNode(name=node_name, op=op, inputs=inputs)])
