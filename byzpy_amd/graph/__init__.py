from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode, MessageSource, graph_input
from byzpy_amd.graph.subtask import SubTask
from byzpy_amd.graph.scheduler import MessageAwareNodeScheduler, NodeScheduler
from byzpy_amd.graph.parallel_scheduler import ParallelScheduler
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.graph.executor import OperatorExecutor, run_operator
from byzpy_amd.graph.lazy import GraphBuilder, LazyNode
from byzpy_amd.graph.session import ExecutionFuture, ExecutionSession

__all__ = [
    "ComputationGraph",
    "GraphNode",
    "GraphInput",
    "MessageSource",
    "graph_input",
    "SubTask",
    "NodeScheduler",
    "MessageAwareNodeScheduler",
    "ParallelScheduler",
    "ActorPool",
    "ActorPoolConfig",
    "OperatorExecutor",
    "run_operator",
    "GraphBuilder",
    "LazyNode",
    "ExecutionFuture",
    "ExecutionSession",
]
