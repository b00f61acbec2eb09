from byzpy_amd.ops.base import Operator, OpContext
from byzpy_amd.graph.subtask import SubTask

__all__ = ["Operator", "OpContext", "SubTask"]
