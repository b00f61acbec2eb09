"""SubTask — the unit of work shipped to an actor-pool worker.

Reference parity: engine/graph/subtask.py:7-18. On the MI355X path a
"subtask" degenerates to a kernel launch on the worker's HIP stream; the
CPU pool path keeps the reference's fan-out semantics for plumbing parity.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Optional


@dataclass
class SubTask:
    fn: Callable[..., Any]
    args: tuple = ()
    kwargs: dict = field(default_factory=dict)
    name: Optional[str] = None
    affinity: Optional[str] = None
    max_retries: int = 0
