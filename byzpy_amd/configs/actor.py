"""Global default actor backend spec.

Reference parity: configs/actor.py:10-30. The reference also carried a
multi-backend ndarray dispatch (configs/backend.py) whose getter was
hard-wired to torch — per the MI355X design there is exactly ONE array
backend (torch-ROCm), so that indirection is deliberately gone
(SURVEY.md §2.4 note).
"""
from __future__ import annotations

from typing import Any

_default_actor: Any = "thread"


def set_actor(spec: Any) -> None:
    global _default_actor
    from byzpy_amd.actor.factory import resolve_backend

    if isinstance(spec, str):
        resolve_backend(spec)  # validate eagerly
    _default_actor = spec


def get_actor() -> Any:
    return _default_actor
