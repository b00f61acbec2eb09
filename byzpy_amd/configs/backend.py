"""Array-backend configuration surface.

Reference parity: configs/backend.py:12-49 — the reference declared a
multi-backend ndarray protocol whose getter was hard-wired to torch
(SURVEY.md §2.4). The MI355X design has exactly ONE array backend
(torch-ROCm), so these functions keep the API shape while stating that
fact: ``set_backend`` accepts only "torch" and ``get_backend`` returns the
torch module.
"""
from __future__ import annotations

import contextlib
from typing import Iterator

_BACKEND = "torch"


def set_backend(name: str) -> None:
    if name != "torch":
        raise ValueError(
            f"byzpy_amd has a single array backend (torch-ROCm); got {name!r}"
        )


def get_backend():
    import torch

    return torch


@contextlib.contextmanager
def use_backend(name: str) -> Iterator[None]:
    set_backend(name)
    yield
