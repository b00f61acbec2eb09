"""Node ABCs.

Reference parity: engine/node/base.py:9-39 — ``Node`` supplies batches and
applies the server update; ``HonestNode`` computes true gradients;
``ByzantineNode`` fabricates adversarial ones (called with empty batches).
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Tuple

import torch


class Node(ABC):
    @abstractmethod
    def next_batch(self) -> Tuple[torch.Tensor, torch.Tensor]: ...

    @abstractmethod
    def apply_server_gradient(self, gradient: Any) -> None: ...


class HonestNode(Node):
    @abstractmethod
    def honest_gradient(self, x: torch.Tensor, y: torch.Tensor) -> Any: ...

    def honest_gradient_for_next_batch(self) -> Any:
        x, y = self.next_batch()
        return self.honest_gradient(x, y)


class ByzantineNode(Node):
    @abstractmethod
    def byzantine_gradient(self, x: torch.Tensor, y: torch.Tensor, **ctx: Any) -> Any: ...

    def byzantine_gradient_for_next_batch(self, **ctx: Any) -> Any:
        return self.byzantine_gradient(torch.empty(0), torch.empty(0), **ctx)
