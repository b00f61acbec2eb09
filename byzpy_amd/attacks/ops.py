"""Byzantine attacks.

Reference parity: attacks/{empire,sign_flip,label_flip,little,gaussian,
inf,mimic}.py. All are cheap elementwise/reduction math; on GPU they run
as device torch ops over the resident honest-gradient matrix (single
launches — no fan-out needed).
"""
from __future__ import annotations

from typing import Any, Dict, Optional

import torch
from torch import nn

from byzpy_amd.attacks.base import Attack
from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops import functional as F
from byzpy_amd.utils.flatten import to_like


class EmpireAttack(Attack):
    """scale * mean(honest gradients) (inner-product-manipulation analogue)."""

    name = "attack/empire"
    uses_honest_grads = True

    def __init__(self, scale: float = -1.0, *, chunk_size: int = 8) -> None:
        self.scale = float(scale)
        self.chunk_size = int(chunk_size)

    def apply(self, *, honest_grads: Any) -> Any:
        X, like = self._stack(honest_grads)
        return to_like(F.empire(X, self.scale), like)


class SignFlipAttack(Attack):
    """scale * base_grad (default -1: flipped sign)."""

    name = "attack/sign-flip"
    uses_base_grad = True

    def __init__(self, scale: float = -1.0, *, chunk_size: int = 8192) -> None:
        self.scale = float(scale)
        self.chunk_size = int(chunk_size)

    def apply(self, *, base_grad: Any) -> Any:
        X, like = self._stack([base_grad])
        return to_like(X[0] * self.scale, like)


class LittleAttack(Attack):
    """'A Little Is Enough' (Baruch et al. 2019): mu + z * sigma with
    s = floor(N/2)+1-f and z = Phi^{-1}((N-s)/N)."""

    name = "attack/little"
    uses_honest_grads = True

    def __init__(
        self, f: int, N: Optional[int] = None, *, chunk_size: int = 8192
    ) -> None:
        if f < 0:
            raise ValueError("f must be >= 0")
        self.f = int(f)
        self.N = None if N is None else int(N)
        self.chunk_size = int(chunk_size)

    def apply(self, *, honest_grads: Any) -> Any:
        X, like = self._stack(honest_grads)
        return to_like(D.little(X, self.f, self.N), like)


class GaussianAttack(Attack):
    """iid N(mu, sigma^2) noise vector; seedable for determinism."""

    name = "attack/gaussian"
    uses_honest_grads = True

    def __init__(
        self,
        mu: float = 0.0,
        sigma: float = 1.0,
        *,
        seed: Optional[int] = None,
        chunk_size: int = 8192,
    ) -> None:
        if sigma < 0:
            raise ValueError("sigma must be >= 0")
        self.mu, self.sigma = float(mu), float(sigma)
        self.seed = seed
        self.chunk_size = int(chunk_size)

    def apply(self, *, honest_grads: Any) -> Any:
        X, like = self._stack(honest_grads)
        return to_like(D.gaussian_attack(X, self.mu, self.sigma, self.seed), like)


class InfAttack(Attack):
    """All-infinity vector."""

    name = "attack/inf"
    uses_honest_grads = True

    def __init__(self, *, chunk_size: int = 8192) -> None:
        self.chunk_size = int(chunk_size)

    def apply(self, *, honest_grads: Any) -> Any:
        X, like = self._stack(honest_grads)
        return to_like(F.inf_attack(X), like)


class MimicAttack(Attack):
    """Copy honest worker #epsilon's vector."""

    name = "attack/mimic"
    uses_honest_grads = True

    def __init__(self, epsilon: int = 0, *, chunk_size: int = 8192) -> None:
        if epsilon < 0:
            raise ValueError("epsilon must be >= 0")
        self.epsilon = int(epsilon)
        self.chunk_size = int(chunk_size)

    def apply(self, *, honest_grads: Any) -> Any:
        X, like = self._stack(honest_grads)
        return to_like(F.mimic(X, self.epsilon), like)


class LabelFlipAttack(Attack):
    """Gradient of the loss on flipped labels (mirror K-1-y or explicit
    mapping); runs a forward+backward on the provided model."""

    name = "attack/label-flip"
    uses_model_batch = True

    def __init__(
        self,
        *,
        num_classes: Optional[int] = None,
        mapping: Optional[Dict[int, int]] = None,
        loss_fn: Optional[nn.Module] = None,
        scale: float = 1.0,
    ) -> None:
        if mapping is None and num_classes is None:
            raise ValueError("Provide either `mapping` or `num_classes`.")
        self.num_classes = num_classes
        self.mapping = mapping
        self.loss_fn = loss_fn or nn.CrossEntropyLoss(reduction="mean")
        self.scale = float(scale)

    def _flip(self, y: torch.Tensor) -> torch.Tensor:
        if self.mapping is not None:
            flipped = y.clone()
            for src, dst in self.mapping.items():
                flipped[y == src] = dst
            return flipped
        return (self.num_classes - 1) - y

    def apply(self, *, model: nn.Module, batch: Any) -> Any:
        x, y = batch
        flipped = self._flip(y)
        model.zero_grad(set_to_none=True)
        loss = self.loss_fn(model(x), flipped) * self.scale
        grads = torch.autograd.grad(loss, [p for p in model.parameters() if p.requires_grad])
        return torch.cat([g.reshape(-1) for g in grads])
