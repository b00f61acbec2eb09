"""P2P node mixins — parameter flatten/write, half-step, aggregate-and-set.

Reference parity: engine/node/mixin.py (p2p_half_step 59-69 = local SGD
half-update returning theta-half; p2p_aggregate_and_set 71-80 =
pre_agg o agg over [self]+neighbors then write params;
P2PByzantineMixin.p2p_broadcast_vector 93-105).
"""
from __future__ import annotations

from typing import Any, List, Optional, Sequence

import torch
from torch import nn


class P2PHonestMixin:
    model: nn.Module
    lr: float

    def p2p_flat_params(self) -> torch.Tensor:
        return torch.cat([p.detach().reshape(-1) for p in self.model.parameters()])

    def p2p_write_params(self, flat: torch.Tensor) -> None:
        off = 0
        with torch.no_grad():
            for p in self.model.parameters():
                num = p.numel()
                p.copy_(flat[off : off + num].reshape(p.shape))
                off += num

    def p2p_local_loss_backward(self) -> None:
        """Override: run one forward+backward populating .grad."""
        raise NotImplementedError

    def p2p_half_step(self, lr: Optional[float] = None) -> torch.Tensor:
        """One local SGD half-update; returns the flattened theta-half."""
        step = lr if lr is not None else getattr(self, "lr", 0.1)
        self.model.zero_grad(set_to_none=True)
        self.p2p_local_loss_backward()
        with torch.no_grad():
            for p in self.model.parameters():
                if p.grad is not None:
                    p.add_(p.grad, alpha=-step)
        return self.p2p_flat_params()

    def p2p_aggregate_and_set(
        self, vectors: Sequence[torch.Tensor], aggregator: Any, pre_aggregator: Any = None
    ) -> torch.Tensor:
        vecs: List[torch.Tensor] = list(vectors)
        if pre_aggregator is not None:
            vecs = pre_aggregator.pre_aggregate(vecs)
        out = aggregator.aggregate(vecs)
        self.p2p_write_params(out)
        return out


class P2PByzantineMixin:
    attack: Any

    def p2p_broadcast_vector(self, neighbor_vectors: Sequence[torch.Tensor]) -> torch.Tensor:
        vecs = list(neighbor_vectors)
        # derive the attack's inputs from its declared uses_* flags
        # (reference mixin.py:93-105): base_grad attacks see the mean of
        # the vectors they observed
        kwargs = {}
        if getattr(self.attack, "uses_honest_grads", False):
            kwargs["honest_grads"] = vecs
        if getattr(self.attack, "uses_base_grad", False):
            kwargs["base_grad"] = torch.stack(vecs).float().mean(dim=0).to(vecs[0].dtype)
        if not kwargs:
            kwargs["honest_grads"] = vecs
        return self.attack.apply(**kwargs)
