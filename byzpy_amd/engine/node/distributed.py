"""Distributed node base classes — wire aggregators/attacks into pipelines.

Reference parity: engine/node/distributed.py (DistributedHonestNode wires
the aggregator into an "aggregate" pipeline and local_honest_gradient into
a CallableOp pipeline 87-134; DistributedByzantineNode detects an
overridden byzantine_gradient or derives inputs from an Attack's uses_*
flags 140-223).
"""
from __future__ import annotations

import inspect
from typing import Any, List, Optional, Sequence

import torch

from byzpy_amd.aggregators.base import Aggregator
from byzpy_amd.attacks.base import Attack
from byzpy_amd.engine.node.application import (
    ByzantineNodeApplication,
    HonestNodeApplication,
)
from byzpy_amd.engine.node.base import ByzantineNode, HonestNode
from byzpy_amd.graph.graph import ComputationGraph, GraphInput, GraphNode
from byzpy_amd.graph.ops import CallableOp, RemoteCallableOp
from byzpy_amd.graph.pool import ActorPool


class DistributedHonestNode(HonestNode):
    """HonestNode with an "aggregate" pipeline (its aggregator) and an
    "honest_gradient" pipeline (its local gradient fn)."""

    def __init__(
        self, aggregator: Aggregator, *, pool: Optional[ActorPool] = None
    ) -> None:
        self.aggregator = aggregator
        self.app = HonestNodeApplication(pool)
        self.app.register_pipeline(
            "aggregate",
            ComputationGraph(
                [
                    GraphNode(
                        "aggregate",
                        aggregator,
                        {"gradients": GraphInput("gradients")},
                    )
                ]
            ),
            _internal=True,
        )
        self.app.register_pipeline(
            "honest_gradient",
            ComputationGraph(
                [
                    GraphNode(
                        "honest_gradient",
                        CallableOp(self.local_honest_gradient, name="honest-grad"),
                        {"x": GraphInput("x"), "y": GraphInput("y")},
                    )
                ]
            ),
            _internal=True,
        )

    # -- overridables -------------------------------------------------------
    def local_honest_gradient(self, x: torch.Tensor, y: torch.Tensor) -> Any:
        raise NotImplementedError

    def honest_gradient(self, x: torch.Tensor, y: torch.Tensor) -> Any:
        return self.local_honest_gradient(x, y)

    async def aggregate(self, gradients: Sequence[Any]) -> Any:
        return await self.app.run_pipeline("aggregate", {"gradients": list(gradients)})


class DistributedByzantineNode(ByzantineNode):
    """ByzantineNode whose "attack" pipeline comes from either a
    user-overridden ``byzantine_gradient`` (wrapped in a RemoteCallableOp
    with signature-derived input keys) or a declarative Attack operator."""

    def __init_subclass__(cls, **kwargs: Any) -> None:
        super().__init_subclass__(**kwargs)
        cls._overrides_byzantine_gradient = (
            "byzantine_gradient" in cls.__dict__
        )

    def __init__(
        self, attack: Optional[Attack] = None, *, pool: Optional[ActorPool] = None
    ) -> None:
        self.attack = attack
        self.app = ByzantineNodeApplication(pool)
        if getattr(self, "_overrides_byzantine_gradient", False):
            fn = self.byzantine_gradient
            sig = inspect.signature(fn)
            keys = [
                p
                for p in sig.parameters
                if p not in ("self", "x", "y") and sig.parameters[p].kind
                in (
                    inspect.Parameter.POSITIONAL_OR_KEYWORD,
                    inspect.Parameter.KEYWORD_ONLY,
                )
            ]

            def _call(**ctx: Any) -> Any:
                return fn(torch.empty(0), torch.empty(0), **ctx)

            self.app.register_pipeline(
                "attack",
                ComputationGraph(
                    [
                        GraphNode(
                            "attack",
                            RemoteCallableOp(_call, name="byz-grad"),
                            {k: GraphInput(k) for k in keys},
                        )
                    ]
                ),
                _internal=True,
            )
            self._attack_keys = keys
        else:
            if attack is None:
                raise ValueError(
                    "DistributedByzantineNode needs an Attack or an overridden "
                    "byzantine_gradient"
                )
            keys: List[str] = []
            inputs = {}
            if attack.uses_base_grad:
                keys.append("base_grad")
                inputs["base_grad"] = GraphInput("base_grad")
            if attack.uses_model_batch:
                keys += ["model", "batch"]
                inputs["model"] = GraphInput("model")
                inputs["batch"] = GraphInput("batch")
            if attack.uses_honest_grads:
                keys.append("honest_grads")
                inputs["honest_grads"] = GraphInput("honest_grads")
            self.app.register_pipeline(
                "attack",
                ComputationGraph([GraphNode("attack", attack, inputs)]),
                _internal=True,
            )
            self._attack_keys = keys

    # default batch/apply behavior so an Attack-only node is concrete
    def next_batch(self):
        return torch.empty(0), torch.empty(0)

    def apply_server_gradient(self, gradient: Any) -> None:
        self.last_server_gradient = gradient

    def byzantine_gradient(self, x: torch.Tensor, y: torch.Tensor, **ctx: Any) -> Any:
        raise NotImplementedError

    async def run_attack(self, **ctx: Any) -> Any:
        inputs = {k: ctx.get(k) for k in self._attack_keys}
        return await self.app.run_pipeline("attack", inputs)
