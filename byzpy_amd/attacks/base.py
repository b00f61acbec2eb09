"""Attack operator base with declarative input flags.

Reference parity: attacks/base.py:47-124 — attacks declare which inputs
they consume (``uses_base_grad`` / ``uses_model_batch`` /
``uses_honest_grads``); ``compute`` collects only the declared inputs and
calls ``apply``.
"""
from __future__ import annotations

from typing import Any

from byzpy_amd.ops.base import Operator, OpContext
from byzpy_amd.utils.flatten import stack_gradients


class Attack(Operator):
    name = "attack"
    input_key = "honest_grads"

    uses_base_grad: bool = False
    uses_model_batch: bool = False
    uses_honest_grads: bool = False

    def apply(self, **kwargs: Any) -> Any:
        raise NotImplementedError

    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        kwargs = {}
        if self.uses_base_grad:
            kwargs["base_grad"] = inputs.get("base_grad")
        if self.uses_model_batch:
            kwargs["model"] = inputs.get("model")
            kwargs["batch"] = inputs.get("batch")
        if self.uses_honest_grads:
            kwargs["honest_grads"] = inputs.get("honest_grads")
        return self.apply(**kwargs)

    @staticmethod
    def _stack(honest_grads: Any):
        return stack_gradients(honest_grads)
