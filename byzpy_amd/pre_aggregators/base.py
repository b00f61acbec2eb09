"""PreAggregator base — transforms a list of gradients into a new list.

Reference parity: pre_aggregators/base.py:9-96 (input_key "vectors",
returns a list).
"""
from __future__ import annotations

from typing import Any, List

import torch

from byzpy_amd.ops.base import Operator, OpContext
from byzpy_amd.aggregators.base import build_matrix_ref, cleanup_handles
from byzpy_amd.utils.flatten import stack_gradients, to_like


class PreAggregator(Operator):
    name = "pre-aggregator"
    input_key = "vectors"

    def pre_aggregate(self, vectors: Any) -> List[Any]:
        X, like = stack_gradients(vectors)
        out = self._pre_aggregate(X)
        return [to_like(row.reshape(-1), like) for row in out]

    def _pre_aggregate(self, X: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def compute(self, ctx: OpContext, **inputs: Any) -> Any:
        return self.pre_aggregate(inputs[self.input_key])

    def _matrix_ref(self, ctx: OpContext, vectors: Any):
        return build_matrix_ref(ctx, vectors)

    _cleanup = staticmethod(cleanup_handles)
