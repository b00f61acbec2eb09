"""hipGraph capture for repeated aggregation (serving/training loops).

The PS/P2P round calls the same aggregate on same-shaped gradients every
round; small-d aggregates are launch-bound (CenteredClipping M=10 is ~20
kernel launches for ~0.03 ms of math). ``CapturedAggregate`` captures the
whole aggregate once into a hipGraph (torch.cuda.CUDAGraph is hipGraph on
ROCm) and replays it per round: copy-in, one replay, copy-out — no
per-kernel launch cost, no Python dispatch in the loop.

Capture-safe ops: everything without data-dependent host sync. The
Weiszfeld convergence poll syncs, so geometric-median capture requires a
fixed iteration count (pass ``fixed_iters`` via the op's max_iter with
tol=0-style usage or use CenteredClipping-like fixed-M ops).
"""
from __future__ import annotations

from typing import Any, Callable

import torch


class CapturedAggregate:
    """Capture ``fn(X) -> Tensor`` on a static input buffer and replay.

    ``fn`` must be capture-safe: no ``.item()``/host sync, shapes fixed.
    """

    def __init__(
        self,
        fn: Callable[[torch.Tensor], torch.Tensor],
        example_input: torch.Tensor,
        *,
        warmup: int = 2,
    ) -> None:
        if not example_input.is_cuda:
            raise ValueError("CapturedAggregate needs a device tensor")
        self.fn = fn
        self.static_input = example_input.clone()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(max(1, warmup)):
                out = fn(self.static_input)
        torch.cuda.current_stream().wait_stream(stream)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_output = fn(self.static_input)

    def run(self, X: torch.Tensor) -> torch.Tensor:
        """Replay on new data (same shape/dtype). Returns the STATIC output
        buffer — clone it if you need it to survive the next replay."""
        self.static_input.copy_(X)
        self.graph.replay()
        return self.static_output

    def run_inplace(self) -> torch.Tensor:
        """Replay on whatever is currently in ``static_input`` (callers may
        write gradients directly into it and skip the copy)."""
        self.graph.replay()
        return self.static_output


def capture_aggregator(
    aggregator: Any, example_matrix: torch.Tensor, *, warmup: int = 2
) -> CapturedAggregate:
    """Capture an Aggregator's matrix path (``_aggregate``)."""
    return CapturedAggregate(
        lambda X: aggregator._aggregate(X), example_matrix, warmup=warmup
    )
