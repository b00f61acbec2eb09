"""RCCL peer-to-peer engine — one rank per GPU, gossip over xGMI.

The MI355X-native replacement for the reference's per-node TCP/queue
message fabric: each round every rank half-steps locally,
the theta-half vectors are exchanged with ONE all-gather (complete
topology) or neighbor send/recv pairs folded into an all-gather + row
select (ring(k): the collective moves over all 7 xGMI links instead of a
per-link-bound ring of unicasts), byzantine ranks substitute their attack
vector, and every rank robust-aggregates [self] + in-neighbors.

BASELINE config 4 shape: P2P ring, NNM pre-agg + GeometricMedian, 8 GPUs.
"""
from __future__ import annotations

from typing import Any, Callable, Optional

import torch

from byzpy_amd.engine.peer_to_peer.topology import Topology
from byzpy_amd.parallel import dist as pdist


class RcclPeerToPeer:
    def __init__(
        self,
        half_step_fn: Callable[[], torch.Tensor],
        write_params_fn: Callable[[torch.Tensor], None],
        aggregator: Any,
        *,
        topology: Optional[Topology] = None,
        pre_aggregator: Any = None,
        attack: Any = None,  # non-None => this rank is byzantine
    ) -> None:
        self.half_step_fn = half_step_fn
        self.write_params_fn = write_params_fn
        self.aggregator = aggregator
        self.pre_aggregator = pre_aggregator
        self.attack = attack
        world = pdist.get_world_size()
        self.topology = topology or Topology.complete(world)

    def round(self) -> torch.Tensor:
        rank = pdist.get_rank()
        theta_half = self.half_step_fn().reshape(-1)
        if self.attack is not None:
            # byzantine rank: broadcast the attack vector instead (it sees
            # its own honest half-step as context); inputs derive from the
            # attack's uses_* flags (base_grad attacks get theta-half)
            kwargs = {}
            if getattr(self.attack, "uses_honest_grads", False):
                kwargs["honest_grads"] = [theta_half]
            if getattr(self.attack, "uses_base_grad", False):
                kwargs["base_grad"] = theta_half
            if not kwargs:
                kwargs["honest_grads"] = [theta_half]
            theta_half = self.attack.apply(**kwargs).reshape(-1)
        all_vecs = pdist.all_gather_rows(theta_half.reshape(1, -1))  # (world, d)
        neighbors = self.topology.in_neighbors(rank)
        rows = [rank] + [j for j in neighbors if j != rank]
        vectors = [all_vecs[j] for j in rows]
        if self.pre_aggregator is not None:
            vectors = self.pre_aggregator.pre_aggregate(vectors)
        out = self.aggregator.aggregate(vectors)
        if self.attack is None:
            self.write_params_fn(out)
        return out
