"""RCCL peer-to-peer engine — one rank per GPU, gossip over xGMI.

The MI355X-native replacement for the reference's per-node TCP/queue
message fabric (reference engine/peer_to_peer/runner.py:284-389). Each
round has the reference's two communication phases:

  phase 1  honest ranks send their theta-half to their out-neighbors —
           ONE all-gather for complete topologies (the collective moves
           over all 7 xGMI links), batched neighbor send/recv pairs for
           sparse ring(k) (traffic proportional to k).
  phase 2  byzantine ranks compute their attack over the RECEIVED honest
           neighbor vectors (reference runner.py:316-367 — the cached
           neighbor vectors, not their own half-step) and send it to
           their honest out-neighbors via p2p pairs.

Honest ranks then robust-aggregate [self] + in-neighbor vectors (honest
theta-halves from phase 1, attack vectors from phase 2) and write their
parameters. The byzantine-rank set is exchanged once at bring-up over the
control plane so every rank can post the matching p2p schedule.

BASELINE config 4 shape: P2P ring, NNM pre-agg + GeometricMedian, 8 GPUs.
"""
from __future__ import annotations

from typing import Any, Callable, Iterable, Optional

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
        byzantine_ranks: Optional[Iterable[int]] = None,
        transport: str = "auto",  # "auto" | "all_gather" | "p2p"
    ) -> None:
        self.half_step_fn = half_step_fn
        self.write_params_fn = write_params_fn
        self.aggregator = aggregator
        self.pre_aggregator = pre_aggregator
        self.attack = attack
        world = pdist.get_world_size()
        self.topology = topology or Topology.complete(world)
        if byzantine_ranks is None:
            # one control-plane exchange so every rank can post the same
            # p2p schedule (phase-2 edges depend on who is byzantine)
            flags = pdist.all_gather_obj(attack is not None)
            byzantine_ranks = [i for i, b in enumerate(flags) if b]
        self.byzantine_ranks = frozenset(int(b) for b in byzantine_ranks)
        if attack is not None and pdist.get_rank() not in self.byzantine_ranks:
            raise ValueError(
                "this rank has an attack but is not in byzantine_ranks"
            )
        if attack is None and pdist.get_rank() in self.byzantine_ranks:
            # a rank the OTHERS treat as byzantine must send an attack
            # vector in phase 2 — otherwise their posted recvs deadlock
            raise ValueError(
                "this rank is in byzantine_ranks but has no attack"
            )
        if transport not in {"auto", "all_gather", "p2p"}:
            raise ValueError(f"bad transport {transport!r}")
        if transport == "auto":
            # all-gather wins when nearly everyone talks to everyone;
            # sparse topologies pay world/k excess traffic under it
            n_edges = sum(len(self.topology.out_neighbors(i)) for i in range(world))
            dense = world <= 1 or n_edges >= world * (world - 1) * 3 // 4
            transport = "all_gather" if dense else "p2p"
        self.transport = transport

    # -- one gossip round ---------------------------------------------------
    def round(self) -> torch.Tensor:
        rank = pdist.get_rank()
        byz = self.byzantine_ranks
        i_am_byz = self.attack is not None
        theta_half = self.half_step_fn().reshape(-1)
        topo = self.topology
        in_nb = [j for j in topo.in_neighbors(rank) if j != rank]
        out_nb = [j for j in topo.out_neighbors(rank) if j != rank]
        honest_in = [j for j in in_nb if j not in byz]
        byz_in = [j for j in in_nb if j in byz]

        # phase 1: honest theta-half movement
        if self.transport == "all_gather":
            all_vecs = pdist.all_gather_rows(theta_half.reshape(1, -1))
            recv_honest = {j: all_vecs[j] for j in honest_in}
        else:
            recv_honest = pdist.neighbor_exchange(
                out_nb if not i_am_byz else [], honest_in, theta_half
            )

        # phase 2: byzantine attack vectors to honest out-neighbors
        send_vec = theta_half
        if i_am_byz:
            ctx = [recv_honest[j] for j in honest_in]
            kwargs: dict = {}
            if getattr(self.attack, "uses_honest_grads", False):
                kwargs["honest_grads"] = ctx if ctx else [theta_half]
            if getattr(self.attack, "uses_base_grad", False):
                kwargs["base_grad"] = theta_half
            if not kwargs:
                kwargs["honest_grads"] = ctx if ctx else [theta_half]
            send_vec = (
                self.attack.apply(**kwargs).reshape(-1).to(theta_half.dtype)
            )
        recv_byz = pdist.neighbor_exchange(
            [j for j in out_nb if j not in byz] if i_am_byz else [],
            byz_in if not i_am_byz else [],
            send_vec,
        )

        if i_am_byz:
            return send_vec  # byzantine ranks do not update parameters

        vectors = [theta_half]
        for j in in_nb:
            vectors.append(recv_honest[j] if j not in byz else recv_byz[j])
        if self.pre_aggregator is not None:
            vectors = self.pre_aggregator.pre_aggregate(vectors)
        out = self.aggregator.aggregate(vectors)
        self.write_params_fn(out)
        return out
