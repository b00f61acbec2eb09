"""Decentralized gossip training demo (ring topology, byzantine peer).

  python examples/p2p_gossip.py --rounds 40
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import asyncio

import torch
from torch import nn

from byzpy_amd.aggregators import GeometricMedian
from byzpy_amd.attacks import EmpireAttack
from byzpy_amd.engine.peer_to_peer.mixin import P2PByzantineMixin, P2PHonestMixin
from byzpy_amd.engine.peer_to_peer.topology import Topology
from byzpy_amd.engine.peer_to_peer.train import PeerToPeer
from byzpy_amd.pre_aggregators import NearestNeighborMixing


class Regressor(P2PHonestMixin):
    def __init__(self, seed: int):
        torch.manual_seed(0)
        self.model = nn.Linear(8, 1, bias=False)
        self.lr = 0.05
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(64, 8, generator=g)
        self.w_true = torch.randn(1, 8, generator=torch.Generator().manual_seed(99))
        self.y = self.x @ self.w_true.T

    def p2p_local_loss_backward(self):
        ((self.model(self.x) - self.y) ** 2).mean().backward()

    def loss(self):
        with torch.no_grad():
            return float(((self.model(self.x) - self.y) ** 2).mean())


class Saboteur(P2PByzantineMixin):
    def __init__(self):
        self.attack = EmpireAttack(scale=-1.5)


async def main(args):
    honest = [Regressor(s) for s in range(args.honest)]
    byz = [Saboteur() for _ in range(args.byzantine)]
    n = len(honest) + len(byz)
    p2p = PeerToPeer(
        honest,
        byz,
        GeometricMedian(),
        pre_aggregator=NearestNeighborMixing(f=args.byzantine),
        topology=Topology.ring(n, 2),
        lr=0.05,
    )
    await p2p.bootstrap()
    print("initial loss:", round(honest[0].loss(), 4))
    for r in range(args.rounds):
        await p2p.round()
    print("final loss:", round(honest[0].loss(), 4))
    await p2p.shutdown()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--honest", type=int, default=6)
    p.add_argument("--byzantine", type=int, default=1)
    p.add_argument("--rounds", type=int, default=40)
    asyncio.run(main(p.parse_args()))
