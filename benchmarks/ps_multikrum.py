"""Reference-table PS workload ("ParameterServer (Multi-Krum)",
10 honest + 3 byzantine, 50 rounds, batch 64, f=3 q=6; reference best 42 ms
per round on an unspecified CPU with a x6 process pool).

Same shape here: 10 honest node actors each run fwd+bwd on a 784-128-10 MLP
(synthetic data, batch 64), 3 SignFlip byzantine workers, Multi-Krum(3, 6)
at the server. Reports mean ms/round.

  python benchmarks/ps_multikrum.py [--rounds 50] [--backend thread]
"""
from __future__ import annotations

import argparse
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
from torch import nn

from byzpy_amd.aggregators import MultiKrum
from byzpy_amd.attacks import SignFlipAttack
from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor
from byzpy_amd.engine.node.base import ByzantineNode, HonestNode
from byzpy_amd.engine.parameter_server.ps import ParameterServer


class MlpHonest(HonestNode):
    def __init__(self, seed: int) -> None:
        torch.manual_seed(seed)
        self.model = nn.Sequential(
            nn.Linear(784, 128), nn.ReLU(), nn.Linear(128, 10)
        )
        self.loss = nn.CrossEntropyLoss()
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(64, 784, generator=g)
        self.y = torch.randint(0, 10, (64,), generator=g)

    def next_batch(self):
        return self.x, self.y

    def honest_gradient(self, x, y):
        self.model.zero_grad(set_to_none=True)
        self.loss(self.model(x), y).backward()
        return torch.cat(
            [p.grad.reshape(-1) for p in self.model.parameters()]
        )

    def apply_server_gradient(self, grad) -> None:
        off = 0
        with torch.no_grad():
            for p in self.model.parameters():
                num = p.numel()
                p.add_(grad[off : off + num].reshape(p.shape), alpha=-0.05)
                off += num


class FlipByz(ByzantineNode):
    def __init__(self) -> None:
        self.attack = SignFlipAttack(scale=-4.0)

    def next_batch(self):
        return torch.empty(0), torch.empty(0)

    def byzantine_gradient(self, x, y, honest_grads=None):
        base = (
            torch.stack(honest_grads).mean(dim=0)
            if honest_grads
            else torch.zeros(1)
        )
        return self.attack.apply(base_grad=base)

    def apply_server_gradient(self, grad) -> None:
        pass


async def run(args: argparse.Namespace) -> float:
    honest = [
        await HonestNodeActor.spawn(MlpHonest, i, backend=args.backend)
        for i in range(10)
    ]
    byz = [
        await ByzantineNodeActor.spawn(FlipByz, backend=args.backend)
        for _ in range(3)
    ]
    ps = ParameterServer(honest, byz, MultiKrum(3, 6))
    for _ in range(args.warmup):
        await ps.round()
    t0 = time.perf_counter()
    for _ in range(args.rounds):
        await ps.round()
    ms = (time.perf_counter() - t0) / args.rounds * 1e3
    for nd in honest + byz:
        await nd.close()
    return ms


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--rounds", type=int, default=50)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--backend", default="thread")
    args = p.parse_args()
    ms = asyncio.run(run(args))
    print(
        f"PS Multi-Krum (10 honest + 3 byz, bs=64, f=3 q=6, "
        f"{args.backend} actors): {ms:.2f} ms/round  (reference best 42 ms)"
    )


if __name__ == "__main__":
    main()
