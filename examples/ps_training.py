"""Parameter-server robust training demo (synthetic data, no downloads).

Mirrors the reference's examples/ps/*/mnist.py structure (SmallCNN +
honest/byzantine node actors + robust aggregation) with a synthetic
classification set. Actor backend selectable: thread | process | stream
(HIP stream workers on a GPU box).

  python examples/ps_training.py --backend thread --rounds 20
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import asyncio

import torch
from torch import nn

from byzpy_amd.aggregators import CoordinateWiseTrimmedMean
from byzpy_amd.attacks import SignFlipAttack
from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor
from byzpy_amd.engine.parameter_server.ps import ParameterServer
from byzpy_amd.utils.training import train_with_progress


class SmallCNN(nn.Module):
    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(1, 8, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),
            nn.Conv2d(8, 16, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),
            nn.Flatten(),
            nn.Linear(16 * 7 * 7, num_classes),
        )

    def forward(self, x):
        return self.net(x)


class HonestWorker:
    def __init__(self, seed: int, device: str = "cpu", batch_size: int = 32):
        torch.manual_seed(0)  # same init on every worker
        self.device = torch.device(device if torch.cuda.is_available() else "cpu")
        self.model = SmallCNN().to(self.device)
        self.loss_fn = nn.CrossEntropyLoss()
        self.batch_size = batch_size
        self.gen = torch.Generator().manual_seed(seed)
        self.lr = 0.05

    def next_batch(self):
        x = torch.randn(self.batch_size, 1, 28, 28, generator=self.gen).to(self.device)
        # synthetic labels correlated with the input so learning happens
        y = (x.mean(dim=(1, 2, 3)) * 20).long().remainder(10).to(self.device)
        return x, y

    def honest_gradient_for_next_batch(self):
        x, y = self.next_batch()
        self.model.zero_grad(set_to_none=True)
        loss = self.loss_fn(self.model(x), y)
        loss.backward()
        return torch.cat(
            [p.grad.reshape(-1) for p in self.model.parameters()]
        ).cpu()

    def apply_server_gradient(self, g):
        g = torch.as_tensor(g).to(self.device)
        off = 0
        with torch.no_grad():
            for p in self.model.parameters():
                num = p.numel()
                p.add_(g[off : off + num].reshape(p.shape), alpha=-self.lr)
                off += num

    def eval_loss(self):
        x, y = self.next_batch()
        with torch.no_grad():
            return float(self.loss_fn(self.model(x), y))


class ByzantineWorker:
    def __init__(self):
        self.attack = SignFlipAttack(scale=-2.0)

    def byzantine_gradient_for_next_batch(self, honest_grads=None):
        base = torch.stack([torch.as_tensor(g) for g in honest_grads]).mean(dim=0)
        return self.attack.apply(base_grad=base)

    def apply_server_gradient(self, g):
        pass


async def main(args: argparse.Namespace) -> None:
    honest = [
        await HonestNodeActor.spawn(HonestWorker, s, args.device, backend=args.backend)
        for s in range(args.honest)
    ]
    byz = [
        await ByzantineNodeActor.spawn(ByzantineWorker, backend=args.backend)
        for _ in range(args.byzantine)
    ]
    ps = ParameterServer(honest, byz, CoordinateWiseTrimmedMean(f=args.byzantine))

    async def _eval(r):
        return {"loss": round(await honest[0].eval_loss(), 4)}

    before = await honest[0].eval_loss()
    await train_with_progress(ps, args.rounds, progress=True)
    after = await honest[0].eval_loss()
    print(f"loss: {before:.4f} -> {after:.4f}")
    for a in honest + byz:
        await a.close()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--backend", default="thread", help="thread|process|stream")
    p.add_argument("--device", default="cpu")
    p.add_argument("--honest", type=int, default=6)
    p.add_argument("--byzantine", type=int, default=2)
    p.add_argument("--rounds", type=int, default=20)
    asyncio.run(main(p.parse_args()))
