"""BASELINE config 1: CoordinateWiseMedian.aggregate on 10 x 1k-dim CPU
tensors via a thread ActorPool — pure plumbing latency, no GPU.

  python benchmarks/config1_thread_pool.py --workers 4
"""
from __future__ import annotations

import argparse
import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.aggregators import CoordinateWiseMedian
from byzpy_amd.graph.executor import OperatorExecutor
from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig


async def main(args: argparse.Namespace) -> None:
    g = torch.Generator().manual_seed(0)
    grads = [torch.randn(1000, generator=g) for _ in range(10)]
    agg = CoordinateWiseMedian(chunk_size=128)

    t0 = time.perf_counter()
    direct = agg.aggregate(grads)
    direct_ms = (time.perf_counter() - t0) * 1000

    pool = ActorPool(ActorPoolConfig(backend="thread", count=args.workers))
    await pool.start()
    ex = OperatorExecutor(agg, pool=pool)
    await ex.run({"gradients": grads})  # warmup
    best = float("inf")
    for _ in range(args.repeat):
        t0 = time.perf_counter()
        out = await ex.run({"gradients": grads})
        best = min(best, time.perf_counter() - t0)
    await pool.close()
    assert torch.allclose(out, direct, atol=1e-5)
    print(
        f"config1 median 10x1k: direct {direct_ms:.3f} ms; "
        f"thread-pool x{args.workers} {best*1000:.3f} ms/call"
    )


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--workers", type=int, default=4)
    p.add_argument("--repeat", type=int, default=20)
    asyncio.run(main(p.parse_args()))
