"""Heterogeneous worker pool: CPU process workers + GPU stream workers in
one pool, with capability/affinity routing (reference README.md:174-180).

  python examples/heterogeneous_pool.py
"""
from __future__ import annotations

import argparse
import asyncio
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
from byzpy_amd.graph.subtask import SubTask


def cpu_preprocess(seed: int):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(4096, generator=g)
    return float(x.clamp(-1, 1).sum())


def gpu_reduce(seed: int):
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(1 << 20, generator=g).to(dev)
    return float((x * x).sum())


async def main() -> None:
    configs = [ActorPoolConfig(backend="process", count=2, name="cpu")]
    if torch.cuda.is_available():
        configs.append(ActorPoolConfig(backend="stream:0", count=2, name="gpu"))
    else:
        print("no GPU visible: gpu-capability tasks run on stream workers in CPU mode")
        configs.append(ActorPoolConfig(backend="stream", count=2, name="gpu"))
    pool = ActorPool(configs)
    await pool.start()
    print("workers:", pool.worker_affinities)

    cpu_tasks = [
        pool.run_subtask(SubTask(fn=cpu_preprocess, args=(s,), affinity="cpu"))
        for s in range(6)
    ]
    gpu_tasks = [
        pool.run_subtask(SubTask(fn=gpu_reduce, args=(s,), affinity="gpu"))
        for s in range(4)
    ]
    cpu_out = await asyncio.gather(*cpu_tasks)
    gpu_out = await asyncio.gather(*gpu_tasks)
    print("cpu results:", [round(v, 2) for v in cpu_out])
    print("gpu results:", [round(v, 2) for v in gpu_out])
    await pool.close()


if __name__ == "__main__":
    asyncio.run(main())
