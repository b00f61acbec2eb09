"""Benchmark suite replicating the reference's committed results table
(benchmarks/README.md of the reference) on CPU or MI355X.

Each workload runs the op's DIRECT path (on GPU: the HIP single-kernel
path — the reference's per-op pool-vs-direct decision disappears there,
BASELINE.md note) with a warmup+repeat harness and reports the best
latency next to the reference's published number (unspecified CPU).

  python benchmarks/run_suite.py --device cuda --repeat 10 --out results.md
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from typing import Any, Callable, Dict, List, Optional, Tuple

import torch

from byzpy_amd.aggregators import (
    CAF,
    CenteredClipping,
    ComparativeGradientElimination,
    CoordinateWiseMedian,
    CoordinateWiseTrimmedMean,
    GeometricMedian,
    MeanOfMedians,
    MinimumDiameterAveraging,
    MoNNA,
    MultiKrum,
    SMEA,
)
from byzpy_amd.attacks import EmpireAttack, GaussianAttack, LittleAttack
from byzpy_amd.pre_aggregators import ARC, Bucketing, Clipping, NearestNeighborMixing

# (name, n, d, runner-builder, reference best ms or None)
Workload = Tuple[str, int, int, Callable[[torch.Tensor], Any], Optional[float]]


def _agg(op):
    return ("agg", op, lambda X: op.aggregate(X))


def _pre(op):
    return ("pre", op, lambda X: op.pre_aggregate(X))


def _atk(op):
    return ("atk", op, lambda X: op.apply(honest_grads=X))


def workloads() -> List[Workload]:
    return [
        ("MDA n=30 d=2048 f=10", 30, 2048, _agg(MinimumDiameterAveraging(10)), 166.0),
        ("SMEA n=16 d=4096 f=5", 16, 4096, _agg(SMEA(5)), 48.0),
        ("ARC n=256 d=65536 f=8", 256, 65536, _pre(ARC(8)), 20.77),
        ("TrimmedMean n=64 d=65536 f=8", 64, 65536, _agg(CoordinateWiseTrimmedMean(8)), 15.15),
        ("Median n=64 d=65536", 64, 65536, _agg(CoordinateWiseMedian()), 37.0),
        ("MultiKrum n=80 d=65536 f=20 q=12", 80, 65536, _agg(MultiKrum(20, 12)), 26.3),
        ("GeoMedian n=64 d=65536", 64, 65536, _agg(GeometricMedian()), 142.97),
        ("CAF n=64 d=65536 f=8", 64, 65536, _agg(CAF(8)), 54.51),
        ("MoNNA n=64 d=65536 f=8", 64, 65536, _agg(MoNNA(8)), 11.0),
        ("CenteredClipping n=64 d=65536 M=10", 64, 65536, _agg(CenteredClipping(c_tau=0.1)), 50.0),
        ("CGE n=64 d=65536 f=8", 64, 65536, _agg(ComparativeGradientElimination(8)), 23.0),
        ("Empire n=64 d=65536", 64, 65536, _atk(EmpireAttack()), 14.0),
        ("Little n=96 d=65536 f=12", 96, 65536, _atk(LittleAttack(12)), 32.86),
        ("Gaussian n=64 d=65536", 64, 65536, _atk(GaussianAttack()), 12.3),
        ("NNM n=196 d=4096 f=32", 196, 4096, _pre(NearestNeighborMixing(32)), 12.0),
        ("MeaMed n=64 d=65536 f=8", 64, 65536, _agg(MeanOfMedians(8)), 59.0),
        ("Bucketing n=512 d=16384 b=32", 512, 16384, _pre(Bucketing(32)), 13.4),
        ("Clipping n=256 d=65536 thr=2", 256, 65536, _pre(Clipping(2.0)), 46.0),
    ]


def bench_one_pooled(op, kind, X, warmup, repeat, pool_backend, pool_n):
    """Reference-style ActorPool xN timing (CPU plumbing path)."""
    import asyncio

    from byzpy_amd.graph.executor import OperatorExecutor
    from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig

    key = {"agg": "gradients", "pre": "vectors", "atk": "honest_grads"}[kind]

    async def go():
        pool = ActorPool(ActorPoolConfig(backend=pool_backend, count=pool_n))
        await pool.start()
        ex = OperatorExecutor(op, pool=pool)
        inputs = {key: list(X)}
        for _ in range(warmup):
            await ex.run(inputs)
        best = float("inf")
        for _ in range(repeat):
            t0 = time.perf_counter()
            await ex.run(inputs)
            best = min(best, time.perf_counter() - t0)
        await pool.close()
        return best * 1000.0

    return asyncio.run(go())


def bench_one(run, X, warmup: int, repeat: int, device) -> float:
    for _ in range(warmup):
        run(X)
    if device.type == "cuda":
        torch.cuda.synchronize()
    best = float("inf")
    for _ in range(repeat):
        t0 = time.perf_counter()
        run(X)
        if device.type == "cuda":
            torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return best * 1000.0


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--repeat", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--dtype", default="f32", choices=["f32", "bf16"])
    p.add_argument("--out", default=None)
    p.add_argument("--only", default=None, help="substring filter")
    p.add_argument("--pool", type=int, default=0, help="ActorPool worker count (0 = direct)")
    p.add_argument("--pool-backend", default="process")
    args = p.parse_args()
    device = torch.device(args.device)
    dtype = torch.float32 if args.dtype == "f32" else torch.bfloat16

    rows = []
    for name, n, d, (kind, op, run), ref_ms in workloads():
        if args.only and args.only.lower() not in name.lower():
            continue
        g = torch.Generator().manual_seed(0)
        X = torch.randn(n, d, generator=g).to(device, dtype)
        try:
            if args.pool > 0:
                ms = bench_one_pooled(op, kind, X, args.warmup, args.repeat,
                                      args.pool_backend, args.pool)
            else:
                ms = bench_one(run, X, args.warmup, args.repeat, device)
            speed = f"{ref_ms / ms:.1f}x" if ref_ms else "-"
            rows.append((name, ms, ref_ms, speed))
            print(f"{name:42s} {ms:9.3f} ms   ref {ref_ms or '-':>8} ms   {speed}")
        except Exception as e:  # noqa: BLE001
            rows.append((name, None, ref_ms, f"ERROR {e!r}"))
            print(f"{name:42s} ERROR {e!r}")

    if args.out:
        with open(args.out, "w") as f:
            f.write(
                f"# byzpy_amd benchmark suite — device={args.device} dtype={args.dtype}\n\n"
                "Reference numbers: ByzPy's committed best-of-table latencies "
                "(BASELINE.md; unspecified CPU hardware).\n\n"
                "| workload | byzpy_amd (ms) | reference best (ms) | speedup |\n"
                "|---|---|---|---|\n"
            )
            for name, ms, ref_ms, speed in rows:
                ms_s = f"{ms:.3f}" if ms is not None else "error"
                f.write(f"| {name} | {ms_s} | {ref_ms or '-'} | {speed} |\n")
        print("wrote", args.out)


if __name__ == "__main__":
    main()
