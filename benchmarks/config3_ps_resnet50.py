"""BASELINE config 3: parameter-server round on ResNet-50-sized gradients
(25.6M params), 8 honest + 3 SignFlip byzantine workers, TrimmedMean f=3.

Single process = single GPU (the full RCCL path is the same code under
torchrun; see tests/test_dist_gloo.py::test_rccl_parameter_server for the
multi-rank collective pattern). Reports ms per PS round, where a round =
11 gradient productions + trimmed-mean aggregate + broadcast-apply.

  python benchmarks/config3_ps_resnet50.py [--rounds 20] [--device cuda]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.engine.parameter_server.rccl import (
    RcclParameterServer,
    trimmed_mean_aggregate,
)

D_RESNET50 = 25_610_152  # ResNet-50 ImageNet parameter count


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--rounds", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--d", type=int, default=D_RESNET50)
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()
    dev = torch.device(args.device)
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    if dev.type == "cpu":
        args.d = min(args.d, 1_000_000)

    g = torch.Generator(device=dev).manual_seed(0)
    base = torch.empty(args.d, dtype=dtype, device=dev).normal_(generator=g)

    def honest(i):
        noise = torch.empty(args.d, dtype=dtype, device=dev).normal_(
            generator=g, std=0.1
        )

        def fn():
            return base + noise

        return fn

    def byz():
        def fn():
            return -8.0 * base  # sign-flip, scaled

        return fn

    fns = [honest(i) for i in range(8)] + [byz() for _ in range(3)]
    ps = RcclParameterServer(fns, trimmed_mean_aggregate(3), gather_result=True)

    def sync():
        if dev.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        ps.round()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.rounds):
        out = ps.round()
    sync()
    ms = (time.perf_counter() - t0) / args.rounds * 1e3
    err = float((out.float() - base.float()).norm() / base.float().norm())
    print(
        f"config3 PS round (8 honest + 3 signflip, trimmed f=3, d={args.d}, "
        f"{dtype}): {ms:.3f} ms/round  rel-err vs honest base {err:.3f}"
    )


if __name__ == "__main__":
    main()
