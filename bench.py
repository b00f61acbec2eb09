"""Flagship benchmark (driver contract).

Measures BASELINE.json's metric — aggregated-grads/sec for Median & Krum —
on config 2: n=64 workers x d=125M-param flat gradients, bf16, synthetic
random data. One step = one CoordinateWiseMedian aggregate + one Multi-Krum
aggregate over the resident (n, d) matrix (2n gradients aggregated/step).

Multi-GPU (launched by torch.distributed.run, one rank per GPU over RCCL):
the gradient matrix is d-sharded across ranks (byzpy_amd/parallel/sharded):
median is communication-free; Multi-Krum all-reduces the (n, n) partial
Gram (16 KB) and the result stays d-sharded. Total work is fixed as N
grows -> "strong" scaling.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--d D] [--n n]
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--n", type=int, default=64)
    # --dim alias: a bare "--d" after the script path trips torchrun's
    # argparse abbreviation check (--duplicate-*); --d kept for humans
    p.add_argument("--d", "--dim", dest="d", type=int, default=125_000_000)
    p.add_argument("--f", type=int, default=16)
    p.add_argument("--q", type=int, default=12)
    p.add_argument("--op", choices=["both", "median", "krum"], default="both")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "f32"])
    p.add_argument(
        "--overlap",
        action="store_true",
        help="run the two aggregates concurrently on two HIP streams "
        "(A/B'd neutral on MI355X: both kernels already fill the chip, so "
        "serial is the default and keeps profiles readable)",
    )
    return p.parse_args()


def main():
    args = parse_args()
    from byzpy_amd.parallel import dist as pdist
    from byzpy_amd.parallel import sharded

    pdist.init_from_env()
    rank = pdist.get_rank()
    world = pdist.get_world_size()
    # self-check: the measured world must be the one
    # the driver asked for — a silent single-rank fallback would report a
    # fake scaling point
    if args.gpus > 1 and world != args.gpus:
        raise SystemExit(
            f"bench: requested --gpus {args.gpus} but world size is {world} "
            "(launch via torch.distributed.run with --nproc-per-node)"
        )
    if "WORLD_SIZE" in os.environ and world != int(os.environ["WORLD_SIZE"]):
        raise SystemExit(
            f"bench: WORLD_SIZE={os.environ['WORLD_SIZE']} but process group "
            f"has {world} ranks"
        )
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device("cuda", rank % torch.cuda.device_count())
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    if not use_cuda and args.d > 2_000_000:
        # CPU smoke (no GPU in the dev container): shrink so it finishes
        args.d = 1_000_000

    d_local = (args.d + world - 1) // world
    n = args.n

    X = torch.empty((n, d_local), dtype=dtype, device=device)
    X.normal_(generator=None)

    overlap = (
        args.op == "both"
        and use_cuda
        and args.overlap
        and world == 1  # collectives and side streams don't mix safely
    )
    if overlap:
        s_med = torch.cuda.Stream()
        s_krum = torch.cuda.Stream()

    def step():
        outs = []
        if overlap:
            cur = torch.cuda.current_stream()
            s_med.wait_stream(cur)
            s_krum.wait_stream(cur)
            with torch.cuda.stream(s_med):
                outs.append(sharded.median(X))
            with torch.cuda.stream(s_krum):
                outs.append(sharded.multi_krum(X, args.f, args.q))
            cur.wait_stream(s_med)
            cur.wait_stream(s_krum)
            return outs
        if args.op == "both":
            med, krum_out = sharded.median_and_multi_krum(X, args.f, args.q)
            return [med, krum_out]
        if args.op in ("both", "median"):
            outs.append(sharded.median(X))
        if args.op in ("both", "krum"):
            outs.append(sharded.multi_krum(X, args.f, args.q))
        return outs

    def sync():
        if use_cuda:
            torch.cuda.synchronize()
        pdist.barrier()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0
    # max over ranks (RCCL needs a DEVICE tensor; gloo takes CPU)
    el = torch.tensor([elapsed], dtype=torch.float64, device=device if use_cuda else "cpu")
    if pdist.is_initialized():
        import torch.distributed as dist

        per_rank = pdist.all_gather_obj(elapsed)
        if rank == 0:
            import sys

            spread = (max(per_rank) - min(per_rank)) / max(per_rank)
            print(
                f"[bench] per-rank seconds: "
                f"{['%.4f' % s for s in per_rank]} (spread {spread:.1%})",
                file=sys.stderr,
            )
        dist.all_reduce(el, op=dist.ReduceOp.MAX)
    elapsed = float(el.item())

    aggs_per_step = n * (2 if args.op == "both" else 1)
    value = aggs_per_step * args.steps / elapsed
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "aggregated-grads/sec (n workers x d-dim, Median+Multi-Krum)",
                    "value": value,
                    "unit": "grads/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "strong",
                    "vs_baseline": None,
                    "dtype": args.dtype,
                    "data": "synthetic",
                    "config": {
                        "model": "Multi-Krum + CoordinateWiseMedian on 64x125M-param flat grads (BASELINE config 2)",
                        "n_workers": n,
                        "d": args.d,
                        "f": args.f,
                        "q": args.q,
                        "op": args.op,
                        "global_batch": n,
                        "seq_len": None,
                        "parallelism": f"dshard{world}",
                    },
                }
            )
        )
    if pdist.is_initialized():
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
