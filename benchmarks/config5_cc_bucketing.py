"""BASELINE config 5 feasibility/perf: Llama-3-8B-sized flat gradients,
CenteredClipping + Bucketing, 32 synthetic workers.

Full config: 32 x 8B bf16 = 512 GB, d-sharded over 8 x 288 GB MI355X
(64 GB per GPU). This harness runs ONE rank's shard (d_local configurable;
the default 1e9 coordinates = 64 GB resident) — under torchrun it runs the
real sharded collectives per iteration.

Single GPU (64 GB shard):
  python benchmarks/config5_cc_bucketing.py --d-local 1000000000
Full 8-rank launch (8 x 64 GB = the 512 GB config):
  python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
      --nnodes=1 --nproc-per-node 8 benchmarks/config5_cc_bucketing.py \
      --d-local 1000000000
With d-sharded checkpoint round-trip (one shard file per rank):
  ... --checkpoint /tmp/c5ckpt
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.parallel import dist as pdist
from byzpy_amd.parallel import sharded


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=32)
    p.add_argument("--d-local", type=int, default=1_000_000_000)
    p.add_argument("--c-tau", type=float, default=1.0)
    p.add_argument("--M", type=int, default=10)
    p.add_argument("--bucket", type=int, default=4)
    p.add_argument("--repeat", type=int, default=3)
    p.add_argument("--checkpoint", default=None,
                   help="directory for a d-sharded aggregate checkpoint "
                        "round-trip (model.rank{r}.safetensors per rank)")
    args = p.parse_args()

    pdist.init_from_env()
    rank, world = pdist.get_rank(), pdist.get_world_size()
    device = torch.device("cuda", rank % max(1, torch.cuda.device_count())) if torch.cuda.is_available() else torch.device("cpu")
    if not torch.cuda.is_available():
        args.d_local = min(args.d_local, 200_000)

    gb = args.n * args.d_local * 2 / 1e9
    if rank == 0:
        print(f"allocating {args.n} x {args.d_local} bf16 = {gb:.1f} GB per rank "
              f"(global d = {args.d_local * world})")
    X = torch.empty((args.n, args.d_local), dtype=torch.bfloat16, device=device)
    X.normal_()

    def sync():
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        pdist.barrier()

    # bucketing pre-agg (identical perm on every rank) then centered clipping
    perm = list(range(args.n))

    def step():
        B = sharded.bucketing(X, args.bucket, perm)
        return sharded.centered_clipping(B, c_tau=args.c_tau, M=args.M)

    step()  # warmup
    sync()
    t0 = time.perf_counter()
    for _ in range(args.repeat):
        out = step()
    sync()
    dt = (time.perf_counter() - t0) / args.repeat
    if args.checkpoint:
        # d-sharded checkpoint: every rank writes its own shard file; no
        # rank ever funnels the full 512 GB state (utils/checkpoint.py)
        from byzpy_amd.utils.checkpoint import load_checkpoint, save_checkpoint

        t1 = time.perf_counter()
        save_checkpoint(
            args.checkpoint,
            round_idx=args.repeat,
            model_state={"aggregate_shard": out},
            rank=rank,
            world_size=world,
            extra_meta={"d_local": args.d_local, "n_workers": args.n},
        )
        pdist.barrier()
        back = load_checkpoint(args.checkpoint, rank=rank)
        got = back["model_state"]["aggregate_shard"].to(out.device)
        assert torch.equal(got, out), "checkpoint round-trip mismatch"
        if rank == 0:
            print(f"checkpoint round-trip ok: "
                  f"{(time.perf_counter() - t1):.2f} s for "
                  f"{out.numel() * out.element_size() / 1e9:.2f} GB/rank")
    if rank == 0:
        reads = (args.M * (args.n // args.bucket) + args.n) * args.d_local * 2 / 1e9
        print(f"bucketing+CC(M={args.M}): {dt*1000:.1f} ms/aggregate "
              f"({args.n / dt:.0f} aggregated-grads/s/rank; ~{reads / dt:.0f} GB/s streamed)")


if __name__ == "__main__":
    main()
