"""Multi-GPU parameter server over RCCL/xGMI (BASELINE config 3 shape).

One process per GPU; each rank hosts a ResNet-50-sized worker set
(8 honest + 3 SignFlip byzantine overall), gradients are d-sharded with
one all-to-all and TrimmedMean runs shard-local.

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/rccl_multigpu_ps.py
"""
from __future__ import annotations

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.engine.parameter_server.rccl import (
    RcclParameterServer,
    trimmed_mean_aggregate,
)
from byzpy_amd.parallel import dist as pdist

D_RESNET50 = 25_557_032  # ResNet-50 parameter count


def main() -> None:
    pdist.init_from_env()
    rank, world = pdist.get_rank(), pdist.get_world_size()
    device = torch.device("cuda", rank % max(1, torch.cuda.device_count())) if torch.cuda.is_available() else torch.device("cpu")
    d = D_RESNET50 if torch.cuda.is_available() else 100_000

    # workers per rank: honest everywhere, byzantine on ranks 0..2.
    # Gradients are generated ON DEVICE (a real worker's fwd/bwd would be
    # too): a CPU randn + H2D here costs ~80 ms per 25M-param worker.
    gen = torch.Generator(device=device).manual_seed(1234 + rank)

    def honest():
        return torch.randn(d, generator=gen, device=device)

    def signflip():
        return -2.0 * torch.randn(d, generator=gen, device=device)

    # 11 workers spread over the ranks: with fewer ranks each hosts more
    # workers; f scales so n > 2f holds at any world size
    per_rank = max(2, (11 + world - 1) // world)
    n_byz_left = 3
    fns = []
    for w in range(per_rank):
        is_byz = rank < 3 and w == 0 and n_byz_left > 0
        fns.append(signflip if is_byz else honest)
    f = min(3, (per_rank * world - 1) // 2)
    # overlap_chunks pipelines chunk i+1's all-to-all under chunk i's
    # aggregation kernels (coordinate-wise ops are chunkable)
    ps = RcclParameterServer(fns, trimmed_mean_aggregate(f=f), overlap_chunks=4)

    for _ in range(3):
        ps.round()  # warmup
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    pdist.barrier()
    t0 = time.perf_counter()
    steps = 10
    for _ in range(steps):
        ps.round()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    pdist.barrier()
    dt = (time.perf_counter() - t0) / steps
    if rank == 0:
        n = per_rank * world
        print(
            f"RCCL PS: {n} workers x d={d} on {world} rank(s): "
            f"{dt * 1000:.2f} ms/round ({n / dt:.0f} aggregated-grads/s)"
        )


if __name__ == "__main__":
    main()
