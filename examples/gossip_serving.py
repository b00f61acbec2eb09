"""Steady-state gossip serving loop on one MI355X (or CPU).

Round-2 features in one place:
  - RcclPeerToPeer semantics run single-process here: 8 node states
    resident on the device, ring(8, 2) topology;
  - poll-free GeometricMedian (`fixed_iters`) so the 8 node updates
    overlap on per-node HIP streams;
  - `init_z` warm starts: each node's center carries over between
    rounds (the fixed point barely moves per gossip round — measured
    2x faster AND tighter consensus than cold 8-iteration rounds).

Run:  python examples/gossip_serving.py [--rounds 50] [--d 1000000]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.engine.peer_to_peer.topology import Topology
from byzpy_amd.hip import dispatch as D


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--nodes", type=int, default=8)
    p.add_argument("--rounds", type=int, default=50)
    p.add_argument("--d", type=int, default=1_000_000)
    p.add_argument("--f", type=int, default=1)
    p.add_argument("--iters", type=int, default=4)
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    dev = torch.device("cuda" if use_cuda else "cpu")
    dtype = torch.bfloat16 if use_cuda else torch.float32
    if not use_cuda:
        args.d = min(args.d, 50_000)

    topo = Topology.ring(args.nodes, 2)
    g = torch.Generator(device=dev).manual_seed(0)
    theta = torch.empty(args.nodes, args.d, dtype=dtype, device=dev)
    theta.normal_(generator=g)
    groups = [
        torch.tensor([i] + topo.in_neighbors(i), device=dev)
        for i in range(args.nodes)
    ]
    streams = [torch.cuda.Stream() for _ in range(args.nodes)] if use_cuda else None
    warm: dict = {}

    def round_once() -> None:
        new = torch.empty_like(theta)

        def update(i: int) -> None:
            mixed = D.nnm(theta[groups[i]], args.f)
            out = D.geometric_median(
                mixed, fixed_iters=args.iters, init_z=warm.get(i)
            )
            warm[i] = out.float()
            new[i] = out

        if streams is None:
            for i in range(args.nodes):
                update(i)
        else:
            cur = torch.cuda.current_stream()
            for s in streams:
                s.wait_stream(cur)
            for i in range(args.nodes):
                with torch.cuda.stream(streams[i]):
                    update(i)
            for s in streams:
                cur.wait_stream(s)
        theta.copy_(new)

    round_once()  # warmup (first-call inits)
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.rounds):
        round_once()
    if use_cuda:
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.rounds
    spread = float(theta.float().std(dim=0).mean())
    print(
        f"gossip serving: {args.nodes} nodes, d={args.d}, "
        f"{dt * 1e3:.2f} ms/round, consensus spread {spread:.5f}"
    )


if __name__ == "__main__":
    main()
