"""BASELINE config 4: P2P ring gossip with NNM pre-aggregation +
GeometricMedian, 8 nodes.

Single process = all 8 node states resident on one GPU; one round = every
node aggregates NNM(GeoMedian) over [self] + its ring neighbors' theta-half
vectors. The multi-rank variant runs the identical aggregate per rank after
an RCCL all-gather (engine/peer_to_peer/rccl.py; gloo-ws2 covered by
tests/test_dist_gloo.py::test_rccl_peer_to_peer). Reports ms per full gossip
round (all 8 node updates).

  python benchmarks/config4_p2p_ring.py [--rounds 10] [--d 25610152]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from byzpy_amd.hip import dispatch as D
from byzpy_amd.engine.peer_to_peer.topology import Topology


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--rounds", type=int, default=10)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--nodes", type=int, default=8)
    p.add_argument("--k", type=int, default=2, help="ring degree (each side)")
    p.add_argument("--d", type=int, default=25_610_152)
    p.add_argument("--f", type=int, default=1)
    p.add_argument("--streams", action="store_true",
                   help="one HIP stream per node update (overlap)")
    p.add_argument("--fixed-iters", type=int, default=16,
                   help="poll-free Weiszfeld iterations (0 = tol/poll mode)")
    p.add_argument("--warm-start", action="store_true",
                   help="carry each node's center across rounds (the "
                        "fixed point barely moves per gossip round)")
    p.add_argument("--graph", action="store_true",
                   help="capture the whole gossip round in one hipGraph")
    p.add_argument("--grouped", action="store_true",
                   help="batch ALL node updates into grouped kernels "
                        "(one launch pair per Weiszfeld iteration)")
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()
    dev = torch.device(args.device)
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    if dev.type == "cpu":
        args.d = min(args.d, 200_000)

    topo = Topology.ring(args.nodes, args.k)
    g = torch.Generator(device=dev).manual_seed(0)
    theta = torch.empty(args.nodes, args.d, dtype=dtype, device=dev).normal_(
        generator=g
    )

    streams = (
        [torch.cuda.Stream() for _ in range(args.nodes)]
        if (args.streams and dev.type == "cuda")
        else None
    )

    fixed = args.fixed_iters if args.fixed_iters > 0 else None
    warm: dict = {}

    def geomed(mixed, i):
        if args.warm_start and i in warm:
            out = D.geometric_median(mixed, fixed_iters=fixed or 4,
                                     init_z=warm[i])
        elif fixed is not None:
            # poll-free: no host sync, so per-node streams truly overlap
            out = D.geometric_median(mixed, fixed_iters=fixed)
        else:
            out = D.geometric_median(mixed, tol=1e-6, max_iter=32)
        if args.warm_start:
            warm[i] = out.float()
        return out

    # index tensors hoisted to device once: a python-list fancy-index does
    # a pageable H2D copy per call, which is also capture-UNSAFE (the
    # --graph mode died on hipErrorStreamCaptureUnsupported without this)
    groups = [
        torch.tensor([i] + topo.in_neighbors(i), device=dev)
        for i in range(args.nodes)
    ]

    group_mat = torch.stack(groups)  # (nodes, 1+2k)
    warm_grouped: dict = {}

    def round_once_grouped():
        Xg = theta[group_mat]  # (G, m, d)
        mixed = D.nnm_grouped(Xg, args.f)
        Z = D.geometric_median_grouped(
            mixed,
            iters=(fixed or 4) if args.warm_start and "Z" in warm_grouped
            else (fixed or 8),
            init_z=warm_grouped.get("Z") if args.warm_start else None,
        )
        if args.warm_start:
            warm_grouped["Z"] = Z
        theta.copy_(Z.to(theta.dtype))

    def round_once():
        if args.grouped:
            round_once_grouped()
            return
        new = torch.empty_like(theta)
        if streams is None:
            for i in range(args.nodes):
                X = theta[groups[i]]  # (1+2k, d) gather
                mixed = D.nnm(X, args.f)
                new[i] = geomed(mixed, i)
        else:
            # node updates are independent: one HIP stream each, so the
            # launch/sync-bound small-kernel chains overlap on the chip
            cur = torch.cuda.current_stream()
            for s in streams:
                s.wait_stream(cur)
            for i in range(args.nodes):
                with torch.cuda.stream(streams[i]):
                    X = theta[groups[i]]
                    mixed = D.nnm(X, args.f)
                    new[i] = geomed(mixed, i)
            for s in streams:
                cur.wait_stream(s)
        theta.copy_(new)

    if args.graph and dev.type == "cuda":
        if fixed is None:
            raise SystemExit("--graph requires --fixed-iters > 0 (capture-safe)")
        if args.grouped:
            raise SystemExit("--graph and --grouped are mutually exclusive")
        # capture one full gossip round (all node updates + the theta swap)
        # and replay it per round: zero Python dispatch in the loop
        s0 = torch.cuda.Stream()
        s0.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s0):
            round_once()
        torch.cuda.current_stream().wait_stream(s0)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            round_once()
        round_replay = g.replay

        def round_once():  # noqa: F811 — replayed capture
            round_replay()

    def sync():
        if dev.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        round_once()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.rounds):
        round_once()
    sync()
    ms = (time.perf_counter() - t0) / args.rounds * 1e3
    spread = float((theta.float().std(dim=0)).mean())
    print(
        f"config4 P2P ring({args.nodes},{args.k}) NNM+GeoMedian d={args.d} "
        f"{dtype}: {ms:.3f} ms/round (all {args.nodes} node updates); "
        f"consensus spread {spread:.4f}"
    )


if __name__ == "__main__":
    main()
