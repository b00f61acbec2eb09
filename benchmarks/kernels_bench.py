"""Per-kernel microbenchmark: achieved bandwidth/latency of each gfx950
kernel at the headline shapes (evidence for profiles/).

  python benchmarks/kernels_bench.py --d 125000000
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch


def timeit(fn, reps=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=64)
    p.add_argument("--d", type=int, default=125_000_000)
    args = p.parse_args()
    assert torch.cuda.is_available()
    from byzpy_amd.hip import dispatch as D, require

    ext = require()
    n, d = args.n, args.d
    X = torch.empty(n, d, dtype=torch.bfloat16, device="cuda").normal_()
    Xf32 = torch.empty(n, min(d, 31_250_000), dtype=torch.float32, device="cuda").normal_()
    gb = n * d * 2 / 1e9
    rows = []

    def add(name, secs, bytes_gb, extra=""):
        rows.append((name, secs * 1e3, bytes_gb / secs, extra))

    add("median (packed-key bitonic, bf16)", timeit(lambda: D.median(X)), gb)
    add("trimmed_mean f=16 (bf16)", timeit(lambda: D.trimmed_mean(X, 16)), gb)
    add("meamed f=16 (bf16)", timeit(lambda: D.mean_of_medians(X, 16)), gb)
    add("gram (MFMA split-K, bf16)", timeit(lambda: ext.gram(X)), gb)
    add("gram (MFMA 16x16x4, f32)", timeit(lambda: ext.gram(Xf32)),
        Xf32.numel() * 4 / 1e9)
    add("row_sqnorms (bf16)", timeit(lambda: ext.row_sqnorms(X)), gb)
    idx = torch.arange(12, dtype=torch.int32, device="cuda")
    add("mean_rows q=12 (bf16)", timeit(lambda: ext.mean_rows(X, idx)),
        12 * d * 2 / 1e9)
    z = torch.zeros(d, dtype=torch.float32, device="cuda")
    sh = torch.zeros((), dtype=torch.float32, device="cuda")
    add("weiszfeld_iter (dist+update, bf16)",
        timeit(lambda: ext.weiszfeld_iter(X, z, 1e-12, sh)),
        (2 * n * d * 2 + d * 8) / 1e9)
    add("cc_iter (dist+update, bf16)",
        timeit(lambda: ext.cc_iter(X, z, 0.5, 1e-12)),
        (2 * n * d * 2 + d * 8) / 1e9)
    perm = torch.arange(n, dtype=torch.int32, device="cuda")
    add("bucket_mean b=4 (bf16)", timeit(lambda: ext.bucket_mean(X, perm, 4)),
        (n * d * 2 + (n // 4) * d * 2) / 1e9)
    G = ext.gram(X)
    add("krum_select n=64", timeit(lambda: ext.krum_select(G, 16, 12)), 0.0, "launch-bound")

    print(f"\nkernel microbenchmarks @ n={n}, d={d} (HBM peak ~8 TB/s, ~6.3 achievable)\n")
    print(f"{'kernel':44s} {'ms':>8s} {'GB/s':>8s}")
    for name, ms, gbs, extra in rows:
        print(f"{name:44s} {ms:8.3f} {gbs:8.0f} {extra}")


if __name__ == "__main__":
    main()
