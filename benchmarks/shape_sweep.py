"""Kernel shape sweep: colsel family + gram across n (8..512) and dtypes —
robustness-of-performance evidence beyond the headline shape.

  python benchmarks/shape_sweep.py
"""
from __future__ import annotations

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
    assert torch.cuda.is_available()
    from byzpy_amd.hip import dispatch as D, require

    ext = require()
    print(f"{'op':16s} {'n':>4s} {'d':>10s} {'dtype':>5s} {'ms':>8s} {'GB/s':>7s}")
    for n, d in [(8, 500_000_000), (16, 250_000_000), (32, 125_000_000),
                 (64, 62_500_000), (128, 16_000_000), (256, 8_000_000),
                 (512, 4_000_000)]:
        for dtype, nm in [(torch.bfloat16, "bf16"), (torch.float32, "f32")]:
            bpe = 2 if dtype == torch.bfloat16 else 4
            X = torch.empty(n, d, dtype=dtype, device="cuda").normal_()
            gb = n * d * bpe / 1e9
            f = max(1, n // 4)
            for op, fn in [
                ("median", lambda X=X: D.median(X)),
                ("trimmed_mean", lambda X=X, f=f: D.trimmed_mean(X, f)),
                ("meamed", lambda X=X, f=f: D.mean_of_medians(X, f)),
                ("gram", lambda X=X: ext.gram(X.contiguous())),
            ]:
                ms = timeit(fn) * 1e3
                print(f"{op:16s} {n:4d} {d:10d} {nm:>5s} {ms:8.3f} {gb / ms * 1e3:7.0f}")
            del X
            torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
