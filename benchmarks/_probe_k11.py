"""Timing probe: device MDA/SMEA (K11) + graph-captured CAF vs round-1
numbers (6.05 / 1.92 / 6.84 ms)."""
import sys, os, time
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from byzpy_amd.hip import dispatch as D

def timeit(fn, reps=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e3

def main():
    g = torch.Generator().manual_seed(0)
    # MDA benchmark shape (n=30 d=2048 f=10)
    X = torch.randn(30, 2048, generator=g).cuda()
    print(f"mda n=30 d=2048 f=10: {timeit(lambda: D.minimum_diameter_averaging(X, 10)):.3f} ms (r01: 6.05)")
    Xb = X.bfloat16()
    print(f"mda bf16 same:        {timeit(lambda: D.minimum_diameter_averaging(Xb, 10)):.3f} ms")
    # SMEA shape from r01 results (n=16 f=3)
    Y = torch.randn(16, 65536, generator=g).cuda()
    print(f"smea n=16 d=65536 f=3: {timeit(lambda: D.smea(Y, 3)):.3f} ms (r01: 1.92)")
    # CAF 64x65k (r01: 6.84; target <= 2)
    Z = torch.randn(64, 65536, generator=g).cuda()
    print(f"caf n=64 d=65536 f=16: {timeit(lambda: D.caf(Z, 16)):.3f} ms (r01: 6.84)")
    os.environ["BYZPY_CAF_GRAPH"] = "0"
    print(f"caf eager (same):      {timeit(lambda: D.caf(Z, 16)):.3f} ms")
    del os.environ["BYZPY_CAF_GRAPH"]

if __name__ == "__main__":
    main()
