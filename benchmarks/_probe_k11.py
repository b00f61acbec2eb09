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
    # host-path on the SAME box for a fair comparison
    from byzpy_amd.hip import require
    from byzpy_amd.ops import functional as FF
    ext = require()
    def host_path():
        D2 = D.pairwise_sq_dists(X)
        idx = ext.mda_search(D2.detach().float().cpu(), 10)
        return D.mean_rows(X, idx.to("cuda", torch.int32))
    print(f"mda host-DFS same box: {timeit(host_path):.3f} ms")
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

def caf_small_d():
    import torch
    from byzpy_amd.hip import dispatch as D
    g = torch.Generator().manual_seed(1)
    for d in (2048, 8192, 32768):
        Z = torch.randn(32, d, generator=g).cuda()
        t_graph = timeit(lambda: D.caf(Z, 8))
        os.environ["BYZPY_CAF_GRAPH"] = "0"
        t_eager = timeit(lambda: D.caf(Z, 8))
        del os.environ["BYZPY_CAF_GRAPH"]
        print(f"caf n=32 d={d}: graph {t_graph:.3f} ms  eager {t_eager:.3f} ms")


def caf_debug():
    import torch
    from byzpy_amd.hip import dispatch as D
    g = torch.Generator().manual_seed(0)
    Z = torch.randn(64, 65536, generator=g).cuda()
    out = D.caf(Z, 16)
    key = (64, 65536, Z.dtype, 16, 3, 8, Z.device.index)
    blk = D._CAF_GRAPHS.get(key)
    if blk is not None:
        print(f"graph replays={blk.last_replays} wsum={float(blk.w.sum()):.2f} "
              f"best_lam={float(blk.best_lambda):.4f} active={bool(blk.active)} "
              f"rounds_left={float(blk.rounds_left)}")
        import time as _t
        torch.cuda.synchronize(); t0 = _t.perf_counter()
        for _ in range(10):
            blk.graph.replay()
        torch.cuda.synchronize()
        print(f"replay-only: {(_t.perf_counter()-t0)/10*1e3:.3f} ms per 8-round replay")
        t0 = _t.perf_counter()
        for _ in range(10):
            sd = torch.randn(4, 65536)
            blk.seeds[:4].copy_(sd)
        torch.cuda.synchronize()
        print(f"seed randn+stage (4 rows): {(_t.perf_counter()-t0)/10*1e3:.3f} ms")
        t0 = _t.perf_counter()
        for _ in range(10):
            _ = bool(blk.active)
        print(f"active sync read: {(_t.perf_counter()-t0)/10*1e3:.3f} ms")
    # eager-equivalent round count with plain torch math
    Xf = Z.float(); n = 64; target = float(n - 32)
    w = torch.ones(n, device="cuda")
    gen = torch.Generator(device="cpu"); gen.manual_seed(0)
    rounds = 0
    for r in range(n):
        if r % 2 == 0:
            seeds = torch.randn(2, 65536, generator=gen).cuda()
        wsum = w.sum()
        mu = (w[:, None] * Xf).sum(0) / wsum
        diffs = Xf - mu
        v = seeds[r % 2]; v = v / v.norm()
        for _ in range(3):
            t = (w * (diffs @ v)) @ diffs / wsum
            lam = t.norm(); v = t / lam.clamp_min(1e-20)
        proj = (diffs @ v) ** 2
        w_next = (w * (1 - proj / proj.max())).clamp_min(0)
        rounds += 1
        if float(wsum) <= target or float(w_next.sum()) <= 0:
            break
        w = w_next
    print(f"eager-equivalent rounds={rounds} final wsum={float(w.sum()):.2f}")


if __name__ == "__main__":
    caf_debug()
    caf_small_d()
