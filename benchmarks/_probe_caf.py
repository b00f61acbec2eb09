"""Time D.caf vs the functional oracle and assert bitwise parity of the
pipelined eager loop (round-2 late: pinned seed staging + one-round-deep
stat readback pipelining)."""
import time

import torch

import byzpy_amd.ops.functional as F
from byzpy_amd.hip import dispatch as D


def t(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    g = torch.Generator().manual_seed(5)
    for (n, d, f) in [(64, 65536, 16), (64, 65536, 8), (128, 65536, 32),
                      (64, 1 << 20, 16), (256, 16384, 64)]:
        X = torch.randn(n, d, generator=g).to("cuda")
        out = D.caf(X, f)
        ref = F.caf(X.cpu(), f)
        err = (out.cpu() - ref).abs().max().item()
        import os
        os.environ["BYZPY_CAF_SYNC"] = "1"
        ms_sync = t(lambda: D.caf(X, f), iters=10)
        out_sync = D.caf(X, f)
        del os.environ["BYZPY_CAF_SYNC"]
        ms = t(lambda: D.caf(X, f), iters=10)
        assert torch.equal(out, out_sync), "pipelined != sync loop"
        print(f"caf n={n} d={d} f={f}: pipe {ms:.3f} ms  sync {ms_sync:.3f} ms  max|err|={err:.2e}")
        assert err < 1e-3, "parity broke"
    print("CAF_PROBE_OK")


if __name__ == "__main__":
    main()
