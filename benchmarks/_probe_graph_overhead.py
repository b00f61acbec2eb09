"""Raw hipGraph (torch.cuda.CUDAGraph) replay overhead on this ROCm:
capture N small kernels, time replay vs eager launches of the same
work — settles whether the CAF graph-block slowness (262 us/round,
~12 nodes/round) was graph-launch overhead or something in our block."""
import time

import torch


def t(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


x = torch.randn(64, 64, device="cuda")
v = torch.randn(64, device="cuda")

for nodes in (12, 96):
    def work():
        a = v
        for _ in range(nodes // 3):
            s = x @ a            # gemv-ish
            a = (x.t() @ s)      # gemv-ish
            a = a / a.norm().clamp_min(1e-20)
        return a

    eager = t(work)
    g = torch.cuda.CUDAGraph()
    # warm + capture
    work()
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        out = work()
    replay = t(lambda: g.replay())
    print(f"nodes~{nodes}: eager {eager:.3f} ms  graph-replay {replay:.3f} ms "
          f"({replay/nodes*1000:.1f} us/node replay)")
