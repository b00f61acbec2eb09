import os
import time

import torch

from byzpy_amd.hip import dispatch as D


def t(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


g = torch.Generator().manual_seed(5)
X = torch.randn(64, 1 << 20, generator=g).to("cuda")
for env, label in [({"BYZPY_CAF_SYNC": "1"}, "sync+pageable"),
                   ({"BYZPY_CAF_NOPIN": "1"}, "pipe+pageable"),
                   ({}, "pipe+pinned")]:
    for k in ("BYZPY_CAF_SYNC", "BYZPY_CAF_NOPIN"):
        os.environ.pop(k, None)
    os.environ.update(env)
    print(label, f"{t(lambda: D.caf(X, 16)):.3f} ms")
