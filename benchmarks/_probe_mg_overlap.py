"""Does median||gram two-stream overlap beat the serial pair now that the
median is VALU-bound (r01 tested this with the old memory-bound median
and found nothing)?"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

from byzpy_amd.hip import dispatch as D


def t(fn, iters=15):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


X = torch.randn(64, 125_000_000, device="cuda").to(torch.bfloat16)
print(f"median alone : {t(lambda: D.median(X)):.3f} ms")
print(f"gram alone   : {t(lambda: D.gram(X)):.3f} ms")
print(f"serial pair  : {t(lambda: (D.median(X), D.gram(X))):.3f} ms")

s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()


def overlapped():
    cur = torch.cuda.current_stream()
    e0 = torch.cuda.Event()
    e0.record(cur)
    with torch.cuda.stream(s1):
        s1.wait_event(e0)
        med = D.median(X)
    with torch.cuda.stream(s2):
        s2.wait_event(e0)
        G = D.gram(X)
    cur.wait_stream(s1)
    cur.wait_stream(s2)
    med.record_stream(cur)
    G.record_stream(cur)
    return med, G


print(f"overlap pair : {t(overlapped):.3f} ms")

m0, g0 = D.median(X), D.gram(X)
m1, g1 = overlapped()
torch.cuda.synchronize()
assert torch.equal(m0, m1) and torch.equal(g0, g1)
print("parity OK")
