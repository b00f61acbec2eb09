"""Seeded random-shape fuzz of the dispatch layer against the functional
oracle (SURVEY.md §4 pattern: every HIP kernel numerics-tested against a
plain torch fp32 reference). Shapes are drawn to cross every dispatch
boundary: register kernels (n<=64, all pad counts), LDS sort (65..512 at
small d), the rsel radix engine (large d), odd/even d (packed pair vs
scalar bf16 path), and both dtypes."""
import pytest
import torch

import byzpy_amd.ops.functional as F
from byzpy_amd.hip import dispatch as D

pytestmark = pytest.mark.gpu

_RNG = torch.Generator().manual_seed(20260914)


def _rand_shape(i):
    # sweep the dispatch boundaries deterministically-ish
    ns = [1, 2, 3, 5, 8, 9, 16, 17, 31, 32, 33, 48, 63, 64, 65, 100, 128,
          129, 200, 256, 400, 512]
    n = ns[i % len(ns)]
    d = int(torch.randint(1, 5000, (1,), generator=_RNG)) + (i % 3)
    return n, d


class TestDispatchFuzz:
    def test_colsel_modes(self):
        for i in range(60):
            n, d = _rand_shape(i)
            dtype = torch.bfloat16 if i % 2 == 0 else torch.float32
            X = torch.randn(n, d, generator=_RNG).to("cuda", dtype)
            f = int(torch.randint(0, max(1, (n - 1) // 2 + 1), (1,), generator=_RNG))
            Xc = X.float().cpu()
            med = D.median(X)
            assert torch.allclose(
                med.float().cpu(), F.median(Xc).to(dtype).float(),
                atol=1e-6, rtol=0
            ), (i, n, d, dtype, "median")
            if n - 2 * f >= 1:
                tm = D.trimmed_mean(X, f)
                ref = F.trimmed_mean(Xc, f)
                tol = 3e-2 if dtype == torch.bfloat16 else 1e-4
                assert torch.allclose(
                    tm.float().cpu(), ref, atol=tol, rtol=tol
                ), (i, n, d, dtype, f, "trimmed")
            if n - f >= 1 and f > 0:
                # meamed keep-set is ambiguous when the (n-f)-th and
                # (n-f+1)-th smallest |v - med| tie (kernel: drop-right-
                # first; oracle: row order) — compare non-tied columns
                mm = D.mean_of_medians(X, f)
                ref = F.mean_of_medians(Xc, f)
                dev = (Xc - F.median(Xc)[None, :]).abs()
                sd, _ = dev.sort(dim=0)
                clear = (sd[n - f] - sd[n - f - 1]) > 1e-5
                tol = 3e-2 if dtype == torch.bfloat16 else 1e-4
                assert torch.allclose(
                    mm.float().cpu()[clear], ref[clear], atol=tol, rtol=tol
                ), (i, n, d, dtype, f, "meamed")

    def test_gram_and_krum(self):
        for i in range(25):
            n = int(torch.randint(2, 300, (1,), generator=_RNG))
            d = int(torch.randint(8, 3000, (1,), generator=_RNG))
            dtype = torch.bfloat16 if i % 2 == 0 else torch.float32
            X = torch.randn(n, d, generator=_RNG).to("cuda", dtype)
            G = D.gram(X)
            Xf = X.float()
            ref = (Xf @ Xf.T).cpu()
            scale = max(1.0, float(ref.abs().max()))
            tol = 2e-2 if dtype == torch.bfloat16 else 2e-5
            assert torch.allclose(
                G.cpu() / scale, ref / scale, atol=tol, rtol=0
            ), (i, n, d, dtype, "gram")
            f = max(0, min((n - 2) // 2, int(torch.randint(0, n, (1,), generator=_RNG)) // 3))
            q = max(1, min(n - f, 1 + i % 5))
            if n - f - 1 >= 1:
                # skip configs where the q-th score boundary is a near-tie
                # (fp association differences could flip the winner set)
                scores = F.multi_krum_scores(X.float().cpu(), f)
                srt = scores.sort().values
                if q < n and float(srt[q] - srt[q - 1]) < 1e-3 * max(
                    1.0, float(srt[q].abs())
                ):
                    continue
                mk = D.multi_krum(X, f, q)
                ref_mk = F.multi_krum(X.float().cpu(), f, q)
                tol = 3e-2 if dtype == torch.bfloat16 else 1e-4
                assert torch.allclose(
                    mk.float().cpu(), ref_mk, atol=tol, rtol=tol
                ), (i, n, d, dtype, f, q, "multi_krum")

    def test_weiszfeld_fixed_iters(self):
        for i in range(8):
            n = int(torch.randint(3, 200, (1,), generator=_RNG))
            d = int(torch.randint(16, 2000, (1,), generator=_RNG))
            X = torch.randn(n, d, generator=_RNG).to("cuda")
            it = 5 + i
            out = D.geometric_median(X, fixed_iters=it)
            ref = F.geometric_median(
                X.float().cpu(), tol=0.0, max_iter=it
            )
            assert torch.allclose(out.cpu(), ref, atol=1e-3, rtol=1e-3), (i, n, d)
