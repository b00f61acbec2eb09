"""Breakdown-point properties on the device kernels: with f byzantine
rows at extreme magnitudes (±1e30, mixed signs), each robust aggregator's
output must stay inside (or within a bounded factor of) the honest hull.
These are the guarantees the aggregator family exists to provide
(reference aggregators' robustness contracts) — checked on the actual
HIP kernels, not the torch oracle."""
import pytest
import torch

from byzpy_amd.hip import dispatch as D

pytestmark = pytest.mark.gpu


def _data(n, f, d, seed, dtype=torch.float32):
    g = torch.Generator().manual_seed(seed)
    honest = torch.randn(n - f, d, generator=g)
    byz = torch.randn(f, d, generator=g).sign() * 1e30
    X = torch.cat([honest, byz]).to("cuda", dtype)
    perm = torch.randperm(n, generator=g)
    return X[perm], honest


class TestBreakdownContainment:
    def test_median_within_honest_range(self):
        for seed in range(4):
            n, f, d = 25, 12, 4096  # f < n/2: median must stay honest
            X, honest = _data(n, f, d, seed)
            out = D.median(X).float().cpu()
            lo, hi = honest.min(dim=0).values, honest.max(dim=0).values
            assert (out >= lo - 1e-4).all() and (out <= hi + 1e-4).all()

    def test_trimmed_mean_within_honest_range(self):
        for seed in range(4):
            n, f, d = 24, 8, 4096
            X, honest = _data(n, f, d, seed)
            out = D.trimmed_mean(X, f).float().cpu()
            lo, hi = honest.min(dim=0).values, honest.max(dim=0).values
            assert (out >= lo - 1e-4).all() and (out <= hi + 1e-4).all()

    def test_meamed_bounded_by_honest_scale(self):
        for seed in range(4):
            n, f, d = 24, 8, 4096
            X, honest = _data(n, f, d, seed)
            out = D.mean_of_medians(X, f).float().cpu()
            assert torch.isfinite(out).all()
            assert float(out.abs().max()) <= float(honest.abs().max()) + 1e-4

    def test_multi_krum_selects_honest(self):
        for seed in range(4):
            n, f, d = 24, 7, 4096
            X, honest = _data(n, f, d, seed)
            out = D.multi_krum(X, f, 3).float().cpu()
            # mean of q honest rows: bounded by honest hull per coordinate
            lo, hi = honest.min(dim=0).values, honest.max(dim=0).values
            assert (out >= lo - 1e-3).all() and (out <= hi + 1e-3).all()

    def test_krum_winner_is_honest_row(self):
        for seed in range(4):
            n, f, d = 24, 7, 4096
            X, honest = _data(n, f, d, seed)
            out = D.krum(X, f).float().cpu()
            dists = (honest - out[None, :]).norm(dim=1)
            assert float(dists.min()) < 1e-3  # matches some honest row

    def test_cge_drops_large_norms(self):
        for seed in range(4):
            n, f, d = 24, 8, 4096
            X, honest = _data(n, f, d, seed)
            out = D.cge(X, f).float().cpu()
            assert float(out.norm()) <= float(honest.norm(dim=1).max()) + 1e-3

    def test_geometric_median_bounded(self):
        for seed in range(4):
            n, f, d = 25, 8, 4096  # f < n/2
            X, honest = _data(n, f, d, seed)
            out = D.geometric_median(X, max_iter=64).float().cpu()
            assert torch.isfinite(out).all()
            # geomed of a set with minority outliers stays within a few
            # honest radii of the honest centroid
            c = honest.mean(dim=0)
            r = (honest - c[None, :]).norm(dim=1).max()
            assert float((out - c).norm()) <= 4 * float(r)

    def test_clip_then_mean_bounded(self):
        for seed in range(2):
            n, f, d = 24, 8, 4096
            X, honest = _data(n, f, d, seed)
            clipped = D.clip_rows(X, 1.0)
            norms = D.row_sqnorms(clipped).sqrt().cpu()
            assert (norms <= 1.0 + 1e-3).all()

    def test_arc_caps_byzantine_norms(self):
        for seed in range(2):
            n, f, d = 24, 6, 4096
            X, honest = _data(n, f, d, seed)
            out = D.arc_clip(X, f)
            norms = D.row_sqnorms(out).sqrt().cpu()
            # the clipping threshold is the (k+1)-th largest norm, which
            # with 2f/n*(n-f) >= f is an honest-scale norm
            assert float(norms.max()) <= float(honest.norm(dim=1).max()) * 1.01
