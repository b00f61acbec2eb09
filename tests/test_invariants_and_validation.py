"""Mathematical invariants and parameter validation for every op family.

Mirrors the reference's exhaustive per-aggregator test style:
scale/translation equivariance, breakdown-point certificates, selection-set
properties, and the ValueError surface of each constructor.
"""
import math

import pytest
import torch

import byzpy_amd.ops.functional as F


def _X(n=9, d=11, seed=3):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, d, generator=g)


class TestEquivariance:
    """Robust aggregators commute with affine reparameterizations of the
    gradient space (textbook sanity properties)."""

    @pytest.mark.parametrize(
        "fn",
        [
            lambda X: F.median(X),
            lambda X: F.trimmed_mean(X, 2),
            lambda X: F.mean_of_medians(X, 2),
            lambda X: F.geometric_median(X),
            lambda X: F.cge(X, 2),
            lambda X: F.multi_krum(X, 2, 3),
        ],
        ids=["median", "trimmed", "meamed", "geomed", "cge", "multikrum"],
    )
    def test_positive_scale_equivariant(self, fn):
        X = _X()
        assert torch.allclose(fn(3.5 * X), 3.5 * fn(X), atol=1e-4)

    @pytest.mark.parametrize(
        "fn",
        [
            lambda X: F.median(X),
            lambda X: F.trimmed_mean(X, 2),
            lambda X: F.geometric_median(X),
            lambda X: F.multi_krum(X, 2, 3),
            lambda X: F.centered_clipping(X, c_tau=0.7),
        ],
        ids=["median", "trimmed", "geomed", "multikrum", "cc"],
    )
    def test_translation_equivariant(self, fn):
        X = _X()
        t = torch.randn(X.shape[1], generator=torch.Generator().manual_seed(7))
        assert torch.allclose(fn(X + t), fn(X) + t, atol=1e-4)

    def test_median_sign_odd(self):
        X = _X()
        assert torch.allclose(F.median(-X), -F.median(X), atol=1e-6)


class TestBreakdownCertificates:
    """With f adversarial rows and valid (n, f), the output must stay inside
    the honest envelope — the defining property of each rule."""

    def test_median_within_honest_envelope(self):
        n, f = 9, 4  # n = 2f + 1: maximal tolerated corruption
        honest = _X(n - f, 6)
        bad = torch.full((f, 6), 1e30)
        out = F.median(torch.cat([honest, bad]))
        lo, hi = honest.min(dim=0).values, honest.max(dim=0).values
        assert (out >= lo - 1e-5).all() and (out <= hi + 1e-5).all()

    def test_trimmed_mean_bounded_by_honest(self):
        n, f = 10, 3
        honest = _X(n - f, 5)
        bad = torch.full((f, 5), -1e28)
        out = F.trimmed_mean(torch.cat([bad, honest]), f)
        lo, hi = honest.min(dim=0).values, honest.max(dim=0).values
        assert (out >= lo - 1e-4).all() and (out <= hi + 1e-4).all()

    def test_cge_drops_all_large_norm_rows(self):
        honest = _X(6, 4)
        bad = 1e6 * torch.ones(2, 4)
        out = F.cge(torch.cat([honest, bad]), 2)
        assert torch.allclose(out, honest.mean(dim=0), atol=1e-4)

    def test_geomed_bounded_displacement(self):
        # geometric median moves < ||sum of adversarial unit pulls|| even
        # when f rows sit at 1e6: its distance to the honest geomed is
        # bounded, unlike the mean which diverges.
        honest = _X(7, 3)
        bad = torch.full((2, 3), 1e6)
        gm_h = F.geometric_median(honest, tol=1e-9)
        gm = F.geometric_median(torch.cat([honest, bad]), tol=1e-9)
        assert (gm - gm_h).norm() < 10 * honest.std()
        assert (torch.cat([honest, bad]).mean(0) - gm_h).norm() > 1e4


class TestSelectionSets:
    def test_multi_krum_winners_are_rows(self):
        X = _X(8, 5)
        q = 3
        out = F.multi_krum(X, 2, q)
        scores = F.multi_krum_scores(X, 2)
        keep = torch.topk(scores, q, largest=False).indices
        assert torch.allclose(out, X[keep].mean(dim=0), atol=1e-5)

    def test_krum_is_multikrum_q1(self):
        X = _X(8, 5)
        assert torch.allclose(F.krum(X, 2), F.multi_krum(X, 2, 1), atol=1e-6)

    def test_monna_averages_reference_neighborhood(self):
        X = _X(7, 4)
        out = F.monna(X, 2, reference_index=3)
        d = ((X - X[3]) ** 2).sum(dim=1)
        keep = torch.topk(d, 5, largest=False).indices
        assert torch.allclose(out, X[keep].mean(dim=0), atol=1e-5)

    def test_mda_subset_size_and_canonical(self):
        X = _X(7, 4)
        D2 = F.pairwise_sq_dists(X)
        sub = F.mda_subset(D2, 2)
        assert len(sub) == 5 and list(sub) == sorted(sub)

    def test_smea_f0_is_mean(self):
        X = _X(6, 3)
        assert torch.allclose(F.smea(X, 0), X.mean(dim=0), atol=1e-5)


class TestLittleVsScipy:
    def test_z_matches_norm_ppf(self):
        scipy_stats = pytest.importorskip("scipy.stats")
        n, f, d = 12, 3, 6
        honest = _X(n - f, d, seed=9)
        out = F.little(honest, f, N=n)
        s = n // 2 + 1 - f
        z = float(scipy_stats.norm.ppf((n - s) / n))
        mu, sigma = honest.mean(dim=0), honest.std(dim=0, unbiased=False)
        assert torch.allclose(out, mu + z * sigma, atol=1e-4)


class TestPreAggInvariants:
    def test_clip_rows_noop_below_threshold(self):
        X = 0.01 * _X()
        assert torch.allclose(F.clip_rows(X, 10.0), X)

    def test_arc_clip_matches_manual_threshold(self):
        n, f = 8, 2
        X = _X(n, 5)
        norms = X.norm(dim=1)
        k = int(2 * f / n * (n - f))  # rows to clip
        order = torch.argsort(norms, descending=True)
        thresh = norms[order[k]].item()
        out = F.arc_clip(X, f)
        out_norms = out.norm(dim=1)
        assert (out_norms <= thresh + 1e-4).all()
        untouched = order[k:]
        assert torch.allclose(out[untouched], X[untouched], atol=1e-6)

    def test_bucketing_remainder_bucket(self):
        X = _X(7, 3)
        out = F.bucketing(X, 3, perm=list(range(7)))
        assert out.shape[0] == 3
        assert torch.allclose(out[2], X[6:].mean(dim=0), atol=1e-6)

    def test_bucketing_b1_identity(self):
        X = _X(5, 4)
        out = F.bucketing(X, 1, perm=list(range(5)))
        assert torch.allclose(out, X)

    def test_nnm_f0_is_global_mean(self):
        X = _X(6, 4)
        out = F.nnm(X, 0)
        m = X.mean(dim=0)
        for row in out:
            assert torch.allclose(row, m, atol=1e-5)

    def test_nnm_preserves_row_count(self):
        X = _X(9, 4)
        assert F.nnm(X, 3).shape == X.shape


class TestCenteredClippingBehavior:
    def test_tau_zero_returns_init(self):
        X = _X(6, 4)
        out = F.centered_clipping(X, c_tau=0.0, init="median")
        assert torch.allclose(out, F.median(X), atol=1e-6)

    def test_large_tau_converges_to_mean(self):
        X = _X(6, 4)
        out = F.centered_clipping(X, c_tau=1e6, M=1, init="zero")
        assert torch.allclose(out, X.mean(dim=0), atol=1e-4)


class TestValidationSurface:
    @pytest.mark.parametrize(
        "call",
        [
            lambda: F.trimmed_mean(_X(4, 3), 2),     # n <= 2f
            lambda: F.mean_of_medians(_X(4, 3), 4),  # f >= n
            lambda: F.multi_krum_scores(_X(3, 3), 2),  # n - f - 1 < 1
            lambda: F.multi_krum(_X(6, 3), 1, 9),    # q > n
            lambda: F.cge(_X(4, 3), 4),
            lambda: F.nnm(_X(4, 3), 4),
            lambda: F.geometric_median(_X(4, 3), init="bogus"),
        ],
        ids=["trimmed", "meamed", "krum-scores", "krum-q", "cge", "nnm", "geomed-init"],
    )
    def test_functional_raises(self, call):
        with pytest.raises(ValueError):
            call()

    def test_operator_ctor_validation(self):
        from byzpy_amd.aggregators import (
            CenteredClipping,
            CoordinateWiseTrimmedMean,
            GeometricMedian,
            MultiKrum,
        )
        from byzpy_amd.attacks import GaussianAttack, LabelFlipAttack, MimicAttack
        from byzpy_amd.pre_aggregators import ARC, Bucketing, Clipping

        for bad in (
            lambda: CoordinateWiseTrimmedMean(-1),
            lambda: MultiKrum(-1, 2),
            lambda: MultiKrum(2, 0),
            lambda: GeometricMedian(tol=-1.0),
            lambda: GeometricMedian(init="nope"),
            lambda: CenteredClipping(c_tau=-0.5),
            lambda: CenteredClipping(c_tau=0.5, M=0),
            lambda: Clipping(-1.0),
            lambda: Bucketing(0),
            lambda: ARC(-1),
            lambda: GaussianAttack(sigma=-1.0),
            lambda: MimicAttack(epsilon=-1),
            lambda: LabelFlipAttack(),
        ):
            with pytest.raises(ValueError):
                bad()


class TestDtypePreservation:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.float64, torch.bfloat16])
    def test_aggregate_returns_input_dtype(self, dtype):
        from byzpy_amd.aggregators import CoordinateWiseMedian, MultiKrum

        grads = [torch.randn(33).to(dtype) for _ in range(6)]
        for agg in (CoordinateWiseMedian(), MultiKrum(1, 2)):
            out = agg.aggregate(grads)
            assert out.dtype == dtype and out.shape == (33,)

    def test_numpy_in_numpy_out(self):
        import numpy as np

        from byzpy_amd.aggregators import CoordinateWiseMedian

        grads = [torch.randn(17).numpy() for _ in range(5)]
        out = CoordinateWiseMedian().aggregate(grads)
        assert isinstance(out, np.ndarray) and out.shape == (17,)
