"""Hand-computed known-answer fixtures for every op family (reference
style: per-op test files with explicit expected values — the expected
tensors here are written out numerically, never computed by the op)."""
import math

import pytest
import torch

import byzpy_amd.ops.functional as F


def T(*rows):
    return torch.tensor(rows, dtype=torch.float32)


class TestCoordinateWiseHand:
    def test_median_3x2(self):
        X = T([1.0, 2.0], [3.0, 4.0], [100.0, 0.0])
        assert torch.equal(F.median(X), torch.tensor([3.0, 2.0]))

    def test_median_even_rows_mean_of_middles(self):
        X = T([1.0], [2.0], [3.0], [10.0])
        assert torch.equal(F.median(X), torch.tensor([2.5]))

    def test_trimmed_mean_f1(self):
        X = T([0.0], [1.0], [2.0], [9.0])
        # drop min 0 and max 9 -> mean(1, 2)
        assert torch.equal(F.trimmed_mean(X, 1), torch.tensor([1.5]))

    def test_meamed_f1(self):
        X = T([0.0], [10.0], [11.0], [12.0])
        # median = 10.5; the 3 closest are 10, 11, 12 -> mean 11
        assert torch.allclose(F.mean_of_medians(X, 1), torch.tensor([11.0]))


class TestGeometricHand:
    def test_krum_scores_3_points(self):
        # points 0, 1, 10 on a line; f=1 -> k = n-f-1 = 1 nearest
        X = T([0.0], [1.0], [10.0])
        s = F.multi_krum_scores(X, 1)
        # score_i = dist^2 to single nearest: [1, 1, 81]
        assert torch.allclose(s, torch.tensor([1.0, 1.0, 81.0]))

    def test_krum_picks_cluster(self):
        X = T([0.0, 0.0], [0.1, 0.0], [0.0, 0.1], [5.0, 5.0])
        w = F.krum(X, 1)
        assert w[0] < 1.0 and w[1] < 1.0  # a cluster member, not (5,5)

    def test_geomed_collinear_is_median_point(self):
        # 1-D geometric median of {0, 1, 10} minimizes sum |x-a| -> x = 1
        X = T([0.0], [1.0], [10.0])
        gm = F.geometric_median(X, tol=1e-10)
        assert abs(float(gm) - 1.0) < 1e-3

    def test_mda_exact_subset(self):
        # 4 points; the tight pair {0, 1} plus nearest third
        X = T([0.0], [0.5], [10.0], [20.0])
        D2 = F.pairwise_sq_dists(X)
        assert F.mda_subset(D2, 2) == (0, 1)

    def test_monna_reference_zero(self):
        X = T([0.0], [1.0], [2.0], [50.0])
        # 2 nearest to x_0 (including itself): {0, 1} -> mean 0.5
        out = F.monna(X, 2, reference_index=0)
        assert torch.allclose(out, torch.tensor([0.5]))

    def test_smea_picks_low_variance(self):
        X = T([0.0], [0.1], [0.2], [9.0])
        out = F.smea(X, 1)
        # subset {0, .1, .2} has much smaller max eigenvalue than any with 9
        assert abs(float(out) - 0.1) < 1e-4


class TestNormWiseHand:
    def test_cge_keeps_smallest_norms(self):
        X = T([1.0, 0.0], [0.0, 1.0], [3.0, 4.0])
        # norms 1, 1, 5; f=1 drops (3,4) -> mean([1,0],[0,1]) = (.5,.5)
        assert torch.allclose(F.cge(X, 1), torch.tensor([0.5, 0.5]))

    def test_cc_single_iteration_hand(self):
        # v0 = 0 (init zero); tau = 1: each clip coef = min(1, 1/|x|)
        # x = {2, -4}: clipped diffs = {1, -1}; v1 = 0 + (1 - 1)/2 = 0
        X = T([2.0], [-4.0])
        out = F.centered_clipping(X, c_tau=1.0, M=1, init="zero")
        assert torch.allclose(out, torch.tensor([0.0]), atol=1e-6)

    def test_cc_one_sided(self):
        # x = {2, 4}, tau=1, v0=0: diffs 2,4 -> coefs .5,.25 -> clipped 1,1
        # v1 = (1+1)/2 = 1
        X = T([2.0], [4.0])
        out = F.centered_clipping(X, c_tau=1.0, M=1, init="zero")
        assert torch.allclose(out, torch.tensor([1.0]), atol=1e-6)


class TestPreAggHand:
    def test_clipping_exact_scale(self):
        X = T([3.0, 4.0], [0.3, 0.4])
        out = F.clip_rows(X, 1.0)
        # row0 norm 5 -> scaled by 1/5; row1 norm .5 untouched
        assert torch.allclose(out, T([0.6, 0.8], [0.3, 0.4]), atol=1e-6)

    def test_bucketing_means(self):
        X = T([1.0], [3.0], [5.0], [7.0])
        out = F.bucketing(X, 2, perm=[0, 1, 2, 3])
        assert torch.allclose(out, T([2.0], [6.0]))

    def test_nnm_k2(self):
        X = T([0.0], [1.0], [10.0])
        out = F.nnm(X, 1)  # k = n-f = 2 nearest incl self
        # x0 -> mean(0,1)=0.5; x1 -> mean(1,0)=0.5; x2 -> mean(10,1)=5.5
        assert torch.allclose(out, T([0.5], [0.5], [5.5]))

    def test_arc_no_clip_when_k0(self):
        X = T([1.0], [100.0])
        # n=2, f=0 -> k=0: identity
        assert torch.allclose(F.arc_clip(X, 0), X)


class TestAttackHand:
    def test_empire_scaled_mean(self):
        honest = T([1.0, 2.0], [3.0, 4.0])
        out = F.empire(honest, scale=-1.0)
        assert torch.allclose(out, torch.tensor([-2.0, -3.0]))

    def test_little_z_value_n12_f3(self):
        # s = floor(12/2)+1-3 = 4; z = Phi^-1((12-4)/12) = Phi^-1(2/3)
        honest = T(*[[float(i)] for i in range(9)])
        out = F.little(honest, 3, N=12)
        mu = honest.mean()
        sigma = honest.std(unbiased=False)
        z = 0.430727299295457  # Phi^-1(2/3), 15 digits
        assert abs(float(out) - float(mu + z * sigma)) < 1e-4

    def test_mimic_copies(self):
        honest = T([1.0, 1.0], [2.0, 2.0])
        assert torch.equal(F.mimic(honest, 1), torch.tensor([2.0, 2.0]))

    def test_gaussian_moments(self):
        out = F.gaussian_attack(torch.zeros(200_000), mu=2.0, sigma=0.5, seed=1)
        assert abs(float(out.mean()) - 2.0) < 0.01
        assert abs(float(out.std()) - 0.5) < 0.01

    def test_inf_is_inf(self):
        out = F.inf_attack(torch.zeros(5))
        assert torch.isinf(out).all() and (out > 0).all()

    def test_sign_flip(self):
        assert torch.equal(
            F.sign_flip(torch.tensor([1.0, -2.0]), scale=-3.0),
            torch.tensor([-3.0, 6.0]),
        )
