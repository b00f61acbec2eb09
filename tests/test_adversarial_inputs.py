"""Adversarial-value robustness (random + adversarial
inputs): aggregators must survive inf-magnitude byzantine rows."""
import pytest
import torch

from byzpy_amd.ops import functional as F


def _poisoned(n=12, d=64, n_bad=2, value=float("inf")):
    g = torch.Generator().manual_seed(5)
    X = torch.randn(n, d, generator=g)
    X[:n_bad] = value
    return X


class TestInfRows:
    def test_median_ignores_inf(self):
        X = _poisoned(n_bad=3)
        out = F.median(X)
        assert torch.isfinite(out).all()

    def test_trimmed_mean_drops_inf(self):
        X = _poisoned(n_bad=3)
        out = F.trimmed_mean(X, 3)
        assert torch.isfinite(out).all()

    def test_trimmed_mean_f0_propagates_inf(self):
        # f=0 keeps everything: the mean IS inf — not silently clipped
        X = _poisoned(n_bad=1)
        out = F.trimmed_mean(X, 0)
        assert torch.isinf(out).all()

    def test_krum_rejects_inf(self):
        X = _poisoned(n_bad=2)
        out = F.multi_krum(X, 2, 3)
        assert torch.isfinite(out).all()

    def test_cge_rejects_inf(self):
        X = _poisoned(n_bad=2)
        assert torch.isfinite(F.cge(X, 2)).all()

    def test_meamed_rejects_inf(self):
        X = _poisoned(n_bad=2)
        assert torch.isfinite(F.mean_of_medians(X, 2)).all()

    def test_neg_inf_rows(self):
        X = _poisoned(n_bad=2, value=float("-inf"))
        assert torch.isfinite(F.median(X)).all()
        assert torch.isfinite(F.trimmed_mean(X, 2)).all()


@pytest.mark.gpu
class TestInfRowsGpu:
    def test_colsel_inf_parity(self):
        for value in (float("inf"), float("-inf")):
            for n, n_bad in [(12, 3), (10, 2), (64, 5)]:
                X = _poisoned(n=n, d=257, n_bad=n_bad, value=value).cuda()
                from byzpy_amd.hip import dispatch as D

                assert torch.allclose(
                    D.median(X).cpu(), F.median(X.cpu()), atol=1e-4
                ), (value, n)
                assert torch.allclose(
                    D.trimmed_mean(X, n_bad).cpu(),
                    F.trimmed_mean(X.cpu(), n_bad),
                    atol=1e-4,
                )

    def test_colsel_f0_inf_propagates(self):
        from byzpy_amd.hip import dispatch as D

        X = _poisoned(n=10, d=128, n_bad=1).cuda()
        assert torch.isinf(D.trimmed_mean(X, 0).cpu()).all()

    def test_krum_gpu_rejects_inf(self):
        from byzpy_amd.hip import dispatch as D

        X = _poisoned(n=16, d=2048, n_bad=3).cuda()
        out = D.multi_krum(X, 3, 4)
        assert torch.isfinite(out).all()

    def test_bf16_packed_median_inf(self):
        from byzpy_amd.hip import dispatch as D

        X = _poisoned(n=12, d=256, n_bad=3).cuda().bfloat16()
        out = D.median(X)
        assert torch.isfinite(out.float()).all()
