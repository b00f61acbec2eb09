"""Property-based invariants (hypothesis) for the robust aggregators."""
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from byzpy_amd.ops import functional as F


def _matrix(n, d, seed):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(n, d, generator=g)


matrix_params = st.tuples(
    st.integers(min_value=3, max_value=24),
    st.integers(min_value=1, max_value=65),
    st.integers(min_value=0, max_value=10_000),
)


@settings(max_examples=25, deadline=None)
@given(matrix_params)
def test_median_permutation_invariant(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    perm = torch.randperm(n, generator=torch.Generator().manual_seed(seed + 1))
    assert torch.allclose(F.median(X), F.median(X[perm]), atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(matrix_params)
def test_trimmed_mean_permutation_invariant(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = (n - 1) // 2
    perm = torch.randperm(n, generator=torch.Generator().manual_seed(seed + 1))
    assert torch.allclose(F.trimmed_mean(X, f), F.trimmed_mean(X[perm], f), atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(matrix_params)
def test_median_between_min_max(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    med = F.median(X)
    assert (med >= X.min(dim=0).values - 1e-6).all()
    assert (med <= X.max(dim=0).values + 1e-6).all()


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_clip_rows_norm_bound_and_idempotent(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    thr = 1.0
    Y = F.clip_rows(X, thr)
    assert (Y.norm(dim=1) <= thr + 1e-4).all()
    Z = F.clip_rows(Y, thr)
    assert torch.allclose(Y, Z, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_krum_winner_is_input_row(p):
    n, d, seed = p
    if n < 4:
        return
    X = _matrix(n, d, seed)
    f = (n - 2) // 2
    if n - f - 1 < 1:
        return
    out = F.krum(X, f)
    assert any(torch.allclose(out, X[i]) for i in range(n))


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_bucketing_mean_preserving(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    b = max(1, n // 3)
    if n % b != 0:
        return  # equal buckets only: mean of bucket-means == global mean
    out = F.bucketing(X, b, perm=list(range(n)))
    assert torch.allclose(out.mean(dim=0), X.mean(dim=0), atol=1e-4)


@settings(max_examples=15, deadline=None)
@given(matrix_params)
def test_geometric_median_translation_equivariant(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    shift = torch.randn(d, generator=torch.Generator().manual_seed(seed + 2))
    a = F.geometric_median(X, tol=1e-8, max_iter=300)
    b = F.geometric_median(X + shift, tol=1e-8, max_iter=300)
    assert torch.allclose(a + shift, b, atol=1e-2)


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_nnm_rows_in_convex_hull_bounds(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = max(0, n // 4)
    out = F.nnm(X, f)
    assert (out.max() <= X.max() + 1e-5) and (out.min() >= X.min() - 1e-5)


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_trimmed_mean_within_envelope(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = (n - 1) // 2
    out = F.trimmed_mean(X, f)
    lo, hi = X.min(dim=0).values, X.max(dim=0).values
    assert (out >= lo - 1e-5).all() and (out <= hi + 1e-5).all()


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_meamed_within_envelope(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = max(0, n // 3)
    out = F.mean_of_medians(X, f)
    lo, hi = X.min(dim=0).values, X.max(dim=0).values
    assert (out >= lo - 1e-5).all() and (out <= hi + 1e-5).all()


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_arc_never_grows_norms(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = max(0, n // 4)
    out = F.arc_clip(X, f)
    assert (out.norm(dim=1) <= X.norm(dim=1) + 1e-4).all()


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_nnm_is_variance_contraction(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = max(0, n // 4)
    out = F.nnm(X, f)
    assert float(out.var(dim=0).sum()) <= float(X.var(dim=0).sum()) + 1e-4


@settings(max_examples=15, deadline=None)
@given(matrix_params)
def test_geomed_optimality_vs_mean_and_rows(p):
    # the geometric median minimizes sum of distances: it must beat both
    # the mean and every input row on that objective
    n, d, seed = p
    X = _matrix(n, d, seed)
    gm = F.geometric_median(X, tol=1e-9, max_iter=500)

    def obj(z):
        return float((X - z[None, :]).norm(dim=1).sum())

    best_row = min(obj(X[i]) for i in range(n))
    assert obj(gm) <= obj(X.mean(dim=0)) + 1e-3
    assert obj(gm) <= best_row + 1e-3


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_cge_subset_of_rows_mean(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = max(0, n // 3)
    out = F.cge(X, f)
    norms = (X * X).sum(dim=1)
    keep = torch.argsort(norms, stable=True)[: n - f]
    assert torch.allclose(out, X[keep].mean(dim=0), atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(matrix_params)
def test_multi_krum_q_equals_n_minus_f_is_selection_mean(p):
    n, d, seed = p
    X = _matrix(n, d, seed)
    f = (n - 2) // 2
    if n - f - 1 < 1:
        return
    q = max(1, n - f)
    scores = F.multi_krum_scores(X, f)
    keep = torch.topk(scores, q, largest=False).indices
    assert torch.allclose(
        F.multi_krum(X, f, q), X[keep].mean(dim=0), atol=1e-4
    )


@settings(max_examples=12, deadline=None)
@given(matrix_params)
def test_pooled_median_fuzz_parity(p):
    """Pooled-vs-direct parity under randomized shapes AND chunk sizes
    (regression net for the chunk-range arithmetic)."""
    import asyncio

    from byzpy_amd.graph.executor import OperatorExecutor
    from byzpy_amd.graph.pool import ActorPool, ActorPoolConfig
    from byzpy_amd.aggregators import CoordinateWiseMedian

    n, d, seed = p
    import random as _r

    chunk = _r.Random(seed).choice([1, 3, 17, 4096])
    grads = list(_matrix(n, d, seed))
    agg = CoordinateWiseMedian(chunk_size=chunk)
    direct = agg.aggregate(grads)

    async def main():
        pool = ActorPool(ActorPoolConfig(backend="thread", count=2))
        await pool.start()
        out = await OperatorExecutor(agg, pool=pool).run({"gradients": grads})
        await pool.close()
        return out

    out = asyncio.run(main())
    assert torch.allclose(out, direct, atol=1e-4)
