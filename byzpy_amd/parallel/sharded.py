"""d-sharded robust aggregation over RCCL (the multi-GPU engine).

Layout: every rank holds the SAME n worker gradients restricted to its own
contiguous d-shard — the 288 GB-HBM-native layout for huge d (SURVEY.md
§5.7; BASELINE config 5 needs it: 32 x 8B-param bf16 grads = 512 GB > one
GPU). Coordinate-wise ops then need NO communication at all; geometry/norm
ops need only tiny (n,) / (n,n) all-reduces; the aggregate stays d-sharded
(all-gather only if the caller wants it assembled).

Every function takes the LOCAL shard (n, d_local) and returns the LOCAL
shard of the aggregate.
"""
from __future__ import annotations

from typing import Sequence

import torch

from byzpy_amd.hip import dispatch as D
from byzpy_amd.ops import functional as F
from byzpy_amd.parallel.dist import all_reduce_


# -- coordinate-wise: pure local --------------------------------------------

def median(X_local: torch.Tensor) -> torch.Tensor:
    return D.median(X_local)


def trimmed_mean(X_local: torch.Tensor, f: int) -> torch.Tensor:
    return D.trimmed_mean(X_local, f)


def mean_of_medians(X_local: torch.Tensor, f: int) -> torch.Tensor:
    return D.mean_of_medians(X_local, f)


# -- geometry: partial Gram + (n,n) all-reduce ------------------------------

def _global_gram(X_local: torch.Tensor) -> torch.Tensor:
    G = D.gram(X_local)
    return all_reduce_(G)


def median_and_multi_krum(X_local: torch.Tensor, f: int, q: int):
    """Both flagship aggregates with ONE collective: the local median
    shard and the partial Gram (two kernels — the single-pass fusion was
    measured slower and removed, see profiles/r02_fusion_negative.md),
    then one (n, n) all-reduce yields a Krum selection identical on
    every rank (RCCL all-reduce returns bitwise-identical sums on all
    ranks, so the winner set never diverges)."""
    from byzpy_amd.hip import dispatch as _D

    n = X_local.shape[0]
    med, G = _D.median_and_gram(X_local)
    G = all_reduce_(G)
    if X_local.is_cuda and n <= 512:
        from byzpy_amd import hip as _h

        winners = _h.require().krum_select(G, int(f), int(q))
        return med, D.mean_rows(X_local, winners)
    norms = torch.diagonal(G)
    D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
    D2 = D2 + torch.diag(torch.full((n,), float("inf"), device=X_local.device))
    k = n - f - 1
    scores = torch.topk(D2, k=k, dim=1, largest=False).values.sum(dim=1)
    winners = torch.topk(scores, k=q, largest=False).indices
    return med, D.mean_rows(X_local, winners.to(torch.int32))


def multi_krum(X_local: torch.Tensor, f: int, q: int) -> torch.Tensor:
    n = X_local.shape[0]
    G = _global_gram(X_local)
    if X_local.is_cuda and n <= 512:
        from byzpy_amd.hip import require

        winners = require().krum_select(G, int(f), int(q))
        return D.mean_rows(X_local, winners)
    norms = torch.diagonal(G)
    D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
    D2 = D2 + torch.diag(torch.full((n,), float("inf"), device=X_local.device))
    k = n - f - 1
    scores = torch.topk(D2, k=k, dim=1, largest=False).values.sum(dim=1)
    winners = torch.topk(scores, k=q, largest=False).indices
    return D.mean_rows(X_local, winners.to(torch.int32))


def krum(X_local: torch.Tensor, f: int) -> torch.Tensor:
    n = X_local.shape[0]
    G = _global_gram(X_local)
    norms = torch.diagonal(G)
    D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
    D2 = D2 + torch.diag(torch.full((n,), float("inf"), device=X_local.device))
    scores = torch.topk(D2, k=n - f - 1, dim=1, largest=False).values.sum(dim=1)
    return X_local[int(torch.argmin(scores))].clone()


def nnm(X_local: torch.Tensor, f: int) -> torch.Tensor:
    n = X_local.shape[0]
    G = _global_gram(X_local)
    norms = torch.diagonal(G)
    D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
    idx = torch.topk(D2, k=n - f, dim=1, largest=False).indices.to(torch.int32)
    if X_local.is_cuda:
        from byzpy_amd.hip import require

        return require().group_mean_rows(X_local.contiguous(), idx)
    return X_local.float()[idx.long()].mean(dim=1).to(X_local.dtype)


# -- norm-wise: (n,) all-reduce ---------------------------------------------

def _global_row_sqnorms(X_local: torch.Tensor) -> torch.Tensor:
    return all_reduce_(D.row_sqnorms(X_local))


def cge(X_local: torch.Tensor, f: int) -> torch.Tensor:
    norms = _global_row_sqnorms(X_local)
    idx = torch.argsort(norms, stable=True)[: X_local.shape[0] - f]
    return D.mean_rows(X_local, idx.to(torch.int32))


def clip_rows(X_local: torch.Tensor, threshold: float) -> torch.Tensor:
    norms = _global_row_sqnorms(X_local).sqrt_().clamp_min_(1e-20)
    scale = torch.clamp(threshold / norms, max=1.0)
    return D.row_scale(X_local, scale)


def arc_clip(X_local: torch.Tensor, f: int) -> torch.Tensor:
    n = X_local.shape[0]
    k = int(2 * f / n * (n - f))
    if k <= 0:
        return X_local.clone()
    norms = _global_row_sqnorms(X_local).sqrt_()
    order = torch.argsort(norms, descending=True)
    threshold = norms[order[k]]
    scale = torch.clamp(threshold / norms.clamp_min(1e-20), max=1.0)
    return D.row_scale(X_local, scale)


def bucketing(
    X_local: torch.Tensor, bucket_size: int, perm: Sequence[int]
) -> torch.Tensor:
    """perm must be identical on all ranks (caller broadcasts the seed)."""
    return D.bucketing(X_local, bucket_size, perm)


# -- iterative fixed-point: per-iteration (n,) all-reduce -------------------

def geometric_median(
    X_local: torch.Tensor,
    *,
    tol: float = 1e-6,
    max_iter: int = 256,
    eps: float = 1e-12,
    init: str = "median",
    fixed_iters: "int | None" = None,
) -> torch.Tensor:
    """``fixed_iters`` runs exactly that many iterations with NO
    convergence polls — drops the per-poll shift all-reduce AND the host
    sync, leaving one (n,) all-reduce per iteration (the minimum for the
    d-sharded layout)."""
    if not X_local.is_cuda:
        return _geometric_median_cpu(
            X_local, tol=tol, max_iter=max_iter, eps=eps, init=init,
            fixed_iters=fixed_iters,
        )
    from byzpy_amd.hip import require

    ext = require()
    Xc = X_local.contiguous()
    z = (D.median(Xc) if init == "median" else Xc.float().mean(dim=0)).float()
    shift = torch.zeros((), device=Xc.device, dtype=torch.float32)
    if fixed_iters is not None:
        for _ in range(int(fixed_iters)):
            dist2 = all_reduce_(ext.row_center_sqdists(Xc, z))
            z = ext.weiszfeld_apply(Xc, z, dist2, float(eps), shift)
        return z.to(X_local.dtype)
    poll = 4
    it = 0
    while it < max_iter:
        for _ in range(min(poll, max_iter - it)):
            dist2 = all_reduce_(ext.row_center_sqdists(Xc, z))
            z = ext.weiszfeld_apply(Xc, z, dist2, float(eps), shift)
            it += 1
        total_shift = all_reduce_(shift.clone())
        if float(total_shift) <= tol * tol:
            break
    return z.to(X_local.dtype)


def _geometric_median_cpu(X, *, tol, max_iter, eps, init, fixed_iters=None):
    Xf = X.float()
    z = (F.median(Xf) if init == "median" else Xf.mean(dim=0)).float()
    iters = int(fixed_iters) if fixed_iters is not None else max_iter
    for _ in range(iters):
        dist2 = all_reduce_(((Xf - z[None, :]) ** 2).sum(dim=1))
        d = dist2.sqrt().clamp_(min=eps)
        w = 1.0 / d
        z_new = (w[:, None] * Xf).sum(dim=0) / w.sum()
        if fixed_iters is None:
            shift2 = all_reduce_(((z_new - z) ** 2).sum())
            z = z_new
            if float(shift2) <= tol * tol:
                break
        else:
            z = z_new
    return z.to(X.dtype)


def centered_clipping(
    X_local: torch.Tensor,
    *,
    c_tau: float,
    M: int = 10,
    eps: float = 1e-12,
    init: str = "mean",
) -> torch.Tensor:
    n = X_local.shape[0]
    if not X_local.is_cuda:
        Xf = X_local.float()
        if init == "mean":
            v = Xf.mean(dim=0)
        elif init == "median":
            v = F.median(Xf).float()
        else:
            v = torch.zeros_like(Xf[0])
        for _ in range(M):
            dist2 = all_reduce_(((Xf - v[None, :]) ** 2).sum(dim=1))
            norms = dist2.sqrt().clamp_(min=eps)
            alpha = torch.clamp(c_tau / norms, max=1.0)
            v = v + (alpha[:, None] * (Xf - v[None, :])).sum(dim=0) / n
        return v.to(X_local.dtype)
    from byzpy_amd.hip import require

    ext = require()
    Xc = X_local.contiguous()
    if init == "mean":
        v = Xc.float().mean(dim=0)
    elif init == "median":
        v = D.median(Xc).float()
    else:
        v = torch.zeros(Xc.shape[1], device=Xc.device, dtype=torch.float32)
    for _ in range(M):
        dist2 = all_reduce_(ext.row_center_sqdists(Xc, v))
        v = ext.cc_apply(Xc, v, dist2, float(c_tau), float(eps))
    return v.to(X_local.dtype)
