"""Pure-torch reference implementations of every robust-aggregation op.

Every function operates on a 2-D ``(n, d)`` torch tensor (one gradient per
row) and returns either a ``(d,)`` aggregate or an ``(n', d)`` matrix.
These run on any device and are the numerical oracle the HIP kernels are
tested against (SURVEY.md §4 pattern 2). Device dispatch to the gfx950
kernels lives in byzpy_amd/hip/dispatch.py — NOT here.

Algorithm parity notes (cites into /root/reference):
- median: true per-coordinate median (mean of the two middles for even n).
  The reference's chunked path returned the lower middle
  (aggregators/coordinate_wise/median.py:160-171) while its direct path did
  not — we deliberately make both paths the true median.
- krum: Blanchard et al. 2017 scoring (krum.py:183-194).
- geometric_median: Weiszfeld fixed point (geometric_median.py:69-104).
- centered_clipping: Karimireddy et al. 2021 (center_clipping.py:107-156).
- mda: exact (n-f)-subset min-diameter search (minimum_diameter_average.py).
- smea: min max-eigenvalue subset (smea.py:63-107).
- caf: covariance-agnostic filter w/ seeded power iteration (caf.py:133-184).
"""
from __future__ import annotations

import itertools
import math
from typing import List, Optional, Sequence, Tuple

import torch

# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------


def pairwise_sq_dists(X: torch.Tensor) -> torch.Tensor:
    """Full n x n matrix of squared euclidean distances via the Gram trick
    ``||a||^2 + ||b||^2 - 2 a.b`` with f32 accumulation."""
    Xf = X.float()
    norms = (Xf * Xf).sum(dim=1)
    G = Xf @ Xf.T
    D2 = norms[:, None] + norms[None, :] - 2.0 * G
    return D2.clamp_(min=0.0)


# ---------------------------------------------------------------------------
# coordinate-wise aggregators
# ---------------------------------------------------------------------------


def median(X: torch.Tensor) -> torch.Tensor:
    """True per-coordinate median; for even n the mean of the two middles."""
    n = X.shape[0]
    s, _ = torch.sort(X.float(), dim=0)
    if n % 2 == 1:
        out = s[n // 2]
    else:
        out = (s[n // 2 - 1] + s[n // 2]) * 0.5
    return out.to(X.dtype)


def trimmed_mean(X: torch.Tensor, f: int) -> torch.Tensor:
    """Sort each coordinate, drop f from each end, mean the middle n-2f
    (Yin et al. 2018; reference trimmed_mean.py:80-115)."""
    n = X.shape[0]
    if 2 * f >= n:
        raise ValueError(f"need n > 2f, got n={n}, f={f}")
    s, _ = torch.sort(X.float(), dim=0)
    out = s[f : n - f].mean(dim=0)
    return out.to(X.dtype)


def mean_of_medians(X: torch.Tensor, f: int) -> torch.Tensor:
    """MeaMed: per coordinate keep the n-f values closest to the median,
    mean them (reference mean_of_medians.py:53-81)."""
    n = X.shape[0]
    if f < 0 or f >= n:
        raise ValueError(f"need 0 <= f < n, got n={n}, f={f}")
    Xf = X.float()
    med = median(Xf)
    dev = (Xf - med[None, :]).abs()
    # indices of the n-f smallest deviations per column
    idx = torch.topk(dev, k=n - f, dim=0, largest=False).indices
    kept = torch.gather(Xf, 0, idx)
    return kept.mean(dim=0).to(X.dtype)


# ---------------------------------------------------------------------------
# geometric aggregators
# ---------------------------------------------------------------------------


def multi_krum_scores(X: torch.Tensor, f: int) -> torch.Tensor:
    """score_i = sum of the n-f-1 smallest squared distances from i to the
    others (self-distance 0 excluded)."""
    n = X.shape[0]
    k = n - f - 1
    if k < 1:
        raise ValueError(f"need n - f - 1 >= 1, got n={n}, f={f}")
    D2 = pairwise_sq_dists(X)
    D2 = D2 + torch.diag(torch.full((n,), float("inf"), device=X.device))
    smallest = torch.topk(D2, k=k, dim=1, largest=False).values
    return smallest.sum(dim=1)


def multi_krum(X: torch.Tensor, f: int, q: int) -> torch.Tensor:
    """Mean of the q vectors with the best (lowest) Krum scores."""
    n = X.shape[0]
    if q < 1 or q > n:
        raise ValueError(f"need 1 <= q <= n, got q={q}, n={n}")
    scores = multi_krum_scores(X, f)
    winners = torch.topk(scores, k=q, largest=False).indices
    return X.float()[winners].mean(dim=0).to(X.dtype)


def krum(X: torch.Tensor, f: int) -> torch.Tensor:
    """The single vector with the best Krum score (= MultiKrum q=1,
    returning the winner itself, reference krum.py:346-368)."""
    scores = multi_krum_scores(X, f)
    return X[int(torch.argmin(scores))].clone()


def geometric_median(
    X: torch.Tensor,
    *,
    tol: float = 1e-6,
    max_iter: int = 256,
    eps: float = 1e-12,
    init: str = "median",
) -> torch.Tensor:
    """Weiszfeld fixed point: z <- sum(x_i/d_i) / sum(1/d_i), d_i clamped
    >= eps; stops when ||dz|| <= tol."""
    if init not in ("median", "mean"):
        raise ValueError("init must be 'median' or 'mean'")
    Xf = X.float()
    z = median(Xf) if init == "median" else Xf.mean(dim=0)
    z = z.float()
    for _ in range(max_iter):
        d = (Xf - z[None, :]).norm(dim=1).clamp_(min=eps)
        w = 1.0 / d
        z_new = (w[:, None] * Xf).sum(dim=0) / w.sum()
        shift = (z_new - z).norm()
        z = z_new
        if shift <= tol:
            break
    return z.to(X.dtype)


def mda_subset(D2: torch.Tensor, f: int) -> Tuple[int, ...]:
    """Exact minimum-diameter (n-f)-subset search on a squared-distance
    matrix. Branch-and-bound DFS with prefix-max pruning — runs on the host
    over the tiny n x n matrix (the D2 itself is computed on device)."""
    n = D2.shape[0]
    m = n - f
    if m < 1:
        raise ValueError(f"need n - f >= 1, got n={n}, f={f}")
    from byzpy_amd import hip as _hip

    ext = _hip.extension()
    if ext is not None:  # native C++ branch-and-bound (host)
        idx = ext.mda_search(D2.detach().float().cpu(), int(f))
        return tuple(int(i) for i in idx)
    D = D2.detach().cpu().double().numpy()
    best_diam = math.inf
    best: Tuple[int, ...] = tuple(range(m))

    # order candidates so tight subsets are found early (better pruning)
    order = list(range(n))

    def dfs(start: int, chosen: List[int], diam: float) -> None:
        nonlocal best_diam, best
        if diam >= best_diam:
            return
        if len(chosen) == m:
            best_diam = diam
            best = tuple(sorted(chosen))
            return
        # not enough remaining to complete the subset
        remaining = n - start
        if remaining < m - len(chosen):
            return
        for j in range(start, n):
            dj = diam
            ok = True
            for c in chosen:
                dc = D[c][order[j]]
                if dc >= best_diam:
                    ok = False
                    break
                if dc > dj:
                    dj = dc
            if ok:
                chosen.append(order[j])
                dfs(j + 1, chosen, dj)
                chosen.pop()

    dfs(0, [], 0.0)

    # canonicalize to the lexicographically smallest optimal subset (ties
    # are generic: every subset containing the binding edge can tie)
    bound = best_diam
    lex: List[int] = []

    def dfs2(start: int) -> bool:
        if len(lex) == m:
            return True
        need = m - len(lex)
        for j in range(start, n - need + 1):
            if all(D[c][j] <= bound for c in lex):
                lex.append(j)
                if dfs2(j + 1):
                    return True
                lex.pop()
        return False

    if dfs2(0):
        return tuple(lex)
    return best


def minimum_diameter_averaging(X: torch.Tensor, f: int) -> torch.Tensor:
    D2 = pairwise_sq_dists(X)
    subset = mda_subset(D2, f)
    idx = torch.tensor(subset, device=X.device, dtype=torch.long)
    return X.float()[idx].mean(dim=0).to(X.dtype)


def monna(X: torch.Tensor, f: int, reference_index: int = 0) -> torch.Tensor:
    """Mean of the n-f nearest neighbours of a trusted reference row
    (reference-first tiebreak, monna.py:142-145)."""
    n = X.shape[0]
    k = n - f
    if k < 1 or reference_index < 0 or reference_index >= n:
        raise ValueError("bad monna parameters")
    Xf = X.float()
    ref = Xf[reference_index]
    d2 = ((Xf - ref[None, :]) ** 2).sum(dim=1)
    # reference row first regardless of float noise in its self-distance
    d2 = d2.clone()
    d2[reference_index] = -1.0
    idx = torch.argsort(d2, stable=True)[:k]
    return Xf[idx].mean(dim=0).to(X.dtype)


def smea(X: torch.Tensor, f: int) -> torch.Tensor:
    """Among all (n-f)-subsets pick the one whose sample covariance has the
    smallest max eigenvalue; return its mean. Eigenvalues come from the
    centered Gram H G H (m x m), a device-batched eigvalsh."""
    n = X.shape[0]
    m = n - f
    if m < 1:
        raise ValueError(f"need n - f >= 1, got n={n}, f={f}")
    Xf = X.float()
    G = Xf @ Xf.T
    # enumerate subsets in bounded blocks: materializing all C(n, m)
    # (m, m) Grams at once OOMs for larger n (the search stays exact and
    # exponential-time by definition — same as the reference smea.py:63-107)
    BLOCK = 1 << 16
    best_ev = math.inf
    best_rows: Optional[torch.Tensor] = None
    it = itertools.combinations(range(n), m)
    while True:
        combos = list(itertools.islice(it, BLOCK))
        if not combos:
            break
        idx = torch.tensor(combos, device=X.device, dtype=torch.long)  # (B, m)
        # batched centered Gram: HGH with H = I - 1/m
        sub = G[idx[:, :, None], idx[:, None, :]]  # (B, m, m)
        row_mean = sub.mean(dim=2, keepdim=True)
        col_mean = sub.mean(dim=1, keepdim=True)
        all_mean = sub.mean(dim=(1, 2), keepdim=True)
        centered = sub - row_mean - col_mean + all_mean
        ev = torch.linalg.eigvalsh(
            centered.cpu() if X.device.type == "cpu" else centered
        )
        max_ev = ev[..., -1]
        b = int(torch.argmin(max_ev))
        v = float(max_ev[b])
        # ties resolve to the lexicographically-smallest subset (blocks are
        # generated in lexicographic order, so strict < suffices)
        if v < best_ev:
            best_ev = v
            best_rows = idx[b]
    assert best_rows is not None
    return Xf[best_rows].mean(dim=0).to(X.dtype)


# ---------------------------------------------------------------------------
# norm-wise aggregators
# ---------------------------------------------------------------------------


def centered_clipping(
    X: torch.Tensor,
    *,
    c_tau: float,
    M: int = 10,
    eps: float = 1e-12,
    init: str = "mean",
) -> torch.Tensor:
    """v <- v + (1/n) sum_i clip(x_i - v, c_tau) for M iterations."""
    Xf = X.float()
    n = Xf.shape[0]
    if init == "mean":
        v = Xf.mean(dim=0)
    elif init == "median":
        v = median(Xf).float()
    elif init == "zero":
        v = torch.zeros_like(Xf[0])
    else:
        raise ValueError("init must be one of {'mean','median','zero'}")
    for _ in range(M):
        diff = Xf - v[None, :]
        norms = diff.norm(dim=1).clamp_(min=eps)
        alpha = torch.clamp(c_tau / norms, max=1.0)
        v = v + (alpha[:, None] * diff).sum(dim=0) / n
    return v.to(X.dtype)


def cge(X: torch.Tensor, f: int) -> torch.Tensor:
    """Comparative gradient elimination: mean of the n-f smallest-L2-norm
    rows."""
    n = X.shape[0]
    k = n - f
    if k < 1:
        raise ValueError(f"need n - f >= 1, got n={n}, f={f}")
    Xf = X.float()
    norms = (Xf * Xf).sum(dim=1)
    idx = torch.argsort(norms, stable=True)[:k]
    return Xf[idx].mean(dim=0).to(X.dtype)


def caf(X: torch.Tensor, f: int, *, power_iters: int = 3) -> torch.Tensor:
    """Covariance-agnostic filter: iteratively downweight along the dominant
    eigvec of the weighted covariance until the weight mass <= n - 2f;
    return the best-lambda weighted mean (reference caf.py:133-184, seeded
    rng(0) power iteration)."""
    Xf = X.float()
    n = Xf.shape[0]
    if n - 2 * f <= 0:
        raise ValueError(f"need n - 2f > 0, got n={n}, f={f}")
    d = Xf.shape[1]
    w = torch.ones(n, device=Xf.device)
    gen = torch.Generator(device="cpu")
    gen.manual_seed(0)
    best_lambda = torch.full((), math.inf, device=Xf.device)
    best_mu = Xf.mean(dim=0)
    target = float(n - 2 * f)
    # power-iteration seeds pre-generated per block (one CPU randn + one
    # H2D per block, capped ~512 MB for huge d; falls back to per-round)
    block_rows = max(1, min(4, (1 << 27) // max(d, 1)))
    seeds: torch.Tensor = torch.empty(0)
    for r in range(n):  # at most n rounds of downweighting
        if r % block_rows == 0:
            rows = min(block_rows, n - r)
            seeds = torch.randn(rows, d, generator=gen).to(Xf.device)
        wsum = w.sum()
        mu = (w[:, None] * Xf).sum(dim=0) / wsum
        diffs = Xf - mu[None, :]
        # dominant eigenpair of (1/wsum) * diffs^T W diffs via power iteration
        # (seeded CPU rng is part of the determinism contract)
        v = seeds[r % block_rows]
        v = v / v.norm().clamp_min(1e-20)
        lam = torch.zeros((), device=Xf.device)
        for _ in range(max(1, power_iters)):
            s = diffs @ v  # (n,)
            t = (w * s)[None, :] @ diffs  # (1, d)
            t = t[0] / wsum
            lam = t.norm()
            v = t / lam.clamp_min(1e-20)
        # device-side best tracking (no extra host sync)
        better = lam < best_lambda
        best_lambda = torch.where(better, lam, best_lambda)
        best_mu = torch.where(better, mu, best_mu)
        # downweight along v proportionally to projection^2
        proj = (diffs @ v) ** 2
        pmax = proj.max().clamp_min(1e-20)
        w_next = (w * (1.0 - proj / pmax)).clamp_min(0.0)
        # one host sync per round for the two break conditions (A/B'd:
        # batching the check 4 rounds deep ran inert full rounds past the
        # break and LOST ~1 ms at 64x65536)
        wsum_f, wnext_f = torch.stack([wsum, w_next.sum()]).tolist()
        if wsum_f <= target or wnext_f <= 0:
            break
        w = w_next
    return best_mu.to(X.dtype)


# ---------------------------------------------------------------------------
# pre-aggregators (list -> list semantics; matrix in, matrix out here)
# ---------------------------------------------------------------------------


def clip_rows(X: torch.Tensor, threshold: float) -> torch.Tensor:
    """Scale each row to have L2 norm at most ``threshold``."""
    Xf = X.float()
    norms = Xf.norm(dim=1).clamp_min(1e-20)
    scale = torch.clamp(threshold / norms, max=1.0)
    return (Xf * scale[:, None]).to(X.dtype)


def arc_clip(X: torch.Tensor, f: int) -> torch.Tensor:
    """Adaptive robust clipping: clip the k = floor(2f/n * (n-f)) largest-norm
    rows to the largest remaining (k+1-th largest) norm (arc.py:36-61)."""
    n = X.shape[0]
    k = int(2 * f / n * (n - f))
    if k <= 0:
        return X.clone()
    Xf = X.float()
    norms = Xf.norm(dim=1)
    order = torch.argsort(norms, descending=True)
    threshold = norms[order[k]]  # largest norm NOT clipped
    scale = torch.clamp(threshold / norms.clamp_min(1e-20), max=1.0)
    return (Xf * scale[:, None]).to(X.dtype)


def bucketing(
    X: torch.Tensor, bucket_size: int, perm: Optional[Sequence[int]] = None
) -> torch.Tensor:
    """Random permutation -> consecutive buckets of ``bucket_size`` ->
    per-bucket mean. ``perm`` injectable for determinism (bucketing.py:93-94)."""
    n = X.shape[0]
    if perm is None:
        perm_t = torch.randperm(n, device=X.device)
    else:
        perm_t = torch.as_tensor(list(perm), device=X.device, dtype=torch.long)
    Xp = X.float()[perm_t]
    nb = (n + bucket_size - 1) // bucket_size
    out = torch.empty((nb, X.shape[1]), device=X.device, dtype=torch.float32)
    for b in range(nb):
        out[b] = Xp[b * bucket_size : min(n, (b + 1) * bucket_size)].mean(dim=0)
    return out.to(X.dtype)


def nnm(X: torch.Tensor, f: int) -> torch.Tensor:
    """Nearest-neighbor mixing: replace x_i by the mean of its n-f nearest
    neighbours (including itself) (nnm.py:82-97)."""
    n = X.shape[0]
    k = n - f
    if k < 1:
        raise ValueError(f"need n - f >= 1, got n={n}, f={f}")
    D2 = pairwise_sq_dists(X)
    idx = torch.topk(D2, k=k, dim=1, largest=False).indices  # (n, k)
    Xf = X.float()
    # mixing as a 0/1-mask matmul (one GEMM) rather than a gather that
    # materializes an (n, k, d) tensor
    mask = torch.zeros((n, n), device=X.device, dtype=torch.float32)
    mask.scatter_(1, idx, 1.0)
    out = (mask @ Xf) / k
    return out.to(X.dtype)


# ---------------------------------------------------------------------------
# attacks
# ---------------------------------------------------------------------------


def empire(honest: torch.Tensor, scale: float = -1.0) -> torch.Tensor:
    return (honest.float().mean(dim=0) * scale).to(honest.dtype)


def sign_flip(base_grad: torch.Tensor, scale: float = -1.0) -> torch.Tensor:
    return base_grad * scale


def little(honest: torch.Tensor, f: int, N: Optional[int] = None) -> torch.Tensor:
    """'A Little Is Enough': mu + z * sigma with s = floor(N/2)+1-f,
    z = Phi^{-1}((N-s)/N) (little.py:81-231)."""
    n = honest.shape[0]
    N_total = N if N is not None else n + f
    s = N_total // 2 + 1 - f
    phi_arg = (N_total - s) / N_total
    z = float(
        torch.distributions.Normal(0.0, 1.0).icdf(torch.tensor(float(phi_arg)))
    )
    Hf = honest.float()
    mu = Hf.mean(dim=0)
    sigma = Hf.std(dim=0, unbiased=False)
    return (mu + z * sigma).to(honest.dtype)


def gaussian_attack(
    like: torch.Tensor, mu: float = 0.0, sigma: float = 1.0, seed: Optional[int] = None
) -> torch.Tensor:
    d = like.shape[-1]
    if seed is not None:
        gen = torch.Generator(device="cpu")
        gen.manual_seed(seed)
        out = torch.normal(mu, sigma, size=(d,), generator=gen)
        return out.to(device=like.device, dtype=like.dtype)
    return torch.normal(
        mu, sigma, size=(d,), device=like.device, dtype=torch.float32
    ).to(like.dtype)


def inf_attack(like: torch.Tensor) -> torch.Tensor:
    return torch.full(
        (like.shape[-1],), float("inf"), device=like.device, dtype=like.dtype
    )


def mimic(honest: torch.Tensor, epsilon: int = 0) -> torch.Tensor:
    return honest[epsilon].clone()
