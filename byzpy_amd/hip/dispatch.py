"""Device dispatch for the robust-aggregation hot ops.

CPU tensors -> byzpy_amd.ops.functional (pure torch, the parity oracle).
CUDA(ROCm) tensors -> hand-written gfx950 kernels in byzpy_amd/_hip_ops
(SURVEY.md §2.7 kernel inventory K1-K14). Ops whose GPU path composes
library GEMMs / batched eig (SMEA combos, attacks) run the functional
torch path on-device — rocBLAS library GEMMs are the sanctioned path for
plain GEMM shapes. CAF runs on the fused K9 matvec/colsum kernel pair.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from byzpy_amd import hip as _hip
from byzpy_amd.ops import functional as F

_COLSEL_MEDIAN = 0
_COLSEL_TRIMMED = 1
_COLSEL_MEAMED = 2

# Column order-statistic caps: registers to n=64, LDS-staged sort to
# n=512, streaming radix-select (every mode x dtype) to n=65535
# (colsel.hip + rsel.hip). Larger n falls back to torch.
COLSEL_MAX_N = 65535


def _gpu(X: torch.Tensor) -> bool:
    return X.is_cuda


# -- coordinate-wise (K1-K3) ------------------------------------------------


def median(X: torch.Tensor) -> torch.Tensor:
    if _gpu(X) and X.shape[0] <= COLSEL_MAX_N:
        return _hip.require().colsel(X.contiguous(), _COLSEL_MEDIAN, 0)
    return F.median(X)


def trimmed_mean(X: torch.Tensor, f: int) -> torch.Tensor:
    if _gpu(X) and X.shape[0] <= COLSEL_MAX_N:
        return _hip.require().colsel(X.contiguous(), _COLSEL_TRIMMED, int(f))
    return F.trimmed_mean(X, f)


def mean_of_medians(X: torch.Tensor, f: int) -> torch.Tensor:
    if _gpu(X) and X.shape[0] <= COLSEL_MAX_N:
        return _hip.require().colsel(X.contiguous(), _COLSEL_MEAMED, int(f))
    return F.mean_of_medians(X, f)


# -- pairwise distances / Krum (K4, K5) -------------------------------------


def gram(X: torch.Tensor) -> torch.Tensor:
    """X @ X.T with f32 accumulation; MFMA split-K kernel on device."""
    if _gpu(X):
        return _hip.require().gram(X.contiguous())
    Xf = X.float()
    return Xf @ Xf.T


def median_and_gram(X: torch.Tensor):
    """Both flagship aggregates: (median, gram). Two separate kernels —
    a fused single-HBM-pass kernel was built and measured across six
    organizations and LOST to this pair every time (32-60 ms vs 6.9 at
    64x125M; see profiles/r02_fusion_negative.md): the median work only
    amortizes with full per-lane column residency, which cannot coexist
    with the Gram's register/occupancy budget."""
    return median(X), gram(X)


def pairwise_sq_dists(X: torch.Tensor) -> torch.Tensor:
    if _gpu(X):
        G = gram(X)
        norms = torch.diagonal(G)
        D2 = norms[:, None] + norms[None, :] - 2.0 * G
        return D2.clamp_(min=0.0)
    return F.pairwise_sq_dists(X)


def multi_krum_scores(X: torch.Tensor, f: int) -> torch.Tensor:
    n = X.shape[0]
    k = n - f - 1
    if k < 1:
        raise ValueError(f"need n - f - 1 >= 1, got n={n}, f={f}")
    if _gpu(X):
        D2 = pairwise_sq_dists(X)
        D2 = D2 + torch.diag(
            torch.full((n,), float("inf"), device=X.device, dtype=D2.dtype)
        )
        smallest = torch.topk(D2, k=k, dim=1, largest=False).values
        return smallest.sum(dim=1)
    return F.multi_krum_scores(X, f)


def multi_krum(X: torch.Tensor, f: int, q: int) -> torch.Tensor:
    if _gpu(X):
        n = X.shape[0]
        G = gram(X)
        if n <= 512:
            idx = _hip.require().krum_select(G, int(f), int(q))
        else:
            scores = _scores_from_gram(G, n, f)
            idx = torch.topk(scores, k=q, largest=False).indices.to(torch.int32)
        return mean_rows(X, idx)
    return F.multi_krum(X, f, q)


def _scores_from_gram(G: torch.Tensor, n: int, f: int) -> torch.Tensor:
    norms = torch.diagonal(G)
    D2 = (norms[:, None] + norms[None, :] - 2.0 * G).clamp_(min=0.0)
    D2 = D2 + torch.diag(torch.full((n,), float("inf"), device=G.device, dtype=D2.dtype))
    return torch.topk(D2, k=n - f - 1, dim=1, largest=False).values.sum(dim=1)


def krum(X: torch.Tensor, f: int) -> torch.Tensor:
    if _gpu(X) and X.shape[0] <= 512:
        G = gram(X)
        idx = _hip.require().krum_select(G, int(f), 1)
        return X[idx.long()[0]].clone()
    scores = multi_krum_scores(X, f)
    return X[int(torch.argmin(scores))].clone()


def mean_rows(X: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """Mean of the selected rows, f32 accumulation (K10 gather-mean)."""
    if _gpu(X):
        return _hip.require().mean_rows(X.contiguous(), idx.to(torch.int32))
    return X.float()[idx].mean(dim=0).to(X.dtype)


# -- row norms / scaling (K8, K12) ------------------------------------------


def row_sqnorms(X: torch.Tensor) -> torch.Tensor:
    if _gpu(X):
        return _hip.require().row_sqnorms(X.contiguous())
    Xf = X.float()
    return (Xf * Xf).sum(dim=1)


def row_scale(X: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    if _gpu(X):
        return _hip.require().row_scale(X.contiguous(), scales.float())
    return (X.float() * scales.float()[:, None]).to(X.dtype)


def clip_rows(X: torch.Tensor, threshold: float) -> torch.Tensor:
    if _gpu(X):
        norms = row_sqnorms(X).sqrt_().clamp_min_(1e-20)
        scale = torch.clamp(threshold / norms, max=1.0)
        return row_scale(X, scale)
    return F.clip_rows(X, threshold)


def arc_clip(X: torch.Tensor, f: int) -> torch.Tensor:
    n = X.shape[0]
    k = int(2 * f / n * (n - f))
    if k <= 0:
        return X.clone()
    if _gpu(X):
        norms = row_sqnorms(X).sqrt_()
        order = torch.argsort(norms, descending=True)
        threshold = norms[order[k]]
        scale = torch.clamp(threshold / norms.clamp_min(1e-20), max=1.0)
        return row_scale(X, scale)
    return F.arc_clip(X, f)


def cge(X: torch.Tensor, f: int) -> torch.Tensor:
    n = X.shape[0]
    k = n - f
    if _gpu(X):
        norms = row_sqnorms(X)
        idx = torch.argsort(norms, stable=True)[:k]
        return mean_rows(X, idx)
    return F.cge(X, f)


# -- iterative fixed-point ops (K6, K7) -------------------------------------


def geometric_median(
    X: torch.Tensor,
    *,
    tol: float = 1e-6,
    max_iter: int = 256,
    eps: float = 1e-12,
    init: str = "median",
    fixed_iters: Optional[int] = None,
    init_z: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Weiszfeld fixed point. ``fixed_iters`` runs exactly that many
    iterations with NO convergence polls — fully async (no host sync), so
    per-node streams overlap and the whole call is hipGraph-capture-safe
    (the polls serialized config-4 gossip in round 1). ``init_z`` warm-
    starts from a caller-provided center (e.g. the previous gossip
    round's output — the fixed point barely moves between rounds, so a
    handful of iterations replaces a cold median init + long descent)."""
    if not _gpu(X) or X.shape[0] > 1024:
        # extension's weiszfeld_iter TORCH_CHECKs n <= 1024; the torch
        # functional path runs fine on-device for larger n
        if init_z is not None:
            Xf = X.float()
            z = init_z.detach().to(device=X.device, dtype=torch.float32)
            iters = int(fixed_iters) if fixed_iters is not None else max_iter
            for _ in range(iters):
                dist = (Xf - z[None, :]).norm(dim=1).clamp_(min=eps)
                w = 1.0 / dist
                z_new = (w[:, None] * Xf).sum(dim=0) / w.sum()
                if fixed_iters is None and float((z_new - z).norm()) <= tol:
                    z = z_new
                    break
                z = z_new
            return z.to(X.dtype)
        if fixed_iters is not None:
            return F.geometric_median(
                X, tol=0.0, max_iter=int(fixed_iters), eps=eps, init=init
            )
        return F.geometric_median(X, tol=tol, max_iter=max_iter, eps=eps, init=init)
    ext = _hip.require()
    Xc = X.contiguous()
    if init_z is not None:
        z = init_z.detach().to(device=X.device, dtype=torch.float32)
    else:
        z = (median(Xc) if init == "median" else Xc.float().mean(dim=0)).float()
    shift = torch.zeros((), device=X.device, dtype=torch.float32)
    if fixed_iters is not None:
        for _ in range(int(fixed_iters)):
            z = ext.weiszfeld_iter(Xc, z, float(eps), shift)
        return z.to(X.dtype)
    # fused per-iteration kernel pair; convergence polled every `poll`
    # iters, with the poll readback PIPELINED one block deep (same
    # pattern as the CAF loop): block b+1's kernels launch while block
    # b's pinned copy of `shift` is in flight, so the host never stalls
    # on a sync — value-identical, speculative blocks past the break
    # point are discarded (each iteration returns a fresh z tensor, so
    # the pre-break z is simply kept by reference).
    # `shift` holds the last iteration's SQUARED step ||dz||^2 (rowops.hip
    # weiszfeld_iter resets + atomicAdds it), so compare against tol^2 to
    # match the CPU oracle's ||dz|| <= tol and parallel/sharded.py.
    poll = 4
    stat_pin = [torch.empty((), pin_memory=True) for _ in range(2)]
    stat_ev = [torch.cuda.Event() for _ in range(2)]
    pending: list = []  # (z_after_block, slot)
    it = 0
    blk = 0
    result_z = None
    while it < max_iter:
        steps = min(poll, max_iter - it)
        for _ in range(steps):
            z = ext.weiszfeld_iter(Xc, z, float(eps), shift)
        it += steps
        slot = blk & 1
        blk += 1
        stat_pin[slot].copy_(shift, non_blocking=True)
        stat_ev[slot].record()
        pending.append((z, slot))
        if len(pending) == 2:
            z_b, sb = pending.pop(0)
            stat_ev[sb].synchronize()
            if float(stat_pin[sb]) <= tol * tol:
                result_z = z_b
                break
    if result_z is None:
        while pending:
            z_b, sb = pending.pop(0)
            stat_ev[sb].synchronize()
            if float(stat_pin[sb]) <= tol * tol:
                result_z = z_b
                break
        if result_z is None:
            result_z = z
    return result_z.to(X.dtype)


def geometric_median_grouped(
    X3: torch.Tensor,
    *,
    iters: int = 8,
    eps: float = 1e-12,
    init_z: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Geometric medians of G independent (m, d) groups in one kernel
    pair per iteration (gossip rounds: all nodes advance together —
    replaces per-node streams + 2*G*iters launches). Fixed iteration
    count (poll-free); warm-start with ``init_z`` (G, d)."""
    G, m, d = X3.shape
    if not _gpu(X3) or m > 32:
        Z = []
        for g in range(G):
            z0 = init_z[g] if init_z is not None else None
            Z.append(
                geometric_median(
                    X3[g], fixed_iters=int(iters), eps=eps, init_z=z0
                ).float()
            )
        return torch.stack(Z)
    ext = _hip.require()
    Xc = X3.contiguous()
    if init_z is not None:
        Z = init_z.detach().to(dtype=torch.float32).contiguous()
    else:
        # per-group colsel medians (G fast launches — torch's dim-median
        # on a (G, m, 25M) tensor measured ~945 ms)
        Z = torch.stack([median(Xc[g]).float() for g in range(G)]).contiguous()
    for _ in range(int(iters)):
        Z = ext.weiszfeld_iter_grouped(Xc, Z, float(eps))
    return Z


def nnm_grouped(X3: torch.Tensor, f: int) -> torch.Tensor:
    """NNM for G independent (m, d) groups: per-group f32 Grams on the
    exactly-once MFMA kernel (tiny launches) + ONE (1/k)-mask bmm that
    mixes neighbors without materializing gathered/f32 copies."""
    G, m, d = X3.shape
    k = m - f
    if _gpu(X3):
        Gm = torch.stack([gram(X3[g]) for g in range(G)])  # (G, m, m) f32
    else:
        Xf = X3.float()
        Gm = torch.bmm(Xf, Xf.transpose(1, 2))
    norms = torch.diagonal(Gm, dim1=1, dim2=2)
    D2 = (norms[:, :, None] + norms[:, None, :] - 2.0 * Gm).clamp_(min=0.0)
    idx = torch.topk(D2, k=k, dim=2, largest=False).indices
    mask = torch.zeros_like(Gm)
    mask.scatter_(2, idx, 1.0 / k)
    return torch.bmm(mask.to(X3.dtype), X3)


def centered_clipping(
    X: torch.Tensor,
    *,
    c_tau: float,
    M: int = 10,
    eps: float = 1e-12,
    init: str = "mean",
) -> torch.Tensor:
    if not _gpu(X) or X.shape[0] > 1024:
        # extension's cc_iter TORCH_CHECKs n <= 1024; fall back on-device
        return F.centered_clipping(X, c_tau=c_tau, M=M, eps=eps, init=init)
    ext = _hip.require()
    Xc = X.contiguous()
    if init == "mean":
        v = Xc.float().mean(dim=0)
    elif init == "median":
        v = median(Xc).float()
    else:
        v = torch.zeros(X.shape[1], device=X.device, dtype=torch.float32)
    for _ in range(M):
        v = ext.cc_iter(Xc, v, float(c_tau), float(eps))
    return v.to(X.dtype)


# -- pre-aggregators --------------------------------------------------------


def bucketing(
    X: torch.Tensor, bucket_size: int, perm: Optional[Sequence[int]] = None
) -> torch.Tensor:
    if _gpu(X):
        n = X.shape[0]
        if perm is None:
            perm_t = torch.randperm(n, device=X.device, dtype=torch.int32)
        else:
            perm_t = torch.as_tensor(list(perm), device=X.device, dtype=torch.int32)
        return _hip.require().bucket_mean(X.contiguous(), perm_t, int(bucket_size))
    return F.bucketing(X, bucket_size, perm)


def nnm(X: torch.Tensor, f: int) -> torch.Tensor:
    n = X.shape[0]
    k = n - f
    if _gpu(X):
        D2 = pairwise_sq_dists(X)
        idx = torch.topk(D2, k=k, dim=1, largest=False).indices.to(torch.int32)
        return _hip.require().group_mean_rows(X.contiguous(), idx)
    return F.nnm(X, f)


# -- attack math (K14) ------------------------------------------------------


def little(X: torch.Tensor, f: int, N: Optional[int] = None) -> torch.Tensor:
    """'A Little Is Enough': mu + z*sigma, fused single pass on device
    (reference little.py:219-224 chunked sum/sum-sq decomposition)."""
    if not _gpu(X):
        return F.little(X, f, N)
    import math

    n = X.shape[0]
    N_total = N if N is not None else n + f
    s = N_total // 2 + 1 - f
    phi_arg = (N_total - s) / N_total
    # inverse normal CDF via erfinv (host scalar math, no torch RNG)
    z = math.sqrt(2.0) * torch.special.erfinv(
        torch.tensor(2.0 * phi_arg - 1.0, dtype=torch.float64)
    ).item()
    return _hip.require().little_fused(X.contiguous(), float(z))


def gaussian_attack(
    like: torch.Tensor,
    mu: float = 0.0,
    sigma: float = 1.0,
    seed: Optional[int] = None,
) -> torch.Tensor:
    """Philox4x32-10 + Box-Muller device fill — no torch RNG on the hot
    path. Seeded calls are deterministic per (seed, index); the stream is
    the kernel's own, not torch's CPU sequence (callers who need the CPU
    sequence use the functional path)."""
    if not _gpu(like):
        return F.gaussian_attack(like, mu, sigma, seed)
    d = like.shape[-1]
    if seed is None:
        seed = int(torch.randint(0, 2**62, (1,)).item())
    return _hip.require().gaussian_fill(int(d), int(seed), float(mu),
                                        float(sigma), like)


# -- remaining ops: torch composition on either device ----------------------


def minimum_diameter_averaging(X: torch.Tensor, f: int) -> torch.Tensor:
    n = X.shape[0]
    m = n - f
    if _gpu(X):
        D2 = pairwise_sq_dists(X)
        if n <= 64 and 2 <= m < n:
            # fully device-side two-pass branch-and-bound (subsets.hip):
            # no host D2 copy, no host sync — the winning (lex-smallest
            # optimal) subset is selected with a cumsum one-hot mask.
            # Seed the shared bound with a cheap anchor-greedy upper
            # bound (diameter of each row's m nearest rows) so pass 1
            # prunes like the host B&B instead of starting from +inf.
            idxn = torch.topk(D2, k=m, dim=1, largest=False).indices
            sub = D2[idxn[:, :, None], idxn[:, None, :]]  # (n, m, m)
            ub = sub.amax(dim=(1, 2)).min()
            found, subsets = _hip.require().mda_select(D2, int(f), ub)
            if not bool(found.any()):  # degenerate (e.g. all-inf D2)
                subset = F.mda_subset(D2, f)
                idx = torch.tensor(subset, device=X.device, dtype=torch.long)
                return mean_rows(X, idx)
            first = ((found.cumsum(0) == 1) & (found == 1)).to(subsets.dtype)
            subset_t = (subsets * first[:, None]).sum(dim=0).to(torch.int32)
            return mean_rows(X, subset_t)
        subset = F.mda_subset(D2, f)
        idx = torch.tensor(subset, device=X.device, dtype=torch.long)
        return mean_rows(X, idx)
    return F.minimum_diameter_averaging(X, f)


def monna(X: torch.Tensor, f: int, reference_index: int = 0) -> torch.Tensor:
    n = X.shape[0]
    k = n - f
    if _gpu(X):
        # ||x - r||^2 = ||x||^2 + ||r||^2 - 2 x.r ; reuse the norm kernel
        norms = row_sqnorms(X)
        dots = (X.float() @ X.float()[reference_index]).flatten()
        d2 = norms + norms[reference_index] - 2.0 * dots
        d2[reference_index] = -1.0
        idx = torch.argsort(d2, stable=True)[:k]
        return mean_rows(X, idx)
    return F.monna(X, f, reference_index)


_SMEA_COMBOS: dict = {}
_SMEA_COMBOS_MAX_CACHED = 4  # FIFO cap: a combos tensor can be ~33 MB
_SMEA_MAX_COMBOS = 1 << 17


def smea(X: torch.Tensor, f: int) -> torch.Tensor:
    import itertools
    import math

    n = X.shape[0]
    m = n - f
    if (
        _gpu(X)
        and 1 <= m <= 64
        and n <= 128
        and math.comb(n, m) <= _SMEA_MAX_COMBOS
    ):
        # device path (subsets.hip): one wave per subset runs a cyclic
        # Jacobi eigensolve on the centered subset Gram in LDS; the
        # packed (eig, combo) atomicMin reproduces the oracle's
        # lex-smallest tie-break. Combos cached per shape.
        key = (n, m, X.device.index)
        combos = _SMEA_COMBOS.get(key)
        if combos is None:
            if len(_SMEA_COMBOS) >= _SMEA_COMBOS_MAX_CACHED:
                _SMEA_COMBOS.pop(next(iter(_SMEA_COMBOS)))
            combos = torch.tensor(
                list(itertools.combinations(range(n), m)),
                dtype=torch.int32,
                device=X.device,
            )
            _SMEA_COMBOS[key] = combos
        G = gram(X)
        best = _hip.require().smea_select(G, combos)
        idx = (best[0] & 0xFFFFFFFF).long()
        rows = combos[idx].to(torch.int32)
        return mean_rows(X, rows)
    return F.smea(X, f)


class _CafGraphBlock:
    """R CAF rounds captured into one hipGraph (torch.cuda.CUDAGraph is
    hipGraph on ROCm) with DEVICE-side best-lambda/active tracking — the
    eager loop's one-host-sync-per-round made CAF only ~8x the CPU
    reference at 64x65k (round-1 weakness; NOTES_R02 item 3). One replay =
    R rounds = zero Python dispatch and one sync per R rounds.

    Round semantics match F.caf exactly: wsum/mu/power-iteration/best
    tracking per reference caf.py:133-184; rounds after the stop condition
    (weight mass <= n-2f, or the n-round cap via ``rounds_left``) are
    masked out with ``active`` so replaying a full block never corrupts
    the result."""

    def __init__(self, n: int, d: int, dtype, f: int, power_iters: int,
                 R: int, device) -> None:
        ext = _hip.require()
        self.R, self.n, self.d = R, n, d
        self.power_iters = max(1, int(power_iters))
        self.target = float(n - 2 * f)
        self.X = torch.zeros(n, d, dtype=dtype, device=device)
        self.seeds = torch.zeros(R, d, dtype=torch.float32, device=device)
        self.w = torch.ones(n, device=device)
        self.best_lambda = torch.full((), float("inf"), device=device)
        self.best_mu = torch.zeros(d, device=device)
        self.active = torch.ones((), dtype=torch.bool, device=device)
        self.rounds_left = torch.full((), float(n), device=device)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._block(ext)  # warmup (allocator pools the intermediates)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._block(ext)

    def _block(self, ext) -> None:
        X = self.X
        for r in range(self.R):
            w = self.w
            active_in = self.active.clone()
            rl = self.rounds_left.clone()
            wsum = w.sum()
            inv_wsum = torch.reciprocal(wsum.clamp_min(1e-30))
            mu = ext.caf_colsum(X, w, None, inv_wsum)
            v = self.seeds[r]
            v = v / v.norm().clamp_min(1e-20)
            lam = torch.zeros((), device=X.device)
            for _ in range(self.power_iters):
                s_ = ext.caf_matvec(X, mu, v)
                t_ = ext.caf_colsum(X, w * s_, mu, inv_wsum)
                lam = t_.norm()
                v = t_ / lam.clamp_min(1e-20)
            proj = ext.caf_matvec(X, mu, v) ** 2
            pmax = proj.max().clamp_min(1e-20)
            w_next = (w * (1.0 - proj / pmax)).clamp_min(0.0)
            better = (lam < self.best_lambda) & active_in
            self.best_mu.copy_(torch.where(better, mu, self.best_mu))
            self.best_lambda.copy_(torch.where(better, lam, self.best_lambda))
            stop = (wsum <= self.target) | (w_next.sum() <= 0)
            active_next = active_in & ~stop & (rl > 1.0)
            self.w.copy_(torch.where(active_next, w_next, w))
            self.active.copy_(active_next)
            self.rounds_left.copy_(rl - 1.0)

    def run(self, X: torch.Tensor) -> torch.Tensor:
        gen = torch.Generator(device="cpu")
        gen.manual_seed(0)
        self.X.copy_(X)
        self.w.fill_(1.0)
        self.best_lambda.fill_(float("inf"))
        self.best_mu.copy_(self.X.mean(dim=0, dtype=torch.float32))
        self.active.fill_(True)
        self.rounds_left.fill_(float(self.n))
        # seeds drawn in the ORACLE's block pattern (rng stream parity with
        # F.caf, which draws min(block_rows, n-r) rows at a time)
        block_rows = max(1, min(4, (1 << 27) // max(self.d, 1)))
        pending: list = []
        drawn = 0
        total = 0
        self.last_replays = 0
        while total < self.n:
            self.last_replays += 1
            need = min(self.R, self.n - total)
            while sum(p.shape[0] for p in pending) < need and drawn < self.n:
                rows = min(block_rows, self.n - drawn)
                pending.append(torch.randn(rows, self.d, generator=gen))
                drawn += rows
            flat = torch.cat(pending, dim=0) if len(pending) > 1 else pending[0]
            self.seeds[:need].copy_(flat[:need])
            pending = [flat[need:]] if flat.shape[0] > need else []
            self.graph.replay()
            total += self.R  # a partial tail block is masked by rounds_left
            if not bool(self.active):  # one sync per R rounds
                break
        return self.best_mu.to(X.dtype)


_CAF_GRAPHS: dict = {}
_CAF_GRAPHS_MAX = 4  # FIFO cap: each block pins X + seeds device buffers

# graph path bounds: seeds are staged per replay (R*d f32 H2D), so cap d;
# n caps the round count (and kernel TORCH_CHECKs n <= 1024)
_CAF_GRAPH_MAX_D = 1 << 21


def _caf_graph_ok(X: torch.Tensor, f: int) -> bool:
    # OFF by default: measured on MI355X, a replayed 8-round block runs
    # ~3x slower per round than the eager loop (2.1 ms/replay at 64x65k
    # vs ~85 us/round eager; graph 1.9 vs eager 0.97 ms even at 32x2048),
    # so capture only pays when Python dispatch is the true bottleneck —
    # opt in with BYZPY_CAF_GRAPH=1. The block stays parity-tested.
    import os

    if os.environ.get("BYZPY_CAF_GRAPH", "0") != "1":
        return False
    n, d = X.shape
    return n <= 256 and d <= _CAF_GRAPH_MAX_D


def caf(X: torch.Tensor, f: int, *, power_iters: int = 3) -> torch.Tensor:
    """Covariance-agnostic filter on fused K9 kernels (reference
    caf.py:133-184): caf_matvec computes s = (X - mu) @ v, caf_colsum
    computes weighted centered column sums — neither materializes the
    (n, d) f32 diffs matrix and both replace rocblas gemvt (~260 GB/s on
    this skinny shape). Same algorithm and seeded-rng contract as the
    functional oracle. Steady-state shapes run the hipGraph block path
    (_CafGraphBlock); others fall back to the eager loop below."""
    import math
    import os

    n = X.shape[0]
    if not (_gpu(X) and n <= 1024 and n - 2 * f > 0):
        return F.caf(X, f, power_iters=power_iters)
    if _caf_graph_ok(X, f):
        d = X.shape[1]
        R = min(8, n)
        key = (n, d, X.dtype, int(f), int(power_iters), R, X.device.index)
        blk = _CAF_GRAPHS.get(key)
        if blk is None:
            if len(_CAF_GRAPHS) >= _CAF_GRAPHS_MAX:
                _CAF_GRAPHS.pop(next(iter(_CAF_GRAPHS)))
            blk = _CafGraphBlock(n, d, X.dtype, int(f), power_iters, R, X.device)
            _CAF_GRAPHS[key] = blk
        return blk.run(X.contiguous())
    ext = _hip.require()
    d = X.shape[1]
    dev = X.device
    w = torch.ones(n, device=dev)
    gen = torch.Generator(device="cpu")
    gen.manual_seed(0)
    best_lambda = math.inf
    best_mu = X.float().mean(dim=0)
    target = float(n - 2 * f)
    block_rows = max(1, min(4, (1 << 27) // max(d, 1)))
    # At the steady-state shapes (n<=1024, d~65k) the per-round kernel
    # work is a few us — the round cost is host latency: the 3-scalar
    # break-condition sync and the pageable seed-block H2D. Both are
    # hidden without changing any computed value: round r+1's kernels
    # launch while round r's stat readback rides an async pinned copy
    # (speculative rounds launched past a break point are discarded —
    # they only ever consumed RNG stream, never affected the output),
    # and seed blocks stage through double-buffered pinned memory.
    sync_mode = os.environ.get("BYZPY_CAF_SYNC", "0") == "1"  # A/B probe aid
    nopin = os.environ.get("BYZPY_CAF_NOPIN", "0") == "1"  # A/B probe aid
    pin_seeds = (
        not sync_mode and not nopin and block_rows * d * 4 <= (64 << 20)
    )
    if pin_seeds:
        seed_pin = [
            torch.empty(block_rows, d, pin_memory=True) for _ in range(2)
        ]
        seed_dev = [torch.empty(block_rows, d, device=dev) for _ in range(2)]
        seed_ev = [torch.cuda.Event() for _ in range(2)]
    stat_pin = [torch.empty(3, pin_memory=True) for _ in range(2)]
    stat_ev = [torch.cuda.Event() for _ in range(2)]
    pending: list = []  # (mu, slot) rounds not yet validated for break

    def consume_oldest() -> bool:
        nonlocal best_lambda, best_mu
        mu_p, sp = pending.pop(0)
        stat_ev[sp].synchronize()
        wsum_f, wnext_f, lam_f = stat_pin[sp].tolist()
        if lam_f < best_lambda:
            best_lambda = lam_f
            best_mu = mu_p
        return wsum_f <= target or wnext_f <= 0

    seeds = torch.empty(0)
    broke = False
    for r in range(n):
        if r % block_rows == 0:
            rows = min(block_rows, n - r)
            if pin_seeds:
                blk = (r // block_rows) % 2
                seed_ev[blk].synchronize()  # prior H2D out of this slot done
                torch.randn(rows, d, generator=gen, out=seed_pin[blk][:rows])
                seed_dev[blk][:rows].copy_(seed_pin[blk][:rows], non_blocking=True)
                seed_ev[blk].record()
                seeds = seed_dev[blk]
            else:
                seeds = torch.randn(rows, d, generator=gen).to(dev)
        wsum = w.sum()
        inv_wsum = torch.reciprocal(wsum.clamp_min(1e-30))
        mu = ext.caf_colsum(X, w, None, inv_wsum)
        v = seeds[r % block_rows]
        v = v / v.norm().clamp_min(1e-20)
        lam = torch.zeros((), device=dev)
        for _ in range(max(1, power_iters)):
            s = ext.caf_matvec(X, mu, v)
            t = ext.caf_colsum(X, w * s, mu, inv_wsum)
            lam = t.norm()
            v = t / lam.clamp_min(1e-20)
        proj = ext.caf_matvec(X, mu, v) ** 2
        pmax = proj.max().clamp_min(1e-20)
        w_next = (w * (1.0 - proj / pmax)).clamp_min(0.0)
        slot = r % 2
        stat_pin[slot].copy_(
            torch.stack([wsum, w_next.sum(), lam]), non_blocking=True
        )
        stat_ev[slot].record()
        pending.append((mu, slot))
        w = w_next
        if len(pending) == (1 if sync_mode else 2) and consume_oldest():
            broke = True  # the still-pending round is speculative: discard
            break
    while not broke and pending:
        if consume_oldest():
            break
    return best_mu.to(X.dtype)
