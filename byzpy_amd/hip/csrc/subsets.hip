// Device-side exact subset searches for gfx950.
//
// SMEA: among C(n, m) row subsets, find the one whose centered subset
// Gram H G_sub H has the smallest max eigenvalue (reference
// smea.py:63-107 runs batched host eigvalsh). Here: ONE WAVE per subset,
// the m x m centered Gram staged in LDS, a fixed-sweep cyclic Jacobi
// eigensolve executed wave-cooperatively, and a packed (eigkey, combo)
// u64 atomicMin so ties resolve to the lexicographically-smallest combo
// index exactly like the oracle's block scan.
//
// MDA: exact minimum-diameter (n-f)-subset (reference
// minimum_diameter_average.py:267-386 runs seeded-DFS subtask batches;
// round 1 copied D2 to the host for a serial C++ branch-and-bound).
// Here the reference's seed_prefix=2 decomposition maps to workgroups:
// one workgroup per (a, b) prefix pair runs a bounded DFS over the
// remaining candidates with per-level incremental max-distance arrays in
// LDS, sharing the global best diameter through an atomicMin every
// improvement. A second bounded pass recovers the lexicographically
// smallest optimal subset (host parity: bind.cpp mda_search's dfs2).
#include "common.h"

namespace {

typedef unsigned int u32;
typedef unsigned long long u64;

// non-negative f32 -> monotone u32 (covers tiny negative rounding too)
DEV u32 f32_key(float v) {
  u32 b = __float_as_uint(v);
  return (b >> 31) ? ~b : (b | 0x80000000u);
}

// ---------------------------------------------------------------------------
// SMEA
// ---------------------------------------------------------------------------

constexpr int SMEA_SWEEPS = 10;

// one 64-lane wave per subset; LDS: A[m*m] + rows[m]
__global__ void __launch_bounds__(64)
smea_select_kernel(const float* __restrict__ G, const int* __restrict__ combos,
                   int n, int m, int C, u64* __restrict__ best) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* A = smem;                 // m*m
  float* rmean = smem + m * m;     // m
  __shared__ int rows[64];
  const int lane = threadIdx.x;

  for (int c = blockIdx.x; c < C; c += gridDim.x) {
    if (lane < m) rows[lane] = combos[(long)c * m + lane];
    __syncthreads();
    // subset Gram
    for (int e = lane; e < m * m; e += 64) {
      const int i = e / m, j = e % m;
      A[e] = G[(long)rows[i] * n + rows[j]];
    }
    __syncthreads();
    // center: A <- A - rmean_i - rmean_j + gmean
    if (lane < m) {
      float s = 0.0f;
      for (int j = 0; j < m; ++j) s += A[lane * m + j];
      rmean[lane] = s / m;
    }
    __syncthreads();
    float gm = 0.0f;
    for (int j = 0; j < m; ++j) gm += rmean[j];
    gm /= m;
    for (int e = lane; e < m * m; e += 64) {
      const int i = e / m, j = e % m;
      A[e] = A[e] - rmean[i] - rmean[j] + gm;
    }
    __syncthreads();

    // cyclic Jacobi, fixed sweeps (quadratic convergence: ~6 sweeps is
    // machine precision at m <= 64)
    for (int sweep = 0; sweep < SMEA_SWEEPS; ++sweep) {
      for (int p = 0; p < m - 1; ++p) {
        for (int q = p + 1; q < m; ++q) {
          const float apq = A[p * m + q];
          if (fabsf(apq) < 1e-12f) continue;
          const float app = A[p * m + p];
          const float aqq = A[q * m + q];
          const float tau = (aqq - app) / (2.0f * apq);
          const float t = (tau >= 0.0f ? 1.0f : -1.0f) /
                          (fabsf(tau) + sqrtf(1.0f + tau * tau));
          const float cth = rsqrtf(1.0f + t * t);
          const float sth = t * cth;
          // rows p, q (each lane owns columns k, k+64, ...)
          for (int k = lane; k < m; k += 64) {
            const float akp = A[p * m + k];
            const float akq = A[q * m + k];
            A[p * m + k] = cth * akp - sth * akq;
            A[q * m + k] = sth * akp + cth * akq;
          }
          __syncthreads();
          // columns p, q
          for (int k = lane; k < m; k += 64) {
            const float apk = A[k * m + p];
            const float aqk = A[k * m + q];
            A[k * m + p] = cth * apk - sth * aqk;
            A[k * m + q] = sth * apk + cth * aqk;
          }
          __syncthreads();
          // exact 2x2 block (reduces accumulated error)
          if (lane == 0) {
            const float d = app - aqq;
            const float app2 =
                cth * cth * app - 2.0f * sth * cth * apq + sth * sth * aqq;
            const float aqq2 =
                sth * sth * app + 2.0f * sth * cth * apq + cth * cth * aqq;
            (void)d;
            A[p * m + p] = app2;
            A[q * m + q] = aqq2;
            A[p * m + q] = 0.0f;
            A[q * m + p] = 0.0f;
          }
          __syncthreads();
        }
      }
    }
    float lam = -3.4e38f;
    for (int j = lane; j < m; j += 64) lam = fmaxf(lam, A[j * m + j]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      lam = fmaxf(lam, __shfl_down(lam, off, 64));
    if (lane == 0) {
      const u64 key = ((u64)f32_key(lam) << 32) | (u32)c;
      atomicMin(best, key);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// MDA
// ---------------------------------------------------------------------------

// One workgroup (one wave) per lexicographic P-element prefix (P=2
// pairs or P=3 triples — triples cut the slowest prefix's serial DFS
// tail by ~n and put ~C(n,3) independent searchers on the chip).
// LDS: D2[n][n] + maxd[m+1][n] level stack.
// PASS 1 (FIND=false): branch-and-bound on the shared best diameter
// (strict <, so tied subsets prune instantly).
// PASS 2 (FIND=true): bounded lex-order DFS; the FIRST completion is the
// prefix's lex-smallest subset with diameter <= bound. The dispatcher
// picks the first found prefix in lex order = the globally lex-smallest
// optimal subset (host parity: bind.cpp mda_search's dfs2).
template <bool FIND>
__global__ void __launch_bounds__(64)
mda_dfs_kernel(const float* __restrict__ D2g, const int* __restrict__ prefixes,
               int P, int n, int m, int npre, u32* __restrict__ best,
               int* __restrict__ out_subsets, int* __restrict__ out_found) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* D2 = smem;                  // n*n
  float* maxd = smem + n * n;        // (m+1) * n level stack
  __shared__ int chosen[66];         // chosen[level] = element pushed
  __shared__ float pdiam[66];        // diameter after level pushes
  __shared__ int cand[66];           // next candidate index per level
  __shared__ int pref[4];
  const int lane = threadIdx.x;

  for (int e = lane; e < n * n; e += 64) D2[e] = D2g[e];
  __syncthreads();

  auto read_best = [&]() -> float {
    const u32 bk = *(volatile u32*)best;
    const u32 bb = (bk & 0x80000000u) ? (bk ^ 0x80000000u) : ~bk;
    return __uint_as_float(bb);
  };
  const float bound = FIND ? read_best() : 0.0f;

  for (int pi = blockIdx.x; pi < npre; pi += gridDim.x) {
    if (lane < P) pref[lane] = prefixes[(long)pi * P + lane];
    if (FIND && lane == 0) out_found[pi] = 0;
    __syncthreads();
    // prefix diameter + level-0 max-distance row
    float diam0 = 0.0f;
    for (int i = 0; i < P; ++i)
      for (int j = i + 1; j < P; ++j)
        diam0 = fmaxf(diam0, D2[pref[i] * n + pref[j]]);
    for (int k = lane; k < n; k += 64) {
      float v = D2[pref[0] * n + k];
      for (int i = 1; i < P; ++i) v = fmaxf(v, D2[pref[i] * n + k]);
      maxd[0 * n + k] = v;
    }
    __syncthreads();
    const int need = m - P;  // elements beyond the prefix
    if (need == 0) {
      if (FIND) {
        if (lane == 0 && diam0 <= bound) {
          out_found[pi] = 1;
          for (int i = 0; i < P; ++i) out_subsets[(long)pi * m + i] = pref[i];
        }
      } else if (lane == 0) {
        atomicMin(best, f32_key(diam0));
      }
      __syncthreads();
      continue;
    }
    {
      const float cb = FIND ? bound : read_best();
      const bool dead = FIND ? (diam0 > cb) : (diam0 >= cb);
      if (dead) { __syncthreads(); continue; }
    }
    pdiam[0] = diam0;
    cand[0] = pref[P - 1] + 1;
    __syncthreads();

    int depth = 0;
    bool done = false;
    while (!done && depth >= 0) {
      // fresh shared-bound read per node: staleness was measured FAR
      // more expensive than the L2 read (8-node caching exploded the
      // node count 12x — collaborative pruning needs tight bounds)
      const float cur_best = FIND ? bound : read_best();
      const int start = cand[depth];
      const int maxj = n - (need - depth - 1);
      // WAVE-PARALLEL candidate scan: n <= 64, so one ballot evaluates
      // every remaining candidate at this level at once
      const int j = start + lane;
      bool ok = false;
      float dj = 0.0f;
      if (j < maxj) {
        dj = fmaxf(pdiam[depth], maxd[depth * n + j]);
        ok = FIND ? (dj <= cur_best) : (dj < cur_best);
      }
      const unsigned long long mask = __ballot(ok);
      if (mask == 0ull) {
        --depth;  // level exhausted: backtrack
        continue;
      }
      if (depth + 1 == need) {
        if (FIND) {
          // lex-first completion = lowest ok lane
          const int sel = __ffsll((long long)mask) - 1;
          if (lane == 0) {
            out_found[pi] = 1;
            for (int i = 0; i < P; ++i)
              out_subsets[(long)pi * m + i] = pref[i];
            for (int t2 = 0; t2 < need - 1; ++t2)
              out_subsets[(long)pi * m + P + t2] = chosen[t2];
            out_subsets[(long)pi * m + P + need - 1] = start + sel;
          }
          done = true;
          break;
        }
        // pass 1: every ok lane is a completion — fold them all in one
        // wave-reduced min and exhaust the level
        float dmin = ok ? dj : 3.4e38f;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
          dmin = fminf(dmin, __shfl_down(dmin, off, 64));
        if (lane == 0) atomicMin(best, f32_key(dmin));
        __syncthreads();
        --depth;
        continue;
      }
      // push the first viable candidate
      const int sel = __ffsll((long long)mask) - 1;
      const int jsel = start + sel;
      const float djsel = fmaxf(pdiam[depth], maxd[depth * n + jsel]);
      cand[depth] = jsel + 1;
      chosen[depth] = jsel;
      pdiam[depth + 1] = djsel;
      for (int k = lane; k < n; k += 64)
        maxd[(depth + 1) * n + k] =
            fmaxf(maxd[depth * n + k], D2[jsel * n + k]);
      __syncthreads();
      cand[depth + 1] = jsel + 1;
      ++depth;
    }
    __syncthreads();
  }
}

}  // namespace

void launch_smea_select(const float* G, const int* combos, int n, int m,
                        int C, unsigned long long* best, hipStream_t stream) {
  const int grid = C < 8192 ? C : 8192;
  const size_t lds = (size_t)(m * m + m) * sizeof(float);
  hipLaunchKernelGGL(smea_select_kernel, dim3(grid), dim3(64), lds, stream, G,
                     combos, n, m, C, best);
}

void launch_mda_pass1(const float* D2, const int* prefixes, int P, int n,
                      int m, int npre, unsigned int* best,
                      hipStream_t stream) {
  const int grid = npre < 16384 ? npre : 16384;
  const size_t lds = (size_t)(n * n + (m + 1) * n) * sizeof(float);
  hipLaunchKernelGGL(mda_dfs_kernel<false>, dim3(grid), dim3(64), lds, stream,
                     D2, prefixes, P, n, m, npre, best, nullptr, nullptr);
}

void launch_mda_pass2(const float* D2, const int* prefixes, int P, int n,
                      int m, int npre, unsigned int* best, int* out_subsets,
                      int* out_found, hipStream_t stream) {
  const int grid = npre < 16384 ? npre : 16384;
  const size_t lds = (size_t)(n * n + (m + 1) * n) * sizeof(float);
  hipLaunchKernelGGL(mda_dfs_kernel<true>, dim3(grid), dim3(64), lds, stream,
                     D2, prefixes, P, n, m, npre, best, out_subsets,
                     out_found);
}
