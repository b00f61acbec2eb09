// Generic streaming radix-select engine for large-n column order
// statistics (SURVEY.md K1-K3 at n > ~192, where the in-LDS sort
// re-touches every element log^2 P times and collapses to ~0.1-0.6 TB/s).
//
// The specialized 2-pass bf16 MEDIAN kernels live in colsel.hip
// (rsel_pass1/2). This file generalizes the idea to every other
// mode x dtype combination as a sequence of STREAMING passes, each
// HBM-bound and coalesced the same way (64 adjacent columns per block,
// 4 row slices per column):
//
//   level pass   : 64-bucket histogram of one 6-bit digit of a MONOTONE
//                  u32 key, restricted to elements whose higher digits
//                  match the per-column prefix resolved so far; a scan
//                  advances the prefix for 1 or 2 target ranks. bf16
//                  value keys need 3 levels, f32 keys 6. 64 bins (not
//                  256) keep the histogram at 16.6 KB LDS so 4 blocks
//                  (32 waves) co-reside per CU — the 256-bin variant's
//                  66-132 KB allowed only 8-16 waves and ran 7x slower
//                  than the identically-striding sum kernels.
//   sum pass     : one more read of X accumulating the value sums /
//                  boundary-tie counts the mode's closed form needs.
//
// Pass counts: f32 MEDIAN 6+0, TRIMMED 3+1 (bf16) / 6+1 (f32),
// MEAMED = median levels + 4 dev-key levels + 1 sum pass (dev = |v - med|
// is an exact f32, so its key needs all 4 digits — a bf16-rounded dev key
// would merge near-ties and push boundary elements across med, a
// 2*rho/(n-f) output error, not an ulp).
//
// Per-column selection state in global scratch, 4 u32 per column:
//   [0] prefix for rank 0   [1] count of keys strictly below prefix0
//   [2] prefix for rank 1   [3] count below prefix1
// Ranks are 1-based. Counts are exact after the last level.
#include "common.h"
#include <cstdlib>

namespace {

typedef unsigned int u32;

constexpr int RS_COLS = 64;

// -- monotone key transforms ------------------------------------------------

// bf16 bits -> sortable u16, placed in the TOP 16 bits of the u32 key
DEV u32 key_from_bf16_bits(u32 bits) {
  const u32 s = (bits >> 15) & 1u;
  return ((bits ^ (0x8000u + s * 0x7FFFu)) & 0xFFFFu) << 16;
}

// f32 -> sortable u32 (sign-magnitude flip; NaN sorts above +inf)
DEV u32 key_from_f32(float v) {
  u32 bits = __float_as_uint(v);
  return (bits >> 31) ? ~bits : (bits | 0x80000000u);
}

DEV float f32_from_key(u32 key) {
  const u32 bits = (key & 0x80000000u) ? (key ^ 0x80000000u) : ~key;
  return __uint_as_float(bits);
}

DEV float bf16val_from_key(u32 key) {
  const u32 k16 = key >> 16;
  unsigned short bits = (k16 & 0x8000u) ? (unsigned short)(k16 ^ 0x8000u)
                                        : (unsigned short)(~k16 & 0xFFFFu);
  union { unsigned short s; __hip_bfloat16 h; } c;
  c.s = bits;
  return __bfloat162float(c.h);
}

// |v - med| as a monotone key: non-negative f32 bits are already ordered;
// NaN (v and med both inf) forced to the top so it is trimmed, not kept
DEV u32 key_from_dev(float v, float med) {
  const float dev = fabsf(v - med);
  if (dev != dev) return 0xFFFFFFFFu;
  return __float_as_uint(dev);
}

enum KeyKind { VAL_BF16 = 0, VAL_F32 = 1, DEV_KEY = 2 };

template <typename T, int KK>
DEV u32 make_key(T raw, float med) {
  if (KK == VAL_BF16) {
    union { T t; unsigned short s; } c;
    c.t = raw;
    return key_from_bf16_bits((u32)c.s);
  } else if (KK == VAL_F32) {
    return key_from_f32(to_f<T>(raw));
  } else {
    return key_from_dev(to_f<T>(raw), med);
  }
}

template <typename T, int KK>
DEV float val_from_key(u32 key) {
  return (KK == VAL_BF16) ? bf16val_from_key(key) : f32_from_key(key);
}

// bf16 value key in the 16-bit domain (no <<16): the level loop's
// per-element fast path — the u32 state convention is translated once
// per column instead (see rsel_level_kernel)
template <typename T>
DEV u32 make_key16(T raw) {
  union { T t; unsigned short s; } c;
  c.t = raw;
  const u32 bits = (u32)c.s;
  const u32 sg = (bits >> 15) & 1u;
  return (bits ^ (0x8000u + sg * 0x7FFFu)) & 0xFFFFu;
}

// -- level pass -------------------------------------------------------------

// TWO=true tracks two ranks with a rank-packed 16/16 histogram; TWO=false
// tracks one rank with a COLUMN-PARITY-packed 16/16 histogram (two
// adjacent columns share one u32). Counts <= n <= 65535 either way.
// Digits may OVERLAP already-resolved bits (the bin then carries a
// constant prefix offset inside the matched set — the scan and the
// prefix OR are unaffected), so a fixed 6-bit digit covers any key
// width with ceil(bits/6) levels.
constexpr int RS_THREADS = 512;
constexpr int RS_BINS = 64;

template <typename T, int KK, bool TWO>
__global__ void __launch_bounds__(RS_THREADS, 4)
rsel_level_kernel(const T* __restrict__ X, const float* __restrict__ med,
                  u32* __restrict__ state, int n, long d, int shift,
                  u32 hi_mask, u32 t0, u32 t1) {
  // LDS words per bin: TWO -> one u32 per column (rank-packed);
  // ONE -> one u32 per column PAIR (parity-packed)
  constexpr int W = TWO ? RS_COLS : RS_COLS / 2;
  extern __shared__ __attribute__((aligned(16))) u32 rs_lds[];
  u32 (*cnt)[W + 1] = reinterpret_cast<u32(*)[W + 1]>(rs_lds);
  const int t = threadIdx.x;
  const int c = t & 63;
  const int slice = t >> 6;
  constexpr int SLICES = RS_THREADS / RS_COLS;
  const long col0 = (long)blockIdx.x * RS_COLS;
  const int cols = (int)min((long)RS_COLS, d - col0);
  for (int i = t; i < RS_BINS * (W + 1); i += RS_THREADS)
    reinterpret_cast<u32*>(cnt)[i] = 0;
  __syncthreads();
  if (c < cols) {
    const long col = col0 + c;
    const u32 p0 = state[col * 4 + 0];
    const u32 p1 = TWO ? state[col * 4 + 2] : 0;
    const float m = (KK == DEV_KEY) ? med[col] : 0.0f;
    const u32 one = TWO ? 1u : ((c & 1) ? 0x10000u : 1u);
    const int w = TWO ? c : (c >> 1);
    const T* xc = X + col;
    // The streaming loop is VALU-bound (PMC: 17.4 VALU/element at 64
    // bins before this rework), so per-element arithmetic is hoisted
    // per COLUMN wherever possible:
    //  - the prefix compare ((key ^ p) & hi_mask) == 0 becomes
    //    (key & hi_mask) == qp with qp = p & hi_mask precomputed, and
    //    the AND is shared between both rank trackers;
    //  - bf16 value keys live in the 16-bit domain (the u16 key is
    //    produced WITHOUT the <<16 normalization; shift/masks/prefixes
    //    are moved down 16 once per column instead).
    constexpr bool K16 = (KK == VAL_BF16);
    const int sh = K16 ? shift - 16 : shift;
    const u32 hm = K16 ? (hi_mask >> 16) : hi_mask;
    const u32 q0 = (K16 ? (p0 >> 16) : p0) & hm;
    const u32 q1 = TWO ? (((K16 ? (p1 >> 16) : p1)) & hm) : 0;
    // 8 loads in flight per slice thread; UNCONDITIONAL zero-capable
    // atomics (an `if (inc)` guard emits per-element exec save/restore)
    int row = slice;
    for (; row + 7 * SLICES < n; row += 8 * SLICES) {
      T raw[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) raw[j] = xc[(long)(row + SLICES * j) * d];
#pragma unroll
      for (int q = 0; q < 8; ++q) {
        const u32 key = K16 ? make_key16<T>(raw[q]) : make_key<T, KK>(raw[q], m);
        const u32 kh = key & hm;
        const bool m0 = kh == q0;
        u32 inc;
        if (TWO) {
          const bool m1 = kh == q1;
          inc = (m0 ? 1u : 0u) | (m1 ? 0x10000u : 0u);
        } else {
          inc = m0 ? one : 0u;
        }
        atomicAdd(&cnt[(key >> sh) & (RS_BINS - 1)][w], inc);
      }
    }
    for (; row < n; row += SLICES) {
      const u32 key = K16 ? make_key16<T>(xc[(long)row * d])
                          : make_key<T, KK>(xc[(long)row * d], m);
      const u32 kh = key & hm;
      const bool m0 = kh == q0;
      u32 inc;
      if (TWO) {
        const bool m1 = kh == q1;
        inc = (m0 ? 1u : 0u) | (m1 ? 0x10000u : 0u);
      } else {
        inc = m0 ? one : 0u;
      }
      atomicAdd(&cnt[(key >> sh) & (RS_BINS - 1)][w], inc);
    }
  }
  __syncthreads();
  // one thread per column: advance prefix by the resolved digit
  if (t < cols) {
    const long col = col0 + t;
    u32 p0 = state[col * 4 + 0], b0 = state[col * 4 + 1];
    u32 p1 = 0, b1 = 0;
    if (TWO) { p1 = state[col * 4 + 2]; b1 = state[col * 4 + 3]; }
    u32 run0 = 0, run1 = 0;
    bool got0 = false, got1 = !TWO;
    const int w = TWO ? t : (t >> 1);
    const int sh = TWO ? 0 : (t & 1) * 16;
#pragma unroll 4
    for (int b = 0; b < RS_BINS; ++b) {
      const u32 packed = cnt[b][w];
      const u32 c0 = TWO ? (packed & 0xFFFFu) : ((packed >> sh) & 0xFFFFu);
      if (!got0 && t0 <= b0 + run0 + c0) {
        p0 |= (u32)b << shift;
        b0 += run0;
        got0 = true;
      }
      run0 += c0;
      if (TWO) {
        const u32 c1 = packed >> 16;
        if (!got1 && t1 <= b1 + run1 + c1) {
          p1 |= (u32)b << shift;
          b1 += run1;
          got1 = true;
        }
        run1 += c1;
      }
    }
    state[col * 4 + 0] = p0;
    state[col * 4 + 1] = b0;
    if (TWO) {
      state[col * 4 + 2] = p1;
      state[col * 4 + 3] = b1;
    }
  }
}

// -- finalize kernels -------------------------------------------------------

// median output (or med staging for MEAMED): 0.5 * (v(rank_lo) + v(rank_hi))
template <typename T, int KK, bool TO_F32>
__global__ void rsel_median_out_kernel(const u32* __restrict__ state,
                                       void* __restrict__ out, long d) {
  const long col = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= d) return;
  const float lo = val_from_key<T, KK>(state[col * 4 + 0]);
  const float hi = val_from_key<T, KK>(state[col * 4 + 2]);
  const float m = 0.5f * (lo + hi);
  if (TO_F32)
    reinterpret_cast<float*>(out)[col] = m;
  else
    reinterpret_cast<T*>(out)[col] = from_f<T>(m);
}

// TRIMMED closed form from boundary keys L (rank f+1) and U (rank n-f):
//   S = sum(L < v < U) + keptL * val(L) + keptU * val(U)
// with keptL = min(belowL + eqL, n-f) - f and keptU = (n-f) - belowU.
template <typename T, int KK>
__global__ void __launch_bounds__(RS_THREADS, 2)
rsel_trimmed_sum_kernel(const T* __restrict__ X, const u32* __restrict__ state,
                        T* __restrict__ out, int n, long d, int f) {
  __shared__ float sMid[RS_COLS];
  __shared__ u32 cEqL[RS_COLS];
  const int t = threadIdx.x;
  const int c = t & 63;
  const int slice = t >> 6;
  const long col0 = (long)blockIdx.x * RS_COLS;
  const int cols = (int)min((long)RS_COLS, d - col0);
  if (t < RS_COLS) { sMid[t] = 0.0f; cEqL[t] = 0; }
  __syncthreads();
  if (c < cols) {
    const long col = col0 + c;
    const u32 keyL = state[col * 4 + 0];
    const u32 keyU = state[col * 4 + 2];
    const T* xc = X + col;
    // middle sum accumulated DIRECTLY (strictly between the boundary
    // keys): a prefix-difference formulation turns -inf rows into
    // (-inf) - (-inf) = NaN
    float my_sMid = 0.0f;
    u32 my_cEqL = 0;
    int row = slice;
    for (; row + 24 < n; row += 32) {
      T raw[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) raw[j] = xc[(long)(row + 8 * j) * d];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const u32 key = make_key<T, KK>(raw[q], 0.0f);
        if (key > keyL && key < keyU) my_sMid += to_f<T>(raw[q]);
        if (key == keyL) ++my_cEqL;
      }
    }
    for (; row < n; row += 8) {
      const T raw = xc[(long)row * d];
      const u32 key = make_key<T, KK>(raw, 0.0f);
      if (key > keyL && key < keyU) my_sMid += to_f<T>(raw);
      if (key == keyL) ++my_cEqL;
    }
    atomicAdd(&sMid[c], my_sMid);
    atomicAdd(&cEqL[c], my_cEqL);
  }
  __syncthreads();
  if (t < cols) {
    const long col = col0 + t;
    const u32 kL = state[col * 4 + 0], kU = state[col * 4 + 2];
    const u32 aL = state[col * 4 + 1], aU = state[col * 4 + 3];
    const float vL = val_from_key<T, KK>(kL);
    const u32 keep = (u32)(n - f);
    float S;
    if (kL == kU) {
      S = (float)(n - 2 * f) * vL;
    } else {
      const float vU = val_from_key<T, KK>(kU);
      const u32 keptL = min(aL + cEqL[t], keep) - (u32)f;  // provably >= 1
      const u32 keptU = keep - aU;                          // provably >= 1
      S = sMid[t] + (float)keptL * vL + (float)keptU * vU;
    }
    out[col] = from_f<T>(S / (float)(n - 2 * f));
  }
}

// MEAMED finish: rho = dev key at rank n-f (state slot 0, ONE-rank run).
//   S = sum(v : dev < rho) + ties, taking ties below the median first
// (matches the LDS kernel's shrink-from-the-worse-end policy, which keeps
// the left end on equal deviation). Tied values sharing an exact dev key
// are taken at their mean.
template <typename T, int KK>
__global__ void __launch_bounds__(RS_THREADS, 2)
rsel_meamed_sum_kernel(const T* __restrict__ X, const float* __restrict__ med,
                       const u32* __restrict__ state, T* __restrict__ out,
                       int n, long d, int f) {
  __shared__ float sKept[RS_COLS], sEqL[RS_COLS], sEqR[RS_COLS];
  __shared__ u32 cEqL[RS_COLS], cEqR[RS_COLS];
  const int t = threadIdx.x;
  const int c = t & 63;
  const int slice = t >> 6;
  const long col0 = (long)blockIdx.x * RS_COLS;
  const int cols = (int)min((long)RS_COLS, d - col0);
  if (t < RS_COLS) {
    sKept[t] = 0.0f; sEqL[t] = 0.0f; sEqR[t] = 0.0f;
    cEqL[t] = 0; cEqR[t] = 0;
  }
  __syncthreads();
  if (c < cols) {
    const long col = col0 + c;
    const u32 rho = state[col * 4 + 0];
    const float m = med[col];
    float my_sKept = 0.0f, my_sEqL = 0.0f, my_sEqR = 0.0f;
    u32 my_cEqL = 0, my_cEqR = 0;
    const T* xc = X + col;
    int row = slice;
    for (; row + 24 < n; row += 32) {
      T raw[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) raw[j] = xc[(long)(row + 8 * j) * d];
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const float v = to_f<T>(raw[q]);
        const u32 key = key_from_dev(v, m);
        if (key < rho) {
          my_sKept += v;
        } else if (key == rho) {
          if (v < m) { my_sEqL += v; ++my_cEqL; }
          else       { my_sEqR += v; ++my_cEqR; }
        }
      }
    }
    for (; row < n; row += 8) {
      const T raw = xc[(long)row * d];
      const float v = to_f<T>(raw);
      const u32 key = key_from_dev(v, m);
      if (key < rho) {
        my_sKept += v;
      } else if (key == rho) {
        if (v < m) { my_sEqL += v; ++my_cEqL; }
        else       { my_sEqR += v; ++my_cEqR; }
      }
    }
    atomicAdd(&sKept[c], my_sKept);
    atomicAdd(&sEqL[c], my_sEqL);
    atomicAdd(&sEqR[c], my_sEqR);
    atomicAdd(&cEqL[c], my_cEqL);
    atomicAdd(&cEqR[c], my_cEqR);
  }
  __syncthreads();
  if (t < cols) {
    const long col = col0 + t;
    const u32 below = state[col * 4 + 1];
    const u32 keep = (u32)(n - f);
    const u32 ties = keep - below;
    const u32 takeL = min(ties, cEqL[t]);
    const u32 takeR = ties - takeL;
    float S = sKept[t];
    if (takeL && cEqL[t]) S += (sEqL[t] / (float)cEqL[t]) * (float)takeL;
    if (takeR && cEqR[t]) S += (sEqR[t] / (float)cEqR[t]) * (float)takeR;
    out[col] = from_f<T>(S / (float)keep);
  }
}

template <typename T, int KK, bool TWO>
void run_levels(const T* X, const float* med, u32* state, int n, long d,
                u32 t0, u32 t1, hipStream_t stream) {
  const long grid = (d + RS_COLS - 1) / RS_COLS;
  const int w = TWO ? RS_COLS : RS_COLS / 2;
  const size_t lds = (size_t)RS_BINS * (w + 1) * sizeof(u32);
  // 6-bit digits, overlapping at the bottom: bf16 keys (top 16 bits)
  // resolve in 3 levels, f32/dev keys in 6
  static const int plan16[] = {26, 20, 16};
  static const int plan32[] = {26, 20, 14, 8, 2, 0};
  const int* plan = (KK == VAL_BF16) ? plan16 : plan32;
  const int nlev = (KK == VAL_BF16) ? 3 : 6;
  int prev_shift = 32;
  for (int li = 0; li < nlev; ++li) {
    const int shift = plan[li];
    // all bits resolved so far must match the per-column prefix
    const u32 hi_mask =
        (prev_shift >= 32) ? 0u : (0xFFFFFFFFu << prev_shift);
    hipLaunchKernelGGL((rsel_level_kernel<T, KK, TWO>), dim3((unsigned)grid),
                       dim3(RS_THREADS), lds, stream, X, med, state, n, d,
                       shift, hi_mask, t0, t1);
    prev_shift = shift;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host-side drivers (called from bind.cpp; all launches enqueue on
// `stream`, no host syncs). `state` is caller-zeroed d*4 u32 scratch;
// `med` is a d-float scratch for MEAMED.
// ---------------------------------------------------------------------------

void launch_rsel_trimmed_bf16(const __hip_bfloat16* X, __hip_bfloat16* out,
                              unsigned int* state, int n, long d, int f,
                              hipStream_t stream) {
  run_levels<__hip_bfloat16, VAL_BF16, true>(X, nullptr, state, n, d,
                                             (u32)(f + 1), (u32)(n - f),
                                             stream);
  const long grid = (d + RS_COLS - 1) / RS_COLS;
  hipLaunchKernelGGL((rsel_trimmed_sum_kernel<__hip_bfloat16, VAL_BF16>),
                     dim3((unsigned)grid), dim3(RS_THREADS), 0, stream, X,
                     state, out, n, d, f);
}

void launch_rsel_trimmed_f32(const float* X, float* out, unsigned int* state,
                             int n, long d, int f, hipStream_t stream) {
  run_levels<float, VAL_F32, true>(X, nullptr, state, n, d, (u32)(f + 1),
                                   (u32)(n - f), stream);
  const long grid = (d + RS_COLS - 1) / RS_COLS;
  hipLaunchKernelGGL((rsel_trimmed_sum_kernel<float, VAL_F32>),
                     dim3((unsigned)grid), dim3(RS_THREADS), 0, stream, X,
                     state, out, n, d, f);
}

void launch_rsel_median_bf16(const __hip_bfloat16* X, __hip_bfloat16* out,
                             unsigned int* state, int n, long d,
                             hipStream_t stream) {
  const u32 t_lo = (u32)((n - 1) >> 1) + 1, t_hi = (u32)(n >> 1) + 1;
  run_levels<__hip_bfloat16, VAL_BF16, true>(X, nullptr, state, n, d, t_lo,
                                             t_hi, stream);
  const long grid = (d + 255) / 256;
  hipLaunchKernelGGL((rsel_median_out_kernel<__hip_bfloat16, VAL_BF16, false>),
                     dim3((unsigned)grid), dim3(256), 0, stream, state, out,
                     d);
}

void launch_rsel_median_f32(const float* X, float* out, unsigned int* state,
                            int n, long d, hipStream_t stream) {
  const u32 t_lo = (u32)((n - 1) >> 1) + 1, t_hi = (u32)(n >> 1) + 1;
  run_levels<float, VAL_F32, true>(X, nullptr, state, n, d, t_lo, t_hi,
                                   stream);
  const long elems = d;
  const long grid = (elems + 255) / 256;
  hipLaunchKernelGGL((rsel_median_out_kernel<float, VAL_F32, false>),
                     dim3((unsigned)grid), dim3(256), 0, stream, state, out,
                     d);
}

// MEAMED: median levels -> med buffer -> dev-key rank (n-f) -> sum pass.
// Caller re-zeroes `state` between the two selections? No — the dev
// selection reuses slot 0/1 only, and must start from prefix 0: the
// med_out kernel is followed by a device-side state reset here.
namespace {
__global__ void rsel_state_reset_kernel(u32* __restrict__ state, long d) {
  const long col = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (col < d) {
    state[col * 4 + 0] = 0;
    state[col * 4 + 1] = 0;
  }
}
}  // namespace

template <typename T, int KK>
static void rsel_meamed_impl(const T* X, T* out, unsigned int* state,
                             float* med, int n, long d, int f,
                             hipStream_t stream) {
  const u32 t_lo = (u32)((n - 1) >> 1) + 1, t_hi = (u32)(n >> 1) + 1;
  run_levels<T, KK, true>(X, nullptr, state, n, d, t_lo, t_hi, stream);
  const long egrid = (d + 255) / 256;
  hipLaunchKernelGGL((rsel_median_out_kernel<T, KK, true>),
                     dim3((unsigned)egrid), dim3(256), 0, stream, state, med,
                     d);
  hipLaunchKernelGGL(rsel_state_reset_kernel, dim3((unsigned)egrid), dim3(256),
                     0, stream, state, d);
  run_levels<T, DEV_KEY, false>(X, med, state, n, d, (u32)(n - f), 0, stream);
  const long grid = (d + RS_COLS - 1) / RS_COLS;
  hipLaunchKernelGGL((rsel_meamed_sum_kernel<T, KK>), dim3((unsigned)grid),
                     dim3(RS_THREADS), 0, stream, X, med, state, out, n, d, f);
}

void launch_rsel_meamed_bf16(const __hip_bfloat16* X, __hip_bfloat16* out,
                             unsigned int* state, float* med, int n, long d,
                             int f, hipStream_t stream) {
  rsel_meamed_impl<__hip_bfloat16, VAL_BF16>(X, out, state, med, n, d, f,
                                             stream);
}

void launch_rsel_meamed_f32(const float* X, float* out, unsigned int* state,
                            float* med, int n, long d, int f,
                            hipStream_t stream) {
  rsel_meamed_impl<float, VAL_F32>(X, out, state, med, n, d, f, stream);
}
