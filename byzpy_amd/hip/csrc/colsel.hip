// Column-wise order-statistic kernels: per-coordinate
// median / trimmed mean / mean-of-medians over the n-axis of an (n, d)
// matrix, one column per thread. Large n (> 64 at d >= 32K, always past
// 512) is served by the generic streaming radix-select engine in
// rsel.hip; this file keeps the small-n register kernels and the
// mid-n/small-d LDS sort.
//
// Two variants:
//  - register kernel (n <= 64): the column lives in a register array; the
//    bitonic network is fully unrolled so every index is compile-time
//    (guide §5.4 rule 20 — runtime-indexed register arrays spill).
//    Order statistics at runtime positions are extracted with predicated
//    unrolled scans for the same reason.
//  - LDS kernel (64 < n <= 512): one wave per block, each lane owns a
//    column staged in LDS with a +1 pad row stride (bank-conflict free,
//    guide §6 G4); runtime bitonic loops.
//
// Loads are coalesced: adjacent lanes read adjacent columns, so each
// row-iteration is one 256 B (f32) / 128 B (bf16) wave transaction.
#include "common.h"
#include <cstdlib>

namespace {

// +inf (not a big-finite sentinel): adversarial rows can BE +inf
// (InfAttack) and pads must never sort below a real value — with inf pads
// a pad/value tie still yields the mathematically-correct inf statistics.
#define PAD __builtin_huge_valf()

// Copy a wave-uniform int into a VGPR. Unrolled predicates like
// `i == pos` / `i >= n` (i compile-time, pos/n uniform) otherwise become
// batched s_cmp/s_cselect_b64 SGPR mask pairs — ~64 live masks spill
// through v_writelane/readlane and dominate the kernel. Against a VGPR
// operand they lower to v_cmp/v_cndmask instead (2 VALU each, no spills).
DEV int vecify(int x) {
  int r;
  asm("v_mov_b32_e32 %0, %1" : "=v"(r) : "s"(x));
  return r;
}

enum Mode { MEDIAN = 0, TRIMMED = 1, MEAMED = 2 };

// ---------------------------------------------------------------------------
// register variant, n <= P, P in {8, 16, 32, 64}
// ---------------------------------------------------------------------------

// KMAX < P stops the network after phase KMAX: with KMAX = P/2 the two
// halves come out sorted ascending/descending — exactly the input the
// median SELECTION epilogue needs (see median_select_reg), at the cost
// of the full sort minus its entire last merge phase (log2(P) substages).
template <int P, int KMAX = P>
DEV void bitonic_sort_reg(float (&v)[P]) {
#pragma unroll
  for (int k = 2; k <= KMAX; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = (i & k) == 0;
          const float a = v[i], b = v[l];
          const float lo = fminf(a, b), hi = fmaxf(a, b);
          v[i] = asc ? lo : hi;
          v[l] = asc ? hi : lo;
        }
      }
    }
  }
}

// key-value variant (sort k ascending, carry v along)
template <int P>
DEV void bitonic_sort_kv_reg(float (&key)[P], float (&val)[P]) {
#pragma unroll
  for (int k = 2; k <= P; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = (i & k) == 0;
          const float a = key[i], b = key[l];
          if (asc ? (a > b) : (a < b)) {
            key[i] = b; key[l] = a;
            const float t = val[i]; val[i] = val[l]; val[l] = t;
          }
        }
      }
    }
  }
}

// Median ranks P/2-1 and P/2 of [asc32 | desc32] (a bitonic sequence):
// one elementwise min/max pass splits it into the smallest and largest
// P/2 multisets, then rank P/2-1 = max(lower), rank P/2 = min(upper) —
// 3P/2 - 2 ops replacing the last merge phase (P/2*log2(P) CEs) PLUS the
// two O(P) predicated rank extractions. The caller pre-pads so the
// median lands exactly on these ranks (see the MEDIAN pad split below).
template <int P>
DEV void median_select_reg(float (&v)[P], float& mlo, float& mhi) {
#pragma unroll
  for (int i = 0; i < P / 2; ++i) {
    const float a = v[i], b = v[i + P / 2];
    v[i] = fminf(a, b);
    v[i + P / 2] = fmaxf(a, b);
  }
#pragma unroll
  for (int s = P / 4; s >= 1; s >>= 1)
#pragma unroll
    for (int i = 0; i < s; ++i) {
      v[i] = fmaxf(v[i], v[i + s]);
      v[P / 2 + i] = fminf(v[P / 2 + i], v[P / 2 + s + i]);
    }
  mlo = v[0];
  mhi = v[P / 2];
}

// standalone half-sort (see bitonic_sort_pk_half for why source order
// matters): UP=true ascending, UP=false the mirrored descending network
template <int P2, bool UP>
DEV void bitonic_sort_reg_half(float (&v)[P2]) {
#pragma unroll
  for (int k = 2; k <= P2; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P2; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = ((i & k) == 0) == UP;
          const float a = v[i], b = v[l];
          const float lo = fminf(a, b), hi = fmaxf(a, b);
          v[i] = asc ? lo : hi;
          v[l] = asc ? hi : lo;
        }
      }
    }
  }
}

template <int P>
DEV float extract_at(const float (&v)[P], int pos) {
  float r = 0.0f;
#pragma unroll
  for (int i = 0; i < P; ++i)
    if (i == pos) r = v[i];
  return r;
}

// MEAMED keeps two live P-float arrays (values + deviation keys); at P=64
// that needs ~140 VGPRs, above the occupancy heuristic's 128 cap — the
// MEAMED instantiations get __launch_bounds__(256, 2) via this trait so the
// allocator may use 256 VGPRs instead of spilling, while MEDIAN/TRIMMED
// keep the default 4-waves/SIMD occupancy.
template <int P, int MODE, typename T>
__global__ void
__launch_bounds__(256, (MODE == 2 && P >= 64) ? 2 : 4)
colsel_reg_kernel(const T* __restrict__ X, T* __restrict__ out,
                                  int n, long d, int f) {
  const long col0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long col = col0; col < d; col += stride) {
    float v[P];
    T raw[P];
    // Load phase. Two rules keep the address walk at ONE live base:
    // (1) raw loads only — any data-dependent convert inside the walk
    //     would force a wait per element; conversion happens after;
    // (2) sched_barrier(0) pins the load/advance alternation — without it
    //     the compiler batches all P addresses into SGPR pairs and spills
    //     them through v_writelane/readlane (~1100 VALU per column).
    // Once a load ISSUES its address registers are free; the loads stay
    // in flight and the converts below wait on counted vmcnt.
    {
      const T* p = X + col;
#pragma unroll
      for (int i = 0; i < P; ++i) {
        raw[i] = *p;
        if (i + 1 < n) p += d;
        __builtin_amdgcn_sched_barrier(0);
      }
    }
    const int nv = vecify(n);

    float result;
    if (MODE == MEDIAN) {
      // selection network, half at a time IN SOURCE ORDER (see the pk
      // kernel): half A's conversion+sort consume only the first P/2
      // loads and overlap the tail loads. MEDIAN pads split: L low pads
      // (-inf) shift the median ranks to exactly P/2-1 / P/2; a -inf
      // pad tying a -inf data value is value-identical.
      const int n_lo = vecify(n + (P / 2 - 1 - ((n - 1) >> 1)));
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int base = h * (P / 2);
#pragma unroll
        for (int i = base; i < base + P / 2; ++i) v[i] = to_f<T>(raw[i]);
        if (n < P) {
#pragma unroll
          for (int i = base; i < base + P / 2; ++i)
            if (i >= nv) v[i] = (i < n_lo) ? -PAD : PAD;
        }
        if (h == 0)
          bitonic_sort_reg_half<P / 2, true>(
              *reinterpret_cast<float(*)[P / 2]>(&v[0]));
        else
          bitonic_sort_reg_half<P / 2, false>(
              *reinterpret_cast<float(*)[P / 2]>(&v[P / 2]));
      }
      float mlo, mhi;
      median_select_reg<P>(v, mlo, mhi);
      result = (n & 1) ? mlo : 0.5f * (mlo + mhi);
    } else if (MODE == TRIMMED) {
#pragma unroll
      for (int i = 0; i < P; ++i) v[i] = to_f<T>(raw[i]);
      if (n < P) {
#pragma unroll
        for (int i = 0; i < P; ++i)
          if (i >= nv) v[i] = PAD;
      }
      bitonic_sort_reg<P>(v);
      const int fv = vecify(f), nfv = vecify(n - f);
      float s = 0.0f;
#pragma unroll
      for (int i = 0; i < P; ++i)
        if (i >= fv && i < nfv) s += v[i];
      result = s / (float)(n - 2 * f);
    } else {  // MEAMED: mean of the n-f values closest to the median
#pragma unroll
      for (int i = 0; i < P; ++i) v[i] = to_f<T>(raw[i]);
      if (n < P) {
#pragma unroll
        for (int i = 0; i < P; ++i)
          if (i >= nv) v[i] = PAD;
      }
      bitonic_sort_reg<P>(v);
      const int pos_lo = vecify((n - 1) >> 1), pos_hi = vecify(n >> 1);
      const float med =
          0.5f * (extract_at<P>(v, pos_lo) + extract_at<P>(v, pos_hi));
      const int nv2 = vecify(n), nfv = vecify(n - f);
      float dev[P];
#pragma unroll
      for (int i = 0; i < P; ++i)
        dev[i] = (i < nv2) ? fabsf(v[i] - med) : PAD;
      bitonic_sort_kv_reg<P>(dev, v);
      float s = 0.0f;
#pragma unroll
      for (int i = 0; i < P; ++i)
        if (i < nfv) s += v[i];
      result = s / (float)(n - f);
    }
    out[col] = from_f<T>(result);
  }
}

// ---------------------------------------------------------------------------
// Packed bf16 median (n <= 64, even d): TWO adjacent columns per thread.
// Order statistics only need a monotone key, so bf16 values become
// sortable u16 keys (sign-flip transform) and the bitonic network runs on
// v_pk_min_u16 / v_pk_max_u16 — one instruction per compare-exchange for
// BOTH columns (no packed f32 min/max exists on gfx950), with half the
// register footprint (P u32 regs for 2 columns) and 4 B/lane loads.
// ---------------------------------------------------------------------------

typedef unsigned int u32;

DEV u32 pk_min_u16(u32 a, u32 b) {
  u32 r;
  asm("v_pk_min_u16 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}
DEV u32 pk_max_u16(u32 a, u32 b) {
  u32 r;
  asm("v_pk_max_u16 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// bf16 bits -> sortable u16 key, two lanes at once: per half
// key = bits ^ (sign ? 0xFFFF : 0x8000). The packed arithmetic shift
// broadcasts each half's sign to a 0x0000/0xFFFF mask in one op, so the
// transform is 3 VALU (ashr, or, xor) for both columns — it runs once
// per element on a VALU-bound kernel, so width matters.
DEV u32 pk_key_from_bf16(u32 bits) {
  u32 t;
  // shift amount lives in a VGPR as {15, 15}: a bare inline constant
  // would shift only the LOW half (VOP3P literals do not replicate)
  asm("v_pk_ashrrev_i16 %0, %2, %1"
      : "=v"(t)
      : "v"(bits), "v"(0x000F000Fu));
  return bits ^ (t | 0x80008000u);
}

DEV float key_to_float(u32 key16) {
  unsigned short bits =
      (key16 & 0x8000u) ? (unsigned short)(key16 ^ 0x8000u)
                        : (unsigned short)(~key16 & 0xFFFFu);
  union { unsigned short s; __hip_bfloat16 h; } c;
  c.s = bits;
  return __bfloat162float(c.h);
}

template <int P, int KMAX = P>
DEV void bitonic_sort_pk(u32 (&v)[P]) {
#pragma unroll
  for (int k = 2; k <= KMAX; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = (i & k) == 0;
          const u32 a = v[i], b = v[l];
          const u32 lo = pk_min_u16(a, b), hi = pk_max_u16(a, b);
          v[i] = asc ? lo : hi;
          v[l] = asc ? hi : lo;
        }
      }
    }
  }
}

// standalone half-sort for the median selection path: UP=true ascending,
// UP=false the mirrored (descending) network. Sorting a half touches only
// its own P2 registers, so in source order the FIRST half's 240 CEs issue
// while the second half's row loads are still in flight (the shared
// <P, P/2> network's k=2 phase touches all P loads up front and forces a
// full vmcnt(0) wait before any compute).
template <int P2, bool UP>
DEV void bitonic_sort_pk_half(u32 (&v)[P2]) {
#pragma unroll
  for (int k = 2; k <= P2; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P2; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = ((i & k) == 0) == UP;
          const u32 a = v[i], b = v[l];
          const u32 lo = pk_min_u16(a, b), hi = pk_max_u16(a, b);
          v[i] = asc ? lo : hi;
          v[l] = asc ? hi : lo;
        }
      }
    }
  }
}

// packed twin of median_select_reg (see its comment): split the bitonic
// [asc P/2 | desc P/2] keys, then max-reduce the lower / min-reduce the
// upper half — both columns of the pair in one packed op throughout
template <int P>
DEV void pk_median_select(u32 (&v)[P], u32& mlo, u32& mhi) {
#pragma unroll
  for (int i = 0; i < P / 2; ++i) {
    const u32 a = v[i], b = v[i + P / 2];
    v[i] = pk_min_u16(a, b);
    v[i + P / 2] = pk_max_u16(a, b);
  }
#pragma unroll
  for (int s = P / 4; s >= 1; s >>= 1)
#pragma unroll
    for (int i = 0; i < s; ++i) {
      v[i] = pk_max_u16(v[i], v[i + s]);
      v[P / 2 + i] = pk_min_u16(v[P / 2 + i], v[P / 2 + s + i]);
    }
  mlo = v[0];
  mhi = v[P / 2];
}

template <int P>
DEV u32 extract_at_pk(const u32 (&v)[P], int pos) {
  u32 r = 0;
#pragma unroll
  for (int i = 0; i < P; ++i)
    if (i == pos) r = v[i];
  return r;
}

// QUADS=true: each thread owns TWO adjacent packed pairs (4 columns,
// 8 B/lane row reads — wider DRAM bursts; ~150 VGPRs at P=64) with two
// independent sorting networks; QUADS=false: one pair (4 B/lane).
// MODE: MEDIAN or TRIMMED (order statistics read straight off the sorted
// keys; TRIMMED unpacks the kept range and sums in f32).
template <int P, bool QUADS, int MODE = MEDIAN>
__global__ void
// QUADS at P=64 holds two 64-u32 arrays (~150 VGPRs): ask for 3 waves/SIMD
// so the allocator doesn't cap at 128 and spill
__launch_bounds__(256, (QUADS && P >= 64) ? 3
                       : (MODE == MEDIAN && P >= 64) ? 5
                                                     : 4)
colsel_pk_median_bf16(const unsigned short* __restrict__ X,
                                      unsigned short* __restrict__ out, int n,
                                      long d, int f) {
  const long npairs = d >> 1;
  const long nunits = QUADS ? (npairs >> 1) : npairs;
  const long unit0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long rowstride = d >> 1;  // in u32 units
  for (long unit = unit0; unit < nunits; unit += stride) {
    const long pair = QUADS ? unit * 2 : unit;
    u32 v[P], v2[QUADS ? P : 1];
    // raw loads first + pinned alternation: see colsel_reg_kernel's load
    // phase for why (one live base, no SGPR spill storm). The walk
    // condition is vecified too — 64 uniform selects otherwise become
    // batched SGPR mask pairs and spill at QUADS register pressure.
    const u32* p = reinterpret_cast<const u32*>(X) + pair;
    if (n == P) {
      // full tile (the flagship n=64 shape): unconditional walk, no pads
      // — drops ~190 predication VALU ops from a VALU-bound kernel
#pragma unroll
      for (int i = 0; i < P; ++i) {
        if (QUADS) {
          const uint2 w = *reinterpret_cast<const uint2*>(p);
          v[i] = w.x;
          v2[i] = w.y;
        } else {
          v[i] = *p;
        }
        if (i + 1 < P) p += rowstride;
        __builtin_amdgcn_sched_barrier(0);
      }
    } else {
      const int n_walk = vecify(n);
#pragma unroll
      for (int i = 0; i < P; ++i) {
        if (QUADS) {
          const uint2 w = *reinterpret_cast<const uint2*>(p);
          v[i] = w.x;
          v2[i] = w.y;
        } else {
          v[i] = *p;
        }
        p += (i + 1 < n_walk) ? rowstride : 0;
        __builtin_amdgcn_sched_barrier(0);
      }
    }
    const int nv = vecify(n);  // conversion/pad loops + MEAMED epilogue
    u32 sel_lo[2] = {0u, 0u}, sel_hi[2] = {0u, 0u};
    if (MODE == MEDIAN) {
      // selection network, half at a time IN SOURCE ORDER: converting and
      // sorting [0,P/2) consumes only the first P/2 row loads, so those
      // 240 CEs overlap the in-flight tail loads; then the (mirrored
      // descending) upper half, one split pass and two reductions.
      // MEDIAN pads split low/high so the fixed ranks P/2-1 / P/2 hit
      // the true median (low-pad key 0 only ties a negative-NaN data
      // key — NaN order statistics are unspecified in the full-sort
      // path too).
      const int n_lo = vecify(n + (P / 2 - 1 - ((n - 1) >> 1)));
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int base = h * (P / 2);
#pragma unroll
        for (int i = base; i < base + P / 2; ++i) {
          v[i] = pk_key_from_bf16(v[i]);
          if (QUADS) v2[i] = pk_key_from_bf16(v2[i]);
        }
        if (n < P) {
#pragma unroll
          for (int i = base; i < base + P / 2; ++i)
            if (i >= nv) {
              const u32 pad = (i < n_lo) ? 0u : 0xFFFFFFFFu;
              v[i] = pad;
              if (QUADS) v2[i] = pad;
            }
        }
        if (h == 0) {
          bitonic_sort_pk_half<P / 2, true>(
              *reinterpret_cast<u32(*)[P / 2]>(&v[0]));
          if (QUADS)
            bitonic_sort_pk_half<P / 2, true>(
                *reinterpret_cast<u32(*)[P / 2]>(&v2[0]));
        } else {
          bitonic_sort_pk_half<P / 2, false>(
              *reinterpret_cast<u32(*)[P / 2]>(&v[P / 2]));
          if (QUADS)
            bitonic_sort_pk_half<P / 2, false>(
                *reinterpret_cast<u32(*)[P / 2]>(&v2[P / 2]));
        }
      }
      pk_median_select<P>(v, sel_lo[0], sel_hi[0]);
      if (QUADS)
        pk_median_select<P>(*reinterpret_cast<u32(*)[P]>(&v2[0]),
                            sel_lo[1], sel_hi[1]);
    } else {
#pragma unroll
      for (int i = 0; i < P; ++i) {
        v[i] = pk_key_from_bf16(v[i]);
        if (QUADS) v2[i] = pk_key_from_bf16(v2[i]);
      }
      if (n < P) {
#pragma unroll
        for (int i = 0; i < P; ++i)
          if (i >= nv) {
            v[i] = 0xFFFFFFFFu;
            if (QUADS) v2[i] = 0xFFFFFFFFu;
          }
      }
      bitonic_sort_pk<P>(v);
      if (QUADS) bitonic_sort_pk<P>(*reinterpret_cast<u32(*)[P]>(&v2[0]));
    }
    const int plo = vecify((n - 1) >> 1), phi = vecify(n >> 1);
    const int fv = vecify(f), nfv = vecify(n - f);
    u32* outw = reinterpret_cast<u32*>(out);
#pragma unroll
    for (int half = 0; half < (QUADS ? 2 : 1); ++half) {
      const u32* arr = half ? v2 : v;
      float m0, m1;
      if (MODE == MEDIAN) {
        const u32 lo = sel_lo[half];
        const u32 hi = (n & 1) ? lo : sel_hi[half];
        m0 = 0.5f * (key_to_float(lo & 0xFFFFu) + key_to_float(hi & 0xFFFFu));
        m1 = 0.5f * (key_to_float(lo >> 16) + key_to_float(hi >> 16));
      } else if (MODE == TRIMMED) {
        // unpack the kept range of the sorted keys and sum
        float s0 = 0.0f, s1 = 0.0f;
#pragma unroll
        for (int i = 0; i < P; ++i)
          if (i >= fv && i < nfv) {
            s0 += key_to_float(arr[i] & 0xFFFFu);
            s1 += key_to_float(arr[i] >> 16);
          }
        const float inv = 1.0f / (float)(n - 2 * f);
        m0 = s0 * inv;
        m1 = s1 * inv;
      } else {  // MEAMED: per half, shrink the sorted window [l, r) from
        // whichever end deviates more from the median. The walk only ever
        // touches positions [0, f] and [n-f-1, n), so those 2(f+1) sorted
        // keys are staged into a per-thread LDS strip and each step reads
        // its two candidate ends by dynamic ds_read in O(1) — the previous
        // O(P) predicated extract per step cost ~f*P*4 VALU ops per column
        // pair and left the kernel 4x slower than MEDIAN.
        extern __shared__ __attribute__((aligned(16))) u32 pk_lds[];
        const int stride_l = 2 * (f + 1) + 1;  // odd => spread banks
        u32* strip = pk_lds + (int)threadIdx.x * stride_l;
        const int roff = vecify(2 * f + 2 - n);  // maps [n-f-1, n) -> [f+1, 2f+2)
#pragma unroll
        for (int i = 0; i < P; ++i) {
          if (i <= fv) strip[i] = arr[i];
          if (i >= nv - fv - 1 && i < nv) strip[i + roff] = arr[i];
        }
        u32 mlo = 0, mhi = 0;
#pragma unroll
        for (int i = 0; i < P; ++i) {
          if (i == plo) mlo = arr[i];
          if (i == phi) mhi = arr[i];
        }
        float t0 = 0.0f, t1 = 0.0f;
#pragma unroll
        for (int i = 0; i < P; ++i)
          if (i < nv) {
            t0 += key_to_float(arr[i] & 0xFFFFu);
            t1 += key_to_float(arr[i] >> 16);
          }
#pragma unroll
        for (int half2 = 0; half2 < 2; ++half2) {
          const int sh = half2 * 16;
          const float med =
              0.5f * (key_to_float((mlo >> sh) & 0xFFFFu) +
                      key_to_float((mhi >> sh) & 0xFFFFu));
          // running total: full first-n sum minus the dropped end each step
          float total = half2 ? t1 : t0;
          int l = vecify(0), r = nv;
          for (int kdrop = 0; kdrop < f; ++kdrop) {
            const float vl = key_to_float((strip[l] >> sh) & 0xFFFFu);
            const float vr =
                key_to_float((strip[r - 1 + roff] >> sh) & 0xFFFFu);
            const bool drop_left = (med - vl) > (vr - med);
            total -= drop_left ? vl : vr;
            l += drop_left ? 1 : 0;
            r -= drop_left ? 0 : 1;
          }
          const float m = total / (float)(n - f);
          if (half2 == 0) m0 = m; else m1 = m;
        }
      }
      union { unsigned short s[2]; u32 w; } o;
      union { unsigned short s; __hip_bfloat16 h; } c0, c1;
      c0.h = __float2bfloat16(m0);
      c1.h = __float2bfloat16(m1);
      o.s[0] = c0.s;
      o.s[1] = c1.s;
      outw[pair + half] = o.w;
    }
  }
}

// (the specialized 2-pass 256-bin bf16 median radix kernels that lived
// here were retired: the generic 64-bin engine in rsel.hip runs the same
// selection in 3 passes at 32 waves/CU and measures 2x faster)

// ---------------------------------------------------------------------------
// LDS variant, 64 < n <= 512: block-COOPERATIVE batched bitonic. One block
// (16 waves) owns 64 columns staged as an LDS plane [P][64]; every
// compare-exchange step spreads its P/2 x 64 sites over all 1024 threads
// (adjacent threads take adjacent columns of one site -> adjacent banks,
// conflict-free), with a block barrier between j-substeps. Replaces the
// original one-wave-per-block version whose single wave left the CU 97%
// idle (n=512 d=1M: 91.7 -> ~1 ms).
// ---------------------------------------------------------------------------

constexpr int LDS_COLS = 64;
// 512 threads (8 waves): halves barrier cost vs 16 waves and lets 2-4
// blocks co-reside per CU at small P (LDS is the binding resource at
// P=512: 128 KB -> 1 block)
constexpr int LDS_THREADS = 512;

template <int MODE, typename T>
__global__ void __launch_bounds__(LDS_THREADS)
colsel_lds_kernel(const T* __restrict__ X, T* __restrict__ out,
                  int n, long d, int f, int P) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* buf = reinterpret_cast<float*>(smem);  // [P][LDS_COLS]
  const int t = threadIdx.x;
  const long col0 = (long)blockIdx.x * LDS_COLS;
  const int cols = (int)min((long)LDS_COLS, d - col0);

  // cooperative stage: element (row, c) by thread index
  for (int idx = t; idx < P * LDS_COLS; idx += LDS_THREADS) {
    const int row = idx / LDS_COLS;
    const int c = idx % LDS_COLS;
    float v = PAD;
    if (row < n && c < cols)
      v = to_f<T>(X[(long)row * d + col0 + c]);
    buf[idx] = v;
  }
  __syncthreads();

  // batched bitonic: P/2 compare sites x LDS_COLS columns per substep.
  // j is always a power of two, so the site decomposition is pure shifts
  // (an integer division per site per substage dominated this kernel).
  for (int k = 2; k <= P; k <<= 1) {
    for (int jl = 31 - __clz(k >> 1); jl >= 0; --jl) {
      const int j = 1 << jl;
      const int k_ = k;
      for (int idx = t; idx < (P >> 1) * LDS_COLS; idx += LDS_THREADS) {
        const int site = idx >> 6;          // LDS_COLS == 64
        const int c = idx & 63;
        // site s enumerates pairs (i, i^j) with i^j > i:
        // i = (s >> jl) << (jl+1) | (s & (j-1))
        const int i = ((site >> jl) << (jl + 1)) | (site & (j - 1));
        const int l = i ^ j;
        const bool asc = (i & k_) == 0;
        float* a = &buf[i * LDS_COLS + c];
        float* b = &buf[l * LDS_COLS + c];
        const float av = *a, bv = *b;
        const float lo = fminf(av, bv), hi = fmaxf(av, bv);
        *a = asc ? lo : hi;
        *b = asc ? hi : lo;
      }
      __syncthreads();
    }
  }

  // per-column epilogue: threads 0..cols-1
  if (t < cols) {
    float* mine = buf + t;
    const int stride = LDS_COLS;
    const float med =
        0.5f * (mine[((n - 1) >> 1) * stride] + mine[(n >> 1) * stride]);
    float result;
    if (MODE == MEDIAN) {
      result = med;
    } else if (MODE == TRIMMED) {
      float s = 0.0f;
      for (int i2 = f; i2 < n - f; ++i2) s += mine[i2 * stride];
      result = s / (float)(n - 2 * f);
    } else {  // MEAMED: contiguous window of the sorted column (two-pointer)
      int l = 0, r = n;
      for (int kk = 0; kk < f; ++kk) {
        const float dl = med - mine[l * stride];
        const float dr = mine[(r - 1) * stride] - med;
        if (dl > dr) ++l; else --r;
      }
      float s = 0.0f;
      for (int i2 = l; i2 < r; ++i2) s += mine[i2 * stride];
      result = s / (float)(n - f);
    }
    out[col0 + t] = from_f<T>(result);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host-side launch (called from bind.cpp)
// ---------------------------------------------------------------------------

template <typename T>
static void launch_colsel_typed(const T* X, T* out, int n, long d, int mode,
                                int f, hipStream_t stream) {
  if (n <= 64) {
    const int block = 256;
    const long want = (d + block - 1) / block;
    const int grid = (int)(want < 8192 ? (want > 0 ? want : 1) : 8192);
#define DISPATCH_ONE(P, MODE)                                                  \
  hipLaunchKernelGGL((colsel_reg_kernel<P, MODE, T>), dim3(grid),              \
                     dim3(block), 0, stream, X, out, n, d, f)
#define DISPATCH_REG(P)                                                        \
  do {                                                                         \
    if (mode == MEDIAN) DISPATCH_ONE(P, MEDIAN);                               \
    else if (mode == TRIMMED) DISPATCH_ONE(P, TRIMMED);                        \
    else DISPATCH_ONE(P, MEAMED);                                              \
  } while (0)
    if (n <= 8) DISPATCH_REG(8);
    else if (n <= 16) DISPATCH_REG(16);
    else if (n <= 32) DISPATCH_REG(32);
    else DISPATCH_REG(64);
#undef DISPATCH_REG
#undef DISPATCH_ONE
  } else {
    int P = 128;
    while (P < n) P <<= 1;  // 128/256/512
    const long grid = (d + LDS_COLS - 1) / LDS_COLS;
    const size_t lds = (size_t)P * LDS_COLS * sizeof(float);
    if (mode == MEDIAN)
      hipLaunchKernelGGL((colsel_lds_kernel<MEDIAN, T>), dim3(grid),
                         dim3(LDS_THREADS), lds, stream, X, out, n, d, f, P);
    else if (mode == TRIMMED)
      hipLaunchKernelGGL((colsel_lds_kernel<TRIMMED, T>), dim3(grid),
                         dim3(LDS_THREADS), lds, stream, X, out, n, d, f, P);
    else
      hipLaunchKernelGGL((colsel_lds_kernel<MEAMED, T>), dim3(grid),
                         dim3(LDS_THREADS), lds, stream, X, out, n, d, f, P);
  }
}

void launch_colsel_f32(const float* X, float* out, int n, long d, int mode,
                       int f, hipStream_t stream) {
  launch_colsel_typed<float>(X, out, n, d, mode, f, stream);
}

void launch_colsel_bf16(const __hip_bfloat16* X, __hip_bfloat16* out, int n,
                        long d, int mode, int f, hipStream_t stream) {
  if (n <= 64 && (d % 2) == 0) {
    const int block = 256;
    // A/B'd on MI355X: QUADS (8 B/lane) TIED the old full-sort variant
    // (3.9 vs 3.8 ms at 64 x 125M) and LOSES 3x to the selection-network
    // pair variant (9.09 vs 3.09 ms — two 64-reg key arrays force the
    // 3-waves/SIMD bound while the pair kernel runs at 5). Single-pair
    // is the shipping path; BYZPY_PK_QUADS=1 re-runs the comparison.
    const long units = d >> 1;
    const long want = (units + block - 1) / block;
    const int grid = (int)(want < 8192 ? (want > 0 ? want : 1) : 8192);
    const unsigned short* Xu = reinterpret_cast<const unsigned short*>(X);
    unsigned short* Ou = reinterpret_cast<unsigned short*>(out);
    // A/B aid: BYZPY_PK_QUADS=1 re-measures the 4-column (8 B/lane)
    // MEDIAN variant (tied pre-selection-network; see the comment above)
    if (mode == MEDIAN && n > 32 && (d % 4) == 0 &&
        std::getenv("BYZPY_PK_QUADS") != nullptr) {
      const long qunits = d >> 2;
      const long qwant = (qunits + block - 1) / block;
      const int qgrid = (int)(qwant < 8192 ? (qwant > 0 ? qwant : 1) : 8192);
      hipLaunchKernelGGL((colsel_pk_median_bf16<64, true, MEDIAN>),
                         dim3(qgrid), dim3(block), 0, stream, Xu, Ou, n, d, f);
      return;
    }
#define PK_LAUNCH(P)                                                          \
  do {                                                                        \
    if (mode == MEDIAN)                                                       \
      hipLaunchKernelGGL((colsel_pk_median_bf16<P, false, MEDIAN>),           \
                         dim3(grid), dim3(block), 0, stream, Xu, Ou, n, d, f);\
    else if (mode == TRIMMED)                                                 \
      hipLaunchKernelGGL((colsel_pk_median_bf16<P, false, TRIMMED>),          \
                         dim3(grid), dim3(block), 0, stream, Xu, Ou, n, d, f);\
    else                                                                      \
      hipLaunchKernelGGL((colsel_pk_median_bf16<P, false, MEAMED>),           \
                         dim3(grid), dim3(block),                             \
                         (size_t)block * (2 * (f + 1) + 1) * sizeof(u32),     \
                         stream, Xu, Ou, n, d, f);                            \
  } while (0)
    if (n <= 8) PK_LAUNCH(8);
    else if (n <= 16) PK_LAUNCH(16);
    else if (n <= 32) PK_LAUNCH(32);
    else PK_LAUNCH(64);
#undef PK_LAUNCH
    return;
  }
  launch_colsel_typed<__hip_bfloat16>(X, out, n, d, mode, f, stream);
}
