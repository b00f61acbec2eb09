// MFMA split-K Gram kernel G = X @ X^T for gfx950.
//
// Shape regime: n small (8..512), d huge (up to 8e9) — the GEMM is skinny,
// so the kernel is HBM-bound (read X once ~= n*d*2B at ~6.3 TB/s) and the
// design goal is exactly-once HBM traffic + enough blocks to fill 256 CUs:
//   - one workgroup = one 64x64 output tile x one K-slab (split-K across
//     the grid; f32 atomicAdd combine into the tiny n x n output),
//   - 16 waves per block (4x4 of 16x16 MFMA tiles). SHIPPING PATH: the
//     K-chunk is STAGED THROUGH LDS (gram_*_lds_kernel: 16 KB double
//     buffers, XOR-swizzled slots, cooperative T14 load/write split,
//     single image for the diagonal tile) so 16 waves re-reading the same
//     tile rows cost LDS bandwidth, not 8x-amplified L2/HBM traffic. The
//     direct kernels below the LDS ones are the alignment fallback,
//   - bf16 path: v_mfma_f32_16x16x32_bf16 (both fragments are 8 contiguous
//     k-elements of a row of X = one 16 B load);
//     f32 path: v_mfma_f32_16x16x4_f32 (exact f32 at the vector rate,
//     guide §3 — no xf32 on gfx950).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int TILE = 64;    // output tile side
constexpr int WAVES = 16;   // 4x4 waves of 16x16
constexpr int SYNC_EVERY = 8;

// Unconditional fragment load from a pre-clamped row pointer. Out-of-range
// rows are CLAMPED to a valid row by the caller and zeroed with a select
// AFTER the load — a branch around the load would serialize the whole
// K-stream behind per-load vmcnt(0) waits (guide §5 ".s-level traps" (c)).
template <bool VEC>
DEV bf16x8 load_frag8(const __hip_bfloat16* p, bool ok) {
  bf16x8 out;
  if (VEC) {
    out = *reinterpret_cast<const bf16x8*>(p);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      out[j] = *reinterpret_cast<const __bf16*>(p + j);
  }
  if (!ok) {
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (__bf16)0.0f;
  }
  return out;
}

template <bool VEC>
__global__ void gram_bf16_kernel(const __hip_bfloat16* __restrict__ X,
                                 float* __restrict__ G, int n, long d,
                                 long k_per_block) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wave >> 2, wc = wave & 3;
  const int row_base = blockIdx.y * TILE + wr * 16;
  const int col_base = blockIdx.z * TILE + wc * 16;

  const long k_lo = (long)blockIdx.x * k_per_block;
  const long k_hi = min(d, k_lo + k_per_block);

  f32x4 acc = {0.0f, 0.0f, 0.0f, 0.0f};
  const int frag_row = lane & 15;
  const long frag_k = (long)(lane >> 4) * 8;
  const int a_row = row_base + frag_row;
  const int b_row = col_base + frag_row;
  const bool a_ok = a_row < n, b_ok = b_row < n;
  // loop-invariant row pointers (clamped): no 64-bit mul in the K-loop
  const __hip_bfloat16* a_base = X + (long)min(a_row, n - 1) * d + frag_k;
  const __hip_bfloat16* b_base = X + (long)min(b_row, n - 1) * d + frag_k;

  // K-unroll x4: issue all 8 fragment loads before the MFMA cluster so the
  // HBM latency of one iteration hides under the previous one's MFMAs.
  // Fully in-range tiles (always, when n % 16 == 0) take the select-free
  // loop — the zero-select costs 32 VALU/iteration otherwise.
  const bool tile_ok = ((int)(blockIdx.y + 1) * TILE <= n) &&
                       ((int)(blockIdx.z + 1) * TILE <= n);
  int step = 0;
  long k0 = k_lo;
  if (tile_ok) {
    for (; k0 + 128 <= k_hi; k0 += 128) {
      bf16x8 a[4], b[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        a[u] = load_frag8<VEC>(a_base + k0 + u * 32, true);
        b[u] = load_frag8<VEC>(b_base + k0 + u * 32, true);
      }
#pragma unroll
      for (int u = 0; u < 4; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc, 0, 0, 0);
      if (++step == SYNC_EVERY) { step = 0; __syncthreads(); }
    }
  }
  for (; k0 + 128 <= k_hi; k0 += 128) {
    bf16x8 a[4], b[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      a[u] = load_frag8<VEC>(a_base + k0 + u * 32, a_ok);
      b[u] = load_frag8<VEC>(b_base + k0 + u * 32, b_ok);
    }
#pragma unroll
    for (int u = 0; u < 4; ++u)
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], b[u], acc, 0, 0, 0);
    if (++step == SYNC_EVERY) { step = 0; __syncthreads(); }
  }
  for (; k0 + 32 <= k_hi; k0 += 32) {
    const bf16x8 a = load_frag8<VEC>(a_base + k0, a_ok);
    const bf16x8 b = load_frag8<VEC>(b_base + k0, b_ok);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  // K tail (< 32): one MFMA on zero-padded fragments.
  if (k0 < k_hi) {
    bf16x8 a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const long k = k0 + frag_k + j;
      const bool in = k < k_hi;
      a[j] = (in && a_ok) ? *reinterpret_cast<const __bf16*>(a_base + k0 + j)
                          : (__bf16)0.0f;
      b[j] = (in && b_ok) ? *reinterpret_cast<const __bf16*>(b_base + k0 + j)
                          : (__bf16)0.0f;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

  // C/D map for 16x16: col = lane&15, row = (lane>>4)*4 + reg
  const int out_col = col_base + (lane & 15);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int out_row = row_base + (lane >> 4) * 4 + r;
    if (out_row < n && out_col < n)
      atomicAdd(&G[(long)out_row * n + out_col], acc[r]);
  }
}

__global__ void gram_f32_kernel(const float* __restrict__ X,
                                float* __restrict__ G, int n, long d,
                                long k_per_block) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = wave >> 2, wc = wave & 3;
  const int row_base = blockIdx.y * TILE + wr * 16;
  const int col_base = blockIdx.z * TILE + wc * 16;

  const long k_lo = (long)blockIdx.x * k_per_block;
  const long k_hi = min(d, k_lo + k_per_block);

  f32x4 acc = {0.0f, 0.0f, 0.0f, 0.0f};
  const int a_row = row_base + (lane & 15);
  const int b_row = col_base + (lane & 15);
  const bool a_ok = a_row < n, b_ok = b_row < n;
  const long lane_k = lane >> 4;  // k = k0 + lane_k, K-step 4
  const float* a_base = X + (long)min(a_row, n - 1) * d + lane_k;
  const float* b_base = X + (long)min(b_row, n - 1) * d + lane_k;

  int step = 0;
  long k0 = k_lo;
  for (; k0 + 32 <= k_hi; k0 += 32) {
    float av[8], bv[8];
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      av[u] = a_base[k0 + u * 4];
      bv[u] = b_base[k0 + u * 4];
    }
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const float a = a_ok ? av[u] : 0.0f;
      const float b = b_ok ? bv[u] : 0.0f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
    if (++step == SYNC_EVERY) { step = 0; __syncthreads(); }
  }
  for (; k0 < k_hi; k0 += 4) {
    const long k = k0 + lane_k;
    const bool in = k < k_hi;
    const float a = (in && a_ok) ? a_base[k0] : 0.0f;
    const float b = (in && b_ok) ? b_base[k0] : 0.0f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }

  const int out_col = col_base + (lane & 15);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int out_row = row_base + (lane >> 4) * 4 + r;
    if (out_row < n && out_col < n)
      atomicAdd(&G[(long)out_row * n + out_col], acc[r]);
  }
}

// ---------------------------------------------------------------------------
// LDS-staged bf16 Gram (v3, the fast path for d % 8 == 0).
//
// Per chunk (64 rows x 128 k = 16 KB): 1024 threads register-load one 16 B
// piece each (T14 split: loads issue BEFORE the compute phase, the ds_write
// lands after the barrier), waves then read MFMA fragments from LDS —
// HBM/L2 sees each line ONCE per block instead of 8x. LDS slots are
// XOR-swizzled (slot ^= row & 15) so a 16-lane ds_read_b128 group touches
// 16 distinct 16 B slots (guide §5.5 T2; linear layout would be 16-way).
// DIAG=true (blockIdx.y == blockIdx.z, always at n <= 64): the A and B
// images coincide — staged once, halving staging traffic.
// ---------------------------------------------------------------------------

template <bool DIAG, int BK = 128, bool MIRROR = false>
__global__ void __launch_bounds__(1024, 2)
gram_bf16_lds_kernel(const __hip_bfloat16* __restrict__ X,
                     float* __restrict__ G, int n, long d, long k_per_block) {
  // MIRROR (multi-tile grids): G is symmetric, so lower-triangle tile
  // blocks exit immediately and upper blocks write both (r,c) and (c,r)
  // — halves the staged HBM traffic (the full grid re-read every row
  // band once per PAIRED tile: n=128 measured 2.45 TB/s vs the ~6.3
  // single-tile ceiling purely from that amplification).
  if (MIRROR && blockIdx.z <= blockIdx.y) return;  // diag tiles run DIAG=true
  constexpr int SLOTS = BK / 8;                 // 16-B slots per row
  constexpr int SUB = SLOTS / 16;               // stage passes per thread
  constexpr int CHUNK_BYTES = TILE * BK * 2;    // 16 KB at BK=128
  __shared__ char smem[(DIAG ? 2 : 4) * CHUNK_BYTES];
  // no pointer ARRAYS into LDS (clang rejects the addrspace-cast array
  // initializer); select buffers with ternaries instead
  char* const bufA0 = smem;
  char* const bufA1 = smem + CHUNK_BYTES;
  char* const bufB0 = DIAG ? smem : smem + 2 * CHUNK_BYTES;
  char* const bufB1 = DIAG ? smem + CHUNK_BYTES : smem + 3 * CHUNK_BYTES;

  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  const int wr = wave >> 2, wc = wave & 3;
  const int row_base = blockIdx.y * TILE;
  // DIAG grids launch with gridDim.z == 1 and enumerate diagonal tiles
  // along y (single shared A==B image per tile)
  const int col_base = DIAG ? row_base : blockIdx.z * TILE;

  const long k_lo = (long)blockIdx.x * k_per_block;
  const long k_hi = min(d, k_lo + k_per_block);
  const long nchunks = (k_hi - k_lo + BK - 1) / BK;

  // stage geometry: thread t handles LDS row st_row, memory slot st_slot;
  // the CONTENT of that slot is global slot (st_slot ^ (st_row & 15)).
  const int st_row = t >> 4;
  const int a_rows = min(TILE, n - row_base);
  const int b_rows = min(TILE, n - col_base);
  // SUB slots per thread: pass u stages slot (t&15) + u*16
  int st_slot[SUB], src_slot[SUB];
#pragma unroll
  for (int u = 0; u < SUB; ++u) {
    st_slot[u] = (t & 15) + u * 16;
    src_slot[u] = st_slot[u] ^ (st_row & 15);
  }
  const __hip_bfloat16* a_row_src =
      X + (long)(row_base + min(st_row, a_rows - 1)) * d;
  const __hip_bfloat16* b_row_src =
      X + (long)(col_base + min(st_row, b_rows - 1)) * d;

  bf16x8 ra[SUB], rb[SUB];
  auto stage_load = [&](long c) {
    const long k0 = k_lo + c * BK;
    const bool full = (k0 + BK <= k_hi);
#pragma unroll
    for (int u = 0; u < SUB; ++u) {
      const long off = k0 + (long)src_slot[u] * 8;
      if (full && a_rows == TILE) {
        ra[u] = *reinterpret_cast<const bf16x8*>(a_row_src + off);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          ra[u][j] = (st_row < a_rows && off + j < k_hi)
                         ? *reinterpret_cast<const __bf16*>(a_row_src + off + j)
                         : (__bf16)0.0f;
      }
      if (!DIAG) {
        if (full && b_rows == TILE) {
          rb[u] = *reinterpret_cast<const bf16x8*>(b_row_src + off);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            rb[u][j] = (st_row < b_rows && off + j < k_hi)
                           ? *reinterpret_cast<const __bf16*>(b_row_src + off + j)
                           : (__bf16)0.0f;
        }
      }
    }
  };
  auto stage_write = [&](int which) {
    char* a = which ? bufA1 : bufA0;
    char* b = which ? bufB1 : bufB0;
#pragma unroll
    for (int u = 0; u < SUB; ++u) {
      *reinterpret_cast<bf16x8*>(a + st_row * (SLOTS * 16) + st_slot[u] * 16) =
          ra[u];
      if (!DIAG)
        *reinterpret_cast<bf16x8*>(b + st_row * (SLOTS * 16) +
                                   st_slot[u] * 16) = rb[u];
    }
  };

  f32x4 acc = {0.0f, 0.0f, 0.0f, 0.0f};
  const int rowA = wr * 16 + (lane & 15);
  const int rowB = wc * 16 + (lane & 15);
  const int grp = lane >> 4;  // 16 B slot group within the K step

  stage_load(0);
  stage_write(0);
  __syncthreads();
  for (long c = 0; c < nchunks; ++c) {
    if (c + 1 < nchunks) stage_load(c + 1);
    const char* A = (c & 1) ? bufA1 : bufA0;
    const char* B = (c & 1) ? bufB1 : bufB0;
#pragma unroll
    for (int step = 0; step < BK / 32; ++step) {
      const int q = step * 4 + grp;
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          A + rowA * (SLOTS * 16) + ((q ^ (rowA & 15)) * 16));
      const bf16x8 b = *reinterpret_cast<const bf16x8*>(
          B + rowB * (SLOTS * 16) + ((q ^ (rowB & 15)) * 16));
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    __syncthreads();
    if (c + 1 < nchunks) {
      stage_write((c + 1) & 1);
      __syncthreads();
    }
  }

  const int out_col = col_base + wc * 16 + (lane & 15);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int out_row = row_base + wr * 16 + (lane >> 4) * 4 + r;
    if (out_row < n && out_col < n) {
      atomicAdd(&G[(long)out_row * n + out_col], acc[r]);
      if (MIRROR)  // strictly-upper tiles only
        atomicAdd(&G[(long)out_col * n + out_row], acc[r]);
    }
  }
}

// (A FUSED gram+median kernel lived here and was removed after SIX
// measured variants all lost to the separate kernels — 32-60 ms vs
// 6.9 ms at 64x125M. The median's per-chunk work only amortizes when
// each lane holds a full column (64 VGPRs of keys), which either
// spills or halves occupancy inside the gram's register/LDS budget;
// register-light organizations serialize on LDS merge chains or waste
// 63/64 lanes on ballot descents. Full writeup:
// profiles/r02_fusion_negative.md.)

// ---------------------------------------------------------------------------
// LDS-staged f32 Gram (fast path for d % 4 == 0): same structure as the
// bf16 v3 kernel with v_mfma_f32_16x16x4_f32 fragments (exact f32 at the
// vector rate, guide §3). Chunk = 64 rows x 64 k f32 = 16 KB. The MFMA
// A/B operand is ONE float per lane (A[l&15][k0 + (l>>4)]), read from LDS
// with ds_read_b32 — a 16-lane group shares k and spans 16 rows, so a
// linear [row][k] image is 16-way bank-conflicted; a slot-granular XOR
// (16 B slot s -> s ^ (row & 7)) spreads the group over 8 slot positions
// (<= 2-way).
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) float f32x4v;

template <bool DIAG, int BKF = 64, bool MIRROR = false>
__global__ void __launch_bounds__(1024, 2)
gram_f32_lds_kernel(const float* __restrict__ X, float* __restrict__ G,
                    int n, long d, long k_per_block) {
  if (MIRROR && blockIdx.z <= blockIdx.y) return;  // diag tiles run DIAG=true
  constexpr int CHUNK_BYTES = TILE * BKF * 4;    // 16 KB
  __shared__ char smem[(DIAG ? 2 : 4) * CHUNK_BYTES];
  char* const bufA0 = smem;
  char* const bufA1 = smem + CHUNK_BYTES;
  char* const bufB0 = DIAG ? smem : smem + 2 * CHUNK_BYTES;
  char* const bufB1 = DIAG ? smem + CHUNK_BYTES : smem + 3 * CHUNK_BYTES;

  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  const int wr = wave >> 2, wc = wave & 3;
  const int row_base = blockIdx.y * TILE;
  const int col_base = DIAG ? row_base : blockIdx.z * TILE;

  const long k_lo = (long)blockIdx.x * k_per_block;
  const long k_hi = min(d, k_lo + k_per_block);
  const long nchunks = (k_hi - k_lo + BKF - 1) / BKF;

  constexpr int SLOTSF = BKF / 4;   // 16-B slots per row
  constexpr int SUBF = SLOTSF / 16; // stage passes per thread
  // stage: thread t owns LDS rows (t>>4) at slots (t&15) + u*16; content
  // of slot s is global slot (s ^ (row & 7)) — both-sides swizzle
  const int st_row = t >> 4;
  const int a_rows = min(TILE, n - row_base);
  const int b_rows = min(TILE, n - col_base);
  int st_slot[SUBF], src_slot[SUBF];
#pragma unroll
  for (int u = 0; u < SUBF; ++u) {
    st_slot[u] = (t & 15) + u * 16;
    src_slot[u] = st_slot[u] ^ (st_row & 7);
  }
  const float* a_row_src = X + (long)(row_base + min(st_row, a_rows - 1)) * d;
  const float* b_row_src = X + (long)(col_base + min(st_row, b_rows - 1)) * d;

  f32x4v ra[SUBF], rb[SUBF];
  auto stage_load = [&](long c) {
    const long k0 = k_lo + c * BKF;
    const bool full = (k0 + BKF <= k_hi);
#pragma unroll
    for (int u = 0; u < SUBF; ++u) {
      const long off = k0 + (long)src_slot[u] * 4;
      if (full && a_rows == TILE) {
        ra[u] = *reinterpret_cast<const f32x4v*>(a_row_src + off);
      } else {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          ra[u][j] =
              (st_row < a_rows && off + j < k_hi) ? a_row_src[off + j] : 0.0f;
      }
      if (!DIAG) {
        if (full && b_rows == TILE) {
          rb[u] = *reinterpret_cast<const f32x4v*>(b_row_src + off);
        } else {
#pragma unroll
          for (int j = 0; j < 4; ++j)
            rb[u][j] =
                (st_row < b_rows && off + j < k_hi) ? b_row_src[off + j] : 0.0f;
        }
      }
    }
  };
  auto stage_write = [&](int which) {
    char* a = which ? bufA1 : bufA0;
    char* b = which ? bufB1 : bufB0;
#pragma unroll
    for (int u = 0; u < SUBF; ++u) {
      *reinterpret_cast<f32x4v*>(a + st_row * (SLOTSF * 16) + st_slot[u] * 16) =
          ra[u];
      if (!DIAG)
        *reinterpret_cast<f32x4v*>(b + st_row * (SLOTSF * 16) +
                                   st_slot[u] * 16) = rb[u];
    }
  };

  // two accumulators break the dependent-MFMA chain (16x16x4 f32 has a
  // 40-cycle dependent-accumulator latency, guide §3) — summed at the end
  f32x4 acc0 = {0.0f, 0.0f, 0.0f, 0.0f};
  f32x4 acc1 = {0.0f, 0.0f, 0.0f, 0.0f};
  const int rowA = wr * 16 + (lane & 15);
  const int rowB = wc * 16 + (lane & 15);
  const int kgrp = lane >> 4;  // k offset within each 4-k step

  stage_load(0);
  stage_write(0);
  __syncthreads();
  for (long c = 0; c < nchunks; ++c) {
    if (c + 1 < nchunks) stage_load(c + 1);
    const char* A = (c & 1) ? bufA1 : bufA0;
    const char* B = (c & 1) ? bufB1 : bufB0;
#pragma unroll
    for (int step = 0; step < BKF / 4; ++step) {
      const int k = step * 4 + kgrp;  // element column within the chunk
      // element k lives in slot (k>>2), swizzled by row
      const int sa = ((k >> 2) ^ (rowA & 7)) * 16 + (k & 3) * 4;
      const int sb = ((k >> 2) ^ (rowB & 7)) * 16 + (k & 3) * 4;
      const float a =
          *reinterpret_cast<const float*>(A + rowA * (SLOTSF * 16) + sa);
      const float b =
          *reinterpret_cast<const float*>(B + rowB * (SLOTSF * 16) + sb);
      if (step & 1)
        acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc1, 0, 0, 0);
      else
        acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc0, 0, 0, 0);
    }
    __syncthreads();
    if (c + 1 < nchunks) {
      stage_write((c + 1) & 1);
      __syncthreads();
    }
  }

  const int out_col = col_base + wc * 16 + (lane & 15);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int out_row = row_base + wr * 16 + (lane >> 4) * 4 + r;
    if (out_row < n && out_col < n) {
      atomicAdd(&G[(long)out_row * n + out_col], acc0[r] + acc1[r]);
      if (MIRROR)  // strictly-upper tiles only
        atomicAdd(&G[(long)out_col * n + out_row], acc0[r] + acc1[r]);
    }
  }
}

// ---------------------------------------------------------------------------
// Small-n Gram (n <= 32): wave-slab kernels. The 64x64-tile kernels above
// amplify HBM traffic up to 8x at n < 64 (the tile is mostly padding:
// r01_shape_sweep measured 937 GB/s at n=8 vs 6,091 at n=64). Here each
// WAVE owns a disjoint K-slab covering ALL n rows, and — because Gram's A
// and B operands are the same matrix — each fragment register is fed to
// the MFMA as BOTH operands, so every X element is loaded exactly once
// grid-wide with no padding traffic at all:
//   - n <= 16 (bf16): v_mfma_f32_16x16x32_bf16 with SLAB PACKING: the 16
//     fragment rows carry P = 16/n copies of the matrix at P different
//     k-sub-slabs (fragment row p*n+i = X row i at sub-slab p). Output
//     quadrant (p,p) accumulates sub-slab p's Gram; cross-slab quadrants
//     are discarded. Load efficiency n*P/16 (100% at n in {8, 16}).
//   - 16 < n <= 32 (bf16): v_mfma_f32_32x32x16_bf16, one slab per wave.
//   - f32: same structure on v_mfma_f32_16x16x4_f32 / _32x32x2_f32 (exact
//     f32 at the vector rate, guide §3).
// Waves reduce their tiny C into an LDS accumulator; one atomicAdd per
// element per block into the (n, n) output.
// ---------------------------------------------------------------------------

constexpr int SMALL_WAVES = 8;  // waves per block (512 threads)

// bf16, n <= 16, packed. Each wave: K range [W*kpw, W*kpw+kpw) split into
// P sub-slabs of `sub` elements (kpw = P * sub, sub % 8 == 0).
__global__ void __launch_bounds__(SMALL_WAVES * 64, 4)
gram_bf16_small16_kernel(const __hip_bfloat16* __restrict__ X,
                         float* __restrict__ G, int n, long d, long kpw,
                         int P) {
  __shared__ float cbuf[16 * 16];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  for (int i = t; i < 16 * 16; i += SMALL_WAVES * 64) cbuf[i] = 0.0f;

  const long W = (long)blockIdx.x * SMALL_WAVES + wave;
  const long base = W * kpw;
  const long sub = kpw / P;

  const int frow = lane & 15;          // fragment row
  const int i = (frow < P * n) ? frow % n : 0;   // X row (clamped)
  const int p = (frow < P * n) ? frow / n : 0;   // sub-slab index
  const long frag_k = (long)(lane >> 4) * 8;     // k offset within step
  const long lane_base = base + (long)p * sub;   // this lane's sub-slab base
  const __hip_bfloat16* src = X + (long)i * d + lane_base + frag_k;

  f32x4 acc = {0.0f, 0.0f, 0.0f, 0.0f};
  if (base + kpw <= d) {
    // whole wave fully in range: unguarded 16 B loads, K-unroll x4
    long k = 0;
    for (; k + 128 <= sub; k += 128) {
      bf16x8 a[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        a[u] = *reinterpret_cast<const bf16x8*>(src + k + u * 32);
#pragma unroll
      for (int u = 0; u < 4; ++u)
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[u], a[u], acc, 0, 0, 0);
    }
    for (; k + 32 <= sub; k += 32) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(src + k);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, a, acc, 0, 0, 0);
    }
  } else if (base < d) {
    // global-tail wave: per-element guards (k limit differs per sub-slab)
    const long hi = d - lane_base;     // may be <= 0 for high-p lanes
    for (long k = 0; k < sub; k += 32) {
      bf16x8 a;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const long kk = k + frag_k + j;
        a[j] = (kk < hi && kk < sub) ? *reinterpret_cast<const __bf16*>(src + k + j)
                                     : (__bf16)0.0f;
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, a, acc, 0, 0, 0);
    }
  }

  __syncthreads();
  // C/D map for 16x16: col = lane&15, row = (lane>>4)*4 + r
  const int cc = lane & 15;
  const int pc = (cc < P * n) ? cc / n : -1;
  const int j = cc - pc * n;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int rr = (lane >> 4) * 4 + r;
    if (pc >= 0 && rr < P * n && rr / n == pc)
      atomicAdd(&cbuf[(rr - pc * n) * n + j], acc[r]);
  }
  __syncthreads();
  for (int idx = t; idx < n * n; idx += SMALL_WAVES * 64)
    atomicAdd(&G[(idx / n) * (long)n + (idx % n)], cbuf[idx]);
}

typedef __attribute__((ext_vector_type(16))) float f32x16;

// bf16, 16 < n <= 32: 32x32x16 MFMA, one slab per wave.
__global__ void __launch_bounds__(SMALL_WAVES * 64, 4)
gram_bf16_small32_kernel(const __hip_bfloat16* __restrict__ X,
                         float* __restrict__ G, int n, long d, long kpw) {
  __shared__ float cbuf[32 * 32];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  for (int i = t; i < 32 * 32; i += SMALL_WAVES * 64) cbuf[i] = 0.0f;

  const long W = (long)blockIdx.x * SMALL_WAVES + wave;
  const long base = W * kpw;

  const int frow = lane & 31;
  const int i = min(frow, n - 1);
  const long frag_k = (long)(lane >> 5) * 8;  // two 8-element halves of K=16
  const __hip_bfloat16* src = X + (long)i * d + base + frag_k;

  f32x16 acc = {};
  if (base + kpw <= d) {
    long k = 0;
    for (; k + 64 <= kpw; k += 64) {
      bf16x8 a[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        a[u] = *reinterpret_cast<const bf16x8*>(src + k + u * 16);
#pragma unroll
      for (int u = 0; u < 4; ++u)
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a[u], a[u], acc, 0, 0, 0);
    }
    for (; k + 16 <= kpw; k += 16) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(src + k);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, a, acc, 0, 0, 0);
    }
  } else if (base < d) {
    const long hi = d - base;
    for (long k = 0; k < kpw && k < hi; k += 16) {
      bf16x8 a;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const long kk = k + frag_k + j;
        a[j] = (kk < hi) ? *reinterpret_cast<const __bf16*>(src + k + j)
                         : (__bf16)0.0f;
      }
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, a, acc, 0, 0, 0);
    }
  }

  __syncthreads();
  // C/D map for 32x32: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int cc = lane & 31;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int rr = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    if (rr < n && cc < n) atomicAdd(&cbuf[rr * n + cc], acc[r]);
  }
  __syncthreads();
  for (int idx = t; idx < n * n; idx += SMALL_WAVES * 64)
    atomicAdd(&G[idx], cbuf[idx]);
}

// f32, n <= 16, packed: v_mfma_f32_16x16x4_f32 (1 float per lane per op).
__global__ void __launch_bounds__(SMALL_WAVES * 64, 4)
gram_f32_small16_kernel(const float* __restrict__ X, float* __restrict__ G,
                        int n, long d, long kpw, int P) {
  __shared__ float cbuf[16 * 16];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  for (int i = t; i < 16 * 16; i += SMALL_WAVES * 64) cbuf[i] = 0.0f;

  const long W = (long)blockIdx.x * SMALL_WAVES + wave;
  const long base = W * kpw;
  const long sub = kpw / P;

  const int frow = lane & 15;
  const int i = (frow < P * n) ? frow % n : 0;
  const int p = (frow < P * n) ? frow / n : 0;
  const long lane_k = lane >> 4;  // k = k0 + lane_k, K-step 4
  const long lane_base = base + (long)p * sub;
  const float* src = X + (long)i * d + lane_base + lane_k;

  // two accumulators break the 32-cycle dependent-accumulator chain
  f32x4 acc0 = {0.0f, 0.0f, 0.0f, 0.0f};
  f32x4 acc1 = {0.0f, 0.0f, 0.0f, 0.0f};
  if (base + kpw <= d) {
    long k = 0;
    for (; k + 32 <= sub; k += 32) {
      float a[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) a[u] = src[k + u * 4];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        if (u & 1)
          acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(a[u], a[u], acc1, 0, 0, 0);
        else
          acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a[u], a[u], acc0, 0, 0, 0);
      }
    }
    for (; k + 4 <= sub; k += 4) {
      const float a = src[k];
      acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, a, acc0, 0, 0, 0);
    }
  } else if (base < d) {
    const long hi = d - lane_base;
    for (long k = 0; k < sub; k += 4) {
      const long kk = k + lane_k;
      const float a = (kk < hi && kk < sub) ? src[k] : 0.0f;
      acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, a, acc0, 0, 0, 0);
    }
  }

  __syncthreads();
  const int cc = lane & 15;
  const int pc = (cc < P * n) ? cc / n : -1;
  const int j = cc - pc * n;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int rr = (lane >> 4) * 4 + r;
    if (pc >= 0 && rr < P * n && rr / n == pc)
      atomicAdd(&cbuf[(rr - pc * n) * n + j], acc0[r] + acc1[r]);
  }
  __syncthreads();
  for (int idx = t; idx < n * n; idx += SMALL_WAVES * 64)
    atomicAdd(&G[idx], cbuf[idx]);
}

// f32, 16 < n <= 32: v_mfma_f32_32x32x2_f32.
__global__ void __launch_bounds__(SMALL_WAVES * 64, 4)
gram_f32_small32_kernel(const float* __restrict__ X, float* __restrict__ G,
                        int n, long d, long kpw) {
  __shared__ float cbuf[32 * 32];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  for (int i = t; i < 32 * 32; i += SMALL_WAVES * 64) cbuf[i] = 0.0f;

  const long W = (long)blockIdx.x * SMALL_WAVES + wave;
  const long base = W * kpw;

  const int frow = lane & 31;
  const int i = min(frow, n - 1);
  const long lane_k = lane >> 5;  // k = k0 + lane_k, K-step 2
  const float* src = X + (long)i * d + base + lane_k;

  f32x16 acc0 = {};
  f32x16 acc1 = {};
  if (base + kpw <= d) {
    long k = 0;
    for (; k + 16 <= kpw; k += 16) {
      float a[8];
#pragma unroll
      for (int u = 0; u < 8; ++u) a[u] = src[k + u * 2];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        if (u & 1)
          acc1 = __builtin_amdgcn_mfma_f32_32x32x2f32(a[u], a[u], acc1, 0, 0, 0);
        else
          acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a[u], a[u], acc0, 0, 0, 0);
      }
    }
    for (; k + 2 <= kpw; k += 2) {
      const float a = src[k];
      acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, a, acc0, 0, 0, 0);
    }
  } else if (base < d) {
    const long hi = d - base;
    for (long k = 0; k < kpw && k < hi; k += 2) {
      const long kk = k + lane_k;
      const float a = (kk < hi) ? src[k] : 0.0f;
      acc0 = __builtin_amdgcn_mfma_f32_32x32x2f32(a, a, acc0, 0, 0, 0);
    }
  }

  __syncthreads();
  // C/D map for 32x32: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int cc = lane & 31;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int rr = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    if (rr < n && cc < n) atomicAdd(&cbuf[rr * n + cc], acc0[r] + acc1[r]);
  }
  __syncthreads();
  for (int idx = t; idx < n * n; idx += SMALL_WAVES * 64)
    atomicAdd(&G[idx], cbuf[idx]);
}

// f32, 32 < n <= 64: SYMMETRIC 3-tile wave-slab. Two 32-row fragments
// (a0 = rows 0..31, a1 = rows 32..63) feed THREE v_mfma_f32_32x32x2_f32
// per K-step — C00 = a0*a0^T, C01 = a0*a1^T, C11 = a1*a1^T — and C10 is
// C01 transposed by Gram symmetry, so the matrix pipe does 3/4 of the
// naive work and every element is loaded once. (The LDS 64-tile f32
// kernel measured 2.4 TB/s here: it is f32-MFMA-issue-bound at 4 full
// tiles; this variant's ceiling is ~6 TB/s.)
__global__ void __launch_bounds__(SMALL_WAVES * 64, 4)
gram_f32_small64_kernel(const float* __restrict__ X, float* __restrict__ G,
                        int n, long d, long kpw) {
  __shared__ float cbuf[64 * 64];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & 63;
  for (int i = t; i < 64 * 64; i += SMALL_WAVES * 64) cbuf[i] = 0.0f;

  const long W = (long)blockIdx.x * SMALL_WAVES + wave;
  const long base = W * kpw;

  const int frow = lane & 31;
  const int i0 = min(frow, n - 1);
  const int i1 = min(32 + frow, n - 1);
  const long lane_k = lane >> 5;  // MFMA operand: k = k0 + lane_k, K-step 2
  // float4 loads (a plain stride-2 dword walk measured 754 GB/s — 8 B
  // per DRAM row-touch): the two k-halves of a wave each load one quad
  // per 8-k window (32 B contiguous per row per instruction) and swap
  // quads with shfl_xor(32) so every MFMA step finds its element
  const float* src0 = X + (long)i0 * d + base + lane_k * 4;
  const float* src1 = X + (long)i1 * d + base + lane_k * 4;

  f32x16 acc00 = {};
  f32x16 acc01 = {};
  f32x16 acc11 = {};
  if (base + kpw <= d) {
    long k = 0;
    for (; k + 8 <= kpw; k += 8) {
      float4 q0 = *reinterpret_cast<const float4*>(src0 + k);
      float4 q1 = *reinterpret_cast<const float4*>(src1 + k);
      // partner half's quad (4 shfl_xor per row stream)
      const float s0x = __shfl_xor(q0.x, 32, 64), s0y = __shfl_xor(q0.y, 32, 64);
      const float s0z = __shfl_xor(q0.z, 32, 64), s0w = __shfl_xor(q0.w, 32, 64);
      const float s1x = __shfl_xor(q1.x, 32, 64), s1y = __shfl_xor(q1.y, 32, 64);
      const float s1z = __shfl_xor(q1.z, 32, 64), s1w = __shfl_xor(q1.w, 32, 64);
      const bool hi_half = lane_k != 0;
      // step s needs element e = 2s + lane_k of the 8-k window; quads
      // are owned low-half = elems 0..3, high-half = 4..7
      const float a0s[4] = {hi_half ? s0y : q0.x, hi_half ? s0w : q0.z,
                            hi_half ? q0.y : s0x, hi_half ? q0.w : s0z};
      const float a1s[4] = {hi_half ? s1y : q1.x, hi_half ? s1w : q1.z,
                            hi_half ? q1.y : s1x, hi_half ? q1.w : s1z};
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0s[s], a0s[s], acc00, 0, 0, 0);
        acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0s[s], a1s[s], acc01, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1s[s], a1s[s], acc11, 0, 0, 0);
      }
    }
  } else if (base < d) {
    const long hi = d - base;
    const float* t0 = X + (long)i0 * d + base + lane_k;
    const float* t1 = X + (long)i1 * d + base + lane_k;
    for (long k = 0; k < kpw && k < hi; k += 2) {
      const long kk = k + lane_k;
      const float a0 = (kk < hi) ? t0[k] : 0.0f;
      const float a1 = (kk < hi) ? t1[k] : 0.0f;
      acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, a0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, a1, acc01, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, a1, acc11, 0, 0, 0);
    }
  }

  __syncthreads();
  // C/D map for 32x32: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  const int cc = lane & 31;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int rr = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    // C00 -> (rr, cc); C01 -> (rr, 32+cc) and its mirror; C11 -> (+32, +32)
    if (rr < n && cc < n) atomicAdd(&cbuf[rr * 64 + cc], acc00[r]);
    if (rr < n && 32 + cc < n) {
      atomicAdd(&cbuf[rr * 64 + 32 + cc], acc01[r]);
      atomicAdd(&cbuf[(32 + cc) * 64 + rr], acc01[r]);
    }
    if (32 + rr < n && 32 + cc < n)
      atomicAdd(&cbuf[(32 + rr) * 64 + 32 + cc], acc11[r]);
  }
  __syncthreads();
  for (int idx = t; idx < n * n; idx += SMALL_WAVES * 64) {
    const int r2 = idx / n, c2 = idx % n;
    atomicAdd(&G[idx], cbuf[r2 * 64 + c2]);
  }
}

// kpw chosen so each wave has >= ~2048 elements and the grid still fills
// the chip (8 XCDs x 32 CUs want >= ~512 blocks when d allows).
inline void small_geometry(long d, int align, long& kpw, long& blocks) {
  long nw = (d + 2047) / 2048;
  if (nw < 1) nw = 1;
  if (nw > 16384) nw = 16384;
  blocks = (nw + SMALL_WAVES - 1) / SMALL_WAVES;
  nw = blocks * SMALL_WAVES;
  kpw = (d + nw - 1) / nw;
  kpw = ((kpw + align - 1) / align) * align;
}

inline void split_geometry(int n, long d, int& splitk, long& k_per_block) {
  const int tiles = (n + TILE - 1) / TILE;
  const long tile_blocks = (long)tiles * tiles;
  long want = 1024 / tile_blocks;
  if (want < 1) want = 1;
  const long max_by_d = (d + 4095) / 4096;  // keep slabs >= 4096 elements
  if (want > max_by_d) want = max_by_d;
  splitk = (int)want;
  k_per_block = (d + splitk - 1) / splitk;
  k_per_block = ((k_per_block + 255) / 256) * 256;  // keep the vector path
}

}  // namespace

void launch_gram_bf16(const __hip_bfloat16* X, float* G, int n, long d,
                      hipStream_t stream) {
  if (n <= 32 && (d % 8) == 0 && d >= 64) {
    long kpw, blocks;
    if (n <= 16) {
      const int P = 16 / n;
      // kpw multiple of P*32 so each sub-slab is a whole number of K=32
      // MFMA steps (the in-range fast loop has no sub-step tail)
      small_geometry(d, P * 32, kpw, blocks);
      hipLaunchKernelGGL(gram_bf16_small16_kernel, dim3((unsigned)blocks),
                         dim3(SMALL_WAVES * 64), 0, stream, X, G, n, d, kpw, P);
    } else {
      small_geometry(d, 16, kpw, blocks);
      hipLaunchKernelGGL(gram_bf16_small32_kernel, dim3((unsigned)blocks),
                         dim3(SMALL_WAVES * 64), 0, stream, X, G, n, d, kpw);
    }
    return;
  }
  int splitk; long kpb;
  split_geometry(n, d, splitk, kpb);
  const int tiles = (n + TILE - 1) / TILE;
  dim3 grid(splitk, tiles, tiles);
  // Symmetric (mirror) tiling reads 25-50% less HBM but costs a second
  // launch and idles the lower-triangle blocks — it pays only when the
  // work is traffic-bound. Measured crossover: MultiKrum n=80 d=65k
  // (5.2M elements) regressed 0.21 -> 0.33 ms under mirror while the
  // sweep's n=128 d=16M improved 855 -> 1,367 GB/s.
  const bool mirror_pays = (double)n * (double)d >= 8.0e6;
  if ((d % 8) == 0 && (kpb % 256) == 0) {
    if (tiles == 1)
      // single tile (n <= 64): BK=256 halves the barrier count (LDS
      // 2 x 32 KB still fits 2 blocks/CU with the single A image)
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 256>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    else if (mirror_pays) {
      // symmetric tiling: diagonal tiles on the single-image DIAG
      // staging (grid z=1, tiles along y), strictly-upper tiles mirror
      // into both halves, lower blocks exit immediately
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 128>),
                         dim3(splitk, tiles, 1), dim3(WAVES * 64), 0, stream,
                         X, G, n, d, kpb);
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128, true>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    } else
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
  } else if ((d % 8) == 0 && (kpb % 128) == 0) {
    if (tiles == 1)
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    else if (mirror_pays) {
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 128>),
                         dim3(splitk, tiles, 1), dim3(WAVES * 64), 0, stream,
                         X, G, n, d, kpb);
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128, true>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    } else
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
  } else {
    hipLaunchKernelGGL((gram_bf16_kernel<false>), grid, dim3(WAVES * 64), 0,
                       stream, X, G, n, d, kpb);
  }
}

void launch_gram_f32(const float* X, float* G, int n, long d,
                     hipStream_t stream) {
  // the n>32 symmetric kernel's float4 loads need d % 4 == 0
  if (n <= 64 && d >= 64 && (n <= 32 || (d % 4) == 0)) {
    long kpw, blocks;
    if (n <= 16) {
      const int P = 16 / n;
      small_geometry(d, P * 4, kpw, blocks);
      hipLaunchKernelGGL(gram_f32_small16_kernel, dim3((unsigned)blocks),
                         dim3(SMALL_WAVES * 64), 0, stream, X, G, n, d, kpw, P);
    } else if (n <= 32) {
      small_geometry(d, 16, kpw, blocks);
      hipLaunchKernelGGL(gram_f32_small32_kernel, dim3((unsigned)blocks),
                         dim3(SMALL_WAVES * 64), 0, stream, X, G, n, d, kpw);
    } else {
      small_geometry(d, 16, kpw, blocks);
      hipLaunchKernelGGL(gram_f32_small64_kernel, dim3((unsigned)blocks),
                         dim3(SMALL_WAVES * 64), 0, stream, X, G, n, d, kpw);
    }
    return;
  }
  int splitk; long kpb;
  split_geometry(n, d, splitk, kpb);
  const int tiles = (n + TILE - 1) / TILE;
  dim3 grid(splitk, tiles, tiles);
  if ((d % 4) == 0 && (kpb % 128) == 0 && tiles == 1) {
    // single tile: BKF=128 halves the barrier count (2 x 32 KB buffers)
    hipLaunchKernelGGL((gram_f32_lds_kernel<true, 128>), grid,
                       dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    return;
  }
  if ((d % 4) == 0 && (kpb % 64) == 0) {
    // same traffic-bound gate as the bf16 launcher (see comment there)
    const bool mirror_pays = (double)n * (double)d >= 8.0e6;
    if (tiles == 1) {
      hipLaunchKernelGGL((gram_f32_lds_kernel<true, 64>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    } else if (mirror_pays) {
      hipLaunchKernelGGL((gram_f32_lds_kernel<true, 64>),
                         dim3(splitk, tiles, 1), dim3(WAVES * 64), 0, stream,
                         X, G, n, d, kpb);
      hipLaunchKernelGGL((gram_f32_lds_kernel<false, 64, true>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    } else {
      hipLaunchKernelGGL((gram_f32_lds_kernel<false, 64>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    }
    return;
  }
  hipLaunchKernelGGL(gram_f32_kernel, grid, dim3(WAVES * 64), 0, stream, X, G,
                     n, d, kpb);
}
