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

template <bool DIAG, int BK = 128>
__global__ void __launch_bounds__(1024, 2)
gram_bf16_lds_kernel(const __hip_bfloat16* __restrict__ X,
                     float* __restrict__ G, int n, long d, long k_per_block) {
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
  const int col_base = blockIdx.z * TILE;

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
    if (out_row < n && out_col < n)
      atomicAdd(&G[(long)out_row * n + out_col], acc[r]);
  }
}

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

template <bool DIAG, int BKF = 64>
__global__ void __launch_bounds__(1024, 2)
gram_f32_lds_kernel(const float* __restrict__ X, float* __restrict__ G,
                    int n, long d, long k_per_block) {
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
  const int col_base = blockIdx.z * TILE;

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
    if (out_row < n && out_col < n)
      atomicAdd(&G[(long)out_row * n + out_col], acc0[r] + acc1[r]);
  }
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
  int splitk; long kpb;
  split_geometry(n, d, splitk, kpb);
  const int tiles = (n + TILE - 1) / TILE;
  dim3 grid(splitk, tiles, tiles);
  if ((d % 8) == 0 && (kpb % 256) == 0) {
    if (tiles == 1)
      // single tile (n <= 64): BK=256 halves the barrier count (LDS
      // 2 x 32 KB still fits 2 blocks/CU with the single A image)
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 256>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    else
      // off-diagonal tiles need separate A/B images; diagonal blocks of a
      // multi-tile grid still produce correct results with DIAG=false.
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
  } else if ((d % 8) == 0 && (kpb % 128) == 0) {
    if (tiles == 1)
      hipLaunchKernelGGL((gram_bf16_lds_kernel<true, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    else
      hipLaunchKernelGGL((gram_bf16_lds_kernel<false, 128>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
  } else {
    hipLaunchKernelGGL((gram_bf16_kernel<false>), grid, dim3(WAVES * 64), 0,
                       stream, X, G, n, d, kpb);
  }
}

void launch_gram_f32(const float* X, float* G, int n, long d,
                     hipStream_t stream) {
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
    if (tiles == 1)
      hipLaunchKernelGGL((gram_f32_lds_kernel<true, 64>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    else
      hipLaunchKernelGGL((gram_f32_lds_kernel<false, 64>), grid,
                         dim3(WAVES * 64), 0, stream, X, G, n, d, kpb);
    return;
  }
  hipLaunchKernelGGL(gram_f32_kernel, grid, dim3(WAVES * 64), 0, stream, X, G,
                     n, d, kpb);
}
