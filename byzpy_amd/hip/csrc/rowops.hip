// Row/column primitives:
//   row_sqnorms        (n,d) -> (n,)   per-row ||x||^2, split over k-slabs
//   row_center_sqdists (n,d),(d,) -> (n,)  per-row ||x - z||^2
//   row_scale          (n,d),(n,) -> (n,d) row scaling (clip apply)
//   mean_rows          gather-mean of selected rows -> (d,)
//   group_mean_rows    per-group gather-mean -> (g,d)   (NNM mixing)
//   bucket_mean        permuted segmented mean -> (nb,d) (Bucketing)
//   weiszfeld_update   fused w=1/max(dist,eps); z' = sum(w x)/sum(w); ||dz||^2
//   cc_update          fused alpha=min(1,c/dist); v' = v + sum(alpha (x-v))/n
//
// All kernels are HBM-bound; loads are vectorized to 16 B/lane (f32x4 /
// bf16x8 — guide G13: hipcc does not auto-vectorize bf16) with a scalar
// fallback when d is not a multiple of the vector width. Row reductions:
// 2D grid (k-slab, row) + block reduce + one f32 atomic per block (G12).
// Column kernels: V consecutive coordinates per thread, row loop inside;
// per-row weights (1/dist etc.) are staged once in LDS per block.
#include "common.h"

namespace {

constexpr int MAX_N_LDS = 1024;  // per-row weight stage (4 KB)

// -- vector load/store ------------------------------------------------------

template <typename T>
struct VecTraits;

template <>
struct VecTraits<float> {
  static constexpr int V = 4;
  static DEV void load(const float* p, float (&o)[4]) {
    const float4 v = *reinterpret_cast<const float4*>(p);
    o[0] = v.x; o[1] = v.y; o[2] = v.z; o[3] = v.w;
  }
  static DEV void store(float* p, const float (&o)[4]) {
    *reinterpret_cast<float4*>(p) = make_float4(o[0], o[1], o[2], o[3]);
  }
};

template <>
struct VecTraits<__hip_bfloat16> {
  static constexpr int V = 8;
  static DEV float unpack(unsigned u) {
    union { unsigned short s; __hip_bfloat16 h; } c;
    c.s = (unsigned short)u;
    return __bfloat162float(c.h);
  }
  static DEV unsigned short pack(float x) {
    union { unsigned short s; __hip_bfloat16 h; } c;
    c.h = __float2bfloat16(x);
    return c.s;
  }
  static DEV void load(const __hip_bfloat16* p, float (&o)[8]) {
    const uint4 v = *reinterpret_cast<const uint4*>(p);
    o[0] = unpack(v.x); o[1] = unpack(v.x >> 16);
    o[2] = unpack(v.y); o[3] = unpack(v.y >> 16);
    o[4] = unpack(v.z); o[5] = unpack(v.z >> 16);
    o[6] = unpack(v.w); o[7] = unpack(v.w >> 16);
  }
  static DEV void store(__hip_bfloat16* p, const float (&o)[8]) {
    uint4 v;
    v.x = pack(o[0]) | ((unsigned)pack(o[1]) << 16);
    v.y = pack(o[2]) | ((unsigned)pack(o[3]) << 16);
    v.z = pack(o[4]) | ((unsigned)pack(o[5]) << 16);
    v.w = pack(o[6]) | ((unsigned)pack(o[7]) << 16);
    *reinterpret_cast<uint4*>(p) = v;
  }
};

// -- row reductions ---------------------------------------------------------

template <typename T, bool CENTER, bool VEC>
__global__ void row_red_kernel(const T* __restrict__ X,
                               const float* __restrict__ z,
                               float* __restrict__ out, long d) {
  __shared__ float lds[16];
  constexpr int V = VecTraits<T>::V;
  const int row = blockIdx.y;
  const T* xr = X + (long)row * d;
  float acc = 0.0f;
  if (VEC) {
    const long dv = d / V;
    const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    long jv = start;
    // 4 grid-stride positions per step: 4 loads in flight
    for (; jv + 3 * stride < dv; jv += 4 * stride) {
      float x0[V], x1[V], x2[V], x3[V];
      VecTraits<T>::load(xr + jv * V, x0);
      VecTraits<T>::load(xr + (jv + stride) * V, x1);
      VecTraits<T>::load(xr + (jv + 2 * stride) * V, x2);
      VecTraits<T>::load(xr + (jv + 3 * stride) * V, x3);
#pragma unroll
      for (int c = 0; c < V; ++c) {
        float v0 = x0[c], v1 = x1[c], v2 = x2[c], v3 = x3[c];
        if (CENTER) {
          v0 -= z[jv * V + c];
          v1 -= z[(jv + stride) * V + c];
          v2 -= z[(jv + 2 * stride) * V + c];
          v3 -= z[(jv + 3 * stride) * V + c];
        }
        acc += (v0 * v0 + v1 * v1) + (v2 * v2 + v3 * v3);
      }
    }
    for (; jv < dv; jv += stride) {
      float x[V];
      VecTraits<T>::load(xr + jv * V, x);
#pragma unroll
      for (int c = 0; c < V; ++c) {
        float v = x[c];
        if (CENTER) v -= z[jv * V + c];
        acc += v * v;
      }
    }
  } else {
    const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (long j = start; j < d; j += stride) {
      float v = to_f<T>(xr[j]);
      if (CENTER) v -= z[j];
      acc += v * v;
    }
  }
  acc = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(&out[row], acc);
}

// -- grouped center distances ----------------------------------------------
// ||x_i - z||^2 for a GROUP of rows per block (blockIdx.y = group of 8):
// z is read once per group instead of once per row — at small n the f32
// center dominates traffic (n=8 bf16: z re-reads were 2/3 of all bytes).

constexpr int DIST_GROUP = 32;

template <typename T, bool VEC>
__global__ void center_sqdists_group_kernel(const T* __restrict__ X,
                                            const float* __restrict__ z,
                                            float* __restrict__ out, int n,
                                            long d) {
  __shared__ float lds[DIST_GROUP][16];
  constexpr int V = VecTraits<T>::V;
  const int g0 = blockIdx.y * DIST_GROUP;
  const int rows = min(DIST_GROUP, n - g0);
  float acc[DIST_GROUP] = {0};
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float zv[V];
      {
        // z is f32; load V floats as V/4 float4s
        float tmp[4];
#pragma unroll
        for (int q4 = 0; q4 < V / 4; ++q4) {
          VecTraits<float>::load(z + jv * V + q4 * 4, tmp);
#pragma unroll
          for (int c = 0; c < 4; ++c) zv[q4 * 4 + c] = tmp[c];
        }
      }
      // full groups of 4 rows keep 4 loads in flight
      int i = 0;
      for (; i + 4 <= rows; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)(g0 + i + 0) * d + jv * V, x0);
        VecTraits<T>::load(X + (long)(g0 + i + 1) * d + jv * V, x1);
        VecTraits<T>::load(X + (long)(g0 + i + 2) * d + jv * V, x2);
        VecTraits<T>::load(X + (long)(g0 + i + 3) * d + jv * V, x3);
#pragma unroll
        for (int c = 0; c < V; ++c) {
          const float d0 = x0[c] - zv[c], d1 = x1[c] - zv[c];
          const float d2 = x2[c] - zv[c], d3 = x3[c] - zv[c];
          acc[i + 0] += d0 * d0;
          acc[i + 1] += d1 * d1;
          acc[i + 2] += d2 * d2;
          acc[i + 3] += d3 * d3;
        }
      }
      for (; i < rows; ++i) {
        float x[V];
        VecTraits<T>::load(X + (long)(g0 + i) * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) {
          const float dd = x[c] - zv[c];
          acc[i] += dd * dd;
        }
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      const float zj = z[j];
      for (int i = 0; i < rows; ++i) {
        const float dd = to_f<T>(X[(long)(g0 + i) * d + j]) - zj;
        acc[i] += dd * dd;
      }
    }
  }
#pragma unroll
  for (int i = 0; i < DIST_GROUP; ++i) {
    if (i < rows) {
      const float s = block_reduce_sum(acc[i], lds[i]);
      if (threadIdx.x == 0) atomicAdd(&out[g0 + i], s);
    }
    __syncthreads();
  }
}

// -- grouped center distances + Weiszfeld update ---------------------------
// G independent small aggregation problems per launch (gossip: every
// node's geomed iteration in ONE kernel pair instead of 2G launches on
// G streams). blockIdx.y = group; each group has its own center.

template <typename T, bool VEC>
__global__ void center_sqdists_grouped_kernel(const T* __restrict__ X,
                                              const float* __restrict__ Z,
                                              float* __restrict__ out, int m,
                                              long d) {
  __shared__ float lds[DIST_GROUP][16];
  constexpr int V = VecTraits<T>::V;
  const int g = blockIdx.y;
  const T* Xg = X + (long)g * m * d;
  const float* z = Z + (long)g * d;
  float acc[DIST_GROUP] = {0};
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float zv[V];
      {
        float tmp[4];
#pragma unroll
        for (int q4 = 0; q4 < V / 4; ++q4) {
          VecTraits<float>::load(z + jv * V + q4 * 4, tmp);
#pragma unroll
          for (int c = 0; c < 4; ++c) zv[q4 * 4 + c] = tmp[c];
        }
      }
      int i = 0;
      for (; i + 4 <= m; i += 4) {  // 4 row loads in flight
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(Xg + (long)(i + 0) * d + jv * V, x0);
        VecTraits<T>::load(Xg + (long)(i + 1) * d + jv * V, x1);
        VecTraits<T>::load(Xg + (long)(i + 2) * d + jv * V, x2);
        VecTraits<T>::load(Xg + (long)(i + 3) * d + jv * V, x3);
#pragma unroll
        for (int c = 0; c < V; ++c) {
          const float d0 = x0[c] - zv[c], d1 = x1[c] - zv[c];
          const float d2 = x2[c] - zv[c], d3 = x3[c] - zv[c];
          acc[i + 0] += d0 * d0;
          acc[i + 1] += d1 * d1;
          acc[i + 2] += d2 * d2;
          acc[i + 3] += d3 * d3;
        }
      }
      for (; i < m; ++i) {
        float x[V];
        VecTraits<T>::load(Xg + (long)i * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) {
          const float dd = x[c] - zv[c];
          acc[i] += dd * dd;
        }
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      const float zj = z[j];
      for (int i = 0; i < m; ++i) {
        const float dd = to_f<T>(Xg[(long)i * d + j]) - zj;
        acc[i] += dd * dd;
      }
    }
  }
#pragma unroll
  for (int i = 0; i < DIST_GROUP; ++i) {
    if (i < m) {
      const float sred = block_reduce_sum(acc[i], lds[i]);
      if (threadIdx.x == 0) atomicAdd(&out[(long)g * m + i], sred);
    }
    __syncthreads();
  }
}

template <typename T, bool VEC>
__global__ void weiszfeld_update_grouped_kernel(
    const T* __restrict__ X, const float* __restrict__ Z,
    const float* __restrict__ dist2, float* __restrict__ Z_new, int m, long d,
    float eps) {
  __shared__ float w_lds[DIST_GROUP];
  constexpr int V = VecTraits<T>::V;
  const int g = blockIdx.y;
  const T* Xg = X + (long)g * m * d;
  const float* z = Z + (long)g * d;
  float* z_new = Z_new + (long)g * d;
  for (int i = threadIdx.x; i < m; i += blockDim.x)
    w_lds[i] = 1.0f / fmaxf(sqrtf(dist2[(long)g * m + i]), eps);
  __syncthreads();
  float den = 0.0f;
  for (int i = 0; i < m; ++i) den += w_lds[i];
  const float inv_den = 1.0f / den;

  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float num[V] = {0};
      int i = 0;
      for (; i + 4 <= m; i += 4) {  // 4 row loads in flight
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(Xg + (long)(i + 0) * d + jv * V, x0);
        VecTraits<T>::load(Xg + (long)(i + 1) * d + jv * V, x1);
        VecTraits<T>::load(Xg + (long)(i + 2) * d + jv * V, x2);
        VecTraits<T>::load(Xg + (long)(i + 3) * d + jv * V, x3);
        const float w0 = w_lds[i], w1 = w_lds[i + 1], w2 = w_lds[i + 2],
                    w3 = w_lds[i + 3];
#pragma unroll
        for (int c = 0; c < V; ++c)
          num[c] += (w0 * x0[c] + w1 * x1[c]) + (w2 * x2[c] + w3 * x3[c]);
      }
      for (; i < m; ++i) {
        const float w = w_lds[i];
        float x[V];
        VecTraits<T>::load(Xg + (long)i * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) num[c] += w * x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) z_new[jv * V + c] = num[c] * inv_den;
    }
  } else {
    for (long j = start; j < d; j += stride) {
      float num = 0.0f;
      for (int i = 0; i < m; ++i)
        num += w_lds[i] * to_f<T>(Xg[(long)i * d + j]);
      z_new[j] = num * inv_den;
    }
  }
}

// -- row scaling ------------------------------------------------------------

template <typename T, bool VEC>
__global__ void row_scale_kernel(const T* __restrict__ X,
                                 const float* __restrict__ s,
                                 T* __restrict__ out, long d) {
  constexpr int V = VecTraits<T>::V;
  const int row = blockIdx.y;
  const float sc = s[row];
  const T* xr = X + (long)row * d;
  T* yr = out + (long)row * d;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float x[V];
      VecTraits<T>::load(xr + jv * V, x);
#pragma unroll
      for (int c = 0; c < V; ++c) x[c] *= sc;
      VecTraits<T>::store(yr + jv * V, x);
    }
  } else {
    for (long j = start; j < d; j += stride)
      yr[j] = from_f<T>(to_f<T>(xr[j]) * sc);
  }
}

// -- gather means -----------------------------------------------------------
// GROUPED=false: idx is (k,) and blockIdx.y==0 -> out (d,).
// GROUPED=true:  idx is (g,k) rows per group g=blockIdx.y -> out (g,d).

template <typename T, bool VEC, bool GROUPED>
__global__ void gather_mean_kernel(const T* __restrict__ X,
                                   const int* __restrict__ idx, int k,
                                   T* __restrict__ out, long d) {
  constexpr int V = VecTraits<T>::V;
  const int g = GROUPED ? blockIdx.y : 0;
  const int* gi = idx + (long)g * k;
  T* og = out + (long)g * d;
  const float inv = 1.0f / (float)k;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float acc[V] = {0};
      // 4 independent row loads in flight per step (a single runtime-n
      // loop leaves one load outstanding and goes latency-bound)
      int i = 0;
      for (; i + 4 <= k; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)gi[i + 0] * d + jv * V, x0);
        VecTraits<T>::load(X + (long)gi[i + 1] * d + jv * V, x1);
        VecTraits<T>::load(X + (long)gi[i + 2] * d + jv * V, x2);
        VecTraits<T>::load(X + (long)gi[i + 3] * d + jv * V, x3);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += (x0[c] + x1[c]) + (x2[c] + x3[c]);
      }
      for (; i < k; ++i) {
        float x[V];
        VecTraits<T>::load(X + (long)gi[i] * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) acc[c] *= inv;
      VecTraits<T>::store(og + jv * V, acc);
    }
  } else {
    for (long j = start; j < d; j += stride) {
      float acc = 0.0f;
      for (int i = 0; i < k; ++i) acc += to_f<T>(X[(long)gi[i] * d + j]);
      og[j] = from_f<T>(acc * inv);
    }
  }
}

template <typename T, bool VEC>
__global__ void bucket_mean_kernel(const T* __restrict__ X,
                                   const int* __restrict__ perm, int n,
                                   int bucket, T* __restrict__ out, long d) {
  constexpr int V = VecTraits<T>::V;
  const int b = blockIdx.y;
  const int lo = b * bucket;
  const int hi = min(n, lo + bucket);
  T* ob = out + (long)b * d;
  const float inv = 1.0f / (float)(hi - lo);
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float acc[V] = {0};
      int i = lo;
      for (; i + 4 <= hi; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)perm[i + 0] * d + jv * V, x0);
        VecTraits<T>::load(X + (long)perm[i + 1] * d + jv * V, x1);
        VecTraits<T>::load(X + (long)perm[i + 2] * d + jv * V, x2);
        VecTraits<T>::load(X + (long)perm[i + 3] * d + jv * V, x3);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += (x0[c] + x1[c]) + (x2[c] + x3[c]);
      }
      for (; i < hi; ++i) {
        float x[V];
        VecTraits<T>::load(X + (long)perm[i] * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) acc[c] *= inv;
      VecTraits<T>::store(ob + jv * V, acc);
    }
  } else {
    for (long j = start; j < d; j += stride) {
      float acc = 0.0f;
      for (int i = lo; i < hi; ++i) acc += to_f<T>(X[(long)perm[i] * d + j]);
      ob[j] = from_f<T>(acc * inv);
    }
  }
}

// -- fused fixed-point iterations ------------------------------------------
// Per-row weights are a function of the global distances only; stage them
// in LDS once per block instead of recomputing per column.

template <typename T, bool VEC>
__global__ void weiszfeld_update_kernel(const T* __restrict__ X,
                                        const float* __restrict__ z,
                                        const float* __restrict__ dist2,
                                        float* __restrict__ z_new,
                                        float* __restrict__ shift2, int n,
                                        long d, float eps) {
  __shared__ float w_lds[MAX_N_LDS];
  __shared__ float red[16];
  constexpr int V = VecTraits<T>::V;
  for (int i = threadIdx.x; i < n; i += blockDim.x)
    w_lds[i] = 1.0f / fmaxf(sqrtf(dist2[i]), eps);
  __syncthreads();
  float den = 0.0f;
  for (int i = 0; i < n; ++i) den += w_lds[i];
  const float inv_den = 1.0f / den;

  float local_shift = 0.0f;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float num[V] = {0};
      int i = 0;
      for (; i + 4 <= n; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)(i + 0) * d + jv * V, x0);
        VecTraits<T>::load(X + (long)(i + 1) * d + jv * V, x1);
        VecTraits<T>::load(X + (long)(i + 2) * d + jv * V, x2);
        VecTraits<T>::load(X + (long)(i + 3) * d + jv * V, x3);
        const float w0 = w_lds[i], w1 = w_lds[i + 1], w2 = w_lds[i + 2],
                    w3 = w_lds[i + 3];
#pragma unroll
        for (int c = 0; c < V; ++c)
          num[c] += (w0 * x0[c] + w1 * x1[c]) + (w2 * x2[c] + w3 * x3[c]);
      }
      for (; i < n; ++i) {
        const float w = w_lds[i];
        float x[V];
        VecTraits<T>::load(X + (long)i * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) num[c] += w * x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) {
        const float zi = num[c] * inv_den;
        const float dz = zi - z[jv * V + c];
        local_shift += dz * dz;
        z_new[jv * V + c] = zi;
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      float num = 0.0f;
      for (int i = 0; i < n; ++i)
        num += w_lds[i] * to_f<T>(X[(long)i * d + j]);
      const float zi = num * inv_den;
      const float dz = zi - z[j];
      local_shift += dz * dz;
      z_new[j] = zi;
    }
  }
  const float s = block_reduce_sum(local_shift, red);
  if (threadIdx.x == 0) atomicAdd(shift2, s);
}

template <typename T, bool VEC>
__global__ void cc_update_kernel(const T* __restrict__ X,
                                 const float* __restrict__ v,
                                 const float* __restrict__ dist2,
                                 float* __restrict__ v_new, int n, long d,
                                 float c_tau, float eps) {
  __shared__ float a_lds[MAX_N_LDS];
  constexpr int V = VecTraits<T>::V;
  for (int i = threadIdx.x; i < n; i += blockDim.x)
    a_lds[i] = fminf(1.0f, c_tau / fmaxf(sqrtf(dist2[i]), eps));
  __syncthreads();
  float alpha_sum = 0.0f;
  for (int i = 0; i < n; ++i) alpha_sum += a_lds[i];
  const float inv_n = 1.0f / (float)n;

  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float acc[V] = {0};
      int i = 0;
      for (; i + 4 <= n; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)(i + 0) * d + jv * V, x0);
        VecTraits<T>::load(X + (long)(i + 1) * d + jv * V, x1);
        VecTraits<T>::load(X + (long)(i + 2) * d + jv * V, x2);
        VecTraits<T>::load(X + (long)(i + 3) * d + jv * V, x3);
        const float a0 = a_lds[i], a1 = a_lds[i + 1], a2 = a_lds[i + 2],
                    a3 = a_lds[i + 3];
#pragma unroll
        for (int c = 0; c < V; ++c)
          acc[c] += (a0 * x0[c] + a1 * x1[c]) + (a2 * x2[c] + a3 * x3[c]);
      }
      for (; i < n; ++i) {
        const float a = a_lds[i];
        float x[V];
        VecTraits<T>::load(X + (long)i * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += a * x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) {
        const float vj = v[jv * V + c];
        v_new[jv * V + c] = vj + (acc[c] - alpha_sum * vj) * inv_n;
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      const float vj = v[j];
      float acc = 0.0f;
      for (int i = 0; i < n; ++i)
        acc += a_lds[i] * (to_f<T>(X[(long)i * d + j]) - vj);
      v_new[j] = vj + acc * inv_n;
    }
  }
}

// -- CAF fused power-iteration pair (reference caf.py:133-184)
// s_i = sum_j (X_ij - mu_j) v_j            (caf_matvec_group_kernel)
// t_j = (*scale) * sum_i a_i (X_ij - mu_j) (caf_colsum_kernel)
// Neither materializes diffs = X - mu (32 GB f32 at 64 x 125M) and both
// replace rocblas gemvt, which runs at ~260 GB/s on this skinny shape.

template <typename T, bool VEC>
__global__ void caf_matvec_group_kernel(const T* __restrict__ X,
                                        const float* __restrict__ mu,
                                        const float* __restrict__ v,
                                        float* __restrict__ out, int n,
                                        long d) {
  __shared__ float lds[DIST_GROUP][16];
  constexpr int V = VecTraits<T>::V;
  const int g0 = blockIdx.y * DIST_GROUP;
  const int rows = min(DIST_GROUP, n - g0);
  float acc[DIST_GROUP] = {0};
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float muv[V], vv[V];
      {
        float tmp[4];
#pragma unroll
        for (int q4 = 0; q4 < V / 4; ++q4) {
          VecTraits<float>::load(mu + jv * V + q4 * 4, tmp);
#pragma unroll
          for (int c = 0; c < 4; ++c) muv[q4 * 4 + c] = tmp[c];
          VecTraits<float>::load(v + jv * V + q4 * 4, tmp);
#pragma unroll
          for (int c = 0; c < 4; ++c) vv[q4 * 4 + c] = tmp[c];
        }
      }
      int i = 0;
      for (; i + 4 <= rows; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)(g0 + i + 0) * d + jv * V, x0);
        VecTraits<T>::load(X + (long)(g0 + i + 1) * d + jv * V, x1);
        VecTraits<T>::load(X + (long)(g0 + i + 2) * d + jv * V, x2);
        VecTraits<T>::load(X + (long)(g0 + i + 3) * d + jv * V, x3);
#pragma unroll
        for (int c = 0; c < V; ++c) {
          acc[i + 0] += (x0[c] - muv[c]) * vv[c];
          acc[i + 1] += (x1[c] - muv[c]) * vv[c];
          acc[i + 2] += (x2[c] - muv[c]) * vv[c];
          acc[i + 3] += (x3[c] - muv[c]) * vv[c];
        }
      }
      for (; i < rows; ++i) {
        float x[V];
        VecTraits<T>::load(X + (long)(g0 + i) * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[i] += (x[c] - muv[c]) * vv[c];
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      const float mj = mu[j], vj = v[j];
      for (int i = 0; i < rows; ++i)
        acc[i] += (to_f<T>(X[(long)(g0 + i) * d + j]) - mj) * vj;
    }
  }
#pragma unroll
  for (int i = 0; i < DIST_GROUP; ++i) {
    if (i < rows) {
      const float s = block_reduce_sum(acc[i], lds[i]);
      if (threadIdx.x == 0) atomicAdd(&out[g0 + i], s);
    }
    __syncthreads();
  }
}

// sum_i a_i X_ij - mu_j * sum_i a_i, scaled: the mu subtraction factors out
// of the row loop, so the inner walk is a plain weighted column sum.
// mu may be null (plain weighted sum: computes the weighted mean when
// a = w and *scale = 1/sum(w)); scale may be null (1.0).
template <typename T, bool VEC>
__global__ void caf_colsum_kernel(const T* __restrict__ X,
                                  const float* __restrict__ a,
                                  const float* __restrict__ mu,
                                  const float* __restrict__ scale,
                                  float* __restrict__ out, int n, long d) {
  __shared__ float a_lds[MAX_N_LDS];
  constexpr int V = VecTraits<T>::V;
  for (int i = threadIdx.x; i < n; i += blockDim.x) a_lds[i] = a[i];
  __syncthreads();
  float a_sum = 0.0f;
  for (int i = 0; i < n; ++i) a_sum += a_lds[i];
  const float sc = scale ? *scale : 1.0f;

  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  if (VEC) {
    const long dv = d / V;
    for (long jv = start; jv < dv; jv += stride) {
      float acc[V] = {0};
      int i = 0;
      for (; i + 4 <= n; i += 4) {
        float x0[V], x1[V], x2[V], x3[V];
        VecTraits<T>::load(X + (long)(i + 0) * d + jv * V, x0);
        VecTraits<T>::load(X + (long)(i + 1) * d + jv * V, x1);
        VecTraits<T>::load(X + (long)(i + 2) * d + jv * V, x2);
        VecTraits<T>::load(X + (long)(i + 3) * d + jv * V, x3);
        const float a0 = a_lds[i], a1 = a_lds[i + 1], a2 = a_lds[i + 2],
                    a3 = a_lds[i + 3];
#pragma unroll
        for (int c = 0; c < V; ++c)
          acc[c] += (a0 * x0[c] + a1 * x1[c]) + (a2 * x2[c] + a3 * x3[c]);
      }
      for (; i < n; ++i) {
        const float ai = a_lds[i];
        float x[V];
        VecTraits<T>::load(X + (long)i * d + jv * V, x);
#pragma unroll
        for (int c = 0; c < V; ++c) acc[c] += ai * x[c];
      }
#pragma unroll
      for (int c = 0; c < V; ++c) {
        const float m = mu ? mu[jv * V + c] : 0.0f;
        out[jv * V + c] = (acc[c] - a_sum * m) * sc;
      }
    }
  } else {
    for (long j = start; j < d; j += stride) {
      float acc = 0.0f;
      for (int i = 0; i < n; ++i)
        acc += a_lds[i] * to_f<T>(X[(long)i * d + j]);
      const float m = mu ? mu[j] : 0.0f;
      out[j] = (acc - a_sum * m) * sc;
    }
  }
}

inline int col_grid(long work, int block) {
  const long want = (work + block - 1) / block;
  return (int)(want < 4096 ? (want > 0 ? want : 1) : 4096);
}

inline int slab_grid(long d, int n, int block) {
  // target ~2048 blocks total across the (slab, row) grid
  long per_row = (d + (long)block * 8 - 1) / ((long)block * 8);
  long cap = 2048 / (n > 0 ? n : 1);
  if (cap < 1) cap = 1;
  return (int)(per_row < cap ? (per_row > 0 ? per_row : 1) : cap);
}

template <typename T>
inline bool vec_ok(long d) {
  return (d % VecTraits<T>::V) == 0;
}

}  // namespace

// ---------------------------------------------------------------------------
// host-side launchers
// ---------------------------------------------------------------------------

#define FOR_TYPES(MACRO) MACRO(float) MACRO(__hip_bfloat16)

template <typename T>
void launch_row_sqnorms(const T* X, float* out, int n, long d,
                        hipStream_t stream) {
  const int block = 256;
  dim3 grid(slab_grid(d, n, block), n);
  if (vec_ok<T>(d))
    hipLaunchKernelGGL((row_red_kernel<T, false, true>), grid, dim3(block), 0,
                       stream, X, nullptr, out, d);
  else
    hipLaunchKernelGGL((row_red_kernel<T, false, false>), grid, dim3(block), 0,
                       stream, X, nullptr, out, d);
}

template <typename T>
void launch_row_center_sqdists(const T* X, const float* z, float* out, int n,
                               long d, hipStream_t stream) {
  const int block = 256;
  const int groups = (n + DIST_GROUP - 1) / DIST_GROUP;
  dim3 grid(slab_grid(d, groups, block), groups);
  if (vec_ok<T>(d))
    hipLaunchKernelGGL((center_sqdists_group_kernel<T, true>), grid,
                       dim3(block), 0, stream, X, z, out, n, d);
  else
    hipLaunchKernelGGL((center_sqdists_group_kernel<T, false>), grid,
                       dim3(block), 0, stream, X, z, out, n, d);
}

template <typename T>
void launch_grouped_weiszfeld(const T* X, const float* Z, float* dist2,
                              float* Z_new, int G, int m, long d, float eps,
                              hipStream_t stream) {
  const int block = 256;
  dim3 grid(slab_grid(d, G, block), G);
  if (vec_ok<T>(d)) {
    hipLaunchKernelGGL((center_sqdists_grouped_kernel<T, true>), grid,
                       dim3(block), 0, stream, X, Z, dist2, m, d);
    hipLaunchKernelGGL((weiszfeld_update_grouped_kernel<T, true>), grid,
                       dim3(block), 0, stream, X, Z, dist2, Z_new, m, d, eps);
  } else {
    hipLaunchKernelGGL((center_sqdists_grouped_kernel<T, false>), grid,
                       dim3(block), 0, stream, X, Z, dist2, m, d);
    hipLaunchKernelGGL((weiszfeld_update_grouped_kernel<T, false>), grid,
                       dim3(block), 0, stream, X, Z, dist2, Z_new, m, d, eps);
  }
}
template void launch_grouped_weiszfeld<float>(const float*, const float*,
                                              float*, float*, int, int, long,
                                              float, hipStream_t);
template void launch_grouped_weiszfeld<__hip_bfloat16>(
    const __hip_bfloat16*, const float*, float*, float*, int, int, long,
    float, hipStream_t);

template <typename T>
void launch_row_scale(const T* X, const float* s, T* out, int n, long d,
                      hipStream_t stream) {
  const int block = 256;
  dim3 grid(slab_grid(d, n, block), n);
  if (vec_ok<T>(d))
    hipLaunchKernelGGL((row_scale_kernel<T, true>), grid, dim3(block), 0,
                       stream, X, s, out, d);
  else
    hipLaunchKernelGGL((row_scale_kernel<T, false>), grid, dim3(block), 0,
                       stream, X, s, out, d);
}

template <typename T>
void launch_mean_rows(const T* X, const int* idx, int k, T* out, long d,
                      hipStream_t stream) {
  const int block = 256;
  const bool v = vec_ok<T>(d);
  const long work = v ? d / VecTraits<T>::V : d;
  if (v)
    hipLaunchKernelGGL((gather_mean_kernel<T, true, false>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       idx, k, out, d);
  else
    hipLaunchKernelGGL((gather_mean_kernel<T, false, false>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       idx, k, out, d);
}

template <typename T>
void launch_group_mean_rows(const T* X, const int* idx, int g, int k, T* out,
                            long d, hipStream_t stream) {
  const int block = 256;
  const bool v = vec_ok<T>(d);
  const long work = (v ? d / VecTraits<T>::V : d + 0) / (g > 4 ? 4 : 1) + 1;
  dim3 grid(col_grid(work, block), g);
  if (v)
    hipLaunchKernelGGL((gather_mean_kernel<T, true, true>), grid, dim3(block),
                       0, stream, X, idx, k, out, d);
  else
    hipLaunchKernelGGL((gather_mean_kernel<T, false, true>), grid, dim3(block),
                       0, stream, X, idx, k, out, d);
}

template <typename T>
void launch_bucket_mean(const T* X, const int* perm, int n, int bucket, int nb,
                        T* out, long d, hipStream_t stream) {
  const int block = 256;
  const bool v = vec_ok<T>(d);
  const long work = (v ? d / VecTraits<T>::V : d + 0) / (nb > 4 ? 4 : 1) + 1;
  dim3 grid(col_grid(work, block), nb);
  if (v)
    hipLaunchKernelGGL((bucket_mean_kernel<T, true>), grid, dim3(block), 0,
                       stream, X, perm, n, bucket, out, d);
  else
    hipLaunchKernelGGL((bucket_mean_kernel<T, false>), grid, dim3(block), 0,
                       stream, X, perm, n, bucket, out, d);
}

template <typename T>
void launch_weiszfeld_update(const T* X, const float* z, const float* dist2,
                             float* z_new, float* shift2, int n, long d,
                             float eps, hipStream_t stream) {
  const int block = 256;
  const bool v = vec_ok<T>(d);
  const long work = v ? d / VecTraits<T>::V : d;
  if (v)
    hipLaunchKernelGGL((weiszfeld_update_kernel<T, true>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       z, dist2, z_new, shift2, n, d, eps);
  else
    hipLaunchKernelGGL((weiszfeld_update_kernel<T, false>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       z, dist2, z_new, shift2, n, d, eps);
}

template <typename T>
void launch_cc_update(const T* X, const float* v, const float* dist2,
                      float* v_new, int n, long d, float c_tau, float eps,
                      hipStream_t stream) {
  const int block = 256;
  const bool vo = vec_ok<T>(d);
  const long work = vo ? d / VecTraits<T>::V : d;
  if (vo)
    hipLaunchKernelGGL((cc_update_kernel<T, true>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       v, dist2, v_new, n, d, c_tau, eps);
  else
    hipLaunchKernelGGL((cc_update_kernel<T, false>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       v, dist2, v_new, n, d, c_tau, eps);
}

template <typename T>
void launch_caf_matvec(const T* X, const float* mu, const float* v, float* out,
                       int n, long d, hipStream_t stream) {
  const int block = 256;
  const int groups = (n + DIST_GROUP - 1) / DIST_GROUP;
  dim3 grid(slab_grid(d, groups, block), groups);
  if (vec_ok<T>(d))
    hipLaunchKernelGGL((caf_matvec_group_kernel<T, true>), grid, dim3(block),
                       0, stream, X, mu, v, out, n, d);
  else
    hipLaunchKernelGGL((caf_matvec_group_kernel<T, false>), grid, dim3(block),
                       0, stream, X, mu, v, out, n, d);
}

template <typename T>
void launch_caf_colsum(const T* X, const float* a, const float* mu,
                       const float* scale, float* out, int n, long d,
                       hipStream_t stream) {
  const int block = 256;
  const bool vo = vec_ok<T>(d);
  const long work = vo ? d / VecTraits<T>::V : d;
  if (vo)
    hipLaunchKernelGGL((caf_colsum_kernel<T, true>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       a, mu, scale, out, n, d);
  else
    hipLaunchKernelGGL((caf_colsum_kernel<T, false>),
                       dim3(col_grid(work, block)), dim3(block), 0, stream, X,
                       a, mu, scale, out, n, d);
}

// explicit instantiations for bind.cpp
#define INSTANTIATE(T)                                                         \
  template void launch_caf_matvec<T>(const T*, const float*, const float*,     \
                                     float*, int, long, hipStream_t);          \
  template void launch_caf_colsum<T>(const T*, const float*, const float*,     \
                                     const float*, float*, int, long,          \
                                     hipStream_t);                             \
  template void launch_row_sqnorms<T>(const T*, float*, int, long,             \
                                      hipStream_t);                            \
  template void launch_row_center_sqdists<T>(const T*, const float*, float*,   \
                                             int, long, hipStream_t);          \
  template void launch_row_scale<T>(const T*, const float*, T*, int, long,     \
                                    hipStream_t);                              \
  template void launch_mean_rows<T>(const T*, const int*, int, T*, long,       \
                                    hipStream_t);                              \
  template void launch_group_mean_rows<T>(const T*, const int*, int, int, T*,  \
                                          long, hipStream_t);                  \
  template void launch_bucket_mean<T>(const T*, const int*, int, int, int, T*, \
                                      long, hipStream_t);                      \
  template void launch_weiszfeld_update<T>(const T*, const float*,             \
                                           const float*, float*, float*, int,  \
                                           long, float, hipStream_t);          \
  template void launch_cc_update<T>(const T*, const float*, const float*,      \
                                    float*, int, long, float, float,           \
                                    hipStream_t);
FOR_TYPES(INSTANTIATE)
#undef INSTANTIATE
