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
// Row reductions: 2D grid (k-slab, row), block-reduce + one f32 atomic per
// block — the (n,) outputs are tiny, contention is nil (guide G12).
// Column kernels: one thread per coordinate, row loop inside; adjacent
// lanes read adjacent coordinates so every row iteration is one coalesced
// wave transaction.
#include "common.h"

namespace {

// -- row reductions ---------------------------------------------------------

template <typename T, bool CENTER>
__global__ void row_red_kernel(const T* __restrict__ X,
                               const float* __restrict__ z,
                               float* __restrict__ out, long d) {
  __shared__ float lds[16];
  const int row = blockIdx.y;
  const T* xr = X + (long)row * d;
  float acc = 0.0f;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long j = start; j < d; j += stride) {
    float v = to_f<T>(xr[j]);
    if (CENTER) v -= z[j];
    acc += v * v;
  }
  acc = block_reduce_sum(acc, lds);
  if (threadIdx.x == 0) atomicAdd(&out[row], acc);
}

// -- row scaling ------------------------------------------------------------

template <typename T>
__global__ void row_scale_kernel(const T* __restrict__ X,
                                 const float* __restrict__ s,
                                 T* __restrict__ out, long d) {
  const int row = blockIdx.y;
  const float sc = s[row];
  const T* xr = X + (long)row * d;
  T* yr = out + (long)row * d;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long j = start; j < d; j += stride)
    yr[j] = from_f<T>(to_f<T>(xr[j]) * sc);
}

// -- gather means -----------------------------------------------------------

template <typename T>
__global__ void mean_rows_kernel(const T* __restrict__ X,
                                 const int* __restrict__ idx, int k,
                                 T* __restrict__ out, long d) {
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv = 1.0f / (float)k;
  for (long j = start; j < d; j += stride) {
    float acc = 0.0f;
    for (int i = 0; i < k; ++i) acc += to_f<T>(X[(long)idx[i] * d + j]);
    out[j] = from_f<T>(acc * inv);
  }
}

template <typename T>
__global__ void group_mean_rows_kernel(const T* __restrict__ X,
                                       const int* __restrict__ idx, int k,
                                       T* __restrict__ out, long d) {
  const int g = blockIdx.y;
  const int* gi = idx + (long)g * k;
  T* og = out + (long)g * d;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv = 1.0f / (float)k;
  for (long j = start; j < d; j += stride) {
    float acc = 0.0f;
    for (int i = 0; i < k; ++i) acc += to_f<T>(X[(long)gi[i] * d + j]);
    og[j] = from_f<T>(acc * inv);
  }
}

template <typename T>
__global__ void bucket_mean_kernel(const T* __restrict__ X,
                                   const int* __restrict__ perm, int n,
                                   int bucket, T* __restrict__ out, long d) {
  const int b = blockIdx.y;
  const int lo = b * bucket;
  const int hi = min(n, lo + bucket);
  T* ob = out + (long)b * d;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv = 1.0f / (float)(hi - lo);
  for (long j = start; j < d; j += stride) {
    float acc = 0.0f;
    for (int i = lo; i < hi; ++i) acc += to_f<T>(X[(long)perm[i] * d + j]);
    ob[j] = from_f<T>(acc * inv);
  }
}

// -- fused fixed-point iterations ------------------------------------------

template <typename T>
__global__ void weiszfeld_update_kernel(const T* __restrict__ X,
                                        const float* __restrict__ z,
                                        const float* __restrict__ dist2,
                                        float* __restrict__ z_new,
                                        float* __restrict__ shift2, int n,
                                        long d, float eps) {
  __shared__ float lds[16];
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  float local_shift = 0.0f;
  for (long j = start; j < d; j += stride) {
    float num = 0.0f, den = 0.0f;
    for (int i = 0; i < n; ++i) {
      const float dist = fmaxf(sqrtf(dist2[i]), eps);
      const float w = 1.0f / dist;
      num += w * to_f<T>(X[(long)i * d + j]);
      den += w;
    }
    const float zi = num / den;
    const float dz = zi - z[j];
    local_shift += dz * dz;
    z_new[j] = zi;
  }
  const float s = block_reduce_sum(local_shift, lds);
  if (threadIdx.x == 0) atomicAdd(shift2, s);
}

template <typename T>
__global__ void cc_update_kernel(const T* __restrict__ X,
                                 const float* __restrict__ v,
                                 const float* __restrict__ dist2,
                                 float* __restrict__ v_new, int n, long d,
                                 float c_tau, float eps) {
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const float inv_n = 1.0f / (float)n;
  for (long j = start; j < d; j += stride) {
    const float vj = v[j];
    float acc = 0.0f;
    for (int i = 0; i < n; ++i) {
      const float dist = fmaxf(sqrtf(dist2[i]), eps);
      const float alpha = fminf(1.0f, c_tau / dist);
      acc += alpha * (to_f<T>(X[(long)i * d + j]) - vj);
    }
    v_new[j] = vj + acc * inv_n;
  }
}

inline int col_grid(long d, int block) {
  const long want = (d + block - 1) / block;
  return (int)(want < 4096 ? (want > 0 ? want : 1) : 4096);
}

inline int slab_grid(long d, int n, int block) {
  // target ~2048 blocks total across the (slab, row) grid
  long per_row = (d + (long)block * 8 - 1) / ((long)block * 8);
  long cap = 2048 / (n > 0 ? n : 1);
  if (cap < 1) cap = 1;
  return (int)(per_row < cap ? (per_row > 0 ? per_row : 1) : cap);
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
  hipLaunchKernelGGL((row_red_kernel<T, false>), grid, dim3(block), 0, stream,
                     X, nullptr, out, d);
}

template <typename T>
void launch_row_center_sqdists(const T* X, const float* z, float* out, int n,
                               long d, hipStream_t stream) {
  const int block = 256;
  dim3 grid(slab_grid(d, n, block), n);
  hipLaunchKernelGGL((row_red_kernel<T, true>), grid, dim3(block), 0, stream,
                     X, z, out, d);
}

template <typename T>
void launch_row_scale(const T* X, const float* s, T* out, int n, long d,
                      hipStream_t stream) {
  const int block = 256;
  dim3 grid(slab_grid(d, n, block), n);
  hipLaunchKernelGGL((row_scale_kernel<T>), grid, dim3(block), 0, stream, X, s,
                     out, d);
}

template <typename T>
void launch_mean_rows(const T* X, const int* idx, int k, T* out, long d,
                      hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL((mean_rows_kernel<T>), dim3(col_grid(d, block)),
                     dim3(block), 0, stream, X, idx, k, out, d);
}

template <typename T>
void launch_group_mean_rows(const T* X, const int* idx, int g, int k, T* out,
                            long d, hipStream_t stream) {
  const int block = 256;
  dim3 grid(col_grid(d, block) / (g > 4 ? 4 : 1) + 1, g);
  hipLaunchKernelGGL((group_mean_rows_kernel<T>), grid, dim3(block), 0, stream,
                     X, idx, k, out, d);
}

template <typename T>
void launch_bucket_mean(const T* X, const int* perm, int n, int bucket, int nb,
                        T* out, long d, hipStream_t stream) {
  const int block = 256;
  dim3 grid(col_grid(d, block) / (nb > 4 ? 4 : 1) + 1, nb);
  hipLaunchKernelGGL((bucket_mean_kernel<T>), grid, dim3(block), 0, stream, X,
                     perm, n, bucket, out, d);
}

template <typename T>
void launch_weiszfeld_update(const T* X, const float* z, const float* dist2,
                             float* z_new, float* shift2, int n, long d,
                             float eps, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL((weiszfeld_update_kernel<T>), dim3(col_grid(d, block)),
                     dim3(block), 0, stream, X, z, dist2, z_new, shift2, n, d,
                     eps);
}

template <typename T>
void launch_cc_update(const T* X, const float* v, const float* dist2,
                      float* v_new, int n, long d, float c_tau, float eps,
                      hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL((cc_update_kernel<T>), dim3(col_grid(d, block)),
                     dim3(block), 0, stream, X, v, dist2, v_new, n, d, c_tau,
                     eps);
}

// explicit instantiations for bind.cpp
#define INSTANTIATE(T)                                                         \
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
