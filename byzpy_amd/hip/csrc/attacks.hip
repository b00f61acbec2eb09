// Attack-math kernels for gfx950.
//
// little_fused: the Little attack's mu + z * sigma per coordinate in ONE
// streaming pass (per-column sum / sum-of-squares, the reference's
// chunked Sigma-x / Sigma-x^2 decomposition little.py:219-224 fused into
// one kernel). Column-per-thread, adjacent lanes on adjacent columns.
//
// gaussian_fill: counter-based Philox4x32-10 + Box-Muller normal fill —
// the Gaussian attack stops consuming torch RNG on the hot path. The
// stream is deterministic per (seed, index) and independent of grid
// shape; it is a DIFFERENT sequence than torch's CPU generator (the
// seeded-CPU contract stays available through the functional path).
#include "common.h"

namespace {

typedef unsigned int u32;
typedef unsigned long long u64;

// -- Little -----------------------------------------------------------------

template <typename T>
__global__ void little_kernel(const T* __restrict__ X, T* __restrict__ out,
                              int n, long d, float z) {
  const long col0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long col = col0; col < d; col += stride) {
    const T* xc = X + col;
    float s = 0.0f, s2 = 0.0f;
    int row = 0;
    for (; row + 3 < n; row += 4) {
      // 4 loads in flight per step
      const float v0 = to_f<T>(xc[(long)(row + 0) * d]);
      const float v1 = to_f<T>(xc[(long)(row + 1) * d]);
      const float v2 = to_f<T>(xc[(long)(row + 2) * d]);
      const float v3 = to_f<T>(xc[(long)(row + 3) * d]);
      s += v0 + v1 + v2 + v3;
      s2 += v0 * v0 + v1 * v1 + v2 * v2 + v3 * v3;
    }
    for (; row < n; ++row) {
      const float v = to_f<T>(xc[(long)row * d]);
      s += v;
      s2 += v * v;
    }
    const float mu = s / n;
    const float var = fmaxf(s2 / n - mu * mu, 0.0f);  // population variance
    out[col] = from_f<T>(mu + z * sqrtf(var));
  }
}

// -- Philox4x32-10 Gaussian fill --------------------------------------------

DEV u32 mulhi32(u32 a, u32 b) { return (u32)(((u64)a * b) >> 32); }

DEV void philox_round(u32& c0, u32& c1, u32& c2, u32& c3, u32 k0, u32 k1) {
  const u32 lo0 = 0xD2511F53u * c0;
  const u32 hi0 = mulhi32(0xD2511F53u, c0);
  const u32 lo1 = 0xCD9E8D57u * c2;
  const u32 hi1 = mulhi32(0xCD9E8D57u, c2);
  const u32 n0 = hi1 ^ c1 ^ k0;
  const u32 n1 = lo1;
  const u32 n2 = hi0 ^ c3 ^ k1;
  const u32 n3 = lo0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

DEV void philox10(u32 ctr_lo, u32 ctr_hi, u32 key_lo, u32 key_hi, u32& r0,
                  u32& r1, u32& r2, u32& r3) {
  u32 c0 = ctr_lo, c1 = ctr_hi, c2 = 0x2E7366CFu, c3 = 0x69C4C6E5u;
  u32 k0 = key_lo, k1 = key_hi;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  r0 = c0; r1 = c1; r2 = c2; r3 = c3;
}

// two Box-Muller normals from two u32s
DEV void box_muller(u32 a, u32 b, float& n0, float& n1) {
  // (a + 1) in (0, 2^32]: avoids log(0)
  const float u1 = ((float)a + 1.0f) * 2.3283064e-10f;
  const float u2 = (float)b * 2.3283064e-10f;
  const float r = sqrtf(-2.0f * __logf(u1));
  const float th = 6.2831853f * u2;
  n0 = r * __cosf(th);
  n1 = r * __sinf(th);
}

template <typename T>
__global__ void gaussian_fill_kernel(T* __restrict__ out, long d, u32 seed_lo,
                                     u32 seed_hi, float mu, float sigma) {
  const long q0 = (long)blockIdx.x * blockDim.x + threadIdx.x;  // quad index
  const long nquads = (d + 3) >> 2;
  const long qstride = (long)gridDim.x * blockDim.x;
  for (long q = q0; q < nquads; q += qstride) {
    u32 r0, r1, r2, r3;
    philox10((u32)q, (u32)(q >> 32) ^ seed_hi, seed_lo, seed_hi, r0, r1, r2,
             r3);
    float n0, n1, n2, n3;
    box_muller(r0, r1, n0, n1);
    box_muller(r2, r3, n2, n3);
    const float v[4] = {n0, n1, n2, n3};
    const long base = q << 2;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      if (base + j < d) out[base + j] = from_f<T>(mu + sigma * v[j]);
  }
}

}  // namespace

template <typename T>
void launch_little(const T* X, T* out, int n, long d, float z,
                   hipStream_t stream) {
  const int block = 256;
  const long want = (d + block - 1) / block;
  const int grid = (int)(want < 8192 ? (want > 0 ? want : 1) : 8192);
  hipLaunchKernelGGL(little_kernel<T>, dim3(grid), dim3(block), 0, stream, X,
                     out, n, d, z);
}
template void launch_little<float>(const float*, float*, int, long, float,
                                   hipStream_t);
template void launch_little<__hip_bfloat16>(const __hip_bfloat16*,
                                            __hip_bfloat16*, int, long, float,
                                            hipStream_t);

template <typename T>
void launch_gaussian_fill(T* out, long d, unsigned long long seed, float mu,
                          float sigma, hipStream_t stream) {
  const int block = 256;
  const long nquads = (d + 3) >> 2;
  const long want = (nquads + block - 1) / block;
  const int grid = (int)(want < 8192 ? (want > 0 ? want : 1) : 8192);
  hipLaunchKernelGGL(gaussian_fill_kernel<T>, dim3(grid), dim3(block), 0,
                     stream, out, d, (u32)seed, (u32)(seed >> 32), mu, sigma);
}
template void launch_gaussian_fill<float>(float*, long, unsigned long long,
                                          float, float, hipStream_t);
template void launch_gaussian_fill<__hip_bfloat16>(__hip_bfloat16*, long,
                                                   unsigned long long, float,
                                                   float, hipStream_t);
