// Shared device helpers for the byzpy_amd gfx950 kernels.
// Wave width is 64 on CDNA4 (guide §1) — all reductions assume it.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define DEV __device__ __forceinline__

constexpr int WAVE = 64;

DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

// Block-wide sum; result valid in thread 0. `lds` must hold >= blockDim/64
// floats. Callers sync before reuse.
DEV float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + 63) >> 6;
  v = (threadIdx.x < (unsigned)nw) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) v = wave_reduce_sum(v);
  return v;
}

// -- dtype conversion -------------------------------------------------------
template <typename T>
DEV float to_f(T x);
template <>
DEV float to_f<float>(float x) { return x; }
template <>
DEV float to_f<__hip_bfloat16>(__hip_bfloat16 x) { return __bfloat162float(x); }

template <typename T>
DEV T from_f(float x);
template <>
DEV float from_f<float>(float x) { return x; }
template <>
DEV __hip_bfloat16 from_f<__hip_bfloat16>(float x) { return __float2bfloat16(x); }
