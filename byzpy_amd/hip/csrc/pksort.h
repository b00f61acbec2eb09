// Packed bf16 order-statistic helpers shared by colsel.hip (K1-K3
// register kernels) and gram.hip (the fused gram+median epilogue).
//
// Order statistics only need a MONOTONE key, so bf16 values become
// sortable u16 keys (sign-flip transform) and bitonic networks run on
// v_pk_min_u16 / v_pk_max_u16 — one VALU instruction per
// compare-exchange for TWO independent columns packed in one u32.
#pragma once
#include "common.h"

typedef unsigned int pk_u32;

DEV pk_u32 pk_min_u16(pk_u32 a, pk_u32 b) {
  pk_u32 r;
  asm("v_pk_min_u16 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}
DEV pk_u32 pk_max_u16(pk_u32 a, pk_u32 b) {
  pk_u32 r;
  asm("v_pk_max_u16 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
  return r;
}

// bf16 bits -> sortable u16 key, two lanes at once (32-bit arithmetic:
// per half key = bits ^ (sign ? 0xFFFF : 0x8000); the multiply cannot
// carry across halves)
DEV pk_u32 pk_key_from_bf16(pk_u32 bits) {
  const pk_u32 s = (bits >> 15) & 0x00010001u;
  return bits ^ (0x80008000u + s * 0x7FFFu);
}

DEV float pk_key_to_float(pk_u32 key16) {
  unsigned short bits =
      (key16 & 0x8000u) ? (unsigned short)(key16 ^ 0x8000u)
                        : (unsigned short)(~key16 & 0xFFFFu);
  union { unsigned short s; __hip_bfloat16 h; } c;
  c.s = bits;
  return __bfloat162float(c.h);
}

template <int P>
DEV void bitonic_sort_pk(pk_u32 (&v)[P]) {
#pragma unroll
  for (int k = 2; k <= P; k <<= 1) {
#pragma unroll
    for (int j = k >> 1; j > 0; j >>= 1) {
#pragma unroll
      for (int i = 0; i < P; ++i) {
        const int l = i ^ j;
        if (l > i) {
          const bool asc = (i & k) == 0;
          const pk_u32 a = v[i], b = v[l];
          const pk_u32 lo = pk_min_u16(a, b), hi = pk_max_u16(a, b);
          v[i] = asc ? lo : hi;
          v[l] = asc ? hi : lo;
        }
      }
    }
  }
}

template <int P>
DEV pk_u32 pk_extract_at(const pk_u32 (&v)[P], int pos) {
  pk_u32 r = 0;
#pragma unroll
  for (int i = 0; i < P; ++i)
    if (i == pos) r = v[i];
  return r;
}

// Copy a wave-uniform int into a VGPR: unrolled predicates against SGPR
// operands otherwise become batched s_cselect_b64 mask pairs that spill
// through v_writelane/readlane (colsel.hip's round-1 lesson).
DEV int pk_vecify(int x) {
  int r;
  asm("v_mov_b32_e32 %0, %1" : "=v"(r) : "s"(x));
  return r;
}
