// common.h — shared types/helpers for the CDNA4 kernel sources.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

typedef unsigned short u16;
typedef short s8v __attribute__((ext_vector_type(8)));    // 16 B of bf16
typedef float f4v __attribute__((ext_vector_type(4)));
typedef uint32_t u4v_ __attribute__((ext_vector_type(4)));  // 16 B raw

#define DEVINL __device__ __forceinline__

DEVINL float b2f(u16 u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)u) << 16;
  return c.f;
}

DEVINL u16 f2b(float f) {
  union { float f; uint32_t i; } c;
  c.f = f;
  uint32_t u = c.i;
  if ((u & 0x7fffffffu) > 0x7f800000u) return (u16)0x7fc0;  // NaN
  u += 0x7fffu + ((u >> 16) & 1u);  // round to nearest even
  return (u16)(u >> 16);
}

DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int m = 1; m < 64; m <<= 1) v += __shfl_xor(v, m);
  return v;
}
