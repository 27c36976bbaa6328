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

typedef float f2v __attribute__((ext_vector_type(2)));

// 8 OCP-e4m3 bytes -> 8 bf16 (exact: e4m3's 3-bit mantissa fits bf16)
DEVINL s8v fp8x8_to_bf16(unsigned long long raw) {
  uint32_t lo = (uint32_t)raw, hi = (uint32_t)(raw >> 32);
  f2v a = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
  f2v b = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
  f2v c = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
  f2v d = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
  u16 o[8];
  o[0] = f2b(a[0]); o[1] = f2b(a[1]); o[2] = f2b(b[0]); o[3] = f2b(b[1]);
  o[4] = f2b(c[0]); o[5] = f2b(c[1]); o[6] = f2b(d[0]); o[7] = f2b(d[1]);
  return *(s8v*)o;
}

// 8 floats (pre-scaled by rs) -> 8 packed e4m3 bytes
DEVINL unsigned long long f32x8_to_fp8(const float* f, float rs) {
  uint32_t lo = 0, hi = 0;
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(f[0] * rs, f[1] * rs, lo, false);
  lo = __builtin_amdgcn_cvt_pk_fp8_f32(f[2] * rs, f[3] * rs, lo, true);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(f[4] * rs, f[5] * rs, hi, false);
  hi = __builtin_amdgcn_cvt_pk_fp8_f32(f[6] * rs, f[7] * rs, hi, true);
  return (unsigned long long)lo | ((unsigned long long)hi << 32);
}
