// Common device helpers for the vantage6_amd CDNA4 (gfx950) kernel library.
// Wave size is 64 on CDNA4; block sizes are multiples of 64 everywhere.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define WAVE 64
#define DEVINL __device__ __forceinline__

using bf16 = __hip_bfloat16;

// vector types for wide loads/stores (16 B per lane where possible)
typedef float  f32x4  __attribute__((ext_vector_type(4)));
typedef float  f32x16 __attribute__((ext_vector_type(16)));
typedef short  s16x8  __attribute__((ext_vector_type(8)));   // 8 bf16 raw
typedef short  s16x4  __attribute__((ext_vector_type(4)));
typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));

DEVINL float bf16_to_f32(short raw) {
  union { float f; unsigned int u; } cvt;
  cvt.u = ((unsigned int)(unsigned short)raw) << 16;
  return cvt.f;
}

DEVINL short f32_to_bf16(float x) {
  union { float f; unsigned int u; } cvt;
  cvt.f = x;
  unsigned int u = cvt.u;
  // round-to-nearest-even like hardware v_cvt
  unsigned int rounding = 0x7fff + ((u >> 16) & 1);
  // NaN-safe: propagate NaN as quiet NaN
  if ((u & 0x7f800000u) == 0x7f800000u && (u & 0x007fffffu)) return (short)0x7fc0;
  return (short)((u + rounding) >> 16);
}

// ---------------------------------------------------------------- Philox4x32
// Counter-based RNG (Philox 4x32-10). Deterministic per (key, counter):
// used for secure-aggregation pairwise masks and DP gaussian noise.
struct Philox4 { uint32_t x, y, z, w; };

DEVINL uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = (uint64_t)a * b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

DEVINL Philox4 philox4x32(uint64_t seed, uint64_t counter) {
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint32_t c0 = (uint32_t)counter, c1 = (uint32_t)(counter >> 32), c2 = 0xdeadbeefu,
           c3 = 0xcafebabeu;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(0xD2511F53u, c0, &hi0);
    uint32_t lo1 = mulhilo(0xCD9E8D57u, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
  }
  return {c0, c1, c2, c3};
}

DEVINL uint64_t philox_u64(uint64_t seed, uint64_t counter) {
  Philox4 p = philox4x32(seed, counter);
  return ((uint64_t)p.x << 32) | p.y;
}

// two independent N(0,1) samples from one philox draw (Box-Muller)
DEVINL void philox_gauss2(uint64_t seed, uint64_t counter, float* g0, float* g1) {
  Philox4 p = philox4x32(seed, counter);
  // (0,1] uniforms
  float u1 = ((float)p.x + 1.0f) * (1.0f / 4294967296.0f);
  float u2 = ((float)p.y) * (1.0f / 4294967296.0f);
  float r = sqrtf(-2.0f * __logf(u1));
  float s, c;
  __sincosf(6.2831853071795864f * u2, &s, &c);
  *g0 = r * c;
  *g1 = r * s;
}

DEVINL int cdiv(int a, int b) { return (a + b - 1) / b; }
