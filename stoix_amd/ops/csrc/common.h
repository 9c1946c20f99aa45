// Common device utilities for stoix_amd CDNA4 (gfx950) kernels.
//
// Philox4x32-10 counter-based RNG: every kernel derives per-env substreams
// from (seed, stream_id, env_id, draw_idx) with no state to carry — the
// MI355X replacement for the reference's explicit JAX PRNG-key threading
// (SURVEY.md §7 "Deterministic-enough PRNG").
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

// ------------------------------------------------------------------ Philox
struct Philox4 {
  uint32_t x, y, z, w;
};

DEV_INLINE uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

DEV_INLINE Philox4 philox4x32_10(uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3,
                                 uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    uint32_t hi0, hi1;
    uint32_t lo0 = mulhilo(M0, c0, &hi0);
    uint32_t lo1 = mulhilo(M1, c2, &hi1);
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}

// uniform in [0, 1)
DEV_INLINE float u32_to_uniform(uint32_t x) {
  return (float)(x >> 8) * (1.0f / 16777216.0f);
}

// 4 uniforms from a counter tuple
struct Rng4 {
  float a, b, c, d;
};

DEV_INLINE Rng4 philox_uniform4(uint64_t seed, uint32_t stream, uint32_t env, uint32_t draw) {
  Philox4 p = philox4x32_10(env, draw, stream, 0u, (uint32_t)seed, (uint32_t)(seed >> 32));
  return {u32_to_uniform(p.x), u32_to_uniform(p.y), u32_to_uniform(p.z), u32_to_uniform(p.w)};
}

// 2 standard normals (Box-Muller) from half a philox block
DEV_INLINE void box_muller(float u1, float u2, float* n1, float* n2) {
  float r = sqrtf(-2.0f * logf(fmaxf(u1, 1.1754944e-38f)));
  float s, c;
  __sincosf(6.283185307179586f * u2, &s, &c);
  *n1 = r * c;
  *n2 = r * s;
}

// step-type codes (stoix_amd.types.StepType)
#define ST_FIRST 0
#define ST_MID 1
#define ST_TERMINATED 2
#define ST_TRUNCATED 3

#define HIP_CHECK(x)                                                           \
  do {                                                                         \
    hipError_t _e = (x);                                                       \
    if (_e != hipSuccess) {                                                    \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, __LINE__); \
    }                                                                          \
  } while (0)
