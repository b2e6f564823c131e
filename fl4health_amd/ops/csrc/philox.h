// Philox4x32-10 counter-based RNG for CDNA4 device code.
// Used by the noise / mask-sampling kernels:
// reference behavior: torch Gaussian noise in strategies/noisy_aggregate.py and
// Bernoulli mask sampling in fl4health/utils/functions.py:10-42 — reimplemented
// here as counter-based RNG so every element's draw is a pure function of
// (seed, offset, index): deterministic, order-independent, replayable across ranks.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

struct Philox4 {
  uint32_t x, y, z, w;
};

__device__ __forceinline__ uint32_t mulhilo32(uint32_t a, uint32_t b, uint32_t* hip) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hip = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

// One Philox4x32-10 block: counter (c0..c3), key (k0,k1) -> 4x uint32
__device__ __forceinline__ Philox4 philox4x32(uint64_t seed, uint64_t counter) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  uint32_t c0 = (uint32_t)counter, c1 = (uint32_t)(counter >> 32), c2 = 0u, c3 = 0u;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    uint32_t h0, h1;
    uint32_t l0 = mulhilo32(M0, c0, &h0);
    uint32_t l1 = mulhilo32(M1, c2, &h1);
    uint32_t n0 = h1 ^ c1 ^ k0;
    uint32_t n1 = l1;
    uint32_t n2 = h0 ^ c3 ^ k1;
    uint32_t n3 = l0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}

__device__ __forceinline__ float u32_to_uniform(uint32_t v) {
  // (0,1]: avoid 0 for log()
  return ((float)v + 1.0f) * 2.3283064e-10f;  // 2^-32
}

// Two standard normals from 4 uniform bits via Box-Muller.
__device__ __forceinline__ void box_muller(uint32_t a, uint32_t b, float* n0, float* n1) {
  float u1 = u32_to_uniform(a);
  float u2 = u32_to_uniform(b);
  float r = sqrtf(-2.0f * logf(u1));
  float s, c;
  __sincosf(6.2831853071795864f * u2, &s, &c);
  *n0 = r * c;
  *n1 = r * s;
}
