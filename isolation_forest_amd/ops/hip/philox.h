// Counter-based Philox4x32-10 device RNG — EXACT mirror of utils/rng.py.
// Every random decision is a pure function of (seed, purpose, tree, index,
// attempt); tests/test_gpu.py enforces bitwise CPU<->GPU forest equality.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

namespace ifa {

// Purpose tags — keep in sync with utils/rng.py
constexpr uint32_t P_BAG = 1;
constexpr uint32_t P_FEATSUB = 2;
constexpr uint32_t P_FEATSEL = 3;
constexpr uint32_t P_SPLIT = 4;
constexpr uint32_t P_EIF_COORD = 5;
constexpr uint32_t P_EIF_NORMAL = 6;
constexpr uint32_t P_EIF_INTERCEPT = 7;
constexpr uint32_t P_DATA = 8;

struct U4 {
  uint32_t x, y, z, w;
};

__device__ __forceinline__ U4 philox4x32(uint32_t c0, uint32_t c1, uint32_t c2,
                                         uint32_t c3, uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0 = __umulhi(M0, c0), lo0 = M0 * c0;
    uint32_t hi1 = __umulhi(M1, c2), lo1 = M1 * c2;
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return U4{c0, c1, c2, c3};
}

__device__ __forceinline__ uint32_t rng_u32(uint64_t seed, uint32_t purpose,
                                            uint32_t tree, uint32_t index,
                                            uint32_t attempt = 0) {
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
  return philox4x32(attempt, index, tree, purpose, k0, k1).x;
}

// float64 uniform in [0,1) with 24-bit resolution (== rng.py uniform)
__device__ __forceinline__ double rng_uniform(uint64_t seed, uint32_t purpose,
                                              uint32_t tree, uint32_t index,
                                              uint32_t attempt = 0) {
  uint32_t r = rng_u32(seed, purpose, tree, index, attempt);
  return (double)(r >> 8) * (1.0 / 16777216.0);
}

__device__ __forceinline__ void rng_uniform2(uint64_t seed, uint32_t purpose,
                                             uint32_t tree, uint32_t index,
                                             double* u1, double* u2,
                                             uint32_t attempt = 0) {
  uint32_t k0 = (uint32_t)(seed & 0xFFFFFFFFull);
  uint32_t k1 = (uint32_t)(seed >> 32);
  U4 r = philox4x32(attempt, index, tree, purpose, k0, k1);
  *u1 = (double)(r.x >> 8) * (1.0 / 16777216.0);
  *u2 = (double)(r.y >> 8) * (1.0 / 16777216.0);
}

// integer in [0, bound): (u32 * bound) >> 32 (== rng.py randint_below)
__device__ __forceinline__ uint32_t rng_below(uint64_t seed, uint32_t purpose,
                                              uint32_t tree, uint32_t index,
                                              uint32_t bound,
                                              uint32_t attempt = 0) {
  uint32_t r = rng_u32(seed, purpose, tree, index, attempt);
  return (uint32_t)(((uint64_t)r * (uint64_t)bound) >> 32);
}

}  // namespace ifa
