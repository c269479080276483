// Philox 4x32-10 counter RNG — same algorithm family as the reference's
// lib/random/philox_random.h (counter-based, splittable, identical results on
// CPU and GPU). Header-only so the HIP kernels include it too.
#pragma once

#include <cstdint>
#include <cmath>

#ifdef __HIPCC__
#include <hip/hip_runtime.h>
#define PHILOX_DEVICE __host__ __device__ __forceinline__
#else
#define PHILOX_DEVICE inline
#endif

namespace stf {
namespace random {

struct Philox4x32 {
  uint32_t counter[4] = {0, 0, 0, 0};
  uint32_t key[2] = {0, 0};

  PHILOX_DEVICE Philox4x32() {}
  PHILOX_DEVICE Philox4x32(uint64_t seed, uint64_t offset) {
    key[0] = (uint32_t)seed;
    key[1] = (uint32_t)(seed >> 32);
    counter[2] = (uint32_t)offset;
    counter[3] = (uint32_t)(offset >> 32);
  }

  PHILOX_DEVICE void Skip(uint64_t n) {
    uint32_t lo = (uint32_t)n, hi = (uint32_t)(n >> 32);
    counter[0] += lo;
    if (counter[0] < lo) ++hi;
    counter[1] += hi;
    if (counter[1] < hi) {
      if (++counter[2] == 0) ++counter[3];
    }
  }

  static PHILOX_DEVICE uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* lo) {
    uint64_t p = (uint64_t)a * b;
    *lo = (uint32_t)p;
    return (uint32_t)(p >> 32);
  }

  // Returns 4 random uint32s and advances the counter by one.
  PHILOX_DEVICE void Next(uint32_t out[4]) {
    uint32_t c0 = counter[0], c1 = counter[1], c2 = counter[2], c3 = counter[3];
    uint32_t k0 = key[0], k1 = key[1];
    for (int round = 0; round < 10; ++round) {
      uint32_t lo0, lo1;
      uint32_t hi0 = mulhilo(0xD2511F53u, c0, &lo0);
      uint32_t hi1 = mulhilo(0xCD9E8D57u, c2, &lo1);
      uint32_t n0 = hi1 ^ c1 ^ k0;
      uint32_t n1 = lo1;
      uint32_t n2 = hi0 ^ c3 ^ k1;
      uint32_t n3 = lo0;
      c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      k0 += 0x9E3779B9u;
      k1 += 0xBB67AE85u;
    }
    out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
    // advance
    if (++counter[0] == 0)
      if (++counter[1] == 0)
        if (++counter[2] == 0) ++counter[3];
  }
};

PHILOX_DEVICE float Uint32ToFloat01(uint32_t x) {
  // [1, 2) mantissa trick -> [0, 1)
  uint32_t bits = (x >> 9) | 0x3f800000u;
  float f;
  __builtin_memcpy(&f, &bits, 4);
  return f - 1.0f;
}

// Box-Muller pair from two uniforms.
PHILOX_DEVICE void BoxMuller(uint32_t a, uint32_t b, float* z0, float* z1) {
  float u1 = Uint32ToFloat01(a);
  float u2 = Uint32ToFloat01(b);
  if (u1 < 1e-7f) u1 = 1e-7f;
  float r = sqrtf(-2.0f * logf(u1));
  float theta = 6.2831853071795864769f * u2;
  *z0 = r * cosf(theta);
  *z1 = r * sinf(theta);
}

}  // namespace random
}  // namespace stf
