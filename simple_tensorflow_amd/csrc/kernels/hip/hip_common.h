// Shared helpers for the CDNA4 (gfx950) HIP kernels.
//
// Design per /opt/skills/guides/cdna_hip_programming.md: wave64, workgroups of
// 256 (4 waves), vectorized bf16 access (short4/short8 reinterpret), grid
// capped at ~2048 blocks with grid-stride for memory-bound ops, XCD-aware
// block swizzle for tiled ops.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define STF_WAVE 64
#define STF_NUM_XCD 8
#define STF_NUM_CU 256

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using s16x8 = __attribute__((ext_vector_type(8))) short;

__device__ __forceinline__ float bf16_to_f32(uint16_t v) {
  union { uint32_t u; float f; } c;
  c.u = ((uint32_t)v) << 16;
  return c.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
  union { uint32_t u; float f; } c;
  c.f = f;
  uint32_t lsb = (c.u >> 16) & 1;
  return (uint16_t)((c.u + 0x7fffu + lsb) >> 16);
}

// Grid sizing for memory-bound grid-stride kernels (guide §6 G11).
inline dim3 ElemwiseGrid(int64_t n, int block = 256, int unroll = 8) {
  int64_t want = (n + (int64_t)block * unroll - 1) / ((int64_t)block * unroll);
  int64_t cap = 2048;
  return dim3((uint32_t)(want < cap ? (want > 0 ? want : 1) : cap));
}

// Bijective XCD-aware block remap (guide §5 T1): contiguous chunks per XCD.
__device__ __forceinline__ int XcdSwizzle(int bid, int nblocks) {
  int q = nblocks / STF_NUM_XCD, r = nblocks % STF_NUM_XCD;
  if (q == 0) return bid;
  int xcd = bid % STF_NUM_XCD, idx = bid / STF_NUM_XCD;
  int base = xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + idx;
}

#define HIP_LAUNCH_CHECK()                                       \
  do {                                                           \
    hipError_t _e = hipGetLastError();                           \
    if (_e != hipSuccess) return _e;                             \
  } while (0)
