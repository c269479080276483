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

// Exact division by a launch-constant divisor via mul+shift (round-up
// magic, Hacker's Delight §10-9: mul = ceil(2^(32+l)/d), exact for all
// n < 2^31 with the 64-bit multiplier — mul can be 2^33, which is why it
// must NOT be stored in 32 bits). Avoids ~30-instruction runtime division
// sequences inside the implicit-GEMM address generators.
struct U32Div {
  unsigned long long mul = 1;
  uint32_t shift = 0;
  uint32_t d = 1;
  void init(uint32_t div) {
    d = div;
    if (d <= 1) { mul = 1; shift = 0; return; }
    uint32_t l = 0;
    while ((1ull << l) < d) ++l;
    shift = 32 + l;
    mul = (unsigned long long)(((__uint128_t(1) << shift) + d - 1) / d);
  }
  __device__ __forceinline__ uint32_t div(uint32_t n) const {
    if (d == 1) return n;
    return (uint32_t)(((unsigned long long)n * mul) >> shift);
  }
  __device__ __forceinline__ uint32_t mod(uint32_t n, uint32_t q) const {
    return n - q * d;
  }
};

// A-operand address generators for the GEMM staging paths: the plain GEMM
// reads a row-major matrix; the implicit-GEMM convolution maps
// (row = output pixel, k = (r,s,c)) onto the NHWC input directly — the
// im2col matrix is never materialized (reference conv_ops.cc would call
// cuDNN implicit-GEMM here). Out-of-bounds (padding / rsc->rscp K padding)
// resolves to a zeroed 16B page.
struct LinearAG {
  const uint16_t* base;
  int64_t ld;
  __device__ __forceinline__ const uint16_t* at(int64_t row,
                                                int64_t k8) const {
    return base + row * ld + k8;
  }
};

struct ConvAG {
  const uint16_t* x;           // NHWC input
  const uint16_t* zero16;      // >=16 zero bytes
  U32Div div_pq, div_q, div_c, div_s;
  int H, W, C, S;
  int sh, sw, ph, pw;
  int64_t rsc;                 // un-padded K extent
  // row = ((n*P)+p)*Q + q; k8 = 8-element-aligned (r,s,c) index.
  __device__ __forceinline__ const uint16_t* at(int64_t row,
                                                int64_t k8) const {
    if (k8 >= rsc) return zero16;
    uint32_t row32 = (uint32_t)row, k32 = (uint32_t)k8;
    uint32_t n = div_pq.div(row32);
    uint32_t pq = div_pq.mod(row32, n);
    uint32_t p = div_q.div(pq);
    uint32_t q = div_q.mod(pq, p);
    uint32_t rs = div_c.div(k32);
    uint32_t c = div_c.mod(k32, rs);
    uint32_t r = div_s.div(rs);
    uint32_t s_ = div_s.mod(rs, r);
    int h = (int)p * sh - ph + (int)r;
    int w = (int)q * sw - pw + (int)s_;
    if (h < 0 || h >= H || w < 0 || w >= W) return zero16;
    return x + ((((int64_t)n * H + h) * W + w) * C + c);
  }
};
