// im2col / col2im for NHWC bf16 conv on gfx950 (conv = im2col + MFMA GEMM;
// replaces the reference's cuDNN path, conv_ops.cc:664). col layout:
// col[(n*P+p)*Q+q][(r*S+s)*C+c] — C-contiguous so warp lanes write
// consecutive bytes; reads from x are C-contiguous too.
#include "hip_common.h"

namespace {

struct ConvGeom {
  int N, H, W, C, R, S, K, sh, sw, ph, pw, P, Q;
};

// one thread per 8 consecutive c (vectorized when C % 8 == 0)
__global__ void Im2ColKernel(const __bf16* __restrict__ x,
                             __bf16* __restrict__ col, ConvGeom g,
                             int64_t total_vec, int vec, int64_t ld) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t cvec = g.C / vec;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total_vec;
       i += stride) {
    int64_t rem = i;
    int64_t cv = rem % cvec; rem /= cvec;
    int s = (int)(rem % g.S); rem /= g.S;
    int r = (int)(rem % g.R); rem /= g.R;
    int q = (int)(rem % g.Q); rem /= g.Q;
    int p = (int)(rem % g.P); rem /= g.P;
    int n = (int)rem;
    int ih = p * g.sh - g.ph + r;
    int iw = q * g.sw - g.pw + s;
    int64_t m = ((int64_t)(n * g.P + p) * g.Q + q);
    int64_t rsc = ((int64_t)(r * g.S + s) * g.C + cv * vec);
    __bf16* dst = col + m * ld + rsc;
    if (ih >= 0 && ih < g.H && iw >= 0 && iw < g.W) {
      const __bf16* src =
          x + (((int64_t)(n * g.H + ih) * g.W + iw) * g.C + cv * vec);
      if (vec == 8) {
        *(ulong2*)dst = *(const ulong2*)src;
      } else {
        for (int e = 0; e < vec; ++e) dst[e] = src[e];
      }
    } else {
      if (vec == 8) {
        *(ulong2*)dst = ulong2{0, 0};
      } else {
        for (int e = 0; e < vec; ++e) dst[e] = (__bf16)0.f;
      }
    }
  }
}

// gather-form col2im (deterministic, no atomics): each dX element sums its
// contributors from dcol.
__global__ void Col2ImKernel(const __bf16* __restrict__ dcol,
                             __bf16* __restrict__ dx, ConvGeom g,
                             int64_t total) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t RSC = (int64_t)g.R * g.S * g.C;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int64_t rem = i;
    int c = (int)(rem % g.C); rem /= g.C;
    int iw = (int)(rem % g.W); rem /= g.W;
    int ih = (int)(rem % g.H); rem /= g.H;
    int n = (int)rem;
    float acc = 0.f;
    for (int r = 0; r < g.R; ++r) {
      int ph = ih + g.ph - r;
      if (ph < 0 || ph % g.sh) continue;
      int p = ph / g.sh;
      if (p >= g.P) continue;
      for (int s = 0; s < g.S; ++s) {
        int pw = iw + g.pw - s;
        if (pw < 0 || pw % g.sw) continue;
        int q = pw / g.sw;
        if (q >= g.Q) continue;
        int64_t m = (int64_t)(n * g.P + p) * g.Q + q;
        acc += (float)dcol[m * RSC + (int64_t)(r * g.S + s) * g.C + c];
      }
    }
    dx[i] = (__bf16)acc;
  }
}

// C%8==0 fast path: one thread per (n, ih, iw, 8-channel group) — 16B
// vector loads per filter tap and a single 16B store (col2im is pure
// HBM-bound gather; vectorization is the whole game).
__global__ void Col2ImKernelV8(const __bf16* __restrict__ dcol,
                               __bf16* __restrict__ dx, ConvGeom g,
                               int64_t total8) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t RSC = (int64_t)g.R * g.S * g.C;
  int cg = g.C / 8;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total8;
       i += stride) {
    int64_t rem = i;
    int c8 = (int)(rem % cg); rem /= cg;
    int iw = (int)(rem % g.W); rem /= g.W;
    int ih = (int)(rem % g.H); rem /= g.H;
    int n = (int)rem;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int r = 0; r < g.R; ++r) {
      int ph = ih + g.ph - r;
      if (ph < 0 || ph % g.sh) continue;
      int p = ph / g.sh;
      if (p >= g.P) continue;
      for (int s = 0; s < g.S; ++s) {
        int pw = iw + g.pw - s;
        if (pw < 0 || pw % g.sw) continue;
        int q = pw / g.sw;
        if (q >= g.Q) continue;
        int64_t m = (int64_t)(n * g.P + p) * g.Q + q;
        __bf16 v[8];
        *(ulong2*)v = *(const ulong2*)(
            dcol + m * RSC + (int64_t)(r * g.S + s) * g.C + c8 * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += (float)v[e];
      }
    }
    __bf16 out[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = (__bf16)acc[e];
    *(ulong2*)(dx + i * 8) = *(ulong2*)out;
  }
}

}  // namespace

extern "C" hipError_t stf_im2col_bf16(const void* x, void* col, int N, int H,
                                      int W, int C, int R, int S, int sh,
                                      int sw, int ph, int pw, int P, int Q,
                                      int64_t ld, hipStream_t stream) {
  ConvGeom g{N, H, W, C, R, S, 0, sh, sw, ph, pw, P, Q};
  int vec = (C % 8 == 0) ? 8 : 1;
  int64_t total = (int64_t)N * P * Q * R * S * (C / vec);
  hipLaunchKernelGGL(Im2ColKernel, ElemwiseGrid(total, 256, 1), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)col, g, total, vec,
                     ld);
  return hipGetLastError();
}

extern "C" hipError_t stf_col2im_bf16(const void* dcol, void* dx, int N, int H,
                                      int W, int C, int R, int S, int sh,
                                      int sw, int ph, int pw, int P, int Q,
                                      hipStream_t stream) {
  ConvGeom g{N, H, W, C, R, S, 0, sh, sw, ph, pw, P, Q};
  int64_t total = (int64_t)N * H * W * C;
  if (C % 8 == 0) {
    int64_t total8 = total / 8;
    hipLaunchKernelGGL(Col2ImKernelV8, ElemwiseGrid(total8, 256, 1),
                       dim3(256), 0, stream, (const __bf16*)dcol,
                       (__bf16*)dx, g, total8);
  } else {
    hipLaunchKernelGGL(Col2ImKernel, ElemwiseGrid(total, 256, 1), dim3(256),
                       0, stream, (const __bf16*)dcol, (__bf16*)dx, g, total);
  }
  return hipGetLastError();
}
