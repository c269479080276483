// Local response normalization for gfx950 (NHWC, window over channels).
// Reference used cuDNN (lrn_op.cc:211 ThenNormalize); this is a direct HIP
// kernel: one thread per output element, channel window re-read from
// registers-resident row segments (C is small for LRN-era nets: 64-384,
// window 2r+1 <= 11 — the row stays in L1/L2 across the window reads).
#include "hip_common.h"

namespace {

__global__ void LrnFwdKernel(const float* __restrict__ x,
                             float* __restrict__ y, int64_t rows, int c,
                             int radius, float bias, float alpha,
                             float beta) {
  int64_t n = rows * c;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t row = i / c;
    int k = (int)(i - row * c);
    const float* xr = x + row * c;
    int lo = k - radius < 0 ? 0 : k - radius;
    int hi = k + radius + 1 > c ? c : k + radius + 1;
    float s = bias;
    for (int j = lo; j < hi; ++j) s += alpha * xr[j] * xr[j];
    y[i] = xr[k] * __powf(s, -beta);
  }
}

// dx_k = dy_k * s_k^-b - 2*a*b * x_k * sum_{j: |j-k|<=r} dy_j * y_j / s_j
__global__ void LrnGradKernel(const float* __restrict__ x,
                              const float* __restrict__ y,
                              const float* __restrict__ dy,
                              float* __restrict__ dx, int64_t rows, int c,
                              int radius, float bias, float alpha,
                              float beta) {
  int64_t n = rows * c;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int64_t row = i / c;
    int k = (int)(i - row * c);
    const float* xr = x + row * c;
    const float* yr = y + row * c;
    const float* dyr = dy + row * c;
    // s_k for own term
    int lo = k - radius < 0 ? 0 : k - radius;
    int hi = k + radius + 1 > c ? c : k + radius + 1;
    float sk = bias;
    for (int j = lo; j < hi; ++j) sk += alpha * xr[j] * xr[j];
    float acc = dyr[k] * __powf(sk, -beta);
    // cross terms: j ranges over outputs whose window contains k
    float cross = 0.f;
    for (int j = lo; j < hi; ++j) {
      int jlo = j - radius < 0 ? 0 : j - radius;
      int jhi = j + radius + 1 > c ? c : j + radius + 1;
      float sj = bias;
      for (int t = jlo; t < jhi; ++t) sj += alpha * xr[t] * xr[t];
      cross += dyr[j] * yr[j] / sj;
    }
    dx[i] = acc - 2.f * alpha * beta * xr[k] * cross;
  }
}

}  // namespace

extern "C" {

hipError_t stf_lrn_fwd(const void* x, void* y, int64_t rows, int c,
                       int radius, float bias, float alpha, float beta,
                       hipStream_t stream) {
  hipLaunchKernelGGL(LrnFwdKernel, ElemwiseGrid(rows * c, 256, 2), dim3(256),
                     0, stream, (const float*)x, (float*)y, rows, c, radius,
                     bias, alpha, beta);
  return hipGetLastError();
}

hipError_t stf_lrn_grad(const void* x, const void* y, const void* dy,
                        void* dx, int64_t rows, int c, int radius, float bias,
                        float alpha, float beta, hipStream_t stream) {
  hipLaunchKernelGGL(LrnGradKernel, ElemwiseGrid(rows * c, 256, 2), dim3(256),
                     0, stream, (const float*)x, (const float*)y,
                     (const float*)dy, (float*)dx, rows, c, radius, bias,
                     alpha, beta);
  return hipGetLastError();
}

}  // extern "C"
