// f32 NT GEMM: C[M,N] = A[M,K] * B[N,K]^T — LDS-tiled VALU kernel.
// f32 matmuls are off the bf16 hot path (guide §3: f32 peak is 157 TF either
// way); this straightforward 64x64-tile kernel keeps f32 graphs on-device.
#include "hip_common.h"

namespace {

// 256 threads = 16x16; each thread computes 4x4 outputs; tile 64x64, BK=16.
__launch_bounds__(256) __global__ void GemmF32NT(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int64_t M, int64_t N, int64_t K) {
  __shared__ float at[16][64 + 1];
  __shared__ float bt[16][64 + 1];
  int64_t nbn = (N + 63) / 64;
  int64_t bm = blockIdx.x / nbn, bn = blockIdx.x % nbn;
  int64_t m0 = bm * 64, n0 = bn * 64;
  int tx = threadIdx.x % 16, ty = threadIdx.x / 16;

  float acc[4][4] = {};
  for (int64_t k0 = 0; k0 < K; k0 += 16) {
    // load A tile: 64 rows x 16 k  (thread (tx,ty) loads A[m0+ty*4+?]...)
    for (int i = threadIdx.x; i < 64 * 16; i += 256) {
      int r = i / 16, kk = i % 16;
      int64_t row = m0 + r, k = k0 + kk;
      at[kk][r] = (row < M && k < K) ? A[row * K + k] : 0.f;
    }
    for (int i = threadIdx.x; i < 64 * 16; i += 256) {
      int r = i / 16, kk = i % 16;
      int64_t row = n0 + r, k = k0 + kk;
      bt[kk][r] = (row < N && k < K) ? B[row * K + k] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < 16; ++kk) {
      float av[4], bv[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) av[i] = at[kk][ty * 4 + i];
#pragma unroll
      for (int j = 0; j < 4; ++j) bv[j] = bt[kk][tx * 4 + j];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] += av[i] * bv[j];
    }
    __syncthreads();
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int64_t row = m0 + ty * 4 + i;
    if (row >= M) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t col = n0 + tx * 4 + j;
      if (col < N) C[row * N + col] = acc[i][j];
    }
  }
}

}  // namespace

extern "C" hipError_t stf_gemm_f32_nt(const void* A, const void* B, void* C,
                                      int64_t M, int64_t N, int64_t K,
                                      hipStream_t stream) {
  int64_t blocks = ((M + 63) / 64) * ((N + 63) / 64);
  hipLaunchKernelGGL(GemmF32NT, dim3((uint32_t)blocks), dim3(256), 0, stream,
                     (const float*)A, (const float*)B, (float*)C, M, N, K);
  return hipGetLastError();
}
