// Data-movement & reduction kernels: 2D transpose (LDS-tiled), generic
// permute, strided broadcast (Tile), reductions (inner-dim row reduce,
// full reduce, generic), concat/split along the last dims, optimizer applies,
// Philox RNG fills. (Replaces reference transpose_functor_gpu,
// reduction_ops_gpu, tile_ops_gpu, training_ops_gpu, random_op_gpu —
// wave64-native designs.)
#include "hip_common.h"

#include "../philox.h"

namespace {

// ---- 2D transpose: out[j,i] = in[i,j]; LDS 64x64 tile with +1-pad ----
template <typename T>
__global__ void Transpose2DKernel(const T* __restrict__ in, T* __restrict__ out,
                                  int64_t rows, int64_t cols) {
  __shared__ T tile[64][65];
  int64_t tiles_c = (cols + 63) / 64;
  int64_t bid = blockIdx.x;
  int64_t br = bid / tiles_c, bc = bid % tiles_c;
  int64_t r0 = br * 64, c0 = bc * 64;
  int tx = threadIdx.x & 63;   // col within tile
  int ty = threadIdx.x >> 6;   // 4 rows per pass
  for (int p = 0; p < 16; ++p) {
    int64_t r = r0 + ty + p * 4;
    int64_t c = c0 + tx;
    if (r < rows && c < cols) tile[ty + p * 4][tx] = in[r * cols + c];
  }
  __syncthreads();
  for (int p = 0; p < 16; ++p) {
    int64_t r = c0 + ty + p * 4;  // output row = input col
    int64_t c = r0 + tx;          // output col = input row
    if (r < cols && c < rows) out[r * rows + c] = tile[tx][ty + p * 4];
  }
}

// ---- generic permute (rank <= 6) ----
struct PermArgs {
  int rank;
  int64_t out_dims[6];
  int64_t src_strides[6];  // stride in src for each out dim
};

template <typename T>
__global__ void PermuteKernel(const T* __restrict__ in, T* __restrict__ out,
                              int64_t n, PermArgs args) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    int64_t rem = i, src = 0;
    for (int d = args.rank - 1; d >= 0; --d) {
      int64_t c = rem % args.out_dims[d];
      rem /= args.out_dims[d];
      src += c * args.src_strides[d];
    }
    out[i] = in[src];
  }
}

// ---- strided broadcast: out[i] = in[map(i)] (Tile / grad broadcast) ----
template <typename T>
__global__ void BcastCopyKernel(const T* __restrict__ in, T* __restrict__ out,
                                int64_t n, PermArgs args) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    int64_t rem = i, src = 0;
    for (int d = args.rank - 1; d >= 0; --d) {
      int64_t c = rem % args.out_dims[d];
      rem /= args.out_dims[d];
      src += c * args.src_strides[d];  // stride 0 on broadcast dims
    }
    out[i] = in[src];
  }
}


// 0.5 * sum(x^2) fused in one pass (L2Loss hot path: clip_by_global_norm
// reduces every gradient tensor; the square-temp + reduce pair costs 3
// memory passes, this costs 1).
template <typename T>
__global__ void SquareSumKernel(const T* __restrict__ x,
                                float* __restrict__ out, int64_t n) {
  __shared__ float lds4[4];
  float acc = 0.f;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = (float)x[i];
    acc += v * v;
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    atomicAdd(out, 0.5f * (lds4[0] + lds4[1] + lds4[2] + lds4[3]));
}


// Blocked channel pad/unpad: view src as [nblocks, in_block] and dst as
// [nblocks, out_block]; elements past in_block zero-fill (pad mode when
// out_block > in_block, truncate when smaller). Lets C%8!=0 convolutions
// (the ResNet/Inception stems, C=3) take the 8-channel implicit-GEMM path.
template <typename T>
__global__ void BlockPadKernel(const T* __restrict__ src, T* __restrict__ dst,
                               int64_t nblocks, int64_t in_block,
                               int64_t out_block) {
  int64_t total = nblocks * out_block;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int64_t b = i / out_block;
    int64_t off = i - b * out_block;
    dst[i] = off < in_block ? src[b * in_block + off] : (T)0;
  }
}

// ---- reductions ----
// full reduce to scalar: two-stage (block partials via atomics on f32)
template <typename T, int RED>  // 0 sum, 1 max, 2 min
__global__ void FullReduceKernel(const T* __restrict__ x,
                                 float* __restrict__ out, int64_t n) {
  __shared__ float lds4[4];
  float acc = RED == 0 ? 0.f : (RED == 1 ? -3.4e38f : 3.4e38f);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = (float)x[i];
    if (RED == 0) acc += v;
    else if (RED == 1) acc = fmaxf(acc, v);
    else acc = fminf(acc, v);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float o = __shfl_down(acc, off, 64);
    if (RED == 0) acc += o;
    else if (RED == 1) acc = fmaxf(acc, o);
    else acc = fminf(acc, o);
  }
  int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float v = lds4[0];
    for (int w = 1; w < 4; ++w) {
      if (RED == 0) v += lds4[w];
      else if (RED == 1) v = fmaxf(v, lds4[w]);
      else v = fminf(v, lds4[w]);
    }
    if (RED == 0) atomicAdd(out, v);
    else if (RED == 1) {
      // atomic max on float via CAS loop
      float old = *out;
      while (v > old) {
        float assumed = old;
        old = __uint_as_float(atomicCAS((unsigned*)out,
                                        __float_as_uint(assumed),
                                        __float_as_uint(v)));
        if (old == assumed) break;
      }
    } else {
      float old = *out;
      while (v < old) {
        float assumed = old;
        old = __uint_as_float(atomicCAS((unsigned*)out,
                                        __float_as_uint(assumed),
                                        __float_as_uint(v)));
        if (old == assumed) break;
      }
    }
  }
}

// row reduce: [rows, inner] -> [rows]; one block per row chunk
template <typename T, int RED>
__global__ void RowReduceKernel(const T* __restrict__ x, float* __restrict__ y,
                                int64_t rows, int64_t inner) {
  __shared__ float lds4[4];
  int64_t r = blockIdx.x;
  if (r >= rows) return;
  const T* row = x + r * inner;
  float acc = RED == 0 ? 0.f : (RED == 1 ? -3.4e38f : 3.4e38f);
  for (int64_t i = threadIdx.x; i < inner; i += blockDim.x) {
    float v = (float)row[i];
    if (RED == 0) acc += v;
    else if (RED == 1) acc = fmaxf(acc, v);
    else acc = fminf(acc, v);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float o = __shfl_down(acc, off, 64);
    if (RED == 0) acc += o;
    else if (RED == 1) acc = fmaxf(acc, o);
    else acc = fminf(acc, o);
  }
  int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float v = lds4[0];
    for (int w = 1; w < 4; ++w) {
      if (RED == 0) v += lds4[w];
      else if (RED == 1) v = fmaxf(v, lds4[w]);
      else v = fminf(v, lds4[w]);
    }
    y[r] = v;
  }
}

// outer reduce: [outer, inner] -> [inner] (column sums; e.g. bias-style)
template <typename T, int RED>
__global__ void ColReduceKernel(const T* __restrict__ x, float* __restrict__ y,
                                int64_t outer, int64_t inner) {
  // grid-stride over inner; each thread owns one column strip
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t c = blockIdx.x * blockDim.x + threadIdx.x; c < inner;
       c += stride) {
    float acc = RED == 0 ? 0.f : (RED == 1 ? -3.4e38f : 3.4e38f);
    for (int64_t r = 0; r < outer; ++r) {
      float v = (float)x[r * inner + c];
      if (RED == 0) acc += v;
      else if (RED == 1) acc = fmaxf(acc, v);
      else acc = fminf(acc, v);
    }
    y[c] = acc;
  }
}

// ---- optimizer applies (f32 params; grad may be f32 or bf16) ----
template <typename G>
__global__ void ApplySgdKernel(float* __restrict__ var,
                               const float* __restrict__ lr,
                               const G* __restrict__ grad, int64_t n) {
  float l = lr[0];
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    var[i] -= l * (float)grad[i];
}

template <typename G>
__global__ void ApplyMomentumKernel(float* __restrict__ var,
                                    float* __restrict__ accum,
                                    const float* __restrict__ lr,
                                    const G* __restrict__ grad,
                                    const float* __restrict__ momentum,
                                    int nesterov, int64_t n) {
  float l = lr[0], mom = momentum[0];
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float g = (float)grad[i];
    float a = accum[i] * mom + g;
    accum[i] = a;
    var[i] -= nesterov ? l * (g + mom * a) : l * a;
  }
}

template <typename G>
__global__ void ApplyAdamKernel(float* __restrict__ var, float* __restrict__ m,
                                float* __restrict__ v,
                                const float* __restrict__ b1p_p,
                                const float* __restrict__ b2p_p,
                                const float* __restrict__ lr_p,
                                const float* __restrict__ b1_p,
                                const float* __restrict__ b2_p,
                                const float* __restrict__ eps_p,
                                const G* __restrict__ grad, int64_t n) {
  float b1p = b1p_p[0], b2p = b2p_p[0], lr = lr_p[0];
  float b1 = b1_p[0], b2 = b2_p[0], eps = eps_p[0];
  float alpha = lr * sqrtf(1.f - b2p) / (1.f - b1p);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float g = (float)grad[i];
    float mi = m[i] + (g - m[i]) * (1.f - b1);
    float vi = v[i] + (g * g - v[i]) * (1.f - b2);
    m[i] = mi;
    v[i] = vi;
    var[i] -= alpha * mi / (sqrtf(vi) + eps);
  }
}

// ---- Philox RNG fills ----
// The philox offset lives in DEVICE memory so a hipGraph replay of the step
// still draws fresh numbers (the advance kernel below bumps it in-graph).
__global__ void RandomUniformKernel(uint64_t seed,
                                    const unsigned long long* __restrict__ ctr,
                                    float* __restrict__ out, int64_t n,
                                    int as_bf16) {
  uint64_t offset = ctr[0];
  int64_t tid = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i * 4 < n; i += stride) {
    stf::random::Philox4x32 rng(seed, offset + i);
    uint32_t r[4];
    rng.Next(r);
    int64_t base = i * 4;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      if (base + e < n) {
        float v = stf::random::Uint32ToFloat01(r[e]);
        if (as_bf16) ((__bf16*)out)[base + e] = (__bf16)v;
        else out[base + e] = v;
      }
    }
  }
}

__global__ void RandomNormalKernel(uint64_t seed,
                                   const unsigned long long* __restrict__ ctr,
                                   float* __restrict__ out, int64_t n,
                                   int as_bf16, int truncated) {
  uint64_t offset = ctr[0];
  int64_t tid = blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i * 4 < n; i += stride) {
    stf::random::Philox4x32 rng(seed, offset + i);
    uint32_t r[4];
    float z[4];
    rng.Next(r);
    stf::random::BoxMuller(r[0], r[1], &z[0], &z[1]);
    stf::random::BoxMuller(r[2], r[3], &z[2], &z[3]);
    if (truncated) {
      // redraw per element until |z| < 2 (bounded attempts)
      for (int e = 0; e < 4; ++e) {
        int tries = 0;
        while (fabsf(z[e]) >= 2.f && tries < 16) {
          rng.Next(r);
          float z2;
          stf::random::BoxMuller(r[0], r[1], &z[e], &z2);
          ++tries;
        }
        if (fabsf(z[e]) >= 2.f) z[e] = 0.f;
      }
    }
    int64_t base = i * 4;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      if (base + e < n) {
        if (as_bf16) ((__bf16*)out)[base + e] = (__bf16)z[e];
        else out[base + e] = z[e];
      }
    }
  }
}

// strided block copy for concat/split: moves [outer, rows, inner] between a
// contiguous side and a strided side (dst_stride rows, dst_off row offset).
template <typename T, bool SCATTER>  // SCATTER: contiguous src -> strided dst
__global__ void StridedCopyKernel(const T* __restrict__ src,
                                  T* __restrict__ dst, int64_t outer,
                                  int64_t rows, int64_t inner,
                                  int64_t big_stride, int64_t row_off,
                                  int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += stride) {
    int64_t i = idx % inner;
    int64_t rem = idx / inner;
    int64_t a = rem % rows;
    int64_t o = rem / rows;
    int64_t small = (o * rows + a) * inner + i;
    int64_t big = (o * big_stride + row_off + a) * inner + i;
    if (SCATTER) dst[big] = src[small];
    else dst[small] = src[big];
  }
}

// embedding gather: out[i, :] = params[indices[i], :]
template <typename T, typename I>
__global__ void GatherRowsKernel(const T* __restrict__ params,
                                 const I* __restrict__ indices,
                                 T* __restrict__ out, int64_t nidx,
                                 int64_t row, int64_t nrows) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t n = nidx * row;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += stride) {
    int64_t i = idx / row, r = idx % row;
    int64_t src = (int64_t)indices[i];
    out[idx] = (src >= 0 && src < nrows) ? params[src * row + r] : (T)0.f;
  }
}

// embedding grad scatter-add into f32 accumulator (zeroed)
template <typename T, typename I>
__global__ void SegmentSumKernel(const T* __restrict__ data,
                                 const I* __restrict__ ids,
                                 float* __restrict__ out, int64_t nidx,
                                 int64_t row, int64_t nseg) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t n = nidx * row;
  for (int64_t idx = blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += stride) {
    int64_t i = idx / row, r = idx % row;
    int64_t seg = (int64_t)ids[i];
    if (seg >= 0 && seg < nseg)
      atomicAdd(&out[seg * row + r], (float)data[idx]);
  }
}

__global__ void AdvanceCtrKernel(unsigned long long* ctr,
                                 unsigned long long delta) {
  if (threadIdx.x == 0 && blockIdx.x == 0) ctr[0] += delta;
}

}  // namespace

extern "C" {

hipError_t stf_transpose2d(int elem_size, const void* in, void* out,
                           int64_t rows, int64_t cols, hipStream_t stream) {
  int64_t blocks = ((rows + 63) / 64) * ((cols + 63) / 64);
  if (elem_size == 2)
    hipLaunchKernelGGL((Transpose2DKernel<uint16_t>), dim3((uint32_t)blocks),
                       dim3(256), 0, stream, (const uint16_t*)in,
                       (uint16_t*)out, rows, cols);
  else if (elem_size == 4)
    hipLaunchKernelGGL((Transpose2DKernel<uint32_t>), dim3((uint32_t)blocks),
                       dim3(256), 0, stream, (const uint32_t*)in,
                       (uint32_t*)out, rows, cols);
  else
    hipLaunchKernelGGL((Transpose2DKernel<uint64_t>), dim3((uint32_t)blocks),
                       dim3(256), 0, stream, (const uint64_t*)in,
                       (uint64_t*)out, rows, cols);
  return hipGetLastError();
}

hipError_t stf_permute(int elem_size, const void* in, void* out, int64_t n,
                       int rank, const int64_t* out_dims,
                       const int64_t* src_strides, hipStream_t stream) {
  PermArgs args;
  args.rank = rank;
  for (int i = 0; i < rank && i < 6; ++i) {
    args.out_dims[i] = out_dims[i];
    args.src_strides[i] = src_strides[i];
  }
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (elem_size == 2)
    hipLaunchKernelGGL((PermuteKernel<uint16_t>), grid, dim3(256), 0, stream,
                       (const uint16_t*)in, (uint16_t*)out, n, args);
  else if (elem_size == 4)
    hipLaunchKernelGGL((PermuteKernel<uint32_t>), grid, dim3(256), 0, stream,
                       (const uint32_t*)in, (uint32_t*)out, n, args);
  else
    hipLaunchKernelGGL((PermuteKernel<uint64_t>), grid, dim3(256), 0, stream,
                       (const uint64_t*)in, (uint64_t*)out, n, args);
  return hipGetLastError();
}

hipError_t stf_block_pad(int dtype, const void* src, void* dst,
                         int64_t nblocks, int64_t in_block,
                         int64_t out_block, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(nblocks * out_block, 256, 4);
  if (dtype == 0)
    hipLaunchKernelGGL(BlockPadKernel<float>, grid, dim3(256), 0, stream,
                       (const float*)src, (float*)dst, nblocks, in_block,
                       out_block);
  else
    hipLaunchKernelGGL(BlockPadKernel<__bf16>, grid, dim3(256), 0, stream,
                       (const __bf16*)src, (__bf16*)dst, nblocks, in_block,
                       out_block);
  return hipGetLastError();
}

hipError_t stf_l2loss(int dtype, const void* x, float* out_f32, int64_t n,
                      hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 8);
  if (dtype == 0)
    hipLaunchKernelGGL((SquareSumKernel<float>), grid, dim3(256), 0, stream,
                       (const float*)x, out_f32, n);
  else
    hipLaunchKernelGGL((SquareSumKernel<__bf16>), grid, dim3(256), 0, stream,
                       (const __bf16*)x, out_f32, n);
  return hipGetLastError();
}

hipError_t stf_full_reduce(int dtype, int red, const void* x, float* out_f32,
                           int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 8);
#define FR(T, R)                                                           \
  hipLaunchKernelGGL((FullReduceKernel<T, R>), grid, dim3(256), 0, stream, \
                     (const T*)x, out_f32, n)
  if (dtype == 0) {
    if (red == 0) FR(float, 0); else if (red == 1) FR(float, 1); else FR(float, 2);
  } else {
    if (red == 0) FR(__bf16, 0); else if (red == 1) FR(__bf16, 1); else FR(__bf16, 2);
  }
#undef FR
  return hipGetLastError();
}

hipError_t stf_row_reduce(int dtype, int red, const void* x, float* y,
                          int64_t rows, int64_t inner, hipStream_t stream) {
#define RR(T, R)                                                            \
  hipLaunchKernelGGL((RowReduceKernel<T, R>), dim3((uint32_t)rows),          \
                     dim3(256), 0, stream, (const T*)x, y, rows, inner)
  if (dtype == 0) {
    if (red == 0) RR(float, 0); else if (red == 1) RR(float, 1); else RR(float, 2);
  } else {
    if (red == 0) RR(__bf16, 0); else if (red == 1) RR(__bf16, 1); else RR(__bf16, 2);
  }
#undef RR
  return hipGetLastError();
}

hipError_t stf_col_reduce(int dtype, int red, const void* x, float* y,
                          int64_t outer, int64_t inner, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(inner, 256, 1);
#define CR(T, R)                                                            \
  hipLaunchKernelGGL((ColReduceKernel<T, R>), grid, dim3(256), 0, stream,   \
                     (const T*)x, y, outer, inner)
  if (dtype == 0) {
    if (red == 0) CR(float, 0); else if (red == 1) CR(float, 1); else CR(float, 2);
  } else {
    if (red == 0) CR(__bf16, 0); else if (red == 1) CR(__bf16, 1); else CR(__bf16, 2);
  }
#undef CR
  return hipGetLastError();
}

hipError_t stf_bcast_copy(int elem_size, const void* in, void* out, int64_t n,
                          int rank, const int64_t* out_dims,
                          const int64_t* src_strides, hipStream_t stream) {
  return stf_permute(elem_size, in, out, n, rank, out_dims, src_strides,
                     stream);
}

hipError_t stf_apply_sgd(int grad_bf16, void* var, const void* lr,
                         const void* grad, int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (grad_bf16)
    hipLaunchKernelGGL((ApplySgdKernel<__bf16>), grid, dim3(256), 0, stream,
                       (float*)var, (const float*)lr, (const __bf16*)grad, n);
  else
    hipLaunchKernelGGL((ApplySgdKernel<float>), grid, dim3(256), 0, stream,
                       (float*)var, (const float*)lr, (const float*)grad, n);
  return hipGetLastError();
}

hipError_t stf_apply_momentum(int grad_bf16, void* var, void* accum,
                              const void* lr, const void* grad,
                              const void* momentum, int nesterov, int64_t n,
                              hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (grad_bf16)
    hipLaunchKernelGGL((ApplyMomentumKernel<__bf16>), grid, dim3(256), 0,
                       stream, (float*)var, (float*)accum, (const float*)lr,
                       (const __bf16*)grad, (const float*)momentum, nesterov,
                       n);
  else
    hipLaunchKernelGGL((ApplyMomentumKernel<float>), grid, dim3(256), 0,
                       stream, (float*)var, (float*)accum, (const float*)lr,
                       (const float*)grad, (const float*)momentum, nesterov,
                       n);
  return hipGetLastError();
}

hipError_t stf_apply_adam(int grad_bf16, void* var, void* m, void* v,
                          const void* b1p, const void* b2p, const void* lr,
                          const void* b1, const void* b2, const void* eps,
                          const void* grad, int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (grad_bf16)
    hipLaunchKernelGGL((ApplyAdamKernel<__bf16>), grid, dim3(256), 0, stream,
                       (float*)var, (float*)m, (float*)v, (const float*)b1p,
                       (const float*)b2p, (const float*)lr, (const float*)b1,
                       (const float*)b2, (const float*)eps,
                       (const __bf16*)grad, n);
  else
    hipLaunchKernelGGL((ApplyAdamKernel<float>), grid, dim3(256), 0, stream,
                       (float*)var, (float*)m, (float*)v, (const float*)b1p,
                       (const float*)b2p, (const float*)lr, (const float*)b1,
                       (const float*)b2, (const float*)eps,
                       (const float*)grad, n);
  return hipGetLastError();
}

hipError_t stf_strided_copy(int elem_size, int scatter, const void* src,
                            void* dst, int64_t outer, int64_t rows,
                            int64_t inner, int64_t big_stride, int64_t row_off,
                            hipStream_t stream) {
  int64_t n = outer * rows * inner;
  dim3 grid = ElemwiseGrid(n, 256, 4);
#define SC(T)                                                                  do {                                                                           if (scatter)                                                                   hipLaunchKernelGGL((StridedCopyKernel<T, true>), grid, dim3(256), 0,                            stream, (const T*)src, (T*)dst, outer, rows, inner,                          big_stride, row_off, n);                                else                                                                           hipLaunchKernelGGL((StridedCopyKernel<T, false>), grid, dim3(256), 0,                           stream, (const T*)src, (T*)dst, outer, rows, inner,                          big_stride, row_off, n);                              } while (0)
  if (elem_size == 2) SC(uint16_t);
  else if (elem_size == 4) SC(uint32_t);
  else SC(uint64_t);
#undef SC
  return hipGetLastError();
}

hipError_t stf_gather_rows(int dtype, int idx_i32, const void* params,
                           const void* indices, void* out, int64_t nidx,
                           int64_t row, int64_t nrows, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(nidx * row, 256, 4);
#define GR(T, I)                                                               hipLaunchKernelGGL((GatherRowsKernel<T, I>), grid, dim3(256), 0, stream,                        (const T*)params, (const I*)indices, (T*)out, nidx,                          row, nrows)
  if (dtype == 0) {
    if (idx_i32) GR(float, int32_t); else GR(float, int64_t);
  } else {
    if (idx_i32) GR(__bf16, int32_t); else GR(__bf16, int64_t);
  }
#undef GR
  return hipGetLastError();
}

hipError_t stf_segment_sum(int dtype, int idx_i32, const void* data,
                           const void* ids, float* out_f32, int64_t nidx,
                           int64_t row, int64_t nseg, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(nidx * row, 256, 4);
#define SS(T, I)                                                               hipLaunchKernelGGL((SegmentSumKernel<T, I>), grid, dim3(256), 0, stream,                        (const T*)data, (const I*)ids, out_f32, nidx, row, nseg)
  if (dtype == 0) {
    if (idx_i32) SS(float, int32_t); else SS(float, int64_t);
  } else {
    if (idx_i32) SS(__bf16, int32_t); else SS(__bf16, int64_t);
  }
#undef SS
  return hipGetLastError();
}

hipError_t stf_random_uniform(uint64_t seed, void* ctr_dev, void* out,
                              int64_t n, int as_bf16, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  hipLaunchKernelGGL(RandomUniformKernel, grid, dim3(256), 0, stream, seed,
                     (const unsigned long long*)ctr_dev, (float*)out, n,
                     as_bf16);
  hipLaunchKernelGGL(AdvanceCtrKernel, dim3(1), dim3(64), 0, stream,
                     (unsigned long long*)ctr_dev,
                     (unsigned long long)((n + 3) / 4 + 1));
  return hipGetLastError();
}

hipError_t stf_random_normal(uint64_t seed, void* ctr_dev, void* out,
                             int64_t n, int as_bf16, int truncated,
                             hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  hipLaunchKernelGGL(RandomNormalKernel, grid, dim3(256), 0, stream, seed,
                     (const unsigned long long*)ctr_dev, (float*)out, n,
                     as_bf16, truncated);
  hipLaunchKernelGGL(AdvanceCtrKernel, dim3(1), dim3(64), 0, stream,
                     (unsigned long long*)ctr_dev,
                     (unsigned long long)((n + 3) / 4 + 16));
  return hipGetLastError();
}

}  // extern "C"
