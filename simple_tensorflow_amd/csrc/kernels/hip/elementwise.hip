// Element-wise kernels (gfx950): unary/binary over f32 and bf16, vectorized
// 8-wide bf16 / 4-wide f32 grid-stride loops (guide App. B: always vectorize
// bf16 as short4/short8; scalar bf16 is 2-2.5x slower). Replaces the
// reference's ~70 Eigen-GPU cwise kernels (cwise_op_gpu_*.cu.cc) with one
// templated family.
#include "hip_common.h"

namespace {

enum class UOp {
  NEG, ABS, SIGN, SQUARE, SQRT, RSQRT, EXP, LOG, LOG1P, TANH, SIGMOID,
  RELU, RELU6, SOFTPLUS, RECIP, FLOOR, CEIL, SIN, COS
};

__device__ __forceinline__ float UEval(UOp op, float a) {
  switch (op) {
    case UOp::NEG: return -a;
    case UOp::ABS: return fabsf(a);
    case UOp::SIGN: return a > 0.f ? 1.f : (a < 0.f ? -1.f : 0.f);
    case UOp::SQUARE: return a * a;
    case UOp::SQRT: return sqrtf(a);
    case UOp::RSQRT: return rsqrtf(a);
    case UOp::EXP: return __expf(a);
    case UOp::LOG: return __logf(a);
    case UOp::LOG1P: return log1pf(a);
    case UOp::TANH: return tanhf(a);
    case UOp::SIGMOID: return 1.f / (1.f + __expf(-a));
    case UOp::RELU: return a > 0.f ? a : 0.f;
    case UOp::RELU6: return a < 0.f ? 0.f : (a > 6.f ? 6.f : a);
    case UOp::SOFTPLUS: return log1pf(__expf(a));
    case UOp::RECIP: return 1.f / a;
    case UOp::FLOOR: return floorf(a);
    case UOp::CEIL: return ceilf(a);
    case UOp::SIN: return __sinf(a);
    case UOp::COS: return __cosf(a);
  }
  return 0.f;
}

enum class BOp {
  ADD, SUB, MUL, DIV, MAX, MIN, POW, SQDIFF,
  // y/dy grads
  SIGMOID_GRAD, TANH_GRAD, RSQRT_GRAD, SQRT_GRAD,
  // grad/feature grads
  RELU_GRAD, RELU6_GRAD, SOFTPLUS_GRAD
};

__device__ __forceinline__ float BEval(BOp op, float a, float b) {
  switch (op) {
    case BOp::ADD: return a + b;
    case BOp::SUB: return a - b;
    case BOp::MUL: return a * b;
    case BOp::DIV: return a / b;
    case BOp::MAX: return a > b ? a : b;
    case BOp::MIN: return a < b ? a : b;
    case BOp::POW: return __powf(a, b);
    case BOp::SQDIFF: return (a - b) * (a - b);
    case BOp::SIGMOID_GRAD: return b * a * (1.f - a);
    case BOp::TANH_GRAD: return b * (1.f - a * a);
    case BOp::RSQRT_GRAD: return -0.5f * b * a * a * a;
    case BOp::SQRT_GRAD: return b / (2.f * a);
    case BOp::RELU_GRAD: return b > 0.f ? a : 0.f;   // a=grad, b=feature
    case BOp::RELU6_GRAD: return (b > 0.f && b < 6.f) ? a : 0.f;
    case BOp::SOFTPLUS_GRAD: return a / (1.f + __expf(-b));
  }
  return 0.f;
}

// ---- unary ----
template <typename T, int VEC>
__global__ void UnaryKernel(UOp op, const T* __restrict__ x, T* __restrict__ y,
                            int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i * VEC < n;
       i += stride) {
    int64_t base = i * VEC;
    if (base + VEC <= n) {
      T vin[VEC], vout[VEC];
      *(ulong2*)vin = *(const ulong2*)(x + base);
#pragma unroll
      for (int e = 0; e < VEC; ++e) vout[e] = (T)UEval(op, (float)vin[e]);
      *(ulong2*)(y + base) = *(ulong2*)vout;
    } else {
      for (int64_t j = base; j < n; ++j) y[j] = (T)UEval(op, (float)x[j]);
    }
  }
}

// ---- binary same-shape ----
template <typename T, int VEC>
__global__ void BinaryKernel(BOp op, const T* __restrict__ a,
                             const T* __restrict__ b, T* __restrict__ y,
                             int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i * VEC < n;
       i += stride) {
    int64_t base = i * VEC;
    if (base + VEC <= n) {
      T va[VEC], vb[VEC], vy[VEC];
      *(ulong2*)va = *(const ulong2*)(a + base);
      *(ulong2*)vb = *(const ulong2*)(b + base);
#pragma unroll
      for (int e = 0; e < VEC; ++e)
        vy[e] = (T)BEval(op, (float)va[e], (float)vb[e]);
      *(ulong2*)(y + base) = *(ulong2*)vy;
    } else {
      for (int64_t j = base; j < n; ++j)
        y[j] = (T)BEval(op, (float)a[j], (float)b[j]);
    }
  }
}

// ---- binary with scalar on one side ----
template <typename T, bool SCALAR_LEFT>
__global__ void BinaryScalarKernel(BOp op, const T* __restrict__ a,
                                   const T* __restrict__ scalar,
                                   T* __restrict__ y, int64_t n) {
  float s = (float)scalar[0];
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float v = (float)a[i];
    y[i] = (T)(SCALAR_LEFT ? BEval(op, s, v) : BEval(op, v, s));
  }
}

// ---- binary with full broadcast strides (rank <= 6) ----
struct BcastArgs {
  int rank;
  int64_t out_dims[6];
  int64_t sa[6], sb[6];
};

template <typename T>
__global__ void BinaryBcastKernel(BOp op, const T* __restrict__ a,
                                  const T* __restrict__ b, T* __restrict__ y,
                                  int64_t n, BcastArgs args) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    int64_t rem = i, ia = 0, ib = 0;
    for (int d = args.rank - 1; d >= 0; --d) {
      int64_t c = rem % args.out_dims[d];
      rem /= args.out_dims[d];
      ia += c * args.sa[d];
      ib += c * args.sb[d];
    }
    y[i] = (T)BEval(op, (float)a[ia], (float)b[ib]);
  }
}

// ---- cast ----
template <typename S, typename D>
__global__ void CastKernel(const S* __restrict__ x, D* __restrict__ y,
                           int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    y[i] = (D)(float)x[i];
}
template <>
__global__ void CastKernel<float, int32_t>(const float* __restrict__ x,
                                           int32_t* __restrict__ y, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    y[i] = (int32_t)x[i];
}

// ---- AddN ----
template <typename T>
__global__ void AddNKernel(const T* const* __restrict__ ptrs, int num,
                           T* __restrict__ y, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    float acc = 0.f;
    for (int k = 0; k < num; ++k) acc += (float)ptrs[k][i];
    y[i] = (T)acc;
  }
}

// ---- scale by immediate ----
template <typename T>
__global__ void ScaleKernel(const T* __restrict__ x, T* __restrict__ y,
                            float f, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    y[i] = (T)((float)x[i] * f);
}

// ---- fill ----
template <typename T>
__global__ void FillKernel(T* __restrict__ y, T v, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    y[i] = v;
}

}  // namespace

// dtype codes: 0 = f32, 1 = bf16 (carried as uint16)

// Fused f32->bf16 cast that RE-ZEROES the source as it reads: restores the
// split-K scratch arena's all-zero invariant in the same pass that drains
// the result (replaces the per-GEMM hipMemset launch).
__global__ void CastF32Bf16ZeroKernel(float* __restrict__ src,
                                      __bf16* __restrict__ dst, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t n8 = n / 8;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    float v[8];
    *(ulong2*)(v + 0) = *(const ulong2*)(src + i * 8);
    *(ulong2*)(v + 4) = *(const ulong2*)(src + i * 8 + 4);
    __bf16 o[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (__bf16)v[e];
    *(ulong2*)(dst + i * 8) = *(ulong2*)o;
    ulong2 z = {0, 0};
    *(ulong2*)(src + i * 8) = z;
    *(ulong2*)(src + i * 8 + 4) = z;
  }
  int64_t t0 = n8 * 8;
  for (int64_t i = t0 + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n; i += stride) {
    dst[i] = (__bf16)src[i];
    src[i] = 0.f;
  }
}


// _FusedElementwise GPU interpreter (kernels/fused_ew.h program format):
// the whole fused DAG runs register-resident — one HBM read per input
// element, one write, nothing else touches memory.
struct FEArgs {
  const void* in[7];
  uint8_t scalar_mask;
  int n_in;
  int n_prog;
  int64_t prog[24];
};

template <typename T>
__global__ void FusedEwKernel(FEArgs a, T* __restrict__ out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += stride) {
    float vals[32];
    for (int i = 0; i < a.n_in; ++i) {
      const T* p = (const T*)a.in[i];
      vals[i] = (float)p[(a.scalar_mask >> i) & 1 ? 0 : idx];
    }
    for (int k = 0; k < a.n_prog; ++k) {
      int64_t ins = a.prog[k];
      int opx = (int)(ins & 0xff);
      float x = vals[(ins >> 8) & 0xff];
      float y = vals[(ins >> 16) & 0xff];
      float r;
      switch (opx) {
        case 1: r = x > 0.f ? x : 0.f; break;                  // relu
        case 2: r = x < 0.f ? 0.f : (x > 6.f ? 6.f : x); break;
        case 3: r = 1.f / (1.f + __expf(-x)); break;           // sigmoid
        case 4: r = tanhf(x); break;
        case 5: r = __expf(x); break;
        case 6: r = __logf(x); break;
        case 7: r = log1pf(x); break;
        case 8: r = -x; break;
        case 9: r = sqrtf(x); break;
        case 10: r = rsqrtf(x); break;
        case 11: r = x * x; break;
        case 12: r = fabsf(x); break;
        case 13: r = log1pf(__expf(-fabsf(x))) + (x > 0.f ? x : 0.f); break;
        case 14: r = x > 0.f ? 1.f : (x < 0.f ? -1.f : 0.f); break;
        case 15: r = floorf(x); break;
        case 16: r = 1.f / x; break;
        case 64: r = x + y; break;
        case 65: r = x - y; break;
        case 66: r = x * y; break;
        case 67: r = x / y; break;
        case 68: r = fmaxf(x, y); break;
        case 69: r = fminf(x, y); break;
        case 70: r = (x - y) * (x - y); break;
        case 71: r = powf(x, y); break;
        default: r = 0.f; break;
      }
      vals[a.n_in + k] = r;
    }
    out[idx] = (T)vals[a.n_in + a.n_prog - 1];
  }
}

extern "C" hipError_t stf_fused_elementwise(
    int dtype, const void** inputs, const uint8_t* scalar_flags, int n_in,
    const int64_t* prog, int n_prog, void* out, int64_t n,
    hipStream_t stream) {
  if (n_in > 7 || n_prog > 24) return hipErrorInvalidValue;
  FEArgs a;
  a.scalar_mask = 0;
  a.n_in = n_in;
  a.n_prog = n_prog;
  for (int i = 0; i < n_in; ++i) {
    a.in[i] = inputs[i];
    if (scalar_flags[i]) a.scalar_mask |= (1 << i);
  }
  for (int k = 0; k < n_prog; ++k) a.prog[k] = prog[k];
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (dtype == 0)
    hipLaunchKernelGGL(FusedEwKernel<float>, grid, dim3(256), 0, stream, a,
                       (float*)out, n);
  else
    hipLaunchKernelGGL(FusedEwKernel<__bf16>, grid, dim3(256), 0, stream, a,
                       (__bf16*)out, n);
  return hipGetLastError();
}

extern "C" hipError_t stf_cast_f32_bf16_zero(void* src_f32, void* dst_bf16,
                                             int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n / 8 + 1, 256, 1);
  hipLaunchKernelGGL(CastF32Bf16ZeroKernel, grid, dim3(256), 0, stream,
                     (float*)src_f32, (__bf16*)dst_bf16, n);
  return hipGetLastError();
}

extern "C" hipError_t stf_unary(int op, int dtype, const void* x, void* y,
                                int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n);
  if (dtype == 0)
    hipLaunchKernelGGL((UnaryKernel<float, 4>), grid, dim3(256), 0, stream,
                       (UOp)op, (const float*)x, (float*)y, n);
  else
    hipLaunchKernelGGL((UnaryKernel<__bf16, 8>), grid, dim3(256), 0, stream,
                       (UOp)op, (const __bf16*)x, (__bf16*)y, n);
  return hipGetLastError();
}

extern "C" hipError_t stf_binary(int op, int dtype, const void* a,
                                 const void* b, void* y, int64_t n,
                                 hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n);
  if (dtype == 0)
    hipLaunchKernelGGL((BinaryKernel<float, 4>), grid, dim3(256), 0, stream,
                       (BOp)op, (const float*)a, (const float*)b, (float*)y, n);
  else
    hipLaunchKernelGGL((BinaryKernel<__bf16, 8>), grid, dim3(256), 0, stream,
                       (BOp)op, (const __bf16*)a, (const __bf16*)b, (__bf16*)y,
                       n);
  return hipGetLastError();
}

extern "C" hipError_t stf_binary_scalar(int op, int dtype, const void* a,
                                        const void* scalar, void* y, int64_t n,
                                        int scalar_left, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (dtype == 0) {
    if (scalar_left)
      hipLaunchKernelGGL((BinaryScalarKernel<float, true>), grid, dim3(256), 0,
                         stream, (BOp)op, (const float*)a, (const float*)scalar,
                         (float*)y, n);
    else
      hipLaunchKernelGGL((BinaryScalarKernel<float, false>), grid, dim3(256),
                         0, stream, (BOp)op, (const float*)a,
                         (const float*)scalar, (float*)y, n);
  } else {
    if (scalar_left)
      hipLaunchKernelGGL((BinaryScalarKernel<__bf16, true>), grid, dim3(256),
                         0, stream, (BOp)op, (const __bf16*)a,
                         (const __bf16*)scalar, (__bf16*)y, n);
    else
      hipLaunchKernelGGL((BinaryScalarKernel<__bf16, false>), grid, dim3(256),
                         0, stream, (BOp)op, (const __bf16*)a,
                         (const __bf16*)scalar, (__bf16*)y, n);
  }
  return hipGetLastError();
}

extern "C" hipError_t stf_binary_bcast(int op, int dtype, const void* a,
                                       const void* b, void* y, int64_t n,
                                       int rank, const int64_t* out_dims,
                                       const int64_t* sa, const int64_t* sb,
                                       hipStream_t stream) {
  BcastArgs args;
  args.rank = rank;
  for (int i = 0; i < rank && i < 6; ++i) {
    args.out_dims[i] = out_dims[i];
    args.sa[i] = sa[i];
    args.sb[i] = sb[i];
  }
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (dtype == 0)
    hipLaunchKernelGGL((BinaryBcastKernel<float>), grid, dim3(256), 0, stream,
                       (BOp)op, (const float*)a, (const float*)b, (float*)y, n,
                       args);
  else
    hipLaunchKernelGGL((BinaryBcastKernel<__bf16>), grid, dim3(256), 0, stream,
                       (BOp)op, (const __bf16*)a, (const __bf16*)b,
                       (__bf16*)y, n, args);
  return hipGetLastError();
}

// src/dst dtype codes: 0 f32, 1 bf16, 2 f16, 3 i32, 4 i64, 5 bool
extern "C" hipError_t stf_cast(int sdt, int ddt, const void* x, void* y,
                               int64_t n, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
#define CASTCASE(SC, DC, S, D)                                              \
  if (sdt == SC && ddt == DC) {                                             \
    hipLaunchKernelGGL((CastKernel<S, D>), grid, dim3(256), 0, stream,      \
                       (const S*)x, (D*)y, n);                              \
    return hipGetLastError();                                               \
  }
  CASTCASE(0, 1, float, __bf16)
  CASTCASE(1, 0, __bf16, float)
  CASTCASE(0, 3, float, int32_t)
  CASTCASE(3, 0, int32_t, float)
  CASTCASE(4, 0, int64_t, float)
  CASTCASE(0, 4, float, int64_t)
  CASTCASE(3, 4, int32_t, int64_t)
  CASTCASE(4, 3, int64_t, int32_t)
  CASTCASE(1, 1, __bf16, __bf16)
  CASTCASE(0, 0, float, float)
  CASTCASE(5, 0, bool, float)
  CASTCASE(5, 1, bool, __bf16)
  CASTCASE(5, 3, bool, int32_t)
  CASTCASE(5, 4, bool, int64_t)
  CASTCASE(0, 5, float, bool)
  CASTCASE(3, 5, int32_t, bool)
  CASTCASE(4, 5, int64_t, bool)
#undef CASTCASE
  return hipErrorInvalidValue;
}

extern "C" hipError_t stf_addn(int dtype, const void* const* dev_ptr_array,
                               int num, void* y, int64_t n,
                               hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (dtype == 0)
    hipLaunchKernelGGL((AddNKernel<float>), grid, dim3(256), 0, stream,
                       (const float* const*)dev_ptr_array, num, (float*)y, n);
  else
    hipLaunchKernelGGL((AddNKernel<__bf16>), grid, dim3(256), 0, stream,
                       (const __bf16* const*)dev_ptr_array, num, (__bf16*)y, n);
  return hipGetLastError();
}

extern "C" hipError_t stf_scale(int dtype, const void* x, void* y, int64_t n,
                                float factor, hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (dtype == 0)
    hipLaunchKernelGGL((ScaleKernel<float>), grid, dim3(256), 0, stream,
                       (const float*)x, (float*)y, factor, n);
  else
    hipLaunchKernelGGL((ScaleKernel<__bf16>), grid, dim3(256), 0, stream,
                       (const __bf16*)x, (__bf16*)y, factor, n);
  return hipGetLastError();
}

extern "C" hipError_t stf_fill_f32(void* y, float v, int64_t n, int as_bf16,
                                   hipStream_t stream) {
  dim3 grid = ElemwiseGrid(n, 256, 4);
  if (as_bf16)
    hipLaunchKernelGGL((FillKernel<__bf16>), grid, dim3(256), 0, stream,
                       (__bf16*)y, (__bf16)v, n);
  else
    hipLaunchKernelGGL((FillKernel<float>), grid, dim3(256), 0, stream,
                       (float*)y, v, n);
  return hipGetLastError();
}
