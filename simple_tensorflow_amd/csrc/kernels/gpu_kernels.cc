// GPU OpKernels: thin wrappers that resolve shapes/attrs and enqueue the
// hand-written CDNA4 HIP kernels (kernels/hip/*.hip) on the device's compute
// stream. These REPLACE library paths (no rocBLAS/MIOpen): MatMul/Conv2D run
// on the in-tree MFMA GEMM (guide-compliant, gfx950-only).
#include <hip/hip_runtime.h>

#include <atomic>
#include <mutex>

#include "kernels/kernel_util.h"
#include "kernels/resource_mgr.h"

namespace stf {

// ---- extern kernels (kernels/hip/*.hip) ----
extern "C" {
hipError_t stf_gemm_bf16_nt(const void*, const void*, void*, const void*,
                            int64_t, int64_t, int64_t, float, int, int,
                            hipStream_t);
hipError_t stf_gemm_bf16_nt_splitk(const void*, const void*, void*, int64_t,
                                   int64_t, int64_t, int, hipStream_t);
hipError_t stf_gemm_bf16(const void*, const void*, void*, const void*,
                         int64_t, int64_t, int64_t, int64_t, int64_t, float,
                         int, int, int, int, hipStream_t);
hipError_t stf_gemm_bf16_splitk(const void*, const void*, void*, int64_t,
                                int64_t, int64_t, int64_t, int64_t, int, int,
                                int, hipStream_t);
int stf_gemm_bf16_8ph_ok(int64_t, int64_t, int64_t);
hipError_t stf_gemm_bf16_8ph_side(const void*, const void*, void*,
                                  const void*, const void*, int64_t, int64_t,
                                  int64_t, int64_t, int64_t, int, int,
                                  hipStream_t);
hipError_t stf_lrn_fwd(const void*, void*, int64_t, int, int, float, float,
                       float, hipStream_t);
hipError_t stf_conv2d_fwd_8ph(const void*, const void*, void*, const void*,
                              const void*, int, int, int, int, int, int, int,
                              int, int, int, int, int, int64_t, int64_t, int,
                              int, hipStream_t);
hipError_t stf_conv2d_dw_splitk(const void*, const void*, void*, const void*,
                                int, int, int, int, int, int, int, int, int,
                                int, int, int, int64_t, int, hipStream_t);
hipError_t stf_gemm_f32_nt(const void*, const void*, void*, int64_t, int64_t,
                           int64_t, hipStream_t);
hipError_t stf_depthwise_fwd(const void*, const void*, void*, int, int, int,
                             int, int, int, int, int, int, int, int, int,
                             int, hipStream_t);
hipError_t stf_depthwise_bwd_input(const void*, const void*, void*, int, int,
                                   int, int, int, int, int, int, int, int,
                                   int, int, int, hipStream_t);
hipError_t stf_depthwise_bwd_filter(const void*, const void*, float*, int,
                                    int, int, int, int, int, int, int, int,
                                    int, int, int, int, hipStream_t);
hipError_t stf_unary(int, int, const void*, void*, int64_t, hipStream_t);
hipError_t stf_binary(int, int, const void*, const void*, void*, int64_t,
                      hipStream_t);
hipError_t stf_binary_scalar(int, int, const void*, const void*, void*,
                             int64_t, int, hipStream_t);
hipError_t stf_binary_bcast(int, int, const void*, const void*, void*, int64_t,
                            int, const int64_t*, const int64_t*,
                            const int64_t*, hipStream_t);
hipError_t stf_cast(int, int, const void*, void*, int64_t, hipStream_t);
hipError_t stf_addn(int, const void* const*, int, void*, int64_t, hipStream_t);
hipError_t stf_fill_f32(void*, float, int64_t, int, hipStream_t);
hipError_t stf_scale(int, const void*, void*, int64_t, float, hipStream_t);
hipError_t stf_im2col_bf16(const void*, void*, int, int, int, int, int, int,
                           int, int, int, int, int, int, int64_t, hipStream_t);
hipError_t stf_col2im_bf16(const void*, void*, int, int, int, int, int, int,
                           int, int, int, int, int, int, hipStream_t);
hipError_t stf_bias_add(int, const void*, const void*, void*, int64_t, int,
                        hipStream_t);
hipError_t stf_bias_grad(int, const void*, void*, int64_t, int, hipStream_t);
hipError_t stf_softmax(int, int, const void*, void*, int64_t, int,
                       hipStream_t);
hipError_t stf_sparse_xent(int, const void*, const void*, int, void*, void*,
                           int64_t, int, hipStream_t);
hipError_t stf_xent(int, const void*, const void*, void*, void*, int64_t, int,
                    hipStream_t);
hipError_t stf_bn_fwd(int, const void*, const void*, const void*, float*,
                      float*, float*, float*, void*, int64_t, int, float, int,
                      hipStream_t);
hipError_t stf_bn_stats_only(int, const void*, float*, float*, float*,
                             float*, int64_t, int, float, hipStream_t);
hipError_t stf_bn_add_relu(const void*, const void*, const float*,
                           const float*, const void*, const void*, void*,
                           int64_t, int, hipStream_t);
hipError_t stf_bn_bwd(int, const void*, const void*, const void*,
                      const float*, const float*, const void*, float*,
                      float*, void*, int64_t, int, int, hipStream_t);
hipError_t stf_pool_fwd(int, int, const void*, void*, int, int, int, int, int,
                        int, int, int, int, int, int, int, hipStream_t);
hipError_t stf_max_pool_bwd_v8(const void*, const void*, void*, int, int,
                               int, int, int, int, int, int, int, int, int,
                               int, hipStream_t);
hipError_t stf_max_pool_bwd(int, const void*, const void*, float*, int, int,
                            int, int, int, int, int, int, int, int, int, int,
                            hipStream_t);
hipError_t stf_avg_pool_bwd(int, const void*, void*, int, int, int, int, int,
                            int, int, int, int, int, int, int, hipStream_t);
hipError_t stf_transpose2d(int, const void*, void*, int64_t, int64_t,
                           hipStream_t);
hipError_t stf_permute(int, const void*, void*, int64_t, int, const int64_t*,
                       const int64_t*, hipStream_t);
hipError_t stf_full_reduce(int, int, const void*, float*, int64_t,
                           hipStream_t);
hipError_t stf_row_reduce(int, int, const void*, float*, int64_t, int64_t,
                          hipStream_t);
hipError_t stf_col_reduce(int, int, const void*, float*, int64_t, int64_t,
                          hipStream_t);
hipError_t stf_bcast_copy(int, const void*, void*, int64_t, int,
                          const int64_t*, const int64_t*, hipStream_t);
hipError_t stf_apply_sgd(int, void*, const void*, const void*, int64_t,
                         hipStream_t);
hipError_t stf_apply_momentum(int, void*, void*, const void*, const void*,
                              const void*, int, int64_t, hipStream_t);
hipError_t stf_apply_adam(int, void*, void*, void*, const void*, const void*,
                          const void*, const void*, const void*, const void*,
                          const void*, int64_t, hipStream_t);
hipError_t stf_random_uniform(uint64_t, void*, void*, int64_t, int,
                              hipStream_t);
hipError_t stf_strided_copy(int, int, const void*, void*, int64_t, int64_t,
                            int64_t, int64_t, int64_t, hipStream_t);
hipError_t stf_gather_rows(int, int, const void*, const void*, void*, int64_t,
                           int64_t, int64_t, hipStream_t);
hipError_t stf_segment_sum(int, int, const void*, const void*, float*,
                           int64_t, int64_t, int64_t, hipStream_t);
hipError_t stf_random_normal(uint64_t, void*, void*, int64_t, int, int,
                             hipStream_t);
hipError_t stf_lstm_gates(int, const void*, const void*, float, void*, void*,
                          void*, void*, void*, void*, void*, int64_t, int,
                          hipStream_t);
hipError_t stf_lstm_gates_grad(int, const void*, const void*, const void*,
                               const void*, const void*, const void*,
                               const void*, const void*, void*, void*,
                               int64_t, int, hipStream_t);
hipError_t stf_l2loss(int, const void*, float*, int64_t, hipStream_t);
hipError_t stf_cast_f32_bf16_zero(void*, void*, int64_t, hipStream_t);
hipError_t stf_block_pad(int, const void*, void*, int64_t, int64_t, int64_t,
                         hipStream_t);
hipError_t stf_fused_elementwise(int, const void**, const uint8_t*, int,
                                 const int64_t*, int, void*, int64_t,
                                 hipStream_t);
hipError_t stf_conv2d_fwd_nt(const void*, const void*, void*, const void*,
                             int, int, int, int, int, int, int, int, int,
                             int, int, int, int64_t, int64_t, hipStream_t);
}

namespace {

#define GPU_STREAM(ctx) ((hipStream_t)(ctx)->device()->compute_stream())
#define OP_HIP_OK(ctx, expr)                                              \
  {                                                                       \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess) {                                               \
      (ctx)->SetStatus(errors::Internal("HIP kernel failure: ",           \
                                        hipGetErrorString(_e)));          \
      return;                                                             \
    }                                                                     \
  }

inline int DtypeCode(DataType dt) { return dt == DT_FLOAT ? 0 : 1; }

// Choose a split-K factor so the grid covers the 256 CUs (guide: blocks
// ~= 0.5-2x CU count); 1 = no split.
inline int PickSplitK(int64_t M, int64_t N, int64_t K) {
  int64_t tiles = ((M + 127) / 128) * ((N + 127) / 128);
  if (tiles >= 256 || K < 1024) return 1;
  // Measured optimum keeps ~48 K-chunks (of 64) per block: enough work to
  // amortize the prologue, enough blocks to hide staging latency
  // (576x64x802816: 725us @ sk=102 -> 605us @ sk~260).
  static const int64_t target = [] {
    const char* e = getenv("STF_SPLITK_TARGET");
    return e ? atoll(e) : 0;
  }();
  int64_t want;
  if (target > 0) {
    want = target / (tiles ? tiles : 1);
  } else {
    want = 512 / (tiles ? tiles : 1);
    // Very skinny outputs with huge K (the 7x7 stem dW: 392x64x3.2M):
    // deepen the split to ~2048 blocks (sweep: 3.14ms @1044 blocks ->
    // 2.39ms @2048; flat beyond). Keep >=48 K-chunks per block.
    int64_t kt = K / 64;
    if (tiles <= 6 && kt / (tiles * want) > 96) {
      want = 2048 / tiles;
      if (want > kt / 48) want = kt / 48;
      if (tiles * want > 4096) want = 4096 / tiles;
    }
  }
  int64_t maxk = K / 512;  // keep >= 8 K-iters per slice
  if (maxk < 1) maxk = 1;
  int64_t sk = std::min(want, maxk);
  return (int)(sk < 1 ? 1 : sk);
}


// Persistent pre-zeroed f32 accumulator arena for split-K (per GPU): every
// split-K GEMM atomically accumulates into it and the fused
// cast-and-rezero epilogue restores the all-zero invariant while draining
// the result — the per-GEMM hipMemset launch disappears. All compute runs
// on one stream, so consecutive uses are ordered; growth leaks the old
// buffer deliberately (in-flight work and captured hipGraphs may still
// reference it).
inline float* SplitKArena(int ordinal, int64_t elems, hipStream_t s) {
  if (elems * 4 > (1ll << 28)) return nullptr;  // >256 MB: use a temp
  static std::mutex mu;
  static std::map<int, std::pair<float*, int64_t>> arenas;
  std::lock_guard<std::mutex> l(mu);
  auto& a = arenas[ordinal];
  if (a.second < elems) {
    int64_t cap = a.second * 2 > elems ? a.second * 2 : elems;
    if (cap < (1 << 20)) cap = 1 << 20;
    float* p = nullptr;
    if (hipMalloc(&p, cap * 4) != hipSuccess) return nullptr;
    if (hipMemsetAsync(p, 0, cap * 4, s) != hipSuccess) return nullptr;
    a = {p, cap};
  }
  return a.first;
}

// GEMM helper: picks plain vs split-K (f32 scratch + cast) automatically.
// a_km/b_km: the operand is stored contraction-major ([K,M]/[K,N], row
// stride lda/ldb) and transposed in-kernel during LDS staging.
inline hipError_t GemmBf16AutoEx(OpKernelContext* ctx, const void* A,
                                 const void* B, void* C_bf16, int64_t M,
                                 int64_t N, int64_t K, int64_t lda,
                                 int64_t ldb, int a_km, int b_km,
                                 hipStream_t s) {
  int sk = PickSplitK(M, N, K);
  if (sk <= 1)
    return stf_gemm_bf16(A, B, C_bf16, nullptr, M, N, K, lda, ldb, 0.f, a_km,
                         b_km, 1, 0, s);
  if (float* arena = SplitKArena(ctx->device()->gpu_ordinal(), M * N, s)) {
    hipError_t e = stf_gemm_bf16_splitk(A, B, arena, M, N, K, lda, ldb, a_km,
                                        b_km, sk, s);
    if (e != hipSuccess) return e;
    return stf_cast_f32_bf16_zero(arena, C_bf16, M * N, s);
  }
  Tensor scratch = ctx->allocate_temp(DT_FLOAT, TensorShape({M, N}));
  hipError_t e = hipMemsetAsync(scratch.raw_data(), 0, M * N * 4, s);
  if (e != hipSuccess) return e;
  e = stf_gemm_bf16_splitk(A, B, scratch.raw_data(), M, N, K, lda, ldb, a_km,
                           b_km, sk, s);
  if (e != hipSuccess) return e;
  return stf_cast(0, 1, scratch.raw_data(), C_bf16, M * N, s);
}

inline hipError_t GemmBf16Auto(OpKernelContext* ctx, const void* A,
                               const void* B, void* C_bf16, int64_t M,
                               int64_t N, int64_t K, hipStream_t s) {
  return GemmBf16AutoEx(ctx, A, B, C_bf16, M, N, K, K, K, 0, 0, s);
}
inline int CastCode(DataType dt) {
  switch (dt) {
    case DT_FLOAT: return 0;
    case DT_BFLOAT16: return 1;
    case DT_HALF: return 2;
    case DT_INT32: return 3;
    case DT_INT64: return 4;
    case DT_BOOL: return 5;
    default: return -1;
  }
}

// 256 zeroed device bytes for the implicit-GEMM address generators'
// out-of-bounds (padding) loads. Allocated once per process (one GPU per
// process under the data-parallel model).
inline const void* ZeroPage() {
  static const void* page = [] {
    void* p = nullptr;
    if (hipMalloc(&p, 256) != hipSuccess) return (void*)nullptr;
    (void)hipMemset(p, 0, 256);
    return p;
  }();
  return page;
}

// zero an f32 device buffer on the stream
inline hipError_t ZeroF32(void* p, int64_t n, hipStream_t s) {
  return hipMemsetAsync(p, 0, n * 4, s);
}

// ---------------------------------------------------------------------------
// elementwise
// ---------------------------------------------------------------------------
class GpuUnaryOp : public OpKernel {
 public:
  GpuUnaryOp(OpKernelConstruction* c, int op) : OpKernel(c), op_(op) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, x.shape());
    OP_HIP_OK(ctx, stf_unary(op_, DtypeCode(x.dtype()), x.raw_data(),
                             y->raw_data(), x.NumElements(), GPU_STREAM(ctx)));
  }

 private:
  int op_;
};

// matches UOp order in elementwise.hip
enum {
  U_NEG, U_ABS, U_SIGN, U_SQUARE, U_SQRT, U_RSQRT, U_EXP, U_LOG, U_LOG1P,
  U_TANH, U_SIGMOID, U_RELU, U_RELU6, U_SOFTPLUS, U_RECIP, U_FLOOR, U_CEIL,
  U_SIN, U_COS
};
enum {
  B_ADD, B_SUB, B_MUL, B_DIV, B_MAX, B_MIN, B_POW, B_SQDIFF, B_SIGMOID_GRAD,
  B_TANH_GRAD, B_RSQRT_GRAD, B_SQRT_GRAD, B_RELU_GRAD, B_RELU6_GRAD,
  B_SOFTPLUS_GRAD
};

#define REG_GPU_UNARY(NAME, CODE)                                             \
  struct NAME##GpuTag {};                                                     \
  class NAME##GpuOp : public GpuUnaryOp {                                     \
   public:                                                                    \
    explicit NAME##GpuOp(OpKernelConstruction* c) : GpuUnaryOp(c, CODE) {}    \
  };                                                                          \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_GPU).TypeConstraint<float>("T"), NAME##GpuOp); \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), NAME##GpuOp);

REG_GPU_UNARY(Neg, U_NEG)
REG_GPU_UNARY(Abs, U_ABS)
REG_GPU_UNARY(Sign, U_SIGN)
REG_GPU_UNARY(Square, U_SQUARE)
REG_GPU_UNARY(Sqrt, U_SQRT)
REG_GPU_UNARY(Rsqrt, U_RSQRT)
REG_GPU_UNARY(Exp, U_EXP)
REG_GPU_UNARY(Log, U_LOG)
REG_GPU_UNARY(Log1p, U_LOG1P)
REG_GPU_UNARY(Tanh, U_TANH)
REG_GPU_UNARY(Sigmoid, U_SIGMOID)
REG_GPU_UNARY(Relu, U_RELU)
REG_GPU_UNARY(Relu6, U_RELU6)
REG_GPU_UNARY(Softplus, U_SOFTPLUS)
REG_GPU_UNARY(Reciprocal, U_RECIP)
REG_GPU_UNARY(Floor, U_FLOOR)
REG_GPU_UNARY(Ceil, U_CEIL)
REG_GPU_UNARY(Sin, U_SIN)
REG_GPU_UNARY(Cos, U_COS)

class GpuBinaryOp : public OpKernel {
 public:
  GpuBinaryOp(OpKernelConstruction* c, int op) : OpKernel(c), op_(op) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    int dt = DtypeCode(a.dtype());
    hipStream_t s = GPU_STREAM(ctx);
    if (a.shape() == b.shape()) {
      Tensor* y = ctx->allocate_output(0, a.shape());
      OP_HIP_OK(ctx, stf_binary(op_, dt, a.raw_data(), b.raw_data(),
                                y->raw_data(), a.NumElements(), s));
      return;
    }
    if (b.NumElements() == 1) {
      Tensor* y = ctx->allocate_output(0, a.shape());
      OP_HIP_OK(ctx, stf_binary_scalar(op_, dt, a.raw_data(), b.raw_data(),
                                       y->raw_data(), a.NumElements(), 0, s));
      return;
    }
    if (a.NumElements() == 1) {
      Tensor* y = ctx->allocate_output(0, b.shape());
      OP_HIP_OK(ctx, stf_binary_scalar(op_, dt, b.raw_data(), a.raw_data(),
                                       y->raw_data(), b.NumElements(), 1, s));
      return;
    }
    BCast bc(a.shape(), b.shape());
    OP_REQUIRES(ctx, bc.valid && bc.out.size() <= 6,
                errors::InvalidArgument("bad broadcast"));
    Tensor* y = ctx->allocate_output(0, bc.out_shape());
    OP_HIP_OK(ctx, stf_binary_bcast(op_, dt, a.raw_data(), b.raw_data(),
                                    y->raw_data(), bc.num_elements,
                                    (int)bc.out.size(), bc.out.data(),
                                    bc.sx.data(), bc.sy.data(), s));
  }

 private:
  int op_;
};

#define REG_GPU_BINARY(NAME, CODE)                                            \
  class NAME##GpuOp : public GpuBinaryOp {                                    \
   public:                                                                    \
    explicit NAME##GpuOp(OpKernelConstruction* c) : GpuBinaryOp(c, CODE) {}   \
  };                                                                          \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_GPU).TypeConstraint<float>("T"), NAME##GpuOp); \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), NAME##GpuOp);

REG_GPU_BINARY(Add, B_ADD)
REG_GPU_BINARY(Sub, B_SUB)
REG_GPU_BINARY(Mul, B_MUL)
REG_GPU_BINARY(RealDiv, B_DIV)
REG_GPU_BINARY(Div, B_DIV)
REG_GPU_BINARY(Maximum, B_MAX)
REG_GPU_BINARY(Minimum, B_MIN)
REG_GPU_BINARY(Pow, B_POW)
REG_GPU_BINARY(SquaredDifference, B_SQDIFF)
REG_GPU_BINARY(SigmoidGrad, B_SIGMOID_GRAD)
REG_GPU_BINARY(TanhGrad, B_TANH_GRAD)
REG_GPU_BINARY(RsqrtGrad, B_RSQRT_GRAD)
REG_GPU_BINARY(SqrtGrad, B_SQRT_GRAD)
REG_GPU_BINARY(ReluGrad, B_RELU_GRAD)
REG_GPU_BINARY(Relu6Grad, B_RELU6_GRAD)
REG_GPU_BINARY(SoftplusGrad, B_SOFTPLUS_GRAD)

class GpuCastOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, x.shape());
    int sc = CastCode(x.dtype()), dc = CastCode(y->dtype());
    OP_REQUIRES(ctx, sc >= 0 && dc >= 0,
                errors::Unimplemented("GPU cast dtype"));
    OP_HIP_OK(ctx, stf_cast(sc, dc, x.raw_data(), y->raw_data(),
                            x.NumElements(), GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("Cast").Device(DEVICE_GPU), GpuCastOp);

class GpuAddNOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs();
    const Tensor& first = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, first.shape());
    hipStream_t s = GPU_STREAM(ctx);
    int dt = DtypeCode(first.dtype());
    if (n == 1) {
      OP_HIP_OK(ctx, hipMemcpyAsync(y->raw_data(), first.raw_data(),
                                    first.TotalBytes(),
                                    hipMemcpyDeviceToDevice, s));
      return;
    }
    // pairwise adds: y = in0 + in1; y += in_k
    OP_HIP_OK(ctx, stf_binary(B_ADD, dt, first.raw_data(),
                              ctx->input(1).raw_data(), y->raw_data(),
                              first.NumElements(), s));
    for (int k = 2; k < n; ++k) {
      OP_HIP_OK(ctx, stf_binary(B_ADD, dt, y->raw_data(),
                                ctx->input(k).raw_data(), y->raw_data(),
                                first.NumElements(), s));
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("AddN").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuAddNOp);
REGISTER_KERNEL_BUILDER(Name("AddN").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuAddNOp);

// ---------------------------------------------------------------------------
// fill-style
// ---------------------------------------------------------------------------
class GpuZerosLikeOp : public OpKernel {
 public:
  GpuZerosLikeOp(OpKernelConstruction* c, float v) : OpKernel(c), v_(v) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, x.shape());
    OP_HIP_OK(ctx, stf_fill_f32(y->raw_data(), v_, x.NumElements(),
                                x.dtype() == DT_BFLOAT16, GPU_STREAM(ctx)));
  }

 private:
  float v_;
};
class ZerosLikeGpu : public GpuZerosLikeOp {
 public:
  explicit ZerosLikeGpu(OpKernelConstruction* c) : GpuZerosLikeOp(c, 0.f) {}
};
class OnesLikeGpu : public GpuZerosLikeOp {
 public:
  explicit OnesLikeGpu(OpKernelConstruction* c) : GpuZerosLikeOp(c, 1.f) {}
};
REGISTER_KERNEL_BUILDER(Name("ZerosLike").Device(DEVICE_GPU).TypeConstraint<float>("T"), ZerosLikeGpu);
REGISTER_KERNEL_BUILDER(Name("ZerosLike").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), ZerosLikeGpu);
REGISTER_KERNEL_BUILDER(Name("OnesLike").Device(DEVICE_GPU).TypeConstraint<float>("T"), OnesLikeGpu);
REGISTER_KERNEL_BUILDER(Name("OnesLike").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), OnesLikeGpu);

class GpuFillOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    auto dims = IntVector(ctx->input(0));
    const Tensor& v = ctx->input(1);  // host scalar
    Tensor* y = ctx->allocate_output(0, TensorShape(dims));
    float val = v.dtype() == DT_BFLOAT16
                    ? (float)v.flat<bfloat16>()[0]
                    : v.flat<float>()[0];
    OP_HIP_OK(ctx, stf_fill_f32(y->raw_data(), val, y->NumElements(),
                                y->dtype() == DT_BFLOAT16, GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("Fill").Device(DEVICE_GPU).TypeConstraint<float>("T").HostMemory("dims").HostMemory("value"), GpuFillOp);
REGISTER_KERNEL_BUILDER(Name("Fill").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("dims").HostMemory("value"), GpuFillOp);

// ---------------------------------------------------------------------------
// MatMul (MFMA GEMM + pre-transposes to NT form)
// ---------------------------------------------------------------------------
class GpuMatMulOp : public OpKernel {
 public:
  explicit GpuMatMulOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("transpose_a", &ta_);
    c->GetAttr("transpose_b", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    hipStream_t s = GPU_STREAM(ctx);
    int64_t m = ta_ ? a.dim_size(1) : a.dim_size(0);
    int64_t k = ta_ ? a.dim_size(0) : a.dim_size(1);
    int64_t n = tb_ ? b.dim_size(0) : b.dim_size(1);
    Tensor* y = ctx->allocate_output(0, TensorShape({m, n}));
    size_t es = DataTypeSize(a.dtype());
    if (a.dtype() == DT_BFLOAT16) {
      // The bf16 GEMM stages either operand layout directly (K-major
      // staging for transposed sides) as long as the row stride keeps 16B
      // global loads aligned; otherwise fall back to a pre-transpose.
      const void* ap = a.raw_data();
      const void* bp = b.raw_data();
      int a_km = ta_ ? 1 : 0;
      int64_t lda = a.dim_size(1);
      Tensor a_tmp;
      if (ta_ && (m & 7) != 0) {
        a_tmp = ctx->allocate_temp(a.dtype(), TensorShape({m, k}));
        OP_HIP_OK(ctx, stf_transpose2d((int)es, a.raw_data(),
                                       a_tmp.raw_data(), a.dim_size(0),
                                       a.dim_size(1), s));
        ap = a_tmp.raw_data();
        a_km = 0;
        lda = k;
      }
      int b_km = tb_ ? 0 : 1;
      int64_t ldb = b.dim_size(1);
      Tensor b_tmp;
      if (!tb_ && (n & 7) != 0) {
        b_tmp = ctx->allocate_temp(b.dtype(), TensorShape({n, k}));
        OP_HIP_OK(ctx, stf_transpose2d((int)es, b.raw_data(),
                                       b_tmp.raw_data(), b.dim_size(0),
                                       b.dim_size(1), s));
        bp = b_tmp.raw_data();
        b_km = 0;
        ldb = k;
      }
      OP_HIP_OK(ctx, GemmBf16AutoEx(ctx, ap, bp, y->raw_data(), m, n, k, lda,
                                    ldb, a_km, b_km, s));
    } else {
      // A effective [M,K]
      Tensor a_eff = a;
      if (ta_) {
        a_eff = ctx->allocate_temp(a.dtype(), TensorShape({m, k}));
        OP_HIP_OK(ctx, stf_transpose2d((int)es, a.raw_data(),
                                       a_eff.raw_data(), a.dim_size(0),
                                       a.dim_size(1), s));
      }
      // B effective [N,K]
      Tensor b_eff = b;
      if (!tb_) {
        b_eff = ctx->allocate_temp(b.dtype(), TensorShape({n, k}));
        OP_HIP_OK(ctx, stf_transpose2d((int)es, b.raw_data(),
                                       b_eff.raw_data(), b.dim_size(0),
                                       b.dim_size(1), s));
      }
      OP_HIP_OK(ctx, stf_gemm_f32_nt(a_eff.raw_data(), b_eff.raw_data(),
                                     y->raw_data(), m, n, k, s));
    }
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_KERNEL_BUILDER(Name("MatMul").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuMatMulOp);
REGISTER_KERNEL_BUILDER(Name("MatMul").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuMatMulOp);

// BatchMatMul on GPU: per-batch MFMA GEMM launches with offset pointers
// (reference batch_matmul_op_impl.h:350 ThenBlasGemmBatchedWithScratch).
// The adj flags map onto the GEMM's contraction-major staging the same way
// MatMul's transpose flags do, so no transpose kernels are launched.
class GpuBatchMatMulOp : public OpKernel {
 public:
  explicit GpuBatchMatMulOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("adj_x", &ta_);
    c->GetAttr("adj_y", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    hipStream_t s = GPU_STREAM(ctx);
    int ra = a.dims(), rb = b.dims();
    OP_REQUIRES(ctx, ra >= 2 && rb == ra,
                errors::InvalidArgument("BatchMatMul rank mismatch"));
    int64_t m = ta_ ? a.dim_size(ra - 1) : a.dim_size(ra - 2);
    int64_t k = ta_ ? a.dim_size(ra - 2) : a.dim_size(ra - 1);
    int64_t n = tb_ ? b.dim_size(rb - 2) : b.dim_size(rb - 1);
    int64_t kb = tb_ ? b.dim_size(rb - 1) : b.dim_size(rb - 2);
    OP_REQUIRES(ctx, k == kb,
                errors::InvalidArgument("BatchMatMul inner dim mismatch"));
    int64_t batch = 1;
    TensorShape out_shape;
    for (int i = 0; i < ra - 2; ++i) {
      batch *= a.dim_size(i);
      out_shape.AddDim(a.dim_size(i));
    }
    out_shape.AddDim(m);
    out_shape.AddDim(n);
    Tensor* y = ctx->allocate_output(0, out_shape);
    int64_t sa = m * k, sb = k * n, sc = m * n;
    if (a.dtype() == DT_BFLOAT16) {
      // adj_x: A [K,M] contraction-major -> a_km; plain B [K,N] -> b_km.
      int a_km = ta_ ? 1 : 0;
      int b_km = tb_ ? 0 : 1;
      int64_t lda = a.dim_size(ra - 1);
      int64_t ldb = b.dim_size(rb - 1);
      const uint16_t* ap = (const uint16_t*)a.raw_data();
      const uint16_t* bp = (const uint16_t*)b.raw_data();
      uint16_t* yp = (uint16_t*)y->raw_data();
      for (int64_t i = 0; i < batch; ++i) {
        OP_HIP_OK(ctx, stf_gemm_bf16(ap + i * sa, bp + i * sb, yp + i * sc,
                                     nullptr, m, n, k, lda, ldb, 0.f, a_km,
                                     b_km, 1, 0, s));
      }
    } else {
      // f32 path: pre-transpose per batch when adjoint flags are set.
      Tensor a_eff = a, b_eff = b;
      const float* ap = (const float*)a.raw_data();
      const float* bp = (const float*)b.raw_data();
      if (ta_) {
        a_eff = ctx->allocate_temp(DT_FLOAT, TensorShape({batch, m, k}));
        for (int64_t i = 0; i < batch; ++i)
          OP_HIP_OK(ctx, stf_transpose2d(4, ap + i * sa,
                                         (float*)a_eff.raw_data() + i * sa,
                                         k, m, s));
        ap = (const float*)a_eff.raw_data();
      }
      // GEMM f32 wants B as [N, K] (NT): transpose unless adj_y.
      if (!tb_) {
        b_eff = ctx->allocate_temp(DT_FLOAT, TensorShape({batch, n, k}));
        for (int64_t i = 0; i < batch; ++i)
          OP_HIP_OK(ctx, stf_transpose2d(4, bp + i * sb,
                                         (float*)b_eff.raw_data() + i * sb,
                                         k, n, s));
        bp = (const float*)b_eff.raw_data();
      }
      float* yp = (float*)y->raw_data();
      for (int64_t i = 0; i < batch; ++i)
        OP_HIP_OK(ctx, stf_gemm_f32_nt(ap + i * sa, bp + i * sb, yp + i * sc,
                                       m, n, k, s));
    }
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_KERNEL_BUILDER(Name("BatchMatMul").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBatchMatMulOp);
REGISTER_KERNEL_BUILDER(Name("BatchMatMul").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuBatchMatMulOp);

// LRN forward on GPU (reference used cuDNN, lrn_op.cc:211). The composite
// python gradient runs on the GPU's elementwise kernels.
class GpuLRNOp : public OpKernel {
 public:
  explicit GpuLRNOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("depth_radius", &radius_);
    c->GetAttr("bias", &bias_);
    c->GetAttr("alpha", &alpha_);
    c->GetAttr("beta", &beta_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, x.shape());
    int64_t c = x.dim_size(x.dims() - 1);
    OP_HIP_OK(ctx, stf_lrn_fwd(x.raw_data(), y->raw_data(),
                               x.NumElements() / c, (int)c, (int)radius_,
                               bias_, alpha_, beta_, GPU_STREAM(ctx)));
  }

 private:
  int64_t radius_ = 5;
  float bias_ = 1.f, alpha_ = 1.f, beta_ = 0.5f;
};
REGISTER_KERNEL_BUILDER(Name("LRN").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuLRNOp);

// ---------------------------------------------------------------------------
// Conv2D family (im2col + MFMA GEMM)
// ---------------------------------------------------------------------------
struct GpuConvGeom {
  int64_t N, H, W, C, R, S, K, sh, sw, ph, pw, P, Q;
  int64_t M() const { return N * P * Q; }
  int64_t RSC() const { return R * S * C; }
  // K-dim padded to the GEMM K-step so glds staging stays 16B-aligned
  // (conv1's RSC=147 would otherwise force the scalar edge path everywhere).
  int64_t RSCp() const { return (RSC() + 63) & ~int64_t(63); }
  bool is_1x1_s1() const {
    return R == 1 && S == 1 && sh == 1 && sw == 1 && ph == 0 && pw == 0;
  }
};

// im2col reuse between Conv2D forward and Conv2DBackpropFilter: the same
// column matrix is needed by both; materializing it once per step halves
// the im2col cost (288 GB HBM3E comfortably holds every layer's columns).
// Keyed by (x buffer, geometry); safe because x stays referenced (and its
// buffer unreusable) until the backward op consumed it.
struct ColCacheResource : public ResourceBase {
  std::mutex mu;
  std::map<std::pair<const void*, uint64_t>, Tensor> m;
};

inline uint64_t ConvGeomHash(const GpuConvGeom& g) {
  uint64_t h = 1469598103934665603ull;
  auto mix = [&h](int64_t v) {
    h ^= (uint64_t)v;
    h *= 1099511628211ull;
  };
  mix(g.N); mix(g.H); mix(g.W); mix(g.C); mix(g.R); mix(g.S);
  mix(g.K); mix(g.sh); mix(g.sw); mix(g.P); mix(g.Q);
  return h;
}

inline ColCacheResource* GetColCache(OpKernelContext* ctx) {
  if (getenv("STF_NO_COL_CACHE")) return nullptr;
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<ColCacheResource>(
      "__conv_col_cache", []() { return new ColCacheResource(); });
}


static Status GetConvGeom(const TensorShape& x, const TensorShape& f,
                          const std::vector<int64_t>& strides,
                          const std::string& padding, GpuConvGeom* g) {
  g->N = x.dim_size(0);
  g->H = x.dim_size(1);
  g->W = x.dim_size(2);
  g->C = x.dim_size(3);
  g->R = f.dim_size(0);
  g->S = f.dim_size(1);
  g->K = f.dim_size(3);
  if (f.dim_size(2) != g->C)
    return errors::InvalidArgument("conv channel mismatch");
  g->sh = strides[1];
  g->sw = strides[2];
  if (padding == "SAME") {
    g->P = (g->H + g->sh - 1) / g->sh;
    g->Q = (g->W + g->sw - 1) / g->sw;
    g->ph = std::max<int64_t>(0, (g->P - 1) * g->sh + g->R - g->H) / 2;
    g->pw = std::max<int64_t>(0, (g->Q - 1) * g->sw + g->S - g->W) / 2;
  } else {
    g->P = (g->H - g->R) / g->sh + 1;
    g->Q = (g->W - g->S) / g->sw + 1;
    g->ph = g->pw = 0;
  }
  return Status::OK();
}

class GpuConv2DOp : public OpKernel {
 public:
  explicit GpuConv2DOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& w = ctx->input(1);
    hipStream_t s = GPU_STREAM(ctx);
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, GetConvGeom(x.shape(), w.shape(), strides_, padding_,
                                    &g));
    Tensor* y = ctx->allocate_output(0, TensorShape({g.N, g.P, g.Q, g.K}));
    // C not a multiple of 8 (the C=3 stem): pad channels once so the
    // implicit-GEMM paths (8-wide k runs) apply — the padded channels are
    // zero in both x and w, contributing nothing.
    Tensor xp, wp;
    if (!g.is_1x1_s1() && (g.C % 8) != 0 && x.dtype() == DT_BFLOAT16) {
      int64_t c8 = (g.C + 7) & ~7ll;
      xp = ctx->allocate_temp(DT_BFLOAT16,
                              TensorShape({g.N, g.H, g.W, c8}));
      OP_HIP_OK(ctx, stf_block_pad(1, x.raw_data(), xp.raw_data(),
                                   (int64_t)g.N * g.H * g.W, g.C, c8, s));
      wp = ctx->allocate_temp(DT_BFLOAT16,
                              TensorShape({g.R * g.S * c8, g.K}));
      OP_HIP_OK(ctx, stf_block_pad(1, w.raw_data(), wp.raw_data(),
                                   g.R * g.S, g.C * g.K, c8 * g.K, s));
      GpuConvGeom gp = g;
      gp.C = c8;
      ComputeWith(ctx, xp, wp, gp, y, s);
      return;
    }
    ComputeWith(ctx, x, w, g, y, s);
  }

 private:
  void ComputeWith(OpKernelContext* ctx, const Tensor& x, const Tensor& w,
                   GpuConvGeom g, Tensor* y, hipStream_t s) {
    int64_t rsc = g.RSC();
    // 1x1/s1 uses x directly as the GEMM A (ld = C) — no padding there.
    int64_t rscp = g.is_1x1_s1() ? rsc : g.RSCp();
    // weights stay [RSC(p), K] and are read contraction-major by the GEMM
    // (b_km staging) — no transpose kernel. Zero-pad the RSC dim only when
    // it is not 64-aligned (conv1's 147 -> 192); rows [rsc, rscp) are zero,
    // matching the zero-padded im2col columns.
    Tensor wsrc = w;
    if (rscp != rsc) {
      wsrc = ctx->allocate_temp(DT_BFLOAT16, TensorShape({rscp, g.K}));
      OP_HIP_OK(ctx, hipMemsetAsync(wsrc.raw_data(), 0, rscp * g.K * 2, s));
      OP_HIP_OK(ctx, hipMemcpyAsync(wsrc.raw_data(), w.raw_data(),
                                    rsc * g.K * 2, hipMemcpyDeviceToDevice,
                                    s));
    }
    // Implicit-GEMM fast path (the reference's cuDNN implicit-GEMM slot,
    // conv_ops.cc:664): column matrix generated inside the GEMM's A
    // staging — no im2col kernel, no column buffer, no column cache.
    static const bool no_implicit = getenv("STF_NO_IMPLICIT_CONV") != nullptr;
    if (!no_implicit && !g.is_1x1_s1() && (g.C % 8) == 0 &&
        stf_gemm_bf16_8ph_ok(g.M(), g.K, rscp) && ZeroPage()) {
      Tensor wt = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.K, rscp}));
      OP_HIP_OK(ctx, stf_transpose2d(2, wsrc.raw_data(), wt.raw_data(), rscp,
                                     g.K, s));
      OP_HIP_OK(ctx, stf_conv2d_fwd_8ph(
                         x.raw_data(), wt.raw_data(), y->raw_data(), nullptr,
                         ZeroPage(), (int)g.N, (int)g.H, (int)g.W, (int)g.C,
                         (int)g.R, (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                         (int)g.pw, (int)g.P, (int)g.Q, g.K, rscp, 1, 0, s));
      return;
    }
    // Implicit-GEMM through the general NT template for shapes the 8-phase
    // kernel cannot take (e.g. Inception's 48/96/288-cout convs).
    if (!no_implicit && !g.is_1x1_s1() && (g.C % 8) == 0 &&
        (g.K & 7) == 0 && ZeroPage()) {
      OP_HIP_OK(ctx, stf_conv2d_fwd_nt(
                         x.raw_data(), wsrc.raw_data(), y->raw_data(),
                         ZeroPage(), (int)g.N, (int)g.H, (int)g.W, (int)g.C,
                         (int)g.R, (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                         (int)g.pw, (int)g.P, (int)g.Q, g.K, rscp, s));
      return;
    }
    const void* col_data = x.raw_data();
    Tensor col;
    if (!g.is_1x1_s1()) {
      col = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.M(), rscp}));
      if (rscp != rsc)
        OP_HIP_OK(ctx, hipMemsetAsync(col.raw_data(), 0, g.M() * rscp * 2, s));
      OP_HIP_OK(ctx, stf_im2col_bf16(x.raw_data(), col.raw_data(), (int)g.N,
                                     (int)g.H, (int)g.W, (int)g.C, (int)g.R,
                                     (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                                     (int)g.pw, (int)g.P, (int)g.Q, rscp, s));
      col_data = col.raw_data();
      if (ColCacheResource* cc = GetColCache(ctx)) {
        std::lock_guard<std::mutex> l(cc->mu);
        cc->m[{x.raw_data(), ConvGeomHash(g)}] = col;
      }
    }
    if (stf_gemm_bf16_8ph_ok(g.M(), g.K, rscp)) {
      // 8-phase-eligible shape: one small weight transpose ([rscp,K] ->
      // [K,rscp]) buys the NT fast path for the big M=NPQ GEMM.
      Tensor wt = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.K, rscp}));
      OP_HIP_OK(ctx, stf_transpose2d(2, wsrc.raw_data(), wt.raw_data(), rscp,
                                     g.K, s));
      OP_HIP_OK(ctx, stf_gemm_bf16_nt(col_data, wt.raw_data(), y->raw_data(),
                                      nullptr, g.M(), g.K, rscp, 0.f, 1, 0,
                                      s));
    } else if ((g.K & 7) == 0) {
      OP_HIP_OK(ctx, stf_gemm_bf16(col_data, wsrc.raw_data(), y->raw_data(),
                                   nullptr, g.M(), g.K, rscp, rscp, g.K, 0.f,
                                   0, 1, 1, 0, s));
    } else {
      Tensor wt = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.K, rscp}));
      OP_HIP_OK(ctx, stf_transpose2d(2, wsrc.raw_data(), wt.raw_data(), rscp,
                                     g.K, s));
      OP_HIP_OK(ctx, stf_gemm_bf16_nt(col_data, wt.raw_data(), y->raw_data(),
                                      nullptr, g.M(), g.K, rscp, 0.f, 1, 0,
                                      s));
    }
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("Conv2D").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuConv2DOp);

class GpuConv2DBackpropInputOp : public OpKernel {
 public:
  explicit GpuConv2DBackpropInputOp(OpKernelConstruction* c, bool side = false)
      : OpKernel(c), side_(side) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto in_sizes = IntVector(ctx->input(0));
    const Tensor& w = ctx->input(1);
    const Tensor& dy = ctx->input(2);
    hipStream_t s = GPU_STREAM(ctx);
    TensorShape x_shape(in_sizes);
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, GetConvGeom(x_shape, w.shape(), strides_, padding_,
                                    &g));
    Tensor* dx = ctx->allocate_output(0, x_shape);
    const void* side = nullptr;
    if (side_) {
      const Tensor& sd = ctx->input(3);
      OP_REQUIRES(ctx, sd.NumElements() == dx->NumElements(),
                  errors::InvalidArgument(
                      "Conv2DBackpropInputAdd: side shape mismatch"));
      side = sd.raw_data();
    }
    // dcol[M, RSC] = dy[M, K] x w[RSC, K]^T  (filter layout is already NT B)
    if (g.is_1x1_s1()) {
      if (side && stf_gemm_bf16_8ph_ok(g.M(), g.RSC(), g.K)) {
        // residual-gradient accumulation fused into the GEMM epilogue: the
        // separate full-tensor add pass (and its 3x-bandwidth cost) vanishes
        OP_HIP_OK(ctx, stf_gemm_bf16_8ph_side(
                           dy.raw_data(), w.raw_data(), dx->raw_data(),
                           nullptr, side, g.M(), g.RSC(), g.K, g.K, g.K, 1, 0,
                           s));
        return;
      }
      OP_HIP_OK(ctx, stf_gemm_bf16_nt(dy.raw_data(), w.raw_data(),
                                      dx->raw_data(), nullptr, g.M(), g.RSC(),
                                      g.K, 0.f, 1, 0, s));
      if (side)
        OP_HIP_OK(ctx, stf_binary(B_ADD, DtypeCode(DT_BFLOAT16),
                                  dx->raw_data(), side, dx->raw_data(),
                                  dx->NumElements(), s));
      return;
    }
    Tensor dcol = ctx->allocate_temp(DT_BFLOAT16,
                                     TensorShape({g.M(), g.RSC()}));
    OP_HIP_OK(ctx, stf_gemm_bf16_nt(dy.raw_data(), w.raw_data(),
                                    dcol.raw_data(), nullptr, g.M(), g.RSC(),
                                    g.K, 0.f, 1, 0, s));
    OP_HIP_OK(ctx, stf_col2im_bf16(dcol.raw_data(), dx->raw_data(), (int)g.N,
                                   (int)g.H, (int)g.W, (int)g.C, (int)g.R,
                                   (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                                   (int)g.pw, (int)g.P, (int)g.Q, s));
    if (side)
      OP_HIP_OK(ctx, stf_binary(B_ADD, DtypeCode(DT_BFLOAT16), dx->raw_data(),
                                side, dx->raw_data(), dx->NumElements(), s));
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
  bool side_;
};
REGISTER_KERNEL_BUILDER(Name("Conv2DBackpropInput").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("input_sizes"), GpuConv2DBackpropInputOp);

class GpuConv2DBackpropInputAddOp : public GpuConv2DBackpropInputOp {
 public:
  explicit GpuConv2DBackpropInputAddOp(OpKernelConstruction* c)
      : GpuConv2DBackpropInputOp(c, true) {}
};
REGISTER_KERNEL_BUILDER(Name("Conv2DBackpropInputAdd").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("input_sizes"), GpuConv2DBackpropInputAddOp);

class GpuConv2DBackpropFilterOp : public OpKernel {
 public:
  explicit GpuConv2DBackpropFilterOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto f_sizes = IntVector(ctx->input(1));
    const Tensor& dy = ctx->input(2);
    hipStream_t s = GPU_STREAM(ctx);
    TensorShape f_shape(f_sizes);
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, GetConvGeom(x.shape(), f_shape, strides_, padding_,
                                    &g));
    Tensor* dw = ctx->allocate_output(0, f_shape);
    int64_t rsc = g.RSC();
    static const bool no_implicit0 = getenv("STF_NO_IMPLICIT_CONV") != nullptr;
    // C%8!=0 stem: channel-pad x, take the implicit dW path at C8, then
    // drop the padded filter rows (their gradient is exactly zero).
    if (!no_implicit0 && !g.is_1x1_s1() && (g.C % 8) != 0 &&
        x.dtype() == DT_BFLOAT16 && (g.K & 7) == 0 && (g.M() & 63) == 0 &&
        ZeroPage()) {
      int64_t c8 = (g.C + 7) & ~7ll;
      int64_t rsc8 = g.R * g.S * c8;
      Tensor xp = ctx->allocate_temp(DT_BFLOAT16,
                                     TensorShape({g.N, g.H, g.W, c8}));
      OP_HIP_OK(ctx, stf_block_pad(1, x.raw_data(), xp.raw_data(),
                                   (int64_t)g.N * g.H * g.W, g.C, c8, s));
      int sk = PickSplitK(rsc8, g.K, g.M());
      if (float* arena =
              SplitKArena(ctx->device()->gpu_ordinal(), rsc8 * g.K, s)) {
        OP_HIP_OK(ctx, stf_conv2d_dw_splitk(
                           xp.raw_data(), dy.raw_data(), arena, ZeroPage(),
                           (int)g.N, (int)g.H, (int)g.W, (int)c8, (int)g.R,
                           (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                           (int)g.pw, (int)g.P, (int)g.Q, g.K, sk, s));
        Tensor dwp = ctx->allocate_temp(DT_BFLOAT16,
                                        TensorShape({rsc8, g.K}));
        OP_HIP_OK(ctx, stf_cast_f32_bf16_zero(arena, dwp.raw_data(),
                                              rsc8 * g.K, s));
        OP_HIP_OK(ctx, stf_block_pad(1, dwp.raw_data(), dw->raw_data(),
                                     g.R * g.S, c8 * g.K, g.C * g.K, s));
        return;
      }
    }
    // dW[RSC, K] = col[M, RSC]^T x dy[M, K]: both operands are contraction
    // (M-)major as stored, so the GEMM's K-major staging reads them directly
    // — no transpose kernels (this was 2 full passes over the im2col matrix).
    // Implicit-GEMM dW: contraction over output pixels with the column
    // matrix generated inside the K-major staging — pairs with the
    // implicit forward (no materialized columns anywhere in the conv).
    static const bool no_implicit = getenv("STF_NO_IMPLICIT_CONV") != nullptr;
    if (!no_implicit && !g.is_1x1_s1() && (g.C % 8) == 0 &&
        (g.K & 7) == 0 && (g.M() & 63) == 0 && ZeroPage()) {
      int sk = PickSplitK(rsc, g.K, g.M());
      if (float* arena =
              SplitKArena(ctx->device()->gpu_ordinal(), rsc * g.K, s)) {
        OP_HIP_OK(ctx, stf_conv2d_dw_splitk(
                           x.raw_data(), dy.raw_data(), arena, ZeroPage(),
                           (int)g.N, (int)g.H, (int)g.W, (int)g.C, (int)g.R,
                           (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                           (int)g.pw, (int)g.P, (int)g.Q, g.K, sk, s));
        OP_HIP_OK(ctx, stf_cast_f32_bf16_zero(arena, dw->raw_data(),
                                              rsc * g.K, s));
        return;
      }
      Tensor scratch = ctx->allocate_temp(DT_FLOAT, TensorShape({rsc, g.K}));
      OP_HIP_OK(ctx, hipMemsetAsync(scratch.raw_data(), 0, rsc * g.K * 4, s));
      OP_HIP_OK(ctx, stf_conv2d_dw_splitk(
                         x.raw_data(), dy.raw_data(), scratch.raw_data(),
                         ZeroPage(), (int)g.N, (int)g.H, (int)g.W, (int)g.C,
                         (int)g.R, (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                         (int)g.pw, (int)g.P, (int)g.Q, g.K, sk, s));
      OP_HIP_OK(ctx, stf_cast(0, 1, scratch.raw_data(), dw->raw_data(),
                              rsc * g.K, s));
      return;
    }
    bool km_ok = (g.K & 7) == 0 && (!g.is_1x1_s1() || (g.C & 7) == 0);
    if (km_ok) {
      const void* col_data = x.raw_data();
      int64_t lda = g.C;
      Tensor col;
      if (!g.is_1x1_s1()) {
        int64_t rscp = g.RSCp();
        lda = rscp;
        bool cached = false;
        if (ColCacheResource* cc = GetColCache(ctx)) {
          std::lock_guard<std::mutex> l(cc->mu);
          auto it = cc->m.find({x.raw_data(), ConvGeomHash(g)});
          if (it != cc->m.end()) {
            col = it->second;  // forward's columns, same step, same x
            cached = true;
          }
        }
        if (!cached) {
          col = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.M(), rscp}));
          if (rscp != rsc)
            OP_HIP_OK(ctx,
                      hipMemsetAsync(col.raw_data(), 0, g.M() * rscp * 2, s));
          OP_HIP_OK(ctx, stf_im2col_bf16(x.raw_data(), col.raw_data(),
                                         (int)g.N, (int)g.H, (int)g.W,
                                         (int)g.C, (int)g.R, (int)g.S,
                                         (int)g.sh, (int)g.sw, (int)g.ph,
                                         (int)g.pw, (int)g.P, (int)g.Q, rscp,
                                         s));
        }
        col_data = col.raw_data();
      }
      OP_HIP_OK(ctx, GemmBf16AutoEx(ctx, col_data, dy.raw_data(),
                                    dw->raw_data(), rsc, g.K, g.M(), lda,
                                    g.K, 1, 1, s));
      return;
    }
    Tensor colT = ctx->allocate_temp(DT_BFLOAT16, TensorShape({rsc, g.M()}));
    if (g.is_1x1_s1()) {
      OP_HIP_OK(ctx, stf_transpose2d(2, x.raw_data(), colT.raw_data(), g.M(),
                                     rsc, s));
    } else {
      Tensor col = ctx->allocate_temp(DT_BFLOAT16,
                                      TensorShape({g.M(), rsc}));
      OP_HIP_OK(ctx, stf_im2col_bf16(x.raw_data(), col.raw_data(), (int)g.N,
                                     (int)g.H, (int)g.W, (int)g.C, (int)g.R,
                                     (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph,
                                     (int)g.pw, (int)g.P, (int)g.Q, rsc, s));
      OP_HIP_OK(ctx, stf_transpose2d(2, col.raw_data(), colT.raw_data(),
                                     g.M(), rsc, s));
    }
    Tensor dyT = ctx->allocate_temp(DT_BFLOAT16, TensorShape({g.K, g.M()}));
    OP_HIP_OK(ctx, stf_transpose2d(2, dy.raw_data(), dyT.raw_data(), g.M(),
                                   g.K, s));
    OP_HIP_OK(ctx, GemmBf16Auto(ctx, colT.raw_data(), dyT.raw_data(),
                                dw->raw_data(), rsc, g.K, g.M(), s));
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("Conv2DBackpropFilter").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("filter_sizes"), GpuConv2DBackpropFilterOp);

// ---------------------------------------------------------------------------
// depthwise conv (reference depthwise_conv_op_gpu.cu.cc family)
// ---------------------------------------------------------------------------
static Status DwGeomFromShapes(const TensorShape& x_shape,
                               const TensorShape& f_shape,
                               const std::vector<int64_t>& strides,
                               const std::string& padding, GpuConvGeom* g) {
  STF_RETURN_IF_ERROR(
      GetConvGeom(x_shape, f_shape, strides, padding, g));
  return Status::OK();
}

class GpuDepthwiseConvOp : public OpKernel {
 public:
  explicit GpuDepthwiseConvOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& w = ctx->input(1);  // [R, S, C, mult]
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, DwGeomFromShapes(x.shape(), w.shape(), strides_,
                                         padding_, &g));
    int mult = (int)w.dim_size(3);
    Tensor* y = ctx->allocate_output(
        0, TensorShape({g.N, g.P, g.Q, g.C * mult}));
    OP_HIP_OK(ctx, stf_depthwise_fwd(
                       x.raw_data(), w.raw_data(), y->raw_data(), (int)g.N,
                       (int)g.H, (int)g.W, (int)g.C, (int)g.R, (int)g.S,
                       (int)g.sh, (int)g.sw, (int)g.ph, (int)g.pw, (int)g.P,
                       (int)g.Q, mult, GPU_STREAM(ctx)));
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("DepthwiseConv2dNative").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuDepthwiseConvOp);

class GpuDepthwiseConvBackpropInputOp : public OpKernel {
 public:
  explicit GpuDepthwiseConvBackpropInputOp(OpKernelConstruction* c)
      : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto in_sizes = IntVector(ctx->input(0));
    const Tensor& w = ctx->input(1);
    const Tensor& dy = ctx->input(2);
    TensorShape x_shape(in_sizes);
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, DwGeomFromShapes(x_shape, w.shape(), strides_,
                                         padding_, &g));
    int mult = (int)w.dim_size(3);
    Tensor* dx = ctx->allocate_output(0, x_shape);
    OP_HIP_OK(ctx, stf_depthwise_bwd_input(
                       dy.raw_data(), w.raw_data(), dx->raw_data(), (int)g.N,
                       (int)g.H, (int)g.W, (int)g.C, (int)g.R, (int)g.S,
                       (int)g.sh, (int)g.sw, (int)g.ph, (int)g.pw, (int)g.P,
                       (int)g.Q, mult, GPU_STREAM(ctx)));
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("DepthwiseConv2dNativeBackpropInput").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("input_sizes"), GpuDepthwiseConvBackpropInputOp);

class GpuDepthwiseConvBackpropFilterOp : public OpKernel {
 public:
  explicit GpuDepthwiseConvBackpropFilterOp(OpKernelConstruction* c)
      : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto f_sizes = IntVector(ctx->input(1));
    const Tensor& dy = ctx->input(2);
    TensorShape f_shape(f_sizes);
    GpuConvGeom g;
    OP_REQUIRES_OK(ctx, DwGeomFromShapes(x.shape(), f_shape, strides_,
                                         padding_, &g));
    int mult = (int)f_shape.dim_size(3);
    hipStream_t s = GPU_STREAM(ctx);
    Tensor* dw = ctx->allocate_output(0, f_shape);
    Tensor scratch = ctx->allocate_temp(
        DT_FLOAT, TensorShape({f_shape.num_elements()}));
    OP_HIP_OK(ctx, ZeroF32(scratch.raw_data(), f_shape.num_elements(), s));
    OP_HIP_OK(ctx, stf_depthwise_bwd_filter(
                       x.raw_data(), dy.raw_data(), scratch.flat<float>(),
                       (int)g.N, (int)g.H, (int)g.W, (int)g.C, (int)g.R,
                       (int)g.S, (int)g.sh, (int)g.sw, (int)g.ph, (int)g.pw,
                       (int)g.P, (int)g.Q, mult, s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(dw->dtype()), scratch.raw_data(),
                            dw->raw_data(), f_shape.num_elements(), s));
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("DepthwiseConv2dNativeBackpropFilter").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("filter_sizes"), GpuDepthwiseConvBackpropFilterOp);

// ---------------------------------------------------------------------------
// bias / softmax / xent
// ---------------------------------------------------------------------------
class GpuBiasAddOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& b = ctx->input(1);
    Tensor* y = ctx->allocate_output(0, x.shape());
    OP_HIP_OK(ctx, stf_bias_add(DtypeCode(x.dtype()), x.raw_data(),
                                b.raw_data(), y->raw_data(), x.NumElements(),
                                (int)b.NumElements(), GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("BiasAdd").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuBiasAddOp);
REGISTER_KERNEL_BUILDER(Name("BiasAdd").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBiasAddOp);

class GpuBiasAddGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& dy = ctx->input(0);
    int c = (int)dy.dim_size(dy.dims() - 1);
    int64_t rows = dy.NumElements() / c;
    Tensor* db = ctx->allocate_output(0, TensorShape({c}));
    hipStream_t s = GPU_STREAM(ctx);
    Tensor scratch = ctx->allocate_temp(DT_FLOAT, TensorShape({c}));
    OP_HIP_OK(ctx, ZeroF32(scratch.raw_data(), c, s));
    OP_HIP_OK(ctx, stf_bias_grad(DtypeCode(dy.dtype()), dy.raw_data(),
                                 scratch.raw_data(), rows, c, s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(db->dtype()), scratch.raw_data(),
                            db->raw_data(), c, s));
  }
};
REGISTER_KERNEL_BUILDER(Name("BiasAddGrad").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuBiasAddGradOp);
REGISTER_KERNEL_BUILDER(Name("BiasAddGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBiasAddGradOp);

class GpuSoftmaxOp : public OpKernel {
 public:
  GpuSoftmaxOp(OpKernelConstruction* c, bool log_sm)
      : OpKernel(c), log_(log_sm) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* y = ctx->allocate_output(0, x.shape());
    int cols = (int)x.dim_size(x.dims() - 1);
    int64_t rows = x.NumElements() / cols;
    OP_HIP_OK(ctx, stf_softmax(DtypeCode(x.dtype()), log_, x.raw_data(),
                               y->raw_data(), rows, cols, GPU_STREAM(ctx)));
  }

 private:
  bool log_;
};
class SoftmaxGpu : public GpuSoftmaxOp {
 public:
  explicit SoftmaxGpu(OpKernelConstruction* c) : GpuSoftmaxOp(c, false) {}
};
class LogSoftmaxGpu : public GpuSoftmaxOp {
 public:
  explicit LogSoftmaxGpu(OpKernelConstruction* c) : GpuSoftmaxOp(c, true) {}
};
REGISTER_KERNEL_BUILDER(Name("Softmax").Device(DEVICE_GPU).TypeConstraint<float>("T"), SoftmaxGpu);
REGISTER_KERNEL_BUILDER(Name("Softmax").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), SoftmaxGpu);
REGISTER_KERNEL_BUILDER(Name("LogSoftmax").Device(DEVICE_GPU).TypeConstraint<float>("T"), LogSoftmaxGpu);

class GpuSparseXentOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& logits = ctx->input(0);
    const Tensor& labels = ctx->input(1);
    int64_t rows = logits.dim_size(0);
    int cols = (int)logits.dim_size(1);
    Tensor* loss = ctx->allocate_output(0, TensorShape({rows}));
    Tensor* bp = ctx->allocate_output(1, logits.shape());
    hipStream_t s = GPU_STREAM(ctx);
    Tensor loss_f32 = ctx->allocate_temp(DT_FLOAT, TensorShape({rows}));
    OP_HIP_OK(ctx,
              stf_sparse_xent(DtypeCode(logits.dtype()), logits.raw_data(),
                              labels.raw_data(), labels.dtype() == DT_INT32,
                              loss_f32.raw_data(), bp->raw_data(), rows, cols,
                              s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(loss->dtype()), loss_f32.raw_data(),
                            loss->raw_data(), rows, s));
  }
};
REGISTER_KERNEL_BUILDER(Name("SparseSoftmaxCrossEntropyWithLogits").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuSparseXentOp);
REGISTER_KERNEL_BUILDER(Name("SparseSoftmaxCrossEntropyWithLogits").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuSparseXentOp);

class GpuXentOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& logits = ctx->input(0);
    const Tensor& labels = ctx->input(1);
    int64_t rows = logits.dim_size(0);
    int cols = (int)logits.dim_size(1);
    Tensor* loss = ctx->allocate_output(0, TensorShape({rows}));
    Tensor* bp = ctx->allocate_output(1, logits.shape());
    hipStream_t s = GPU_STREAM(ctx);
    Tensor loss_f32 = ctx->allocate_temp(DT_FLOAT, TensorShape({rows}));
    OP_HIP_OK(ctx, stf_xent(DtypeCode(logits.dtype()), logits.raw_data(),
                            labels.raw_data(), loss_f32.raw_data(),
                            bp->raw_data(), rows, cols, s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(loss->dtype()), loss_f32.raw_data(),
                            loss->raw_data(), rows, s));
  }
};
REGISTER_KERNEL_BUILDER(Name("SoftmaxCrossEntropyWithLogits").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuXentOp);
REGISTER_KERNEL_BUILDER(Name("SoftmaxCrossEntropyWithLogits").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuXentOp);

// ---------------------------------------------------------------------------
// batch norm
// ---------------------------------------------------------------------------
class GpuBatchNormMiOp : public OpKernel {
 public:
  explicit GpuBatchNormMiOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("epsilon", &eps_);
    c->GetAttr("fuse_relu", &fuse_relu_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& scale = ctx->input(1);
    const Tensor& offset = ctx->input(2);
    int c = (int)x.dim_size(x.dims() - 1);
    int64_t rows = x.NumElements() / c;
    Tensor* y = ctx->allocate_output(0, x.shape());
    Tensor* mean = ctx->allocate_output(1, TensorShape({c}));
    Tensor* var = ctx->allocate_output(2, TensorShape({c}));
    Tensor* inv_std = ctx->allocate_output(3, TensorShape({c}));
    hipStream_t s = GPU_STREAM(ctx);
    Tensor acc = ctx->allocate_temp(DT_FLOAT, TensorShape({2 * c}));
    OP_HIP_OK(ctx, ZeroF32(acc.raw_data(), 2 * c, s));
    OP_HIP_OK(ctx, stf_bn_fwd(DtypeCode(x.dtype()), x.raw_data(),
                              scale.raw_data(), offset.raw_data(),
                              acc.flat<float>(), mean->flat<float>(),
                              var->flat<float>(), inv_std->flat<float>(),
                              y->raw_data(), rows, c, eps_,
                              fuse_relu_ ? 1 : 0, s));
  }

 private:
  float eps_ = 1e-4f;
  bool fuse_relu_ = false;
};
REGISTER_KERNEL_BUILDER(Name("BatchNormMi").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuBatchNormMiOp);
REGISTER_KERNEL_BUILDER(Name("BatchNormMi").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBatchNormMiOp);

// BN + residual-add + relu fused (reference has no equivalent; the CDNA4
// win is one elementwise pass instead of three over a [N,H,W,C] tensor).
class GpuBatchNormAddReluMiOp : public OpKernel {
 public:
  explicit GpuBatchNormAddReluMiOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("epsilon", &eps_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& scale = ctx->input(1);
    const Tensor& offset = ctx->input(2);
    const Tensor& side = ctx->input(3);
    int c = (int)x.dim_size(x.dims() - 1);
    int64_t rows = x.NumElements() / c;
    Tensor* y = ctx->allocate_output(0, x.shape());
    Tensor* mean = ctx->allocate_output(1, TensorShape({c}));
    Tensor* var = ctx->allocate_output(2, TensorShape({c}));
    Tensor* inv_std = ctx->allocate_output(3, TensorShape({c}));
    hipStream_t s = GPU_STREAM(ctx);
    Tensor acc = ctx->allocate_temp(DT_FLOAT, TensorShape({2 * c}));
    OP_HIP_OK(ctx, ZeroF32(acc.raw_data(), 2 * c, s));
    // stats + finalize from the plain BN path, then the fused normalize
    OP_HIP_OK(ctx, stf_bn_stats_only(DtypeCode(x.dtype()), x.raw_data(),
                                     acc.flat<float>(), mean->flat<float>(),
                                     var->flat<float>(),
                                     inv_std->flat<float>(), rows, c, eps_,
                                     s));
    OP_HIP_OK(ctx, stf_bn_add_relu(x.raw_data(), side.raw_data(),
                                   mean->flat<float>(),
                                   inv_std->flat<float>(), scale.raw_data(),
                                   offset.raw_data(), y->raw_data(),
                                   rows * (int64_t)c, c, s));
  }

 private:
  float eps_ = 1e-4f;
};
REGISTER_KERNEL_BUILDER(Name("BatchNormAddReluMi").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBatchNormAddReluMiOp);

class GpuBatchNormMiGradOp : public OpKernel {
 public:
  explicit GpuBatchNormMiGradOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("fuse_relu", &fuse_relu_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& dy = ctx->input(0);
    const Tensor& x = ctx->input(1);
    const Tensor& scale = ctx->input(2);
    const Tensor& mean = ctx->input(3);
    const Tensor& inv_std = ctx->input(4);
    const Tensor& y_relu = ctx->input(5);
    int c = (int)x.dim_size(x.dims() - 1);
    int64_t rows = x.NumElements() / c;
    Tensor* dx = ctx->allocate_output(0, x.shape());
    Tensor* dscale = ctx->allocate_output(1, TensorShape({c}));
    Tensor* doffset = ctx->allocate_output(2, TensorShape({c}));
    hipStream_t s = GPU_STREAM(ctx);
    // stats accumulate straight into the gradient outputs (sum_dy ==
    // doffset, sum_dy_xhat == dscale) — no scratch, no d2d copies.
    OP_HIP_OK(ctx, ZeroF32(doffset->raw_data(), c, s));
    OP_HIP_OK(ctx, ZeroF32(dscale->raw_data(), c, s));
    OP_HIP_OK(ctx, stf_bn_bwd(DtypeCode(x.dtype()), dy.raw_data(),
                              x.raw_data(), y_relu.raw_data(),
                              mean.flat<float>(), inv_std.flat<float>(),
                              scale.raw_data(), doffset->flat<float>(),
                              dscale->flat<float>(), dx->raw_data(), rows, c,
                              fuse_relu_ ? 1 : 0, s));
  }

 private:
  bool fuse_relu_ = false;
};
class GpuBatchNormAddReluMiGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& dy = ctx->input(0);
    const Tensor& x = ctx->input(1);
    const Tensor& scale = ctx->input(2);
    const Tensor& mean = ctx->input(3);
    const Tensor& inv_std = ctx->input(4);
    const Tensor& y_relu = ctx->input(5);
    int c = (int)x.dim_size(x.dims() - 1);
    int64_t rows = x.NumElements() / c;
    Tensor* dx = ctx->allocate_output(0, x.shape());
    Tensor* dscale = ctx->allocate_output(1, TensorShape({c}));
    Tensor* doffset = ctx->allocate_output(2, TensorShape({c}));
    Tensor* dside = ctx->allocate_output(3, x.shape());
    hipStream_t s = GPU_STREAM(ctx);
    OP_HIP_OK(ctx, ZeroF32(doffset->raw_data(), c, s));
    OP_HIP_OK(ctx, ZeroF32(dscale->raw_data(), c, s));
    OP_HIP_OK(ctx, stf_bn_bwd(DtypeCode(x.dtype()), dy.raw_data(),
                              x.raw_data(), y_relu.raw_data(),
                              mean.flat<float>(), inv_std.flat<float>(),
                              scale.raw_data(), doffset->flat<float>(),
                              dscale->flat<float>(), dx->raw_data(), rows, c,
                              1, s));
    // dside = relu-masked dy
    OP_HIP_OK(ctx, stf_binary(B_RELU_GRAD, DtypeCode(x.dtype()),
                              dy.raw_data(), y_relu.raw_data(),
                              dside->raw_data(), x.NumElements(), s));
  }
};
REGISTER_KERNEL_BUILDER(Name("BatchNormAddReluMiGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBatchNormAddReluMiGradOp);

REGISTER_KERNEL_BUILDER(Name("BatchNormMiGrad").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuBatchNormMiGradOp);
REGISTER_KERNEL_BUILDER(Name("BatchNormMiGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuBatchNormMiGradOp);

// ---------------------------------------------------------------------------
// pooling
// ---------------------------------------------------------------------------
struct PoolAttrs {
  std::vector<int64_t> ksize, strides;
  std::string padding;
};
static void GetPool(OpKernelConstruction* c, PoolAttrs* p) {
  c->GetAttr("ksize", &p->ksize);
  c->GetAttr("strides", &p->strides);
  c->GetAttr("padding", &p->padding);
}
static void PoolDims(const TensorShape& x, const PoolAttrs& p, int64_t* P,
                     int64_t* Q, int64_t* ph, int64_t* pw) {
  int64_t H = x.dim_size(1), W = x.dim_size(2);
  int64_t kh = p.ksize[1], kw = p.ksize[2], sh = p.strides[1],
          sw = p.strides[2];
  if (p.padding == "SAME") {
    *P = (H + sh - 1) / sh;
    *Q = (W + sw - 1) / sw;
    *ph = std::max<int64_t>(0, (*P - 1) * sh + kh - H) / 2;
    *pw = std::max<int64_t>(0, (*Q - 1) * sw + kw - W) / 2;
  } else {
    *P = (H - kh) / sh + 1;
    *Q = (W - kw) / sw + 1;
    *ph = *pw = 0;
  }
}

class GpuPoolOp : public OpKernel {
 public:
  GpuPoolOp(OpKernelConstruction* c, bool is_max)
      : OpKernel(c), is_max_(is_max) {
    GetPool(c, &p_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t P, Q, ph, pw;
    PoolDims(x.shape(), p_, &P, &Q, &ph, &pw);
    Tensor* y = ctx->allocate_output(
        0, TensorShape({x.dim_size(0), P, Q, x.dim_size(3)}));
    OP_HIP_OK(ctx, stf_pool_fwd(DtypeCode(x.dtype()), is_max_, x.raw_data(),
                                y->raw_data(), (int)x.dim_size(0),
                                (int)x.dim_size(1), (int)x.dim_size(2),
                                (int)x.dim_size(3), (int)p_.ksize[1],
                                (int)p_.ksize[2], (int)p_.strides[1],
                                (int)p_.strides[2], (int)ph, (int)pw, (int)P,
                                (int)Q, GPU_STREAM(ctx)));
  }

 private:
  bool is_max_;
  PoolAttrs p_;
};
class MaxPoolGpu : public GpuPoolOp {
 public:
  explicit MaxPoolGpu(OpKernelConstruction* c) : GpuPoolOp(c, true) {}
};
class AvgPoolGpu : public GpuPoolOp {
 public:
  explicit AvgPoolGpu(OpKernelConstruction* c) : GpuPoolOp(c, false) {}
};
REGISTER_KERNEL_BUILDER(Name("MaxPool").Device(DEVICE_GPU).TypeConstraint<float>("T"), MaxPoolGpu);
REGISTER_KERNEL_BUILDER(Name("MaxPool").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), MaxPoolGpu);
REGISTER_KERNEL_BUILDER(Name("AvgPool").Device(DEVICE_GPU).TypeConstraint<float>("T"), AvgPoolGpu);
REGISTER_KERNEL_BUILDER(Name("AvgPool").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), AvgPoolGpu);

class GpuMaxPoolGradOp : public OpKernel {
 public:
  explicit GpuMaxPoolGradOp(OpKernelConstruction* c) : OpKernel(c) {
    GetPool(c, &p_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& dy = ctx->input(2);
    int64_t P, Q, ph, pw;
    PoolDims(x.shape(), p_, &P, &Q, &ph, &pw);
    Tensor* dx = ctx->allocate_output(0, x.shape());
    hipStream_t s = GPU_STREAM(ctx);
    if (x.dtype() == DT_BFLOAT16 && x.dim_size(3) % 8 == 0) {
      // per-input direct path: no f32 scratch, no zero fill, no cast
      OP_HIP_OK(ctx, stf_max_pool_bwd_v8(
                         x.raw_data(), dy.raw_data(), dx->raw_data(),
                         (int)x.dim_size(0), (int)x.dim_size(1),
                         (int)x.dim_size(2), (int)x.dim_size(3),
                         (int)p_.ksize[1], (int)p_.ksize[2],
                         (int)p_.strides[1], (int)p_.strides[2], (int)ph,
                         (int)pw, (int)P, (int)Q, s));
      return;
    }
    Tensor scratch = ctx->allocate_temp(DT_FLOAT, x.shape());
    OP_HIP_OK(ctx, ZeroF32(scratch.raw_data(), x.NumElements(), s));
    OP_HIP_OK(ctx, stf_max_pool_bwd(DtypeCode(x.dtype()), x.raw_data(),
                                    dy.raw_data(), scratch.flat<float>(),
                                    (int)x.dim_size(0), (int)x.dim_size(1),
                                    (int)x.dim_size(2), (int)x.dim_size(3),
                                    (int)p_.ksize[1], (int)p_.ksize[2],
                                    (int)p_.strides[1], (int)p_.strides[2],
                                    (int)ph, (int)pw, (int)P, (int)Q, s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(dx->dtype()), scratch.raw_data(),
                            dx->raw_data(), x.NumElements(), s));
  }

 private:
  PoolAttrs p_;
};
REGISTER_KERNEL_BUILDER(Name("MaxPoolGrad").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuMaxPoolGradOp);
REGISTER_KERNEL_BUILDER(Name("MaxPoolGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuMaxPoolGradOp);

class GpuAvgPoolGradOp : public OpKernel {
 public:
  explicit GpuAvgPoolGradOp(OpKernelConstruction* c) : OpKernel(c) {
    GetPool(c, &p_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto in_sizes = IntVector(ctx->input(0));
    const Tensor& dy = ctx->input(1);
    TensorShape x_shape(in_sizes);
    int64_t P, Q, ph, pw;
    PoolDims(x_shape, p_, &P, &Q, &ph, &pw);
    Tensor* dx = ctx->allocate_output(0, x_shape);
    OP_HIP_OK(ctx, stf_avg_pool_bwd(DtypeCode(dy.dtype()), dy.raw_data(),
                                    dx->raw_data(), (int)x_shape.dim_size(0),
                                    (int)x_shape.dim_size(1),
                                    (int)x_shape.dim_size(2),
                                    (int)x_shape.dim_size(3),
                                    (int)p_.ksize[1], (int)p_.ksize[2],
                                    (int)p_.strides[1], (int)p_.strides[2],
                                    (int)ph, (int)pw, (int)P, (int)Q,
                                    GPU_STREAM(ctx)));
  }

 private:
  PoolAttrs p_;
};
REGISTER_KERNEL_BUILDER(Name("AvgPoolGrad").Device(DEVICE_GPU).TypeConstraint<float>("T").HostMemory("orig_input_shape"), GpuAvgPoolGradOp);
REGISTER_KERNEL_BUILDER(Name("AvgPoolGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("orig_input_shape"), GpuAvgPoolGradOp);

// ---------------------------------------------------------------------------
// reductions: Sum / Mean / Max (all-axes, trailing-axes, or leading-axes)
// ---------------------------------------------------------------------------
// L2Loss: 0.5 * sum(x^2) — square into a temp then full-reduce (the
// reference's l2loss_op_gpu; without this PTB's clip_by_global_norm dragged
// multi-MB gradients through a CPU-placed reduction at ~22 s/step).
class GpuL2LossOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    hipStream_t s = GPU_STREAM(ctx);
    int dt = DtypeCode(x.dtype());
    int64_t n = x.NumElements();
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    if (x.dtype() == DT_FLOAT) {
      OP_HIP_OK(ctx, ZeroF32(out->raw_data(), 1, s));
      OP_HIP_OK(ctx, stf_l2loss(dt, x.raw_data(), out->flat<float>(), n, s));
    } else {
      Tensor acc = ctx->allocate_temp(DT_FLOAT, TensorShape({}));
      OP_HIP_OK(ctx, ZeroF32(acc.raw_data(), 1, s));
      OP_HIP_OK(ctx, stf_l2loss(dt, x.raw_data(), acc.flat<float>(), n, s));
      OP_HIP_OK(ctx, stf_cast(0, CastCode(x.dtype()), acc.raw_data(),
                              out->raw_data(), 1, s));
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("L2Loss").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuL2LossOp);
REGISTER_KERNEL_BUILDER(Name("L2Loss").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuL2LossOp);


// _FusedElementwise: register-resident interpreter over the fused DAG
// (graph/optimizer.cc FuseElementwise; program format kernels/fused_ew.h).
class GpuFusedElementwiseOp : public OpKernel {
 public:
  explicit GpuFusedElementwiseOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("program", &prog_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n_in = ctx->num_inputs();
    const Tensor& root = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, root.shape());
    const void* ins[8];
    uint8_t scalar[8];
    for (int i = 0; i < n_in; ++i) {
      ins[i] = ctx->input(i).raw_data();
      scalar[i] = ctx->input(i).NumElements() == 1 ? 1 : 0;
    }
    OP_HIP_OK(ctx, stf_fused_elementwise(
                       DtypeCode(root.dtype()), ins, scalar, n_in,
                       prog_.data(), (int)prog_.size(), out->raw_data(),
                       root.NumElements(), GPU_STREAM(ctx)));
  }

 private:
  std::vector<int64_t> prog_;
};
REGISTER_KERNEL_BUILDER(Name("_FusedElementwise").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuFusedElementwiseOp);
REGISTER_KERNEL_BUILDER(Name("_FusedElementwise").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuFusedElementwiseOp);

// Fused LSTM cell pointwise (nn_kernels.hip LstmGatesKernel); one launch
// replaces the composed cell's ~17 elementwise/split launches per timestep.
class GpuLSTMGatesOp : public OpKernel {
 public:
  explicit GpuLSTMGatesOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("forget_bias", &forget_bias_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& gates = ctx->input(0);
    const Tensor& c_prev = ctx->input(1);
    int64_t B = c_prev.shape().dim_size(0);
    int64_t H = c_prev.shape().dim_size(1);
    void* outs[7];
    for (int k = 0; k < 7; ++k)
      outs[k] = ctx->allocate_output(k, c_prev.shape())->raw_data();
    OP_HIP_OK(ctx, stf_lstm_gates(DtypeCode(gates.dtype()), gates.raw_data(),
                                  c_prev.raw_data(), forget_bias_, outs[0],
                                  outs[1], outs[2], outs[3], outs[4], outs[5],
                                  outs[6], B, (int)H, GPU_STREAM(ctx)));
  }

 private:
  float forget_bias_ = 1.f;
};
REGISTER_KERNEL_BUILDER(Name("LSTMGates").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuLSTMGatesOp);
REGISTER_KERNEL_BUILDER(Name("LSTMGates").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuLSTMGatesOp);

class GpuLSTMGatesGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& c_prev = ctx->input(0);
    int64_t B = c_prev.shape().dim_size(0);
    int64_t H = c_prev.shape().dim_size(1);
    Tensor* dgates = ctx->allocate_output(0, TensorShape({B, 4 * H}));
    Tensor* dc_prev = ctx->allocate_output(1, c_prev.shape());
    OP_HIP_OK(ctx, stf_lstm_gates_grad(
                       DtypeCode(c_prev.dtype()), c_prev.raw_data(),
                       ctx->input(1).raw_data(), ctx->input(2).raw_data(),
                       ctx->input(3).raw_data(), ctx->input(4).raw_data(),
                       ctx->input(5).raw_data(), ctx->input(6).raw_data(),
                       ctx->input(7).raw_data(), dgates->raw_data(),
                       dc_prev->raw_data(), B, (int)H, GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("LSTMGatesGrad").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuLSTMGatesGradOp);
REGISTER_KERNEL_BUILDER(Name("LSTMGatesGrad").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T"), GpuLSTMGatesGradOp);

class GpuReduceOp : public OpKernel {
 public:
  GpuReduceOp(OpKernelConstruction* c, int red, bool mean)
      : OpKernel(c), red_(red), mean_(mean) {
    c->GetAttr("keep_dims", &keep_dims_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto axes_v = IntVector(ctx->input(1));
    int rank = x.dims();
    std::vector<bool> reduce(rank, false);
    for (auto a : axes_v) reduce[a < 0 ? a + rank : a] = true;
    hipStream_t s = GPU_STREAM(ctx);
    int dt = DtypeCode(x.dtype());
    TensorShape out_shape;
    int64_t count = 1;
    for (int i = 0; i < rank; ++i) {
      if (reduce[i]) {
        count *= x.dim_size(i);
        if (keep_dims_) out_shape.AddDim(1);
      } else {
        out_shape.AddDim(x.dim_size(i));
      }
    }
    int64_t out_n = out_shape.num_elements();
    if (axes_v.empty()) {
      // no reduction: reshape copy
      Tensor* out = ctx->allocate_output(0, out_shape);
      OP_HIP_OK(ctx, hipMemcpyAsync(out->raw_data(), x.raw_data(),
                                    x.TotalBytes(), hipMemcpyDeviceToDevice,
                                    s));
      return;
    }
    Tensor f32_out = ctx->allocate_temp(DT_FLOAT, TensorShape({out_n}));
    // classify
    bool all_red = count == x.NumElements();
    int first_keep = -1, last_red = -1, first_red = rank, contiguous = 1;
    for (int i = 0; i < rank; ++i) {
      if (reduce[i]) {
        if (i < first_red) first_red = i;
        last_red = i;
      }
    }
    bool trailing = last_red == rank - 1;
    bool leading = first_red == 0;
    for (int i = first_red; i <= last_red && i >= 0; ++i)
      if (!reduce[i]) contiguous = 0;
    if (all_red) {
      float init = red_ == 0 ? 0.f : (red_ == 1 ? -3.4e38f : 3.4e38f);
      OP_HIP_OK(ctx, stf_fill_f32(f32_out.raw_data(), init, 1, 0, s));
      OP_HIP_OK(ctx, stf_full_reduce(dt, red_, x.raw_data(),
                                     f32_out.flat<float>(), x.NumElements(),
                                     s));
    } else if (trailing && contiguous) {
      OP_HIP_OK(ctx, stf_row_reduce(dt, red_, x.raw_data(),
                                    f32_out.flat<float>(), out_n, count, s));
    } else if (leading && contiguous) {
      OP_HIP_OK(ctx, stf_col_reduce(dt, red_, x.raw_data(),
                                    f32_out.flat<float>(), count, out_n, s));
    } else {
      OP_REQUIRES(ctx, false,
                  errors::Unimplemented(
                      "GPU reduce supports all/leading/trailing axes"));
    }
    if (mean_ && count > 0) {
      OP_HIP_OK(ctx, stf_scale(0, f32_out.raw_data(), f32_out.raw_data(),
                               out_n, 1.f / (float)count, s));
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    OP_HIP_OK(ctx, stf_cast(0, CastCode(out->dtype()), f32_out.raw_data(),
                            out->raw_data(), out_n, s));
  }

 private:
  int red_;
  bool mean_;
  bool keep_dims_ = false;
};
class SumGpu : public GpuReduceOp {
 public:
  explicit SumGpu(OpKernelConstruction* c) : GpuReduceOp(c, 0, false) {}
};
class MeanGpu : public GpuReduceOp {
 public:
  explicit MeanGpu(OpKernelConstruction* c) : GpuReduceOp(c, 0, true) {}
};
class MaxGpu : public GpuReduceOp {
 public:
  explicit MaxGpu(OpKernelConstruction* c) : GpuReduceOp(c, 1, false) {}
};
class MinGpu : public GpuReduceOp {
 public:
  explicit MinGpu(OpKernelConstruction* c) : GpuReduceOp(c, 2, false) {}
};
#define REG_GPU_REDUCE(OP, CLS)                                                \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<float>("T").HostMemory("reduction_indices"), CLS); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("reduction_indices"), CLS);
REG_GPU_REDUCE("Sum", SumGpu)
REG_GPU_REDUCE("Mean", MeanGpu)
REG_GPU_REDUCE("Max", MaxGpu)
REG_GPU_REDUCE("Min", MinGpu)
#undef REG_GPU_REDUCE

// ---------------------------------------------------------------------------
// Tile (broadcast only: tiled dims must have input size 1) + Transpose
// ---------------------------------------------------------------------------
class GpuTileOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto mult = IntVector(ctx->input(1));
    int rank = x.dims();
    TensorShape out_shape;
    std::vector<int64_t> strides(rank), dims(rank);
    int64_t s_acc = 1;
    for (int i = rank - 1; i >= 0; --i) {
      strides[i] = x.dim_size(i) == 1 ? 0 : s_acc;
      s_acc *= x.dim_size(i);
    }
    for (int i = 0; i < rank; ++i) {
      OP_REQUIRES(ctx, mult[i] == 1 || x.dim_size(i) == 1,
                  errors::Unimplemented("GPU Tile: only broadcast tiling"));
      dims[i] = x.dim_size(i) * mult[i];
      out_shape.AddDim(dims[i]);
    }
    Tensor* y = ctx->allocate_output(0, out_shape);
    OP_HIP_OK(ctx, stf_bcast_copy((int)DataTypeSize(x.dtype()), x.raw_data(),
                                  y->raw_data(), out_shape.num_elements(),
                                  rank, dims.data(), strides.data(),
                                  GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("Tile").Device(DEVICE_GPU).HostMemory("multiples"), GpuTileOp);

class GpuTransposeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto perm = IntVector(ctx->input(1));
    int rank = x.dims();
    TensorShape out_shape;
    std::vector<int64_t> in_strides(rank, 1), src_strides(rank), dims(rank);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * x.dim_size(i + 1);
    for (int i = 0; i < rank; ++i) {
      dims[i] = x.dim_size((int)perm[i]);
      src_strides[i] = in_strides[(int)perm[i]];
      out_shape.AddDim(dims[i]);
    }
    Tensor* y = ctx->allocate_output(0, out_shape);
    if (rank == 2 && perm[0] == 1) {
      OP_HIP_OK(ctx, stf_transpose2d((int)DataTypeSize(x.dtype()),
                                     x.raw_data(), y->raw_data(),
                                     x.dim_size(0), x.dim_size(1),
                                     GPU_STREAM(ctx)));
      return;
    }
    OP_HIP_OK(ctx, stf_permute((int)DataTypeSize(x.dtype()), x.raw_data(),
                               y->raw_data(), out_shape.num_elements(), rank,
                               dims.data(), src_strides.data(),
                               GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("Transpose").Device(DEVICE_GPU).HostMemory("perm"), GpuTransposeOp);

// ---------------------------------------------------------------------------
// state: Assign / AssignAdd / AssignSub
// ---------------------------------------------------------------------------
class GpuAssignOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor ref = ctx->input(0);
    const Tensor& value = ctx->input(1);
    OP_REQUIRES(ctx, ref.shape() == value.shape(),
                errors::InvalidArgument("Assign shape mismatch"));
    OP_HIP_OK(ctx, hipMemcpyAsync(ref.raw_data(), value.raw_data(),
                                  value.TotalBytes(), hipMemcpyDeviceToDevice,
                                  GPU_STREAM(ctx)));
    ctx->set_output(0, ref);
  }
};
REGISTER_KERNEL_BUILDER(Name("Assign").Device(DEVICE_GPU), GpuAssignOp);

template <int BOP>
class GpuAssignUpdateOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor ref = ctx->input(0);
    const Tensor& value = ctx->input(1);
    OP_HIP_OK(ctx, stf_binary(BOP, DtypeCode(ref.dtype()), ref.raw_data(),
                              value.raw_data(), ref.raw_data(),
                              ref.NumElements(), GPU_STREAM(ctx)));
    ctx->set_output(0, ref);
  }
};
REGISTER_KERNEL_BUILDER(Name("AssignAdd").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuAssignUpdateOp<B_ADD>);
REGISTER_KERNEL_BUILDER(Name("AssignSub").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuAssignUpdateOp<B_SUB>);

// ---------------------------------------------------------------------------
// optimizer applies (f32 vars)
// ---------------------------------------------------------------------------
class GpuApplySgdOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    const Tensor& lr = ctx->input(1);
    const Tensor& grad = ctx->input(2);
    OP_HIP_OK(ctx, stf_apply_sgd(grad.dtype() == DT_BFLOAT16, var.raw_data(),
                                 lr.raw_data(), grad.raw_data(),
                                 var.NumElements(), GPU_STREAM(ctx)));
    ctx->set_output(0, var);
  }
};
REGISTER_KERNEL_BUILDER(Name("ApplyGradientDescent").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuApplySgdOp);

class GpuApplyMomentumOp : public OpKernel {
 public:
  explicit GpuApplyMomentumOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("use_nesterov", &nesterov_);
  }
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor accum = ctx->input(1);
    const Tensor& lr = ctx->input(2);
    const Tensor& grad = ctx->input(3);
    const Tensor& mom = ctx->input(4);
    OP_HIP_OK(ctx, stf_apply_momentum(grad.dtype() == DT_BFLOAT16,
                                      var.raw_data(), accum.raw_data(),
                                      lr.raw_data(), grad.raw_data(),
                                      mom.raw_data(), nesterov_,
                                      var.NumElements(), GPU_STREAM(ctx)));
    ctx->set_output(0, var);
  }

 private:
  bool nesterov_ = false;
};
REGISTER_KERNEL_BUILDER(Name("ApplyMomentum").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuApplyMomentumOp);

class GpuApplyAdamOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor m = ctx->input(1);
    Tensor v = ctx->input(2);
    OP_HIP_OK(ctx, stf_apply_adam(
                       ctx->input(9).dtype() == DT_BFLOAT16, var.raw_data(),
                       m.raw_data(), v.raw_data(), ctx->input(3).raw_data(),
                       ctx->input(4).raw_data(), ctx->input(5).raw_data(),
                       ctx->input(6).raw_data(), ctx->input(7).raw_data(),
                       ctx->input(8).raw_data(), ctx->input(9).raw_data(),
                       var.NumElements(), GPU_STREAM(ctx)));
    ctx->set_output(0, var);
  }
};
REGISTER_KERNEL_BUILDER(Name("ApplyAdam").Device(DEVICE_GPU).TypeConstraint<float>("T"), GpuApplyAdamOp);

// ---------------------------------------------------------------------------
// random
// ---------------------------------------------------------------------------
class GpuRandomOp : public OpKernel {
 public:
  GpuRandomOp(OpKernelConstruction* c, int kind) : OpKernel(c), kind_(kind) {
    int64_t seed = 0, seed2 = 0;
    c->GetAttr("seed", &seed);
    c->GetAttr("seed2", &seed2);
    if (seed == 0 && seed2 == 0) seed = 0x9E3779B9;
    seed_ = ((uint64_t)seed << 32) | (uint32_t)seed2;
  }
  void Compute(OpKernelContext* ctx) override {
    auto dims = IntVector(ctx->input(0));
    Tensor* y = ctx->allocate_output(0, TensorShape(dims));
    int64_t n = y->NumElements();
    int bf16 = y->dtype() == DT_BFLOAT16;
    hipStream_t s = GPU_STREAM(ctx);
    {
      std::lock_guard<std::mutex> l(mu_);
      if (!ctr_.IsInitialized()) {
        // device-resident philox offset (advanced in-graph so hipGraph
        // replays still draw fresh numbers)
        ctr_ = Tensor(ctx->device()->allocator(), DT_INT64, TensorShape({1}));
        OP_HIP_OK(ctx, hipMemsetAsync(ctr_.raw_data(), 0, 8, s));
      }
    }
    if (kind_ == 0) {
      OP_HIP_OK(ctx, stf_random_uniform(seed_, ctr_.raw_data(), y->raw_data(),
                                        n, bf16, s));
    } else {
      OP_HIP_OK(ctx, stf_random_normal(seed_, ctr_.raw_data(), y->raw_data(),
                                       n, bf16, kind_ == 2, s));
    }
  }

 private:
  int kind_;
  uint64_t seed_;
  std::mutex mu_;
  Tensor ctr_;
};
class RandomUniformGpu : public GpuRandomOp {
 public:
  explicit RandomUniformGpu(OpKernelConstruction* c) : GpuRandomOp(c, 0) {}
};
class RandomNormalGpu : public GpuRandomOp {
 public:
  explicit RandomNormalGpu(OpKernelConstruction* c) : GpuRandomOp(c, 1) {}
};
class TruncatedNormalGpu : public GpuRandomOp {
 public:
  explicit TruncatedNormalGpu(OpKernelConstruction* c) : GpuRandomOp(c, 2) {}
};
REGISTER_KERNEL_BUILDER(Name("RandomUniform").Device(DEVICE_GPU).HostMemory("shape"), RandomUniformGpu);
REGISTER_KERNEL_BUILDER(Name("RandomStandardNormal").Device(DEVICE_GPU).HostMemory("shape"), RandomNormalGpu);
REGISTER_KERNEL_BUILDER(Name("TruncatedNormal").Device(DEVICE_GPU).HostMemory("shape"), TruncatedNormalGpu);

// ---------------------------------------------------------------------------
// concat / split / pack / unpack / gather / segment sum / slice
// ---------------------------------------------------------------------------
class GpuConcatV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs() - 1;
    int64_t axis = IntVector(ctx->input(n))[0];
    const Tensor& first = ctx->input(0);
    if (axis < 0) axis += first.dims();
    TensorShape out_shape = first.shape();
    int64_t total = 0;
    for (int k = 0; k < n; ++k) total += ctx->input(k).dim_size((int)axis);
    out_shape.set_dim((int)axis, total);
    Tensor* y = ctx->allocate_output(0, out_shape);
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= first.dim_size(i);
    for (int i = (int)axis + 1; i < first.dims(); ++i)
      inner *= first.dim_size(i);
    hipStream_t s = GPU_STREAM(ctx);
    int es = (int)DataTypeSize(first.dtype());
    int64_t off = 0;
    for (int k = 0; k < n; ++k) {
      const Tensor& t = ctx->input(k);
      int64_t rows = t.dim_size((int)axis);
      OP_HIP_OK(ctx, stf_strided_copy(es, 1, t.raw_data(), y->raw_data(),
                                      outer, rows, inner, total, off, s));
      off += rows;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("ConcatV2").Device(DEVICE_GPU).HostMemory("axis"), GpuConcatV2Op);

class GpuSplitOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    int64_t axis = IntVector(ctx->input(0))[0];
    const Tensor& in = ctx->input(1);
    if (axis < 0) axis += in.dims();
    int n = num_outputs();
    OP_REQUIRES(ctx, in.dim_size((int)axis) % n == 0,
                errors::InvalidArgument("Split axis not divisible"));
    int64_t part = in.dim_size((int)axis) / n;
    TensorShape out_shape = in.shape();
    out_shape.set_dim((int)axis, part);
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= in.dim_size(i);
    for (int i = (int)axis + 1; i < in.dims(); ++i) inner *= in.dim_size(i);
    hipStream_t s = GPU_STREAM(ctx);
    int es = (int)DataTypeSize(in.dtype());
    for (int k = 0; k < n; ++k) {
      Tensor* y = ctx->allocate_output(k, out_shape);
      OP_HIP_OK(ctx, stf_strided_copy(es, 0, in.raw_data(), y->raw_data(),
                                      outer, part, inner,
                                      in.dim_size((int)axis), k * part, s));
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Split").Device(DEVICE_GPU).HostMemory("split_dim"), GpuSplitOp);

class GpuPackOp : public OpKernel {
 public:
  explicit GpuPackOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("axis", &axis_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs();
    const Tensor& first = ctx->input(0);
    int axis = axis_ < 0 ? (int)(axis_ + first.dims() + 1) : (int)axis_;
    TensorShape out_shape = first.shape();
    out_shape.InsertDim(axis, n);
    Tensor* y = ctx->allocate_output(0, out_shape);
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= first.dim_size(i);
    for (int i = axis; i < first.dims(); ++i) inner *= first.dim_size(i);
    hipStream_t s = GPU_STREAM(ctx);
    int es = (int)DataTypeSize(first.dtype());
    for (int k = 0; k < n; ++k) {
      OP_HIP_OK(ctx, stf_strided_copy(es, 1, ctx->input(k).raw_data(),
                                      y->raw_data(), outer, 1, inner, n, k,
                                      s));
    }
  }

 private:
  int64_t axis_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("Pack").Device(DEVICE_GPU), GpuPackOp);

class GpuUnpackOp : public OpKernel {
 public:
  explicit GpuUnpackOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("axis", &axis_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    int axis = axis_ < 0 ? (int)(axis_ + in.dims()) : (int)axis_;
    int n = num_outputs();
    TensorShape out_shape = in.shape();
    out_shape.RemoveDim(axis);
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= in.dim_size(i);
    for (int i = axis + 1; i < in.dims(); ++i) inner *= in.dim_size(i);
    hipStream_t s = GPU_STREAM(ctx);
    int es = (int)DataTypeSize(in.dtype());
    for (int k = 0; k < n; ++k) {
      Tensor* y = ctx->allocate_output(k, out_shape);
      OP_HIP_OK(ctx, stf_strided_copy(es, 0, in.raw_data(), y->raw_data(),
                                      outer, 1, inner, n, k, s));
    }
  }

 private:
  int64_t axis_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("Unpack").Device(DEVICE_GPU), GpuUnpackOp);

class GpuGatherOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& params = ctx->input(0);
    const Tensor& idx = ctx->input(1);
    TensorShape out_shape = idx.shape();
    int64_t row = 1;
    for (int i = 1; i < params.dims(); ++i) {
      out_shape.AddDim(params.dim_size(i));
      row *= params.dim_size(i);
    }
    Tensor* y = ctx->allocate_output(0, out_shape);
    OP_HIP_OK(ctx, stf_gather_rows(DtypeCode(params.dtype()),
                                   idx.dtype() == DT_INT32, params.raw_data(),
                                   idx.raw_data(), y->raw_data(),
                                   idx.NumElements(), row, params.dim_size(0),
                                   GPU_STREAM(ctx)));
  }
};
REGISTER_KERNEL_BUILDER(Name("Gather").Device(DEVICE_GPU).TypeConstraint<float>("Tparams"), GpuGatherOp);
REGISTER_KERNEL_BUILDER(Name("Gather").Device(DEVICE_GPU).TypeConstraint<bfloat16>("Tparams"), GpuGatherOp);

class GpuUnsortedSegmentSumOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& data = ctx->input(0);
    const Tensor& ids = ctx->input(1);
    int64_t nseg = IntVector(ctx->input(2))[0];
    TensorShape out_shape({nseg});
    int64_t row = 1;
    for (int i = ids.dims(); i < data.dims(); ++i) {
      out_shape.AddDim(data.dim_size(i));
      row *= data.dim_size(i);
    }
    Tensor* y = ctx->allocate_output(0, out_shape);
    hipStream_t s = GPU_STREAM(ctx);
    Tensor scratch = ctx->allocate_temp(DT_FLOAT, TensorShape({nseg * row}));
    OP_HIP_OK(ctx, ZeroF32(scratch.raw_data(), nseg * row, s));
    OP_HIP_OK(ctx, stf_segment_sum(DtypeCode(data.dtype()),
                                   ids.dtype() == DT_INT32, data.raw_data(),
                                   ids.raw_data(), scratch.flat<float>(),
                                   ids.NumElements(), row, nseg, s));
    OP_HIP_OK(ctx, stf_cast(0, CastCode(y->dtype()), scratch.raw_data(),
                            y->raw_data(), nseg * row, s));
  }
};
REGISTER_KERNEL_BUILDER(Name("UnsortedSegmentSum").Device(DEVICE_GPU).TypeConstraint<float>("T").HostMemory("num_segments"), GpuUnsortedSegmentSumOp);
REGISTER_KERNEL_BUILDER(Name("UnsortedSegmentSum").Device(DEVICE_GPU).TypeConstraint<bfloat16>("T").HostMemory("num_segments"), GpuUnsortedSegmentSumOp);

// Slice on GPU (single sliced dim; degenerate = plain copy)
class GpuSliceOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto begin = IntVector(ctx->input(1));
    auto size = IntVector(ctx->input(2));
    int rank = in.dims();
    for (int i = 0; i < rank; ++i)
      if (size[i] == -1) size[i] = in.dim_size(i) - begin[i];
    Tensor* y = ctx->allocate_output(0, TensorShape(size));
    int cut = -1;
    for (int i = 0; i < rank; ++i) {
      if (begin[i] != 0 || size[i] != in.dim_size(i)) {
        OP_REQUIRES(ctx, cut == -1,
                    errors::Unimplemented("GPU Slice: one sliced dim"));
        cut = i;
      }
    }
    hipStream_t s = GPU_STREAM(ctx);
    if (cut == -1) {
      OP_HIP_OK(ctx, hipMemcpyAsync(y->raw_data(), in.raw_data(),
                                    in.TotalBytes(), hipMemcpyDeviceToDevice,
                                    s));
      return;
    }
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < cut; ++i) outer *= in.dim_size(i);
    for (int i = cut + 1; i < rank; ++i) inner *= in.dim_size(i);
    OP_HIP_OK(ctx, stf_strided_copy((int)DataTypeSize(in.dtype()), 0,
                                    in.raw_data(), y->raw_data(), outer,
                                    size[cut], inner, in.dim_size(cut),
                                    begin[cut], s));
  }
};
REGISTER_KERNEL_BUILDER(Name("Slice").Device(DEVICE_GPU).HostMemory("begin").HostMemory("size"), GpuSliceOp);

}  // namespace
}  // namespace stf
