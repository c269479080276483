// CPU math kernels: elementwise (broadcasting), reductions, MatMul, softmax,
// cross-entropy, bias. (capability analog of reference core/kernels/
// {cwise_ops,reduction_ops,matmul_op,softmax_op,xent_op,bias_op,aggregate_ops,
// argmax_op}.cc; CPU performance is not the hot path — the MI355X HIP kernels
// are — so these are straightforward loops used for tests and small graphs.)
#include <cmath>
#include <limits>

#include "kernels/fused_ew.h"
#include "kernels/kernel_util.h"

namespace stf {

// --------------------------- elementwise binary -----------------------------
template <typename T, typename F, typename OutT = T>
class BinaryOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& y = ctx->input(1);
    BCast b(x.shape(), y.shape());
    OP_REQUIRES(ctx, b.valid,
                errors::InvalidArgument("Incompatible shapes: ",
                                        x.shape().DebugString(), " vs ",
                                        y.shape().DebugString()));
    Tensor* out = ctx->allocate_output(0, b.out_shape());
    const T* xp = x.flat<T>();
    const T* yp = y.flat<T>();
    OutT* op = out->flat<OutT>();
    F f;
    if (x.shape() == y.shape()) {
      for (int64_t i = 0; i < b.num_elements; ++i) op[i] = f(xp[i], yp[i]);
    } else if (y.NumElements() == 1) {
      T s = yp[0];
      for (int64_t i = 0; i < b.num_elements; ++i) op[i] = f(xp[i], s);
    } else if (x.NumElements() == 1) {
      T s = xp[0];
      for (int64_t i = 0; i < b.num_elements; ++i) op[i] = f(s, yp[i]);
    } else {
      for (int64_t i = 0; i < b.num_elements; ++i) {
        int64_t xi, yi;
        b.Map(i, &xi, &yi);
        op[i] = f(xp[xi], yp[yi]);
      }
    }
  }
};

#define DEFINE_FUNCTOR(NAME, EXPR)                         \
  struct NAME {                                            \
    template <typename T>                                  \
    T operator()(T a, T b) const { return EXPR; }          \
  };
DEFINE_FUNCTOR(FAdd, a + b)
DEFINE_FUNCTOR(FSub, a - b)
DEFINE_FUNCTOR(FMul, a * b)
DEFINE_FUNCTOR(FDiv, a / b)
DEFINE_FUNCTOR(FMax, a > b ? a : b)
DEFINE_FUNCTOR(FMin, a < b ? a : b)
DEFINE_FUNCTOR(FSqDiff, (a - b) * (a - b))
#undef DEFINE_FUNCTOR
struct FPow {
  template <typename T>
  T operator()(T a, T b) const { return (T)std::pow((double)a, (double)b); }
};
struct FFloorDiv {
  float operator()(float a, float b) const { return std::floor(a / b); }
  double operator()(double a, double b) const { return std::floor(a / b); }
  int32_t operator()(int32_t a, int32_t b) const {
    int32_t q = a / b; return q * b != a && ((a < 0) != (b < 0)) ? q - 1 : q;
  }
  int64_t operator()(int64_t a, int64_t b) const {
    int64_t q = a / b; return q * b != a && ((a < 0) != (b < 0)) ? q - 1 : q;
  }
};
struct FFloorMod {
  template <typename T>
  T operator()(T a, T b) const {
    double m = std::fmod((double)a, (double)b);
    if (m != 0 && ((m < 0) != ((double)b < 0))) m += (double)b;
    return (T)m;
  }
};
#define DEFINE_CMP(NAME, OPR)                              \
  struct NAME {                                            \
    template <typename T>                                  \
    bool operator()(T a, T b) const { return a OPR b; }    \
  };
DEFINE_CMP(FLess, <)
DEFINE_CMP(FLessEq, <=)
DEFINE_CMP(FGreater, >)
DEFINE_CMP(FGreaterEq, >=)
DEFINE_CMP(FEq, ==)
DEFINE_CMP(FNe, !=)
#undef DEFINE_CMP

#define REG_BINARY(OP, F)                                                     \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), BinaryOp<float, F>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), BinaryOp<double, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), BinaryOp<int32_t, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), BinaryOp<int64_t, F>);
#define REG_BINARY_BF16(OP, F)                                                \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), BinaryOp<bfloat16, F>);
REG_BINARY_BF16("Add", FAdd)
REG_BINARY_BF16("Sub", FSub)
REG_BINARY_BF16("Mul", FMul)
REG_BINARY_BF16("RealDiv", FDiv)
REG_BINARY_BF16("Maximum", FMax)
REG_BINARY_BF16("Minimum", FMin)
#undef REG_BINARY_BF16
REG_BINARY("Add", FAdd)
REG_BINARY("Sub", FSub)
REG_BINARY("Mul", FMul)
REG_BINARY("Div", FDiv)
REG_BINARY("RealDiv", FDiv)
REG_BINARY("Maximum", FMax)
REG_BINARY("Minimum", FMin)
REG_BINARY("SquaredDifference", FSqDiff)
REG_BINARY("Pow", FPow)
REG_BINARY("FloorDiv", FFloorDiv)
REG_BINARY("FloorMod", FFloorMod)
#undef REG_BINARY
#define REG_CMP(OP, F)                                                        \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), BinaryOp<float, F, bool>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), BinaryOp<double, F, bool>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), BinaryOp<int32_t, F, bool>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), BinaryOp<int64_t, F, bool>);
REG_CMP("Less", FLess)
REG_CMP("LessEqual", FLessEq)
REG_CMP("Greater", FGreater)
REG_CMP("GreaterEqual", FGreaterEq)
REG_CMP("Equal", FEq)
REG_CMP("NotEqual", FNe)
#undef REG_CMP
// Host-compute int32/int64 registrations under the GPU device (all args in
// host memory). These keep while-loop counter arithmetic in the SAME
// partition as the GPU loop body — an in-frame cross-partition edge would
// need the reference's per-device control-loop machinery
// (graph_partition.cc:801), which round 1 does not implement. The analog of
// the reference's int32-on-GPU host kernels.
#define REG_GPU_HOST_INT(OP, F)                                               \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<int32_t>("T").HostMemory("x").HostMemory("y").HostMemory("z"), BinaryOp<int32_t, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<int64_t>("T").HostMemory("x").HostMemory("y").HostMemory("z"), BinaryOp<int64_t, F>);
REG_GPU_HOST_INT("Add", FAdd)
REG_GPU_HOST_INT("Sub", FSub)
REG_GPU_HOST_INT("Mul", FMul)
REG_GPU_HOST_INT("Maximum", FMax)
REG_GPU_HOST_INT("Minimum", FMin)
REG_GPU_HOST_INT("FloorDiv", FFloorDiv)
REG_GPU_HOST_INT("FloorMod", FFloorMod)
#undef REG_GPU_HOST_INT
#define REG_GPU_HOST_CMP(OP, F)                                               \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<int32_t>("T").HostMemory("x").HostMemory("y").HostMemory("z"), BinaryOp<int32_t, F, bool>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_GPU).TypeConstraint<int64_t>("T").HostMemory("x").HostMemory("y").HostMemory("z"), BinaryOp<int64_t, F, bool>);
REG_GPU_HOST_CMP("Less", FLess)
REG_GPU_HOST_CMP("LessEqual", FLessEq)
REG_GPU_HOST_CMP("Greater", FGreater)
REG_GPU_HOST_CMP("GreaterEqual", FGreaterEq)
REG_GPU_HOST_CMP("Equal", FEq)
REG_GPU_HOST_CMP("NotEqual", FNe)
#undef REG_GPU_HOST_CMP

struct FAnd { bool operator()(bool a, bool b) const { return a && b; } };
struct FOr { bool operator()(bool a, bool b) const { return a || b; } };
REGISTER_KERNEL_BUILDER(Name("LogicalAnd").Device(DEVICE_CPU), BinaryOp<bool, FAnd>);
REGISTER_KERNEL_BUILDER(Name("LogicalOr").Device(DEVICE_CPU), BinaryOp<bool, FOr>);

// --------------------------- elementwise unary ------------------------------
template <typename T, typename F, typename OutT = T>
class UnaryOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* xp = x.flat<T>();
    OutT* op = out->flat<OutT>();
    F f;
    for (int64_t i = 0; i < x.NumElements(); ++i) op[i] = f(xp[i]);
  }
};

#define DEFINE_UFUNC(NAME, EXPR)                          \
  struct NAME {                                           \
    template <typename T>                                 \
    T operator()(T a) const { return EXPR; }              \
  };
DEFINE_UFUNC(FNeg, -a)
DEFINE_UFUNC(FAbs, a < T(0) ? -a : a)
DEFINE_UFUNC(FSign, a > T(0) ? T(1) : (a < T(0) ? T(-1) : T(0)))
DEFINE_UFUNC(FSquare, a * a)
DEFINE_UFUNC(FRecip, T(1) / a)
#undef DEFINE_UFUNC
#define DEFINE_MATHFN(NAME, FN)                                   \
  struct NAME {                                                   \
    float operator()(float a) const { return FN##f(a); }          \
    double operator()(double a) const { return FN(a); }           \
  };
DEFINE_MATHFN(FSqrt, sqrt)
DEFINE_MATHFN(FExp, exp)
DEFINE_MATHFN(FLog, log)
DEFINE_MATHFN(FLog1p, log1p)
DEFINE_MATHFN(FTanh, tanh)
DEFINE_MATHFN(FSin, sin)
DEFINE_MATHFN(FCos, cos)
DEFINE_MATHFN(FFloor, floor)
DEFINE_MATHFN(FCeil, ceil)
DEFINE_MATHFN(FRound, rint)
#undef DEFINE_MATHFN
struct FRsqrt {
  float operator()(float a) const { return 1.0f / sqrtf(a); }
  double operator()(double a) const { return 1.0 / sqrt(a); }
};
struct FSigmoid {
  float operator()(float a) const { return 1.0f / (1.0f + expf(-a)); }
  double operator()(double a) const { return 1.0 / (1.0 + exp(-a)); }
};
struct FIsNan { template <typename T> bool operator()(T a) const { return std::isnan((double)a); } };
struct FIsInf { template <typename T> bool operator()(T a) const { return std::isinf((double)a); } };
struct FIsFinite { template <typename T> bool operator()(T a) const { return std::isfinite((double)a); } };

#define REG_UNARY_ALL(OP, F)                                                  \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), UnaryOp<float, F>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), UnaryOp<double, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), UnaryOp<int32_t, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), UnaryOp<int64_t, F>);
#define REG_UNARY_F(OP, F)                                                    \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), UnaryOp<float, F>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), UnaryOp<double, F>);
REG_UNARY_ALL("Neg", FNeg)
REG_UNARY_ALL("Abs", FAbs)
REG_UNARY_ALL("Sign", FSign)
REG_UNARY_ALL("Square", FSquare)
REG_UNARY_ALL("Reciprocal", FRecip)
#define REG_UNARY_BF16(OP, F)                                                 \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), UnaryOp<bfloat16, F>);
REG_UNARY_BF16("Tanh", FTanh)
REG_UNARY_BF16("Sigmoid", FSigmoid)
REG_UNARY_BF16("Exp", FExp)
REG_UNARY_BF16("Neg", FNeg)
REG_UNARY_BF16("Square", FSquare)
REG_UNARY_BF16("Sqrt", FSqrt)
REG_UNARY_BF16("Rsqrt", FRsqrt)
#undef REG_UNARY_BF16
REG_UNARY_F("Sqrt", FSqrt)
REG_UNARY_F("Rsqrt", FRsqrt)
REG_UNARY_F("Exp", FExp)
REG_UNARY_F("Log", FLog)
REG_UNARY_F("Log1p", FLog1p)
REG_UNARY_F("Tanh", FTanh)
REG_UNARY_F("Sigmoid", FSigmoid)
REG_UNARY_F("Sin", FSin)
REG_UNARY_F("Cos", FCos)
REG_UNARY_F("Floor", FFloor)
REG_UNARY_F("Ceil", FCeil)
REG_UNARY_F("Round", FRound)
struct FNot { bool operator()(bool a) const { return !a; } };
REGISTER_KERNEL_BUILDER(Name("LogicalNot").Device(DEVICE_CPU), UnaryOp<bool, FNot>);
#define REG_PRED(OP, F)                                                       \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), UnaryOp<float, F, bool>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), UnaryOp<double, F, bool>);
REG_PRED("IsNan", FIsNan)
REG_PRED("IsInf", FIsInf)
REG_PRED("IsFinite", FIsFinite)
#undef REG_PRED
#undef REG_UNARY_ALL
#undef REG_UNARY_F

// y/dy-style grads.
template <typename T>
struct FSigGrad { T operator()(T y, T dy) const { return dy * y * (T(1) - y); } };
template <typename T>
struct FTanhGrad { T operator()(T y, T dy) const { return dy * (T(1) - y * y); } };
template <typename T>
struct FRsqrtGrad { T operator()(T y, T dy) const { return T(-0.5) * dy * y * y * y; } };
template <typename T>
struct FSqrtGrad { T operator()(T y, T dy) const { return dy / (T(2) * y); } };
template <typename T>
struct FRecipGrad { T operator()(T y, T dy) const { return -dy * y * y; } };
template <typename T, template <typename> class F>
class Grad2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& y = ctx->input(0);
    const Tensor& dy = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, y.shape());
    const T* yp = y.flat<T>();
    const T* dp = dy.flat<T>();
    T* op = out->flat<T>();
    F<T> f;
    for (int64_t i = 0; i < y.NumElements(); ++i) op[i] = f(yp[i], dp[i]);
  }
};
#define REG_GRAD2(OP, F)                                                      \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), Grad2Op<float, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), Grad2Op<double, F>);
REGISTER_KERNEL_BUILDER(Name("SigmoidGrad").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), Grad2Op<bfloat16, FSigGrad>);
REGISTER_KERNEL_BUILDER(Name("TanhGrad").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), Grad2Op<bfloat16, FTanhGrad>);
REG_GRAD2("SigmoidGrad", FSigGrad)
REG_GRAD2("TanhGrad", FTanhGrad)
REG_GRAD2("RsqrtGrad", FRsqrtGrad)
REG_GRAD2("SqrtGrad", FSqrtGrad)
REG_GRAD2("ReciprocalGrad", FRecipGrad)
#undef REG_GRAD2

// -------------------------------- Select ------------------------------------
class SelectOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& c = ctx->input(0);
    const Tensor& t = ctx->input(1);
    const Tensor& e = ctx->input(2);
    Tensor* out = ctx->allocate_output(0, t.shape());
    const bool* cp = c.flat<bool>();
    size_t es = DataTypeSize(t.dtype());
    int64_t n = t.NumElements();
    // cond may be scalar, vector (batch), or same-shape.
    int64_t batch = c.NumElements();
    DispatchBySize(es, [&](auto tag) {
      using U = decltype(tag);
      const U* tp = t.flat<U>();
      const U* ep = e.flat<U>();
      U* op = out->flat<U>();
      if (batch == n) {
        for (int64_t i = 0; i < n; ++i) op[i] = cp[i] ? tp[i] : ep[i];
      } else if (batch == 1) {
        std::memcpy(op, cp[0] ? tp : ep, n * es);
      } else {
        int64_t row = n / batch;
        for (int64_t b = 0; b < batch; ++b)
          std::memcpy(op + b * row, (cp[b] ? tp : ep) + b * row, row * es);
      }
    });
  }
};
REGISTER_KERNEL_BUILDER(Name("Select").Device(DEVICE_CPU), SelectOp);


// ----------------------- fused elementwise (bytecode) -----------------------
// CPU interpreter for _FusedElementwise (see kernels/fused_ew.h).
template <typename T>
class FusedElementwiseOp : public OpKernel {
 public:
  explicit FusedElementwiseOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("program", &prog_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n_in = ctx->num_inputs();
    const Tensor& rootT = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, rootT.shape());
    int64_t n = rootT.NumElements();
    const T* in[fused_ew::kMaxInputs];
    bool scalar[fused_ew::kMaxInputs];
    for (int i = 0; i < n_in; ++i) {
      in[i] = ctx->input(i).flat<T>();
      scalar[i] = ctx->input(i).NumElements() == 1;
    }
    T* op = out->flat<T>();
    int np = (int)prog_.size();
    for (int64_t idx = 0; idx < n; ++idx) {
      float vals[fused_ew::kMaxInputs + fused_ew::kMaxInstr];
      for (int i = 0; i < n_in; ++i)
        vals[i] = (float)in[i][scalar[i] ? 0 : idx];
      for (int k = 0; k < np; ++k) {
        int opx, a, b;
        fused_ew::Unpack(prog_[k], &opx, &a, &b);
        float x = vals[a], y = vals[b], r = 0.f;
        using namespace fused_ew;
        switch (opx) {
          case kRelu: r = x > 0 ? x : 0; break;
          case kRelu6: r = x < 0 ? 0 : (x > 6 ? 6 : x); break;
          case kSigmoid: r = 1.f / (1.f + std::exp(-x)); break;
          case kTanh: r = std::tanh(x); break;
          case kExp: r = std::exp(x); break;
          case kLog: r = std::log(x); break;
          case kLog1p: r = std::log1p(x); break;
          case kNeg: r = -x; break;
          case kSqrt: r = std::sqrt(x); break;
          case kRsqrt: r = 1.f / std::sqrt(x); break;
          case kSquare: r = x * x; break;
          case kAbs: r = std::fabs(x); break;
          case kSoftplus: r = std::log1p(std::exp(-std::fabs(x))) +
                              (x > 0 ? x : 0); break;
          case kSign: r = x > 0 ? 1.f : (x < 0 ? -1.f : 0.f); break;
          case kFloor: r = std::floor(x); break;
          case kReciprocal: r = 1.f / x; break;
          case kAdd: r = x + y; break;
          case kSub: r = x - y; break;
          case kMul: r = x * y; break;
          case kDiv: r = x / y; break;
          case kMaximum: r = x > y ? x : y; break;
          case kMinimum: r = x < y ? x : y; break;
          case kSquaredDifference: r = (x - y) * (x - y); break;
          case kPow: r = std::pow(x, y); break;
        }
        vals[n_in + k] = r;
      }
      op[idx] = (T)vals[n_in + np - 1];
    }
  }

 private:
  std::vector<int64_t> prog_;
};
REGISTER_KERNEL_BUILDER(Name("_FusedElementwise").Device(DEVICE_CPU).TypeConstraint<float>("T"), FusedElementwiseOp<float>);
REGISTER_KERNEL_BUILDER(Name("_FusedElementwise").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), FusedElementwiseOp<bfloat16>);

// --------------------------------- AddN -------------------------------------
template <typename T>
class AddNOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& first = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, first.shape());
    T* op = out->flat<T>();
    int64_t n = first.NumElements();
    std::memcpy(op, first.raw_data(), first.TotalBytes());
    for (int k = 1; k < num_inputs(); ++k) {
      const T* p = ctx->input(k).flat<T>();
      for (int64_t i = 0; i < n; ++i) op[i] += p[i];
    }
  }
};
REGISTER_CPU_KERNEL_TYPES("AddN", AddNOp)
REGISTER_KERNEL_BUILDER(Name("AddN").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), AddNOp<bfloat16>);
REGISTER_KERNEL_BUILDER(Name("AddN").Device(DEVICE_CPU).TypeConstraint<std::complex<float>>("T"), AddNOp<std::complex<float>>);

// -------------------------------- MatMul ------------------------------------
template <typename T>
class MatMulOp : public OpKernel {
 public:
  explicit MatMulOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("transpose_a", &ta_);
    ctx->GetAttr("transpose_b", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    OP_REQUIRES(ctx, a.dims() == 2 && b.dims() == 2,
                errors::InvalidArgument("MatMul needs rank-2 inputs"));
    int64_t m = ta_ ? a.dim_size(1) : a.dim_size(0);
    int64_t k = ta_ ? a.dim_size(0) : a.dim_size(1);
    int64_t k2 = tb_ ? b.dim_size(1) : b.dim_size(0);
    int64_t n = tb_ ? b.dim_size(0) : b.dim_size(1);
    OP_REQUIRES(ctx, k == k2,
                errors::InvalidArgument("MatMul inner dims mismatch: ",
                                        a.shape().DebugString(), " x ",
                                        b.shape().DebugString()));
    Tensor* out = ctx->allocate_output(0, TensorShape({m, n}));
    const T* ap = a.flat<T>();
    const T* bp = b.flat<T>();
    T* cp = out->flat<T>();
    std::memset(cp, 0, out->TotalBytes());
    int64_t lda = a.dim_size(1), ldb = b.dim_size(1);
    // ikj loop, decent cache behavior without blocking.
    for (int64_t i = 0; i < m; ++i) {
      for (int64_t kk = 0; kk < k; ++kk) {
        T av = ta_ ? ap[kk * lda + i] : ap[i * lda + kk];
        if (av == T(0)) continue;
        if (!tb_) {
          const T* brow = bp + kk * ldb;
          T* crow = cp + i * n;
          for (int64_t j = 0; j < n; ++j) crow[j] += av * brow[j];
        } else {
          T* crow = cp + i * n;
          for (int64_t j = 0; j < n; ++j) crow[j] += av * bp[j * ldb + kk];
        }
      }
    }
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_CPU_KERNEL_FLOATS("MatMul", MatMulOp)

// bf16 CPU matmul: widen to f32, run the f32 loop, narrow (numerics match
// the GPU MFMA path's f32 accumulation).
class Bf16MatMulOp : public OpKernel {
 public:
  explicit Bf16MatMulOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("transpose_a", &ta_);
    ctx->GetAttr("transpose_b", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    int64_t m = ta_ ? a.dim_size(1) : a.dim_size(0);
    int64_t k = ta_ ? a.dim_size(0) : a.dim_size(1);
    int64_t n = tb_ ? b.dim_size(0) : b.dim_size(1);
    Tensor* out = ctx->allocate_output(0, TensorShape({m, n}));
    std::vector<float> af(a.NumElements()), bf(b.NumElements()),
        cf(m * n, 0.f);
    const bfloat16* ap = a.flat<bfloat16>();
    const bfloat16* bp = b.flat<bfloat16>();
    for (int64_t i = 0; i < a.NumElements(); ++i) af[i] = (float)ap[i];
    for (int64_t i = 0; i < b.NumElements(); ++i) bf[i] = (float)bp[i];
    int64_t lda = a.dim_size(1), ldb = b.dim_size(1);
    for (int64_t i = 0; i < m; ++i)
      for (int64_t kk = 0; kk < k; ++kk) {
        float av = ta_ ? af[kk * lda + i] : af[i * lda + kk];
        for (int64_t j = 0; j < n; ++j)
          cf[i * n + j] += av * (tb_ ? bf[j * ldb + kk] : bf[kk * ldb + j]);
      }
    bfloat16* cp = out->flat<bfloat16>();
    for (int64_t i = 0; i < m * n; ++i) cp[i] = bfloat16(cf[i]);
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_KERNEL_BUILDER(Name("MatMul").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), Bf16MatMulOp);

// BatchMatMul: [..., M, K] x [..., K, N].
template <typename T>
class BatchMatMulOp : public OpKernel {
 public:
  explicit BatchMatMulOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("adj_x", &ta_);
    ctx->GetAttr("adj_y", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    int ra = a.dims();
    int64_t m = ta_ ? a.dim_size(ra - 1) : a.dim_size(ra - 2);
    int64_t k = ta_ ? a.dim_size(ra - 2) : a.dim_size(ra - 1);
    int64_t n = tb_ ? b.dim_size(b.dims() - 2) : b.dim_size(b.dims() - 1);
    int64_t batch = 1;
    TensorShape out_shape;
    for (int i = 0; i < ra - 2; ++i) {
      batch *= a.dim_size(i);
      out_shape.AddDim(a.dim_size(i));
    }
    out_shape.AddDim(m);
    out_shape.AddDim(n);
    Tensor* out = ctx->allocate_output(0, out_shape);
    const T* ap = a.flat<T>();
    const T* bp = b.flat<T>();
    T* cp = out->flat<T>();
    std::memset(cp, 0, out->TotalBytes());
    for (int64_t bi = 0; bi < batch; ++bi) {
      const T* A = ap + bi * m * k;
      const T* B = bp + bi * k * n;
      T* C = cp + bi * m * n;
      for (int64_t i = 0; i < m; ++i)
        for (int64_t kk = 0; kk < k; ++kk) {
          T av = ta_ ? A[kk * m + i] : A[i * k + kk];
          for (int64_t j = 0; j < n; ++j)
            C[i * n + j] += av * (tb_ ? B[j * k + kk] : B[kk * n + j]);
        }
    }
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_CPU_KERNEL_FLOATS("BatchMatMul", BatchMatMulOp)

// ------------------------------- reductions ---------------------------------
enum class Red { SUM, MEAN, MAX, MIN, PROD };

template <typename T, Red R>
class ReduceOp : public OpKernel {
 public:
  explicit ReduceOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("keep_dims", &keep_dims_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto axes_v = IntVector(ctx->input(1));
    int rank = in.dims();
    // Empty axes reduces nothing (TF semantics: axis=None is expanded by the
    // Python layer before reaching the kernel).
    std::vector<bool> reduce(rank, false);
    for (auto a : axes_v) reduce[a < 0 ? a + rank : a] = true;
    TensorShape out_shape, full_shape;
    for (int i = 0; i < rank; ++i) {
      full_shape.AddDim(reduce[i] ? 1 : in.dim_size(i));
      if (!reduce[i]) out_shape.AddDim(in.dim_size(i));
      else if (keep_dims_) out_shape.AddDim(1);
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    T* op = out->flat<T>();
    int64_t out_n = out_shape.num_elements();
    T init = R == Red::MAX   ? std::numeric_limits<T>::lowest()
             : R == Red::MIN ? std::numeric_limits<T>::max()
             : R == Red::PROD ? T(1)
                              : T(0);
    for (int64_t i = 0; i < out_n; ++i) op[i] = init;
    // out strides in "full" (keep-dims) space
    std::vector<int64_t> ostr(rank, 0);
    int64_t s = 1;
    for (int i = rank - 1; i >= 0; --i) {
      if (!reduce[i]) {
        ostr[i] = s;
        s *= in.dim_size(i);
      }
    }
    const T* ip = in.flat<T>();
    std::vector<int64_t> idx(rank, 0);
    int64_t n = in.NumElements();
    int64_t count = 1;
    for (int i = 0; i < rank; ++i)
      if (reduce[i]) count *= in.dim_size(i);
    for (int64_t e = 0; e < n; ++e) {
      int64_t o = 0;
      for (int i = 0; i < rank; ++i) o += idx[i] * ostr[i];
      T v = ip[e];
      switch (R) {
        case Red::SUM:
        case Red::MEAN: op[o] += v; break;
        case Red::MAX: op[o] = op[o] > v ? op[o] : v; break;
        case Red::MIN: op[o] = op[o] < v ? op[o] : v; break;
        case Red::PROD: op[o] *= v; break;
      }
      for (int i = rank - 1; i >= 0; --i) {
        if (++idx[i] < in.dim_size(i)) break;
        idx[i] = 0;
      }
    }
    if (R == Red::MEAN && count > 0)
      for (int64_t i = 0; i < out_n; ++i) op[i] = op[i] / (T)count;
  }

 private:
  bool keep_dims_ = false;
};
#define REG_REDUCE(OP, R)                                                     \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), ReduceOp<float, R>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), ReduceOp<double, R>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), ReduceOp<int32_t, R>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), ReduceOp<int64_t, R>);
REGISTER_KERNEL_BUILDER(Name("Sum").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), ReduceOp<bfloat16, Red::SUM>);
// bool All/Any: bool is a 1-byte carrier, so min/max over uint8 IS
// logical and/or (flat<T> reinterprets; op def pins dtype to bool).
REGISTER_KERNEL_BUILDER(Name("All").Device(DEVICE_CPU), ReduceOp<uint8_t, Red::MIN>);
REGISTER_KERNEL_BUILDER(Name("Any").Device(DEVICE_CPU), ReduceOp<uint8_t, Red::MAX>);
REGISTER_KERNEL_BUILDER(Name("Mean").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), ReduceOp<bfloat16, Red::MEAN>);
REG_REDUCE("Sum", Red::SUM)
REG_REDUCE("Mean", Red::MEAN)
REG_REDUCE("Max", Red::MAX)
REG_REDUCE("Min", Red::MIN)
REG_REDUCE("Prod", Red::PROD)
#undef REG_REDUCE

// ------------------------------- ArgMax/Min ---------------------------------
template <typename T, bool is_max>
class ArgMaxOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    int64_t axis = IntVector(ctx->input(1))[0];
    int rank = in.dims();
    if (axis < 0) axis += rank;
    TensorShape out_shape;
    int64_t outer = 1, axis_n = in.dim_size((int)axis), inner = 1;
    for (int i = 0; i < rank; ++i) {
      if (i < axis) outer *= in.dim_size(i);
      else if (i > axis) inner *= in.dim_size(i);
      if (i != axis) out_shape.AddDim(in.dim_size(i));
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    const T* ip = in.flat<T>();
    bool i32 = output_type(0) == DT_INT32;
    for (int64_t o = 0; o < outer; ++o)
      for (int64_t in_i = 0; in_i < inner; ++in_i) {
        T best = ip[o * axis_n * inner + in_i];
        int64_t best_i = 0;
        for (int64_t a = 1; a < axis_n; ++a) {
          T v = ip[(o * axis_n + a) * inner + in_i];
          if (is_max ? v > best : v < best) {
            best = v;
            best_i = a;
          }
        }
        if (i32) out->flat<int32_t>()[o * inner + in_i] = (int32_t)best_i;
        else out->flat<int64_t>()[o * inner + in_i] = best_i;
      }
  }
};
#define REG_ARG(OP, M)                                                        \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), ArgMaxOp<float, M>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), ArgMaxOp<double, M>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), ArgMaxOp<int32_t, M>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), ArgMaxOp<int64_t, M>);
REG_ARG("ArgMax", true)
REG_ARG("ArgMin", false)
#undef REG_ARG

// ------------------------------ softmax family ------------------------------
template <typename T, bool log_sm>
class SoftmaxOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    int64_t c = in.dim_size(in.dims() - 1);
    int64_t rows = in.NumElements() / c;
    Tensor* out = ctx->allocate_output(0, in.shape());
    const T* ip = in.flat<T>();
    T* op = out->flat<T>();
    for (int64_t r = 0; r < rows; ++r) {
      const T* x = ip + r * c;
      T* y = op + r * c;
      T mx = x[0];
      for (int64_t i = 1; i < c; ++i) mx = x[i] > mx ? x[i] : mx;
      T sum = 0;
      for (int64_t i = 0; i < c; ++i) {
        y[i] = (T)std::exp((double)(x[i] - mx));
        sum += y[i];
      }
      if (log_sm) {
        T lsum = (T)std::log((double)sum);
        for (int64_t i = 0; i < c; ++i) y[i] = x[i] - mx - lsum;
      } else {
        for (int64_t i = 0; i < c; ++i) y[i] /= sum;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Softmax").Device(DEVICE_CPU).TypeConstraint<float>("T"), SoftmaxOp<float, false>);
REGISTER_KERNEL_BUILDER(Name("Softmax").Device(DEVICE_CPU).TypeConstraint<double>("T"), SoftmaxOp<double, false>);
REGISTER_KERNEL_BUILDER(Name("LogSoftmax").Device(DEVICE_CPU).TypeConstraint<float>("T"), SoftmaxOp<float, true>);
REGISTER_KERNEL_BUILDER(Name("LogSoftmax").Device(DEVICE_CPU).TypeConstraint<double>("T"), SoftmaxOp<double, true>);

template <typename T>
class XentOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& logits = ctx->input(0);
    const Tensor& labels = ctx->input(1);
    int64_t c = logits.dim_size(1);
    int64_t rows = logits.dim_size(0);
    Tensor* loss = ctx->allocate_output(0, TensorShape({rows}));
    Tensor* grad = ctx->allocate_output(1, logits.shape());
    const T* xp = logits.flat<T>();
    const T* lp = labels.flat<T>();
    T* lo = loss->flat<T>();
    T* gp = grad->flat<T>();
    for (int64_t r = 0; r < rows; ++r) {
      const T* x = xp + r * c;
      const T* l = lp + r * c;
      T mx = x[0];
      for (int64_t i = 1; i < c; ++i) mx = x[i] > mx ? x[i] : mx;
      T sum = 0;
      for (int64_t i = 0; i < c; ++i) sum += (T)std::exp((double)(x[i] - mx));
      T lsum = (T)std::log((double)sum);
      T loss_v = 0;
      for (int64_t i = 0; i < c; ++i) {
        T logp = x[i] - mx - lsum;
        loss_v -= l[i] * logp;
        gp[r * c + i] = (T)std::exp((double)logp) - l[i];
      }
      lo[r] = loss_v;
    }
  }
};
REGISTER_CPU_KERNEL_FLOATS("SoftmaxCrossEntropyWithLogits", XentOp)

template <typename T>
class SparseXentOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& logits = ctx->input(0);
    auto labels = IntVector(ctx->input(1));
    int64_t c = logits.dim_size(1);
    int64_t rows = logits.dim_size(0);
    Tensor* loss = ctx->allocate_output(0, TensorShape({rows}));
    Tensor* grad = ctx->allocate_output(1, logits.shape());
    const T* xp = logits.flat<T>();
    T* lo = loss->flat<T>();
    T* gp = grad->flat<T>();
    for (int64_t r = 0; r < rows; ++r) {
      const T* x = xp + r * c;
      T mx = x[0];
      for (int64_t i = 1; i < c; ++i) mx = x[i] > mx ? x[i] : mx;
      T sum = 0;
      for (int64_t i = 0; i < c; ++i) sum += (T)std::exp((double)(x[i] - mx));
      T lsum = (T)std::log((double)sum);
      int64_t lbl = labels[r];
      lo[r] = -(x[lbl] - mx - lsum);
      for (int64_t i = 0; i < c; ++i)
        gp[r * c + i] =
            (T)std::exp((double)(x[i] - mx - lsum)) - (i == lbl ? T(1) : T(0));
    }
  }
};
REGISTER_CPU_KERNEL_FLOATS("SparseSoftmaxCrossEntropyWithLogits", SparseXentOp)

// ------------------------------ Bias / L2Loss -------------------------------
template <typename T>
class BiasAddOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& bias = ctx->input(1);
    int64_t c = bias.NumElements();
    Tensor* out = ctx->allocate_output(0, in.shape());
    const T* ip = in.flat<T>();
    const T* bp = bias.flat<T>();
    T* op = out->flat<T>();
    int64_t n = in.NumElements();
    for (int64_t i = 0; i < n; ++i) op[i] = ip[i] + bp[i % c];
  }
};
REGISTER_CPU_KERNEL_TYPES("BiasAdd", BiasAddOp)
REGISTER_KERNEL_BUILDER(Name("BiasAdd").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), BiasAddOp<bfloat16>);

template <typename T>
class BiasAddGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& dy = ctx->input(0);
    int64_t c = dy.dim_size(dy.dims() - 1);
    Tensor* out = ctx->allocate_output(0, TensorShape({c}));
    T* op = out->flat<T>();
    std::memset(op, 0, out->TotalBytes());
    const T* dp = dy.flat<T>();
    int64_t n = dy.NumElements();
    for (int64_t i = 0; i < n; ++i) op[i % c] += dp[i];
  }
};
REGISTER_CPU_KERNEL_TYPES("BiasAddGrad", BiasAddGradOp)
REGISTER_KERNEL_BUILDER(Name("BiasAddGrad").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), BiasAddGradOp<bfloat16>);

template <typename T>
class L2LossOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    const T* p = in.flat<T>();
    T sum = 0;
    for (int64_t i = 0; i < in.NumElements(); ++i) sum += p[i] * p[i];
    out->flat<T>()[0] = sum / T(2);
  }
};
REGISTER_CPU_KERNEL_FLOATS("L2Loss", L2LossOp)

// ------------------------------- Relu family --------------------------------
template <typename T>
struct FRelu { T operator()(T a) const { return a > T(0) ? a : T(0); } };
template <typename T>
struct FRelu6 { T operator()(T a) const { return a < T(0) ? T(0) : (a > T(6) ? T(6) : a); } };
template <typename T>
struct FSoftplus { T operator()(T a) const { return (T)std::log1p(std::exp((double)a)); } };
template <typename T, template <typename> class F>
class UnaryTplOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* xp = x.flat<T>();
    T* op = out->flat<T>();
    F<T> f;
    for (int64_t i = 0; i < x.NumElements(); ++i) op[i] = f(xp[i]);
  }
};
#define REG_NN_UNARY(OP, F)                                                   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), UnaryTplOp<float, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), UnaryTplOp<double, F>);
REG_NN_UNARY("Relu", FRelu)
REG_NN_UNARY("Relu6", FRelu6)
REG_NN_UNARY("Softplus", FSoftplus)
#undef REG_NN_UNARY

// grad/features-style grads.
template <typename T>
struct FReluGrad { T operator()(T g, T x) const { return x > T(0) ? g : T(0); } };
template <typename T>
struct FRelu6Grad { T operator()(T g, T x) const { return x > T(0) && x < T(6) ? g : T(0); } };
template <typename T>
struct FSoftplusGrad { T operator()(T g, T x) const { return g / (T(1) + (T)std::exp(-(double)x)); } };
template <typename T, template <typename> class F>
class GradFeatOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& g = ctx->input(0);
    const Tensor& x = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* gp = g.flat<T>();
    const T* xp = x.flat<T>();
    T* op = out->flat<T>();
    F<T> f;
    for (int64_t i = 0; i < x.NumElements(); ++i) op[i] = f(gp[i], xp[i]);
  }
};
#define REG_NN_GRAD(OP, F)                                                    \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), GradFeatOp<float, F>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), GradFeatOp<double, F>);
REG_NN_GRAD("ReluGrad", FReluGrad)
REG_NN_GRAD("Relu6Grad", FRelu6Grad)
REG_NN_GRAD("SoftplusGrad", FSoftplusGrad)
#undef REG_NN_GRAD

}  // namespace stf
