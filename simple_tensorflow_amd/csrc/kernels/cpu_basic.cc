// CPU kernels: constants, identity, variables, shape/array manipulation.
// (capability analog of reference core/kernels/{constant_op,variable_ops,
// shape_ops,reshape_op,concat_op,split_op,slice_op,pad_op,transpose_op,
// gather_op,one_hot_op,...}.cc — re-implemented compactly, no Eigen).
#include <atomic>
#include <cstring>
#include <mutex>

#include "kernels/kernel_util.h"

namespace stf {

// ------------------------------- Const -------------------------------------
class ConstOp : public OpKernel {
 public:
  explicit ConstOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    auto it = ctx->def().attr.find("value");
    if (it == ctx->def().attr.end() || it->second.kind != 'e') {
      ctx->SetStatus(errors::InvalidArgument("Const missing value"));
      return;
    }
    Status s = Tensor::FromProto(it->second.tensor, &host_value_);
    if (!s.ok()) ctx->SetStatus(s);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    if (ctx->device()->is_gpu() && !output_mem.empty() &&
        output_mem[0] == MemSpace::DEVICE) {
      std::lock_guard<std::mutex> l(mu_);
      if (!dev_value_.IsInitialized()) {
        Status s = ctx->device()->CopyHostTensorToDevice(host_value_, &dev_value_);
        if (!s.ok()) {
          ctx->SetStatus(s);
          return;
        }
      }
      ctx->set_output(0, dev_value_);
    } else {
      ctx->set_output(0, host_value_);
    }
  }

 private:
  Tensor host_value_;
  Tensor dev_value_;  // cached device copy
  std::mutex mu_;
};
REGISTER_KERNEL_BUILDER(Name("Const").Device(DEVICE_CPU), ConstOp);
REGISTER_KERNEL_BUILDER(Name("Const").Device(DEVICE_GPU), ConstOp);

// ----------------------------- Placeholder ----------------------------------
class PlaceholderOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    ctx->SetStatus(errors::InvalidArgument(
        "You must feed a value for placeholder '", name(), "'"));
  }
};
REGISTER_KERNEL_BUILDER(Name("Placeholder").Device(DEVICE_CPU), PlaceholderOp);
REGISTER_KERNEL_BUILDER(Name("Placeholder").Device(DEVICE_GPU), PlaceholderOp);

class PlaceholderWithDefaultOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    ctx->set_output(0, ctx->input(0));
  }
};
REGISTER_KERNEL_BUILDER(Name("PlaceholderWithDefault").Device(DEVICE_CPU),
                        PlaceholderWithDefaultOp);

// ------------------------------ Identity ------------------------------------
class IdentityOp : public OpKernel {
 public:
  explicit IdentityOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    ctx->set_output(0, ctx->input(0));
  }
};
REGISTER_KERNEL_BUILDER(Name("Identity").Device(DEVICE_CPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("Identity").Device(DEVICE_GPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("PlaceholderWithDefault").Device(DEVICE_CPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("PlaceholderWithDefault").Device(DEVICE_GPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("StopGradient").Device(DEVICE_CPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("StopGradient").Device(DEVICE_GPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("PreventGradient").Device(DEVICE_CPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("PreventGradient").Device(DEVICE_GPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("LoopCond").Device(DEVICE_CPU), IdentityOp);
REGISTER_KERNEL_BUILDER(Name("LoopCond").Device(DEVICE_GPU).HostMemory("input").HostMemory("output"), IdentityOp);


// ---- Assert / Print (reference logging_ops.cc) ----
class AssertOp : public OpKernel {
 public:
  explicit AssertOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("summarize", &summarize_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& cond = ctx->input(0);
    bool ok = true;
    for (int64_t i = 0; i < cond.NumElements(); ++i)
      if (!cond.flat<bool>()[i]) ok = false;
    if (ok) return;
    std::string msg = "assertion failed: [";
    for (int k = 1; k < ctx->num_inputs(); ++k) {
      const Tensor& t = ctx->input(k);
      int64_t n = std::min<int64_t>(t.NumElements(), summarize_);
      for (int64_t i = 0; i < n; ++i) {
        if (i) msg += " ";
        switch (t.dtype()) {
          case DT_FLOAT: msg += std::to_string(t.flat<float>()[i]); break;
          case DT_DOUBLE: msg += std::to_string(t.flat<double>()[i]); break;
          case DT_INT32: msg += std::to_string(t.flat<int32_t>()[i]); break;
          case DT_INT64: msg += std::to_string(t.flat<int64_t>()[i]); break;
          case DT_BOOL:
            msg += t.flat<bool>()[i] ? "true" : "false";
            break;
          case DT_STRING: msg += t.flat<std::string>()[i]; break;
          default: msg += "?";
        }
      }
      msg += k + 1 < ctx->num_inputs() ? "] [" : "]";
    }
    ctx->SetStatus(errors::InvalidArgument(msg));
  }

 private:
  int64_t summarize_ = 3;
};
REGISTER_KERNEL_BUILDER(Name("Assert").Device(DEVICE_CPU), AssertOp);
REGISTER_KERNEL_BUILDER(Name("Assert").Device(DEVICE_GPU).HostMemory("condition").HostMemory("data"), AssertOp);

class PrintOp : public OpKernel {
 public:
  explicit PrintOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("message", &message_);
    c->GetAttr("first_n", &first_n_);
    c->GetAttr("summarize", &summarize_);
  }
  void Compute(OpKernelContext* ctx) override {
    int64_t count = ++count_;
    if (first_n_ < 0 || count <= first_n_) {
      std::string msg = message_;
      for (int k = 1; k < ctx->num_inputs(); ++k) {
        const Tensor& t = ctx->input(k);
        msg += "[";
        int64_t n = std::min<int64_t>(t.NumElements(), summarize_);
        for (int64_t i = 0; i < n; ++i) {
          if (i) msg += " ";
          switch (t.dtype()) {
            case DT_FLOAT: msg += std::to_string(t.flat<float>()[i]); break;
            case DT_DOUBLE: msg += std::to_string(t.flat<double>()[i]); break;
            case DT_INT32: msg += std::to_string(t.flat<int32_t>()[i]); break;
            case DT_INT64: msg += std::to_string(t.flat<int64_t>()[i]); break;
            case DT_BOOL: msg += t.flat<bool>()[i] ? "true" : "false"; break;
            case DT_STRING: msg += t.flat<std::string>()[i]; break;
            default: msg += "?";
          }
        }
        if (t.NumElements() > summarize_) msg += "...";
        msg += "]";
      }
      fprintf(stderr, "%s\n", msg.c_str());
    }
    ctx->set_output(0, ctx->input(0));
  }

 private:
  std::string message_;
  int64_t first_n_ = -1, summarize_ = 3;
  std::atomic<int64_t> count_{0};
};
REGISTER_KERNEL_BUILDER(Name("Print").Device(DEVICE_CPU), PrintOp);

// ------------------------------ Variable ------------------------------------
// The kernel instance owns the storage; instances are shared across executors
// via the session OpSegment (analog of reference OpSegment + LegacyVar).
class VariableOp : public OpKernel {
 public:
  explicit VariableOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    Status s = ctx->GetAttr("shape", &shape_);
    if (!s.ok()) ctx->SetStatus(s);
    dtype_ = output_type(0);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    std::lock_guard<std::mutex> l(mu_);
    if (!value_.IsInitialized() || value_.NumElements() == 0) {
      Allocator* a = ctx->device()->is_gpu() ? ctx->device()->allocator()
                                             : cpu_allocator();
      value_ = Tensor(a, dtype_, shape_);
    }
    ctx->set_output(0, value_);
  }
  Tensor* value() { return &value_; }

 private:
  std::mutex mu_;
  TensorShape shape_;
  DataType dtype_;
  Tensor value_;
};
REGISTER_KERNEL_BUILDER(Name("VariableV2").Device(DEVICE_CPU), VariableOp);
REGISTER_KERNEL_BUILDER(Name("VariableV2").Device(DEVICE_GPU), VariableOp);
REGISTER_KERNEL_BUILDER(Name("Variable").Device(DEVICE_CPU), VariableOp);
REGISTER_KERNEL_BUILDER(Name("Variable").Device(DEVICE_GPU), VariableOp);

// Assign / AssignAdd / AssignSub (CPU; GPU versions in gpu kernels file).
class AssignOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor ref = ctx->input(0);
    const Tensor& value = ctx->input(1);
    OP_REQUIRES(ctx, ref.shape() == value.shape(),
                errors::InvalidArgument("Assign shape mismatch: ",
                                        ref.shape().DebugString(), " vs ",
                                        value.shape().DebugString()));
    std::memcpy(ref.raw_data(), value.raw_data(), value.TotalBytes());
    ctx->set_output(0, ref);
  }
};
REGISTER_KERNEL_BUILDER(Name("Assign").Device(DEVICE_CPU), AssignOp);

template <typename T, bool add>
class AssignUpdateOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor ref = ctx->input(0);
    const Tensor& value = ctx->input(1);
    T* p = ref.flat<T>();
    const T* v = value.flat<T>();
    for (int64_t i = 0; i < ref.NumElements(); ++i)
      p[i] = add ? p[i] + v[i] : p[i] - v[i];
    ctx->set_output(0, ref);
  }
};
#define REG_ASSIGN_UPDATE(T)                                                   \
  REGISTER_KERNEL_BUILDER(Name("AssignAdd").Device(DEVICE_CPU).TypeConstraint<T>("T"), AssignUpdateOp<T, true>); \
  REGISTER_KERNEL_BUILDER(Name("AssignSub").Device(DEVICE_CPU).TypeConstraint<T>("T"), AssignUpdateOp<T, false>);
REG_ASSIGN_UPDATE(float)
REG_ASSIGN_UPDATE(double)
REG_ASSIGN_UPDATE(int32_t)
REG_ASSIGN_UPDATE(int64_t)
#undef REG_ASSIGN_UPDATE

class IsVariableInitializedOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<bool>()[0] = ctx->input(0).IsInitialized();
  }
};
REGISTER_KERNEL_BUILDER(Name("IsVariableInitialized").Device(DEVICE_CPU),
                        IsVariableInitializedOp);

// ------------------------------- Shape etc. ---------------------------------
class ShapeOp : public OpKernel {
 public:
  explicit ShapeOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    const TensorShape& s = ctx->input(0).shape();
    Tensor* out = ctx->allocate_output(0, TensorShape({s.dims()}));
    if (output_type(0) == DT_INT32)
      for (int i = 0; i < s.dims(); ++i)
        out->flat<int32_t>()[i] = (int32_t)s.dim_size(i);
    else
      for (int i = 0; i < s.dims(); ++i)
        out->flat<int64_t>()[i] = s.dim_size(i);
  }
};
REGISTER_KERNEL_BUILDER(Name("Shape").Device(DEVICE_CPU), ShapeOp);
REGISTER_KERNEL_BUILDER(Name("Shape").Device(DEVICE_GPU).HostMemory("output"), ShapeOp);

// ConcatOffset: given concat_dim and N input shapes, emit each input's start
// offset vector along concat_dim (reference concat_op.cc; used by the
// dynamic-shape _ConcatGrad).
class ConcatOffsetOp : public OpKernel {
 public:
  explicit ConcatOffsetOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    int32_t cdim = ctx->input(0).flat<int32_t>()[0];
    int n = num_inputs() - 1;
    int rank = (int)ctx->input(1).NumElements();
    if (cdim < 0) cdim += rank;
    OP_REQUIRES(ctx, cdim >= 0 && cdim < rank,
                errors::InvalidArgument("concat_dim out of range"));
    int32_t running = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& shp = ctx->input(1 + i);
      Tensor* out = ctx->allocate_output(i, shp.shape());
      for (int d = 0; d < rank; ++d)
        out->flat<int32_t>()[d] = (d == cdim) ? running : 0;
      running += shp.flat<int32_t>()[cdim];
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("ConcatOffset").Device(DEVICE_CPU),
                        ConcatOffsetOp);
REGISTER_KERNEL_BUILDER(
    Name("ConcatOffset").Device(DEVICE_GPU).HostMemory("concat_dim")
        .HostMemory("shape").HostMemory("offset"), ConcatOffsetOp);

class InvertPermutationOp : public OpKernel {
 public:
  explicit InvertPermutationOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t n = x.NumElements();
    Tensor* out = ctx->allocate_output(0, x.shape());
    if (x.dtype() == DT_INT32) {
      for (int64_t i = 0; i < n; ++i) {
        int32_t p = x.flat<int32_t>()[i];
        OP_REQUIRES(ctx, p >= 0 && p < n,
                    errors::InvalidArgument("permutation out of range"));
        out->flat<int32_t>()[p] = (int32_t)i;
      }
    } else {
      for (int64_t i = 0; i < n; ++i) {
        int64_t p = x.flat<int64_t>()[i];
        OP_REQUIRES(ctx, p >= 0 && p < n,
                    errors::InvalidArgument("permutation out of range"));
        out->flat<int64_t>()[p] = i;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("InvertPermutation").Device(DEVICE_CPU),
                        InvertPermutationOp);
REGISTER_KERNEL_BUILDER(Name("InvertPermutation").Device(DEVICE_GPU)
                            .HostMemory("x").HostMemory("y"),
                        InvertPermutationOp);

class RankOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int32_t>()[0] = ctx->input(0).dims();
  }
};
REGISTER_KERNEL_BUILDER(Name("Rank").Device(DEVICE_CPU), RankOp);
REGISTER_KERNEL_BUILDER(Name("Rank").Device(DEVICE_GPU).HostMemory("output"), RankOp);

class SizeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    if (output_type(0) == DT_INT32)
      out->flat<int32_t>()[0] = (int32_t)ctx->input(0).NumElements();
    else
      out->flat<int64_t>()[0] = ctx->input(0).NumElements();
  }
};
REGISTER_KERNEL_BUILDER(Name("Size").Device(DEVICE_CPU), SizeOp);
REGISTER_KERNEL_BUILDER(Name("Size").Device(DEVICE_GPU).HostMemory("output"), SizeOp);

// Reshape: shares the buffer.
class ReshapeOp : public OpKernel {
 public:
  explicit ReshapeOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto dims = IntVector(ctx->input(1));
    int64_t known = 1;
    int infer = -1;
    for (size_t i = 0; i < dims.size(); ++i) {
      if (dims[i] == -1) {
        OP_REQUIRES(ctx, infer < 0,
                    errors::InvalidArgument("multiple -1 dims in Reshape"));
        infer = (int)i;
      } else {
        known *= dims[i];
      }
    }
    if (infer >= 0) dims[infer] = known ? in.NumElements() / known : 0;
    TensorShape shape(dims);
    OP_REQUIRES(ctx, shape.num_elements() == in.NumElements(),
                errors::InvalidArgument("Reshape size mismatch: input ",
                                        in.shape().DebugString(), " to ",
                                        shape.DebugString()));
    ctx->set_output(0, in.Reshaped(shape));
  }
};
REGISTER_KERNEL_BUILDER(Name("Reshape").Device(DEVICE_CPU), ReshapeOp);
REGISTER_KERNEL_BUILDER(Name("Reshape").Device(DEVICE_GPU).HostMemory("shape"), ReshapeOp);

class ExpandDimsOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    int64_t axis = IntVector(ctx->input(1))[0];
    TensorShape s = in.shape();
    if (axis < 0) axis += s.dims() + 1;
    s.InsertDim((int)axis, 1);
    ctx->set_output(0, in.Reshaped(s));
  }
};
REGISTER_KERNEL_BUILDER(Name("ExpandDims").Device(DEVICE_CPU), ExpandDimsOp);
REGISTER_KERNEL_BUILDER(Name("ExpandDims").Device(DEVICE_GPU).HostMemory("dim"), ExpandDimsOp);

class SqueezeOp : public OpKernel {
 public:
  explicit SqueezeOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("squeeze_dims", &dims_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    TensorShape s;
    for (int i = 0; i < in.dims(); ++i) {
      bool listed = false;
      for (auto d : dims_) {
        int64_t dd = d < 0 ? d + in.dims() : d;
        if (dd == i) listed = true;
      }
      bool squeeze = dims_.empty() ? (in.dim_size(i) == 1)
                                   : (listed && in.dim_size(i) == 1);
      if (!squeeze) s.AddDim(in.dim_size(i));
    }
    ctx->set_output(0, in.Reshaped(s));
  }

 private:
  std::vector<int64_t> dims_;
};
REGISTER_KERNEL_BUILDER(Name("Squeeze").Device(DEVICE_CPU), SqueezeOp);
REGISTER_KERNEL_BUILDER(Name("Squeeze").Device(DEVICE_GPU), SqueezeOp);

// ------------------------------ Fill / *Like --------------------------------
class FillOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    auto dims = IntVector(ctx->input(0));
    const Tensor& v = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, TensorShape(dims));
    size_t es = DataTypeSize(v.dtype());
    DispatchBySize(es, [&](auto tag) {
      using U = decltype(tag);
      U val = v.flat<U>()[0];
      U* p = out->flat<U>();
      for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = val;
    });
  }
};
REGISTER_KERNEL_BUILDER(Name("Fill").Device(DEVICE_CPU), FillOp);

template <bool ones>
class LikeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    std::memset(out->raw_data(), 0, out->TotalBytes());
    if (ones) {
      switch (in.dtype()) {
        case DT_FLOAT: {
          float* p = out->flat<float>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = 1.0f;
          break;
        }
        case DT_DOUBLE: {
          double* p = out->flat<double>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = 1.0;
          break;
        }
        case DT_INT32: {
          int32_t* p = out->flat<int32_t>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = 1;
          break;
        }
        case DT_INT64: {
          int64_t* p = out->flat<int64_t>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = 1;
          break;
        }
        case DT_BFLOAT16: {
          bfloat16* p = out->flat<bfloat16>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = bfloat16(1.0f);
          break;
        }
        case DT_HALF: {
          uint16_t* p = out->flat<uint16_t>();
          for (int64_t i = 0; i < out->NumElements(); ++i) p[i] = 0x3C00;  // fp16 1.0
          break;
        }
        default:
          ctx->SetStatus(errors::Unimplemented("OnesLike dtype"));
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("ZerosLike").Device(DEVICE_CPU), LikeOp<false>);
// int32/int64 variants under the GPU device compute on host (loop counters
// stay in the GPU partition — see cpu_math.cc REG_GPU_HOST_INT).
REGISTER_KERNEL_BUILDER(Name("ZerosLike").Device(DEVICE_GPU).TypeConstraint<int32_t>("T").HostMemory("x").HostMemory("y"), LikeOp<false>);
REGISTER_KERNEL_BUILDER(Name("ZerosLike").Device(DEVICE_GPU).TypeConstraint<int64_t>("T").HostMemory("x").HostMemory("y"), LikeOp<false>);
REGISTER_KERNEL_BUILDER(Name("OnesLike").Device(DEVICE_GPU).TypeConstraint<int32_t>("T").HostMemory("x").HostMemory("y"), LikeOp<true>);
REGISTER_KERNEL_BUILDER(Name("OnesLike").Device(DEVICE_GPU).TypeConstraint<int64_t>("T").HostMemory("x").HostMemory("y"), LikeOp<true>);
REGISTER_KERNEL_BUILDER(Name("OnesLike").Device(DEVICE_CPU), LikeOp<true>);

// --------------------------------- Cast -------------------------------------
template <typename S, typename D>
static void CastLoop(const S* s, D* d, int64_t n) {
  for (int64_t i = 0; i < n; ++i) d[i] = (D)(float)s[i];
}
class CastOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    int64_t n = in.NumElements();
    DataType st = in.dtype(), dt = out->dtype();
    auto from = [&](auto stag) {
      using S = decltype(stag);
      const S* sp = in.flat<S>();
      switch (dt) {
        case DT_FLOAT: CastLoop(sp, out->flat<float>(), n); break;
        case DT_DOUBLE: { double* d = out->flat<double>(); for (int64_t i=0;i<n;++i) d[i]=(double)sp[i]; break; }
        case DT_INT32: { int32_t* d = out->flat<int32_t>(); for (int64_t i=0;i<n;++i) d[i]=(int32_t)sp[i]; break; }
        case DT_INT64: { int64_t* d = out->flat<int64_t>(); for (int64_t i=0;i<n;++i) d[i]=(int64_t)sp[i]; break; }
        case DT_BFLOAT16: { bfloat16* d = out->flat<bfloat16>(); for (int64_t i=0;i<n;++i) d[i]=bfloat16((float)sp[i]); break; }
        case DT_UINT8: { uint8_t* d = out->flat<uint8_t>(); for (int64_t i=0;i<n;++i) d[i]=(uint8_t)sp[i]; break; }
        case DT_BOOL: { bool* d = out->flat<bool>(); for (int64_t i=0;i<n;++i) d[i]=sp[i]!=S(0); break; }
        default: ctx->SetStatus(errors::Unimplemented("Cast to ", DataTypeString(dt)));
      }
    };
    switch (st) {
      case DT_FLOAT: from(float{}); break;
      case DT_DOUBLE: from(double{}); break;
      case DT_INT32: from(int32_t{}); break;
      case DT_INT64: from(int64_t{}); break;
      case DT_UINT8: from(uint8_t{}); break;
      case DT_BFLOAT16: {
        const bfloat16* sp = in.flat<bfloat16>();
        switch (dt) {
          case DT_FLOAT: { float* d = out->flat<float>(); for (int64_t i=0;i<n;++i) d[i]=(float)sp[i]; break; }
          case DT_DOUBLE: { double* d = out->flat<double>(); for (int64_t i=0;i<n;++i) d[i]=(double)(float)sp[i]; break; }
          default: ctx->SetStatus(errors::Unimplemented("Cast bf16 to ", DataTypeString(dt)));
        }
        break;
      }
      case DT_BOOL: {
        const bool* sp = in.flat<bool>();
        switch (dt) {
          case DT_FLOAT: { float* d = out->flat<float>(); for (int64_t i=0;i<n;++i) d[i]=sp[i]?1.f:0.f; break; }
          case DT_INT32: { int32_t* d = out->flat<int32_t>(); for (int64_t i=0;i<n;++i) d[i]=sp[i]?1:0; break; }
          case DT_INT64: { int64_t* d = out->flat<int64_t>(); for (int64_t i=0;i<n;++i) d[i]=sp[i]?1:0; break; }
          default: ctx->SetStatus(errors::Unimplemented("Cast bool to ", DataTypeString(dt)));
        }
        break;
      }
      default:
        ctx->SetStatus(errors::Unimplemented("Cast from ", DataTypeString(st)));
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Cast").Device(DEVICE_CPU), CastOp);

// ------------------------------ Pack / Unpack -------------------------------
class PackOp : public OpKernel {
 public:
  explicit PackOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("axis", &axis_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs();
    const Tensor& first = ctx->input(0);
    int axis = axis_ < 0 ? axis_ + first.dims() + 1 : (int)axis_;
    TensorShape out_shape = first.shape();
    out_shape.InsertDim(axis, n);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(first.dtype());
    // outer = prod(dims[:axis]), inner = prod(dims[axis:]) of the input.
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= first.dim_size(i);
    for (int i = axis; i < first.dims(); ++i) inner *= first.dim_size(i);
    char* dst = (char*)out->raw_data();
    for (int64_t o = 0; o < outer; ++o) {
      for (int k = 0; k < n; ++k) {
        const char* src = (const char*)ctx->input(k).raw_data() + o * inner * es;
        std::memcpy(dst, src, inner * es);
        dst += inner * es;
      }
    }
  }

 private:
  int64_t axis_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("Pack").Device(DEVICE_CPU), PackOp);

class UnpackOp : public OpKernel {
 public:
  explicit UnpackOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("axis", &axis_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    int axis = axis_ < 0 ? (int)(axis_ + in.dims()) : (int)axis_;
    int n = num_outputs();
    TensorShape out_shape = in.shape();
    out_shape.RemoveDim(axis);
    size_t es = DataTypeSize(in.dtype());
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= in.dim_size(i);
    for (int i = axis + 1; i < in.dims(); ++i) inner *= in.dim_size(i);
    for (int k = 0; k < n; ++k) {
      Tensor* out = ctx->allocate_output(k, out_shape);
      char* dst = (char*)out->raw_data();
      for (int64_t o = 0; o < outer; ++o) {
        const char* src = (const char*)in.raw_data() +
                          ((o * n + k) * inner) * es;
        std::memcpy(dst + o * inner * es, src, inner * es);
      }
    }
  }

 private:
  int64_t axis_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("Unpack").Device(DEVICE_CPU), UnpackOp);

// ------------------------------ Concat / Split ------------------------------
// values at inputs [first_value, first_value+N); axis at `axis_index`.
class ConcatBaseOp : public OpKernel {
 public:
  ConcatBaseOp(OpKernelConstruction* ctx, int axis_index, int first_value)
      : OpKernel(ctx), axis_index_(axis_index), first_value_(first_value) {}
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs() - 1;
    int64_t axis = IntVector(ctx->input(axis_index_))[0];
    const Tensor& first = ctx->input(first_value_);
    if (axis < 0) axis += first.dims();
    TensorShape out_shape = first.shape();
    int64_t axis_total = 0;
    for (int k = 0; k < n; ++k)
      axis_total += ctx->input(first_value_ + k).dim_size((int)axis);
    out_shape.set_dim((int)axis, axis_total);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(first.dtype());
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= first.dim_size(i);
    for (int i = (int)axis + 1; i < first.dims(); ++i) inner *= first.dim_size(i);
    char* dst = (char*)out->raw_data();
    int64_t out_row = axis_total * inner * es;
    int64_t off = 0;
    for (int k = 0; k < n; ++k) {
      const Tensor& t = ctx->input(first_value_ + k);
      int64_t row = t.dim_size((int)axis) * inner * es;
      const char* src = (const char*)t.raw_data();
      for (int64_t o = 0; o < outer; ++o)
        std::memcpy(dst + o * out_row + off, src + o * row, row);
      off += row;
    }
  }

 private:
  int axis_index_, first_value_;
};
class ConcatV2Op : public ConcatBaseOp {
 public:
  explicit ConcatV2Op(OpKernelConstruction* ctx)
      : ConcatBaseOp(ctx, ctx->input_types().size() - 1, 0) {}
};
class ConcatOp : public ConcatBaseOp {
 public:
  explicit ConcatOp(OpKernelConstruction* ctx) : ConcatBaseOp(ctx, 0, 1) {}
};
REGISTER_KERNEL_BUILDER(Name("ConcatV2").Device(DEVICE_CPU), ConcatV2Op);
REGISTER_KERNEL_BUILDER(Name("Concat").Device(DEVICE_CPU), ConcatOp);

class SplitOp : public OpKernel {
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
    size_t es = DataTypeSize(in.dtype());
    int64_t outer = 1, inner = 1;
    for (int i = 0; i < axis; ++i) outer *= in.dim_size(i);
    for (int i = (int)axis + 1; i < in.dims(); ++i) inner *= in.dim_size(i);
    int64_t in_row = in.dim_size((int)axis) * inner * es;
    int64_t out_row = part * inner * es;
    for (int k = 0; k < n; ++k) {
      Tensor* out = ctx->allocate_output(k, out_shape);
      char* dst = (char*)out->raw_data();
      const char* src = (const char*)in.raw_data() + k * out_row;
      for (int64_t o = 0; o < outer; ++o)
        std::memcpy(dst + o * out_row, src + o * in_row, out_row);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Split").Device(DEVICE_CPU), SplitOp);

// ------------------------------- Slice / Pad --------------------------------
class SliceOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto begin = IntVector(ctx->input(1));
    auto size = IntVector(ctx->input(2));
    int rank = in.dims();
    for (int i = 0; i < rank; ++i)
      if (size[i] == -1) size[i] = in.dim_size(i) - begin[i];
    Tensor* out = ctx->allocate_output(0, TensorShape(size));
    size_t es = DataTypeSize(in.dtype());
    std::vector<int64_t> in_strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * in.dim_size(i + 1);
    // iterate output elements row-wise on last dim
    int64_t n = out->NumElements();
    if (n == 0) return;
    int64_t last = rank ? size[rank - 1] : 1;
    char* dst = (char*)out->raw_data();
    std::vector<int64_t> idx(rank, 0);
    for (int64_t o = 0; o < n / (last ? last : 1); ++o) {
      int64_t src_off = 0;
      for (int i = 0; i < rank; ++i)
        src_off += (begin[i] + idx[i]) * in_strides[i];
      std::memcpy(dst, (const char*)in.raw_data() + src_off * es, last * es);
      dst += last * es;
      for (int i = rank - 2; i >= 0; --i) {
        if (++idx[i] < size[i]) break;
        idx[i] = 0;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Slice").Device(DEVICE_CPU), SliceOp);

class PadOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& pads = ctx->input(1);
    int rank = in.dims();
    auto pv = IntVector(pads);
    TensorShape out_shape;
    for (int i = 0; i < rank; ++i)
      out_shape.AddDim(in.dim_size(i) + pv[2 * i] + pv[2 * i + 1]);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(in.dtype());
    std::memset(out->raw_data(), 0, out->TotalBytes());
    if (in.NumElements() == 0) return;
    std::vector<int64_t> out_strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      out_strides[i] = out_strides[i + 1] * out_shape.dim_size(i + 1);
    int64_t last = rank ? in.dim_size(rank - 1) : 1;
    const char* src = (const char*)in.raw_data();
    std::vector<int64_t> idx(rank, 0);
    int64_t rows = in.NumElements() / (last ? last : 1);
    for (int64_t o = 0; o < rows; ++o) {
      int64_t dst_off = 0;
      for (int i = 0; i < rank; ++i)
        dst_off += (pv[2 * i] + idx[i]) * out_strides[i];
      // idx[rank-1] is always 0 here; add left pad of last dim
      std::memcpy((char*)out->raw_data() + (dst_off)*es, src, last * es);
      src += last * es;
      for (int i = rank - 2; i >= 0; --i) {
        if (++idx[i] < in.dim_size(i)) break;
        idx[i] = 0;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Pad").Device(DEVICE_CPU), PadOp);

// ------------------------------- Transpose ----------------------------------
class TransposeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto perm = IntVector(ctx->input(1));
    int rank = in.dims();
    TensorShape out_shape;
    for (int i = 0; i < rank; ++i) out_shape.AddDim(in.dim_size((int)perm[i]));
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(in.dtype());
    std::vector<int64_t> in_strides(rank, 1), out_dims(rank);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * in.dim_size(i + 1);
    std::vector<int64_t> src_stride_for_out(rank);
    for (int i = 0; i < rank; ++i)
      src_stride_for_out[i] = in_strides[(int)perm[i]];
    int64_t n = in.NumElements();
    DispatchBySize(es, [&](auto tag) {
      using U = decltype(tag);
      const U* src = in.flat<U>();
      U* dst = out->flat<U>();
      std::vector<int64_t> idx(rank, 0);
      for (int64_t o = 0; o < n; ++o) {
        int64_t so = 0;
        for (int i = 0; i < rank; ++i) so += idx[i] * src_stride_for_out[i];
        dst[o] = src[so];
        for (int i = rank - 1; i >= 0; --i) {
          if (++idx[i] < out_shape.dim_size(i)) break;
          idx[i] = 0;
        }
      }
    });
  }
};
REGISTER_KERNEL_BUILDER(Name("Transpose").Device(DEVICE_CPU), TransposeOp);

// ------------------------------- Gather -------------------------------------
class GatherOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& params = ctx->input(0);
    const Tensor& indices = ctx->input(1);
    auto idx = IntVector(indices);
    TensorShape out_shape = indices.shape();
    for (int i = 1; i < params.dims(); ++i) out_shape.AddDim(params.dim_size(i));
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(params.dtype());
    int64_t row = params.NumElements() / std::max<int64_t>(1, params.dim_size(0));
    for (size_t i = 0; i < idx.size(); ++i) {
      OP_REQUIRES(ctx, idx[i] >= 0 && idx[i] < params.dim_size(0),
                  errors::InvalidArgument("Gather index out of range"));
      std::memcpy((char*)out->raw_data() + i * row * es,
                  (const char*)params.raw_data() + idx[i] * row * es, row * es);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Gather").Device(DEVICE_CPU), GatherOp);

// ----------------------------- UnsortedSegmentSum ----------------------------
template <typename T>
class UnsortedSegmentSumOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& data = ctx->input(0);
    auto ids = IntVector(ctx->input(1));
    int64_t num_segments = IntVector(ctx->input(2))[0];
    TensorShape out_shape({num_segments});
    int64_t row = 1;
    for (int i = ctx->input(1).dims(); i < data.dims(); ++i) {
      out_shape.AddDim(data.dim_size(i));
      row *= data.dim_size(i);
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    std::memset(out->raw_data(), 0, out->TotalBytes());
    const T* src = data.flat<T>();
    T* dst = out->flat<T>();
    for (size_t i = 0; i < ids.size(); ++i) {
      if (ids[i] < 0 || ids[i] >= num_segments) continue;
      for (int64_t j = 0; j < row; ++j)
        dst[ids[i] * row + j] += src[i * row + j];
    }
  }
};
REGISTER_CPU_KERNEL_TYPES("UnsortedSegmentSum", UnsortedSegmentSumOp)
REGISTER_KERNEL_BUILDER(Name("UnsortedSegmentSum").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), UnsortedSegmentSumOp<bfloat16>);

// ------------------------------- OneHot -------------------------------------
template <typename T>
class OneHotOp : public OpKernel {
 public:
  explicit OneHotOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("axis", &axis_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto idx = IntVector(ctx->input(0));
    int64_t depth = IntVector(ctx->input(1))[0];
    T on = ctx->input(2).flat<T>()[0];
    T off = ctx->input(3).flat<T>()[0];
    OP_REQUIRES(ctx, axis_ == -1 || axis_ == ctx->input(0).dims(),
                errors::Unimplemented("OneHot supports trailing axis only"));
    TensorShape out_shape = ctx->input(0).shape();
    out_shape.AddDim(depth);
    Tensor* out = ctx->allocate_output(0, out_shape);
    T* p = out->flat<T>();
    for (size_t i = 0; i < idx.size(); ++i)
      for (int64_t d = 0; d < depth; ++d)
        p[i * depth + d] = (d == idx[i]) ? on : off;
  }

 private:
  int64_t axis_ = -1;
};
REGISTER_CPU_KERNEL_TYPES("OneHot", OneHotOp)

// -------------------------------- Range -------------------------------------
class RangeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    DataType dt = output_type(0);
    if (dt == DT_INT32 || dt == DT_INT64) {
      auto start = IntVector(ctx->input(0))[0];
      auto limit = IntVector(ctx->input(1))[0];
      auto delta = IntVector(ctx->input(2))[0];
      int64_t n = delta != 0 ? std::max<int64_t>(0, (limit - start + delta +
                                                     (delta > 0 ? -1 : 1)) /
                                                        delta)
                             : 0;
      Tensor* out = ctx->allocate_output(0, TensorShape({n}));
      for (int64_t i = 0; i < n; ++i) {
        if (dt == DT_INT32)
          out->flat<int32_t>()[i] = (int32_t)(start + i * delta);
        else
          out->flat<int64_t>()[i] = start + i * delta;
      }
    } else {
      float start = ctx->input(0).flat<float>()[0];
      float limit = ctx->input(1).flat<float>()[0];
      float delta = ctx->input(2).flat<float>()[0];
      int64_t n = (int64_t)std::ceil((limit - start) / delta);
      if (n < 0) n = 0;
      Tensor* out = ctx->allocate_output(0, TensorShape({n}));
      for (int64_t i = 0; i < n; ++i) out->flat<float>()[i] = start + i * delta;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Range").Device(DEVICE_CPU), RangeOp);
REGISTER_KERNEL_BUILDER(Name("Range").Device(DEVICE_GPU).HostMemory("start").HostMemory("limit").HostMemory("delta").HostMemory("output"), RangeOp);

// ------------------------ BroadcastGradientArgs ------------------------------
// Returns the reduction axes for each input of a broadcasting binary op
// (reference: core/ops/array_ops.cc BroadcastGradientArgs + bcast.cc).
class BroadcastGradientArgsOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    auto s0 = IntVector(ctx->input(0));
    auto s1 = IntVector(ctx->input(1));
    int rank = (int)std::max(s0.size(), s1.size());
    std::vector<int64_t> x(rank, 1), y(rank, 1), r0, r1;
    for (size_t i = 0; i < s0.size(); ++i) x[rank - s0.size() + i] = s0[i];
    for (size_t i = 0; i < s1.size(); ++i) y[rank - s1.size() + i] = s1[i];
    for (int i = 0; i < rank; ++i) {
      if (x[i] == 1 && y[i] != 1) r0.push_back(i);
      if (y[i] == 1 && x[i] != 1) r1.push_back(i);
    }
    Tensor* o0 = ctx->allocate_output(0, TensorShape({(int64_t)r0.size()}));
    Tensor* o1 = ctx->allocate_output(1, TensorShape({(int64_t)r1.size()}));
    for (size_t i = 0; i < r0.size(); ++i) o0->flat<int32_t>()[i] = (int32_t)r0[i];
    for (size_t i = 0; i < r1.size(); ++i) o1->flat<int32_t>()[i] = (int32_t)r1[i];
  }
};
REGISTER_KERNEL_BUILDER(Name("BroadcastGradientArgs").Device(DEVICE_CPU),
                        BroadcastGradientArgsOp);
REGISTER_KERNEL_BUILDER(Name("BroadcastGradientArgs").Device(DEVICE_GPU).HostMemory("s0").HostMemory("s1").HostMemory("r0").HostMemory("r1"),
                        BroadcastGradientArgsOp);

// ------------------------------- Tile ---------------------------------------
class TileOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto mult = IntVector(ctx->input(1));
    int rank = in.dims();
    TensorShape out_shape;
    for (int i = 0; i < rank; ++i) out_shape.AddDim(in.dim_size(i) * mult[i]);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(in.dtype());
    std::vector<int64_t> in_strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * in.dim_size(i + 1);
    int64_t n = out->NumElements();
    DispatchBySize(es, [&](auto tag) {
      using U = decltype(tag);
      const U* src = in.flat<U>();
      U* dst = out->flat<U>();
      std::vector<int64_t> idx(rank, 0);
      for (int64_t o = 0; o < n; ++o) {
        int64_t so = 0;
        for (int i = 0; i < rank; ++i)
          so += (idx[i] % in.dim_size(i)) * in_strides[i];
        dst[o] = src[so];
        for (int i = rank - 1; i >= 0; --i) {
          if (++idx[i] < out_shape.dim_size(i)) break;
          idx[i] = 0;
        }
      }
    });
  }
};
REGISTER_KERNEL_BUILDER(Name("Tile").Device(DEVICE_CPU), TileOp);

// ----------------------------- CheckNumerics --------------------------------
template <typename T>
class CheckNumericsOp : public OpKernel {
 public:
  explicit CheckNumericsOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("message", &message_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const T* p = in.flat<T>();
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      if (std::isnan((double)p[i]))
        OP_REQUIRES(ctx, false,
                    errors::InvalidArgument(message_, " : Tensor had NaN values"));
      if (std::isinf((double)p[i]))
        OP_REQUIRES(ctx, false,
                    errors::InvalidArgument(message_, " : Tensor had Inf values"));
    }
    ctx->set_output(0, in);
  }

 private:
  std::string message_;
};
REGISTER_CPU_KERNEL_FLOATS("CheckNumerics", CheckNumericsOp)

// ----------------------------- SparseToDense --------------------------------
// (reference core/kernels/sparse_to_dense_op.cc)
class SparseToDenseOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& indices = ctx->input(0);
    const Tensor& shape_t = ctx->input(1);
    const Tensor& values = ctx->input(2);
    const Tensor& default_v = ctx->input(3);
    TensorShape out_shape;
    for (int64_t i = 0; i < shape_t.NumElements(); ++i)
      out_shape.AddDim(shape_t.dtype() == DT_INT64
                           ? shape_t.flat<int64_t>()[i]
                           : shape_t.flat<int32_t>()[i]);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(out->dtype());
    char* base = (char*)out->raw_data();
    const char* dv = (const char*)default_v.raw_data();
    for (int64_t i = 0; i < out->NumElements(); ++i)
      std::memcpy(base + i * es, dv, es);
    int64_t nnz = indices.dims() == 0 ? 1 : indices.dim_size(0);
    int rank = out_shape.dims();
    auto idx_at = [&](int64_t n, int d) -> int64_t {
      int64_t flat_i = indices.dims() <= 1 ? n : n * rank + d;
      return indices.dtype() == DT_INT64 ? indices.flat<int64_t>()[flat_i]
                                         : indices.flat<int32_t>()[flat_i];
    };
    const char* vals = (const char*)values.raw_data();
    bool scalar_val = values.NumElements() == 1;
    for (int64_t n = 0; n < nnz; ++n) {
      int64_t off = 0;
      for (int d = 0; d < rank; ++d)
        off = off * out_shape.dim_size(d) + idx_at(n, d);
      std::memcpy(base + off * es, vals + (scalar_val ? 0 : n) * es, es);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("SparseToDense").Device(DEVICE_CPU),
                        SparseToDenseOp);

}  // namespace stf
