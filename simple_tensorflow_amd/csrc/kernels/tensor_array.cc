// TensorArray kernels (reference core/kernels/tensor_array_ops.cc +
// core/ops/data_flow_ops.cc:1080 TensorArrayV3 family). One implementation
// registered for CPU and GPU: elements are whole Tensors living on the
// op's device; slice copies go through memcpy on CPU and stream-ordered
// hipMemcpyAsync on the GPU compute stream.
#include <hip/hip_runtime.h>

#include <cstring>
#include <mutex>
#include <vector>

#include "kernels/kernel_util.h"
#include "kernels/resource_mgr.h"

namespace stf {
namespace {

struct TensorArrayResource : public ResourceBase {
  std::mutex mu;
  DataType dtype = DT_FLOAT;
  bool dynamic_size = false;
  std::vector<Tensor> elems;
  std::vector<bool> written;
  void Reset(int64_t size) {
    elems.assign((size_t)size, Tensor());
    written.assign((size_t)size, false);
  }
};

TensorArrayResource* GetTA(OpKernelContext* ctx, const std::string& handle) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<TensorArrayResource>(
      handle, [&]() { return new TensorArrayResource(); });
}

hipError_t DevCopy(OpKernelContext* ctx, void* dst, const void* src,
                   size_t bytes) {
  if (ctx->device()->is_gpu()) {
    return hipMemcpyAsync(dst, src, bytes, hipMemcpyDeviceToDevice,
                          (hipStream_t)ctx->device()->compute_stream());
  }
  std::memcpy(dst, src, bytes);
  return hipSuccess;
}

#define TA_HIP_OK(ctx, expr)                                             \
  {                                                                      \
    hipError_t _e = (expr);                                              \
    if (_e != hipSuccess) {                                              \
      (ctx)->SetStatus(errors::Internal("TensorArray hip error: ",       \
                                        hipGetErrorString(_e)));         \
      return;                                                            \
    }                                                                    \
  }

class TensorArrayOp : public OpKernel {
 public:
  explicit TensorArrayOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("dtype", &dtype_);
    c->GetAttr("dynamic_size", &dynamic_);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    int64_t size = ctx->input(0).dtype() == DT_INT32
                       ? ctx->input(0).flat<int32_t>()[0]
                       : ctx->input(0).flat<int64_t>()[0];
    TensorArrayResource* ta = GetTA(ctx, name());
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    {
      std::lock_guard<std::mutex> l(ta->mu);
      ta->dtype = dtype_;
      ta->dynamic_size = dynamic_;
      ta->Reset(size);  // a new run restarts the array
    }
    Tensor* h = ctx->allocate_output(0, TensorShape({}));
    h->flat<std::string>()[0] = name();
    Tensor* flow = ctx->allocate_output(1, TensorShape({}));
    flow->flat<float>()[0] = 0.f;
  }

 private:
  DataType dtype_;
  bool dynamic_ = false;
};

class TensorArrayGradOp : public OpKernel {
 public:
  explicit TensorArrayGradOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("source", &source_);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    std::string h = ctx->input(0).flat<std::string>()[0];
    TensorArrayResource* primary = GetTA(ctx, h);
    std::string gh = h + "@" + source_;
    TensorArrayResource* g = GetTA(ctx, gh);
    OP_REQUIRES(ctx, primary && g, errors::Internal("no resource manager"));
    {
      std::lock_guard<std::mutex> l1(primary->mu);
      std::lock_guard<std::mutex> l2(g->mu);
      g->dtype = primary->dtype;
      if (g->elems.size() != primary->elems.size())
        g->Reset((int64_t)primary->elems.size());
    }
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = gh;
    Tensor* flow = ctx->allocate_output(1, TensorShape({}));
    flow->flat<float>()[0] = 0.f;
  }

 private:
  std::string source_;
};

class TensorArrayWriteOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    TensorArrayResource* ta =
        GetTA(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    int64_t idx = ctx->input(1).flat<int32_t>()[0];
    const Tensor& val = ctx->input(2);
    std::lock_guard<std::mutex> l(ta->mu);
    if (idx >= (int64_t)ta->elems.size()) {
      OP_REQUIRES(ctx, ta->dynamic_size,
                  errors::InvalidArgument("TensorArray write index ", idx,
                                          " out of bounds (size ",
                                          ta->elems.size(), ")"));
      ta->elems.resize(idx + 1);
      ta->written.resize(idx + 1, false);
    }
    // Gradient arrays (handle "<primary>@<source>") ACCUMULATE on duplicate
    // indices — a source slot read k times contributes k gradients
    // (reference tensor_array.h TensorAndState multiple_writes_aggregate).
    // Primary arrays overwrite (while-loop grad rematerialization relies
    // on that).
    bool is_grad =
        ctx->input(0).flat<std::string>()[0].find('@') != std::string::npos;
    if (is_grad && ta->written[idx] &&
        ta->elems[idx].shape() == val.shape()) {
      Tensor sum(val.dtype(), val.shape());
      int64_t n = val.NumElements();
      if (val.dtype() == DT_FLOAT) {
        const float* a = ta->elems[idx].flat<float>();
        const float* b = val.flat<float>();
        float* o = sum.flat<float>();
        for (int64_t i = 0; i < n; ++i) o[i] = a[i] + b[i];
        ta->elems[idx] = sum;
      } else if (val.dtype() == DT_DOUBLE) {
        const double* a = ta->elems[idx].flat<double>();
        const double* b = val.flat<double>();
        double* o = sum.flat<double>();
        for (int64_t i = 0; i < n; ++i) o[i] = a[i] + b[i];
        ta->elems[idx] = sum;
      } else {
        ta->elems[idx] = val;  // non-float grads: keep last (unused path)
      }
    } else {
      ta->elems[idx] = val;
    }
    ta->written[idx] = true;
    Tensor* flow = ctx->allocate_output(0, TensorShape({}));
    flow->flat<float>()[0] = 0.f;
  }
};

class TensorArrayReadOp : public OpKernel {
 public:
  explicit TensorArrayReadOp(OpKernelConstruction* c) : OpKernel(c) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    const std::string handle = ctx->input(0).flat<std::string>()[0];
    TensorArrayResource* ta = GetTA(ctx, handle);
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    int64_t idx = ctx->input(1).flat<int32_t>()[0];
    {
      std::lock_guard<std::mutex> l(ta->mu);
      if (idx >= 0 && idx < (int64_t)ta->elems.size() && ta->written[idx]) {
        ctx->set_output(0, ta->elems[idx]);
        return;
      }
    }
    // A GRADIENT array (handle "<primary>@<source>") reads zeros for
    // positions nothing contributed to — shaped like the primary's element
    // (reference tensor_array.h TensorAndState zero-fill semantics).
    auto at = handle.rfind('@');
    if (at != std::string::npos) {
      TensorArrayResource* primary = GetTA(ctx, handle.substr(0, at));
      if (primary) {
        std::lock_guard<std::mutex> l(primary->mu);
        if (idx >= 0 && idx < (int64_t)primary->elems.size() &&
            primary->written[idx]) {
          const Tensor& like = primary->elems[idx];
          Tensor* out = ctx->allocate_output(0, like.shape());
          std::memset(out->raw_data(), 0, out->TotalBytes());
          return;
        }
      }
    }
    ctx->SetStatus(errors::InvalidArgument(
        "TensorArray read of unwritten index ", idx));
  }
};

class TensorArraySizeOp : public OpKernel {
 public:
  explicit TensorArraySizeOp(OpKernelConstruction* c) : OpKernel(c) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    TensorArrayResource* ta =
        GetTA(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(ta->mu);
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int32_t>()[0] = (int32_t)ta->elems.size();
  }
};

class TensorArrayGatherOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    TensorArrayResource* ta =
        GetTA(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    const Tensor& indices = ctx->input(1);
    std::lock_guard<std::mutex> l(ta->mu);
    int64_t n = indices.NumElements();
    TensorShape eshape;
    for (int64_t i = 0; i < n; ++i) {
      int32_t idx = indices.flat<int32_t>()[i];
      OP_REQUIRES(ctx, idx >= 0 && idx < (int64_t)ta->elems.size() &&
                           ta->written[idx],
                  errors::InvalidArgument(
                      "TensorArray gather of unwritten index ", idx));
      if (i == 0) eshape = ta->elems[idx].shape();
    }
    TensorShape out_shape({n});
    for (auto d : eshape.dim_sizes()) out_shape.AddDim(d);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t ebytes = n ? ta->elems[indices.flat<int32_t>()[0]].TotalBytes()
                      : 0;
    for (int64_t i = 0; i < n; ++i) {
      int32_t idx = indices.flat<int32_t>()[i];
      TA_HIP_OK(ctx, DevCopy(ctx, (char*)out->raw_data() + i * ebytes,
                             ta->elems[idx].raw_data(), ebytes));
    }
  }
};

class TensorArrayScatterOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    TensorArrayResource* ta =
        GetTA(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, ta, errors::Internal("no resource manager"));
    const Tensor& indices = ctx->input(1);
    const Tensor& value = ctx->input(2);
    std::lock_guard<std::mutex> l(ta->mu);
    int64_t n = indices.NumElements();
    OP_REQUIRES(ctx, value.dims() >= 1 && value.dim_size(0) == n,
                errors::InvalidArgument("scatter value dim0 != indices"));
    TensorShape eshape;
    for (int i = 1; i < value.dims(); ++i) eshape.AddDim(value.dim_size(i));
    size_t ebytes = value.TotalBytes() / (n ? n : 1);
    for (int64_t i = 0; i < n; ++i) {
      int32_t idx = indices.flat<int32_t>()[i];
      if (idx >= (int64_t)ta->elems.size()) {
        OP_REQUIRES(ctx, ta->dynamic_size,
                    errors::InvalidArgument("scatter index out of bounds"));
        ta->elems.resize(idx + 1);
        ta->written.resize(idx + 1, false);
      }
      Tensor e(ctx->device()->allocator(), value.dtype(), eshape);
      TA_HIP_OK(ctx, DevCopy(ctx, e.raw_data(),
                             (const char*)value.raw_data() + i * ebytes,
                             ebytes));
      ta->elems[idx] = e;
      ta->written[idx] = true;
    }
    Tensor* flow = ctx->allocate_output(0, TensorShape({}));
    flow->flat<float>()[0] = 0.f;
  }
};

class TensorArrayCloseOp : public OpKernel {
 public:
  explicit TensorArrayCloseOp(OpKernelConstruction* c) : OpKernel(c) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    TensorArrayResource* ta =
        GetTA(ctx, ctx->input(0).flat<std::string>()[0]);
    if (ta) {
      std::lock_guard<std::mutex> l(ta->mu);
      ta->Reset(0);
    }
  }
};

#define REG_TA(NAME, OP)                                                    \
  REGISTER_KERNEL_BUILDER(Name(NAME).Device(DEVICE_CPU), OP);               \
  REGISTER_KERNEL_BUILDER(Name(NAME).Device(DEVICE_GPU)                     \
                              .HostMemory("handle")                         \
                              .HostMemory("grad_handle")                    \
                              .HostMemory("flow")                         \
                              .HostMemory("flow_in")                        \
                              .HostMemory("flow_out")                       \
                              .HostMemory("index")                          \
                              .HostMemory("indices")                        \
                              .HostMemory("size"),                          \
                          OP)

REG_TA("TensorArrayV3", TensorArrayOp);
REG_TA("TensorArrayGradV3", TensorArrayGradOp);
REG_TA("TensorArrayWriteV3", TensorArrayWriteOp);
REG_TA("TensorArrayReadV3", TensorArrayReadOp);
REG_TA("TensorArraySizeV3", TensorArraySizeOp);
REG_TA("TensorArrayGatherV3", TensorArrayGatherOp);
REG_TA("TensorArrayScatterV3", TensorArrayScatterOp);
REG_TA("TensorArrayCloseV3", TensorArrayCloseOp);
#undef REG_TA

}  // namespace
}  // namespace stf
