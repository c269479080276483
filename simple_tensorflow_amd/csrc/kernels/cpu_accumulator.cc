// ConditionalAccumulator kernels for synchronous data-parallel training
// (reference core/kernels/conditional_accumulator*.cc + ops at
// core/ops/data_flow_ops.cc:853): gradients with stale local_step are
// dropped; TakeGradient blocks until num_required fresh gradients arrived,
// returns their average, resets the aggregate and bumps global_step.
#include <condition_variable>
#include <cstring>
#include <deque>
#include <functional>
#include <mutex>
#include <vector>

#include "kernels/kernel_util.h"
#include "kernels/resource_mgr.h"

namespace stf {
namespace {

struct AccumResource : public ResourceBase {
  std::mutex mu;
  DataType dtype = DT_FLOAT;
  int64_t global_step = 0;
  int64_t count = 0;
  Tensor sum;  // host tensor, allocated on first apply
  struct Waiter {
    int64_t num_required;
    std::function<void(Tensor)> deliver;
  };
  std::deque<Waiter> waiters;

  // call with mu held; returns deliveries to run outside the lock
  std::vector<std::pair<std::function<void(Tensor)>, Tensor>> MaybeServe() {
    std::vector<std::pair<std::function<void(Tensor)>, Tensor>> out;
    while (!waiters.empty() && count >= waiters.front().num_required &&
           count > 0) {
      Tensor avg(sum.dtype(), sum.shape());
      int64_t n = sum.NumElements();
      if (sum.dtype() == DT_FLOAT) {
        const float* s = sum.flat<float>();
        float* d = avg.flat<float>();
        for (int64_t i = 0; i < n; ++i) d[i] = s[i] / (float)count;
      } else {
        const double* s = sum.flat<double>();
        double* d = avg.flat<double>();
        for (int64_t i = 0; i < n; ++i) d[i] = s[i] / (double)count;
      }
      out.emplace_back(waiters.front().deliver, avg);
      waiters.pop_front();
      // reset
      std::memset(sum.raw_data(), 0, sum.TotalBytes());
      count = 0;
      global_step++;
    }
    return out;
  }
};

AccumResource* GetAccum(OpKernelContext* ctx, const std::string& handle) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<AccumResource>(
      handle, [&]() { return new AccumResource(); });
}

class ConditionalAccumulatorOp : public OpKernel {
 public:
  explicit ConditionalAccumulatorOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("dtype", &dtype_);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    AccumResource* a = GetAccum(ctx, name());
    OP_REQUIRES(ctx, a, errors::Internal("no resource manager"));
    {
      std::lock_guard<std::mutex> l(a->mu);
      a->dtype = dtype_;
    }
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = name();
  }

 private:
  DataType dtype_;
};

class AccumulatorApplyGradientOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    AccumResource* a = GetAccum(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, a, errors::Internal("no resource manager"));
    int64_t local_step = ctx->input(1).flat<int64_t>()[0];
    const Tensor& grad = ctx->input(2);
    std::vector<std::pair<std::function<void(Tensor)>, Tensor>> serve;
    {
      std::lock_guard<std::mutex> l(a->mu);
      if (local_step >= a->global_step) {
        if (!a->sum.IsInitialized() ||
            !(a->sum.shape() == grad.shape())) {
          a->sum = Tensor(grad.dtype(), grad.shape());
          std::memset(a->sum.raw_data(), 0, a->sum.TotalBytes());
          a->count = 0;
        }
        int64_t n = grad.NumElements();
        if (grad.dtype() == DT_FLOAT) {
          float* s = a->sum.flat<float>();
          const float* g = grad.flat<float>();
          for (int64_t i = 0; i < n; ++i) s[i] += g[i];
        } else if (grad.dtype() == DT_DOUBLE) {
          double* s = a->sum.flat<double>();
          const double* g = grad.flat<double>();
          for (int64_t i = 0; i < n; ++i) s[i] += g[i];
        } else {
          ctx->SetStatus(errors::InvalidArgument(
              "accumulator supports float/double gradients"));
          return;
        }
        a->count++;
        serve = a->MaybeServe();
      }
      // stale gradient: silently dropped (reference semantics)
    }
    for (auto& s : serve) s.first(s.second);
  }
};

class AccumulatorTakeGradientOp : public AsyncOpKernel {
 public:
  using AsyncOpKernel::AsyncOpKernel;
  void ComputeAsync(OpKernelContext* ctx, DoneCallback done) override {
    AccumResource* a = GetAccum(ctx, ctx->input(0).flat<std::string>()[0]);
    if (!a) {
      ctx->SetStatus(errors::Internal("no resource manager"));
      done();
      return;
    }
    int64_t need = ctx->input(1).flat<int32_t>()[0];
    std::vector<std::pair<std::function<void(Tensor)>, Tensor>> serve;
    {
      std::lock_guard<std::mutex> l(a->mu);
      a->waiters.push_back(
          {need, [ctx, done](Tensor avg) {
             ctx->set_output(0, avg);
             done();
           }});
      serve = a->MaybeServe();
    }
    for (auto& s : serve) s.first(s.second);
  }
};

class AccumulatorSetGlobalStepOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    AccumResource* a = GetAccum(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, a, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(a->mu);
    a->global_step = ctx->input(1).flat<int64_t>()[0];
  }
};

class AccumulatorNumAccumulatedOp : public OpKernel {
 public:
  explicit AccumulatorNumAccumulatedOp(OpKernelConstruction* c)
      : OpKernel(c) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    AccumResource* a = GetAccum(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, a, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(a->mu);
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int32_t>()[0] = (int32_t)a->count;
  }
};

REGISTER_KERNEL_BUILDER(Name("ConditionalAccumulator").Device(DEVICE_CPU),
                        ConditionalAccumulatorOp);
REGISTER_KERNEL_BUILDER(Name("AccumulatorApplyGradient").Device(DEVICE_CPU),
                        AccumulatorApplyGradientOp);
REGISTER_KERNEL_BUILDER(Name("AccumulatorTakeGradient").Device(DEVICE_CPU),
                        AccumulatorTakeGradientOp);
REGISTER_KERNEL_BUILDER(Name("AccumulatorSetGlobalStep").Device(DEVICE_CPU),
                        AccumulatorSetGlobalStepOp);
REGISTER_KERNEL_BUILDER(Name("AccumulatorNumAccumulated").Device(DEVICE_CPU),
                        AccumulatorNumAccumulatedOp);

}  // namespace
}  // namespace stf
