// CPU optimizer-apply kernels (analog of reference core/kernels/training_ops.cc).
#include <cmath>

#include "kernels/kernel_util.h"

namespace stf {

template <typename T>
class ApplyGradientDescentOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    T lr = ctx->input(1).flat<T>()[0];
    const Tensor& grad = ctx->input(2);
    T* v = var.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) v[i] -= lr * g[i];
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyGradientDescent", ApplyGradientDescentOp)

template <typename T>
class ApplyMomentumOp : public OpKernel {
 public:
  explicit ApplyMomentumOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("use_nesterov", &nesterov_);
  }
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor accum = ctx->input(1);
    T lr = ctx->input(2).flat<T>()[0];
    const Tensor& grad = ctx->input(3);
    T mom = ctx->input(4).flat<T>()[0];
    T* v = var.flat<T>();
    T* a = accum.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      a[i] = a[i] * mom + g[i];
      if (nesterov_) v[i] -= lr * (g[i] + mom * a[i]);
      else v[i] -= lr * a[i];
    }
    ctx->set_output(0, var);
  }

 private:
  bool nesterov_ = false;
};
REGISTER_CPU_KERNEL_FLOATS("ApplyMomentum", ApplyMomentumOp)

template <typename T>
class ApplyAdamOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor m = ctx->input(1);
    Tensor v = ctx->input(2);
    T b1p = ctx->input(3).flat<T>()[0];
    T b2p = ctx->input(4).flat<T>()[0];
    T lr = ctx->input(5).flat<T>()[0];
    T b1 = ctx->input(6).flat<T>()[0];
    T b2 = ctx->input(7).flat<T>()[0];
    T eps = ctx->input(8).flat<T>()[0];
    const Tensor& grad = ctx->input(9);
    T alpha = lr * (T)std::sqrt((double)(T(1) - b2p)) / (T(1) - b1p);
    T* vp = var.flat<T>();
    T* mp = m.flat<T>();
    T* vv = v.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      mp[i] += (g[i] - mp[i]) * (T(1) - b1);
      vv[i] += (g[i] * g[i] - vv[i]) * (T(1) - b2);
      vp[i] -= alpha * mp[i] / ((T)std::sqrt((double)vv[i]) + eps);
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyAdam", ApplyAdamOp)

template <typename T>
class ApplyRMSPropOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor ms = ctx->input(1);
    Tensor mom = ctx->input(2);
    T lr = ctx->input(3).flat<T>()[0];
    T rho = ctx->input(4).flat<T>()[0];
    T momentum = ctx->input(5).flat<T>()[0];
    T eps = ctx->input(6).flat<T>()[0];
    const Tensor& grad = ctx->input(7);
    T* v = var.flat<T>();
    T* m = ms.flat<T>();
    T* mo = mom.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      m[i] = rho * m[i] + (T(1) - rho) * g[i] * g[i];
      mo[i] = momentum * mo[i] + lr * g[i] / (T)std::sqrt((double)(m[i] + eps));
      v[i] -= mo[i];
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyRMSProp", ApplyRMSPropOp)

template <typename T>
class ApplyAdagradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor accum = ctx->input(1);
    T lr = ctx->input(2).flat<T>()[0];
    const Tensor& grad = ctx->input(3);
    T* v = var.flat<T>();
    T* a = accum.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      a[i] += g[i] * g[i];
      v[i] -= lr * g[i] / (T)std::sqrt((double)a[i]);
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyAdagrad", ApplyAdagradOp)

template <typename T>
class ApplyAdadeltaOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor accum = ctx->input(1);
    Tensor accum_update = ctx->input(2);
    T lr = ctx->input(3).flat<T>()[0];
    T rho = ctx->input(4).flat<T>()[0];
    T eps = ctx->input(5).flat<T>()[0];
    const Tensor& grad = ctx->input(6);
    T* v = var.flat<T>();
    T* a = accum.flat<T>();
    T* au = accum_update.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      a[i] = rho * a[i] + (T(1) - rho) * g[i] * g[i];
      T update = (T)(std::sqrt((double)(au[i] + eps)) /
                     std::sqrt((double)(a[i] + eps))) * g[i];
      au[i] = rho * au[i] + (T(1) - rho) * update * update;
      v[i] -= lr * update;
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyAdadelta", ApplyAdadeltaOp)

}  // namespace stf
