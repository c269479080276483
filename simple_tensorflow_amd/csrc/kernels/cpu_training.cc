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

// FTRL-proximal (reference training_ops.cc ApplyFtrl)
template <typename T>
class ApplyFtrlOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    Tensor accum = ctx->input(1);
    Tensor linear = ctx->input(2);
    const Tensor& grad = ctx->input(3);
    T lr = ctx->input(4).flat<T>()[0];
    T l1 = ctx->input(5).flat<T>()[0];
    T l2 = ctx->input(6).flat<T>()[0];
    T lr_power = ctx->input(7).flat<T>()[0];
    T* v = var.flat<T>();
    T* a = accum.flat<T>();
    T* l = linear.flat<T>();
    const T* g = grad.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      double new_a = (double)a[i] + (double)g[i] * g[i];
      double sigma = (std::pow(new_a, -(double)lr_power) -
                      std::pow((double)a[i], -(double)lr_power)) / lr;
      l[i] = (T)((double)l[i] + g[i] - sigma * v[i]);
      a[i] = (T)new_a;
      double quad = std::pow(new_a, -(double)lr_power) / lr + 2.0 * l2;
      double lv = (double)l[i];
      if (std::fabs(lv) > (double)l1)
        v[i] = (T)((lv > 0 ? (double)l1 - lv : -(double)l1 - lv) / quad);
      else
        v[i] = (T)0;
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyFtrl", ApplyFtrlOp)

template <typename T>
class ApplyProximalSgdOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor var = ctx->input(0);
    T alpha = ctx->input(1).flat<T>()[0];
    T l1 = ctx->input(2).flat<T>()[0];
    T l2 = ctx->input(3).flat<T>()[0];
    const Tensor& delta = ctx->input(4);
    T* v = var.flat<T>();
    const T* d = delta.flat<T>();
    for (int64_t i = 0; i < var.NumElements(); ++i) {
      double prox = (double)v[i] - (double)alpha * d[i];
      double shrink = std::fabs(prox) - (double)alpha * l1;
      if (shrink < 0) shrink = 0;
      double sgn = prox > 0 ? 1.0 : (prox < 0 ? -1.0 : 0.0);
      v[i] = (T)(sgn * shrink / (1.0 + (double)alpha * l2));
    }
    ctx->set_output(0, var);
  }
};
REGISTER_CPU_KERNEL_FLOATS("ApplyProximalGradientDescent", ApplyProximalSgdOp)

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
