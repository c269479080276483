// CPU NN kernels: Conv2D (NHWC direct loops), pooling, batch norm.
// These are the fp32 reference implementations the HIP kernels are validated
// against (SURVEY.md §4 test strategy); CPU speed is not a goal.
#include <cmath>

#include "kernels/kernel_util.h"

namespace stf {

struct Conv2DParams {
  int64_t N, H, W, C, R, S, K, stride_h, stride_w, pad_h, pad_w, P, Q;
};

static Status ComputeConvParams(const TensorShape& input,
                                const TensorShape& filter,
                                const std::vector<int64_t>& strides,
                                const std::string& padding, Conv2DParams* p) {
  p->N = input.dim_size(0);
  p->H = input.dim_size(1);
  p->W = input.dim_size(2);
  p->C = input.dim_size(3);
  p->R = filter.dim_size(0);
  p->S = filter.dim_size(1);
  if (filter.dim_size(2) != p->C)
    return errors::InvalidArgument("Conv2D channel mismatch");
  p->K = filter.dim_size(3);
  p->stride_h = strides[1];
  p->stride_w = strides[2];
  if (padding == "SAME") {
    p->P = (p->H + p->stride_h - 1) / p->stride_h;
    p->Q = (p->W + p->stride_w - 1) / p->stride_w;
    int64_t pad_rows = std::max<int64_t>(
        0, (p->P - 1) * p->stride_h + p->R - p->H);
    int64_t pad_cols = std::max<int64_t>(
        0, (p->Q - 1) * p->stride_w + p->S - p->W);
    p->pad_h = pad_rows / 2;
    p->pad_w = pad_cols / 2;
  } else if (padding == "VALID") {
    p->P = (p->H - p->R) / p->stride_h + 1;
    p->Q = (p->W - p->S) / p->stride_w + 1;
    p->pad_h = p->pad_w = 0;
  } else {
    return errors::InvalidArgument("bad padding: ", padding);
  }
  return Status::OK();
}

template <typename T>
class Conv2DOp : public OpKernel {
 public:
  explicit Conv2DOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& filter = ctx->input(1);
    Conv2DParams p;
    OP_REQUIRES_OK(ctx, ComputeConvParams(in.shape(), filter.shape(), strides_,
                                          padding_, &p));
    Tensor* out = ctx->allocate_output(0, TensorShape({p.N, p.P, p.Q, p.K}));
    const T* x = in.flat<T>();
    const T* w = filter.flat<T>();
    T* y = out->flat<T>();
    std::memset(y, 0, out->TotalBytes());
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw) {
          T* yrow = y + ((n * p.P + ph) * p.Q + pw) * p.K;
          for (int64_t r = 0; r < p.R; ++r) {
            int64_t ih = ph * p.stride_h - p.pad_h + r;
            if (ih < 0 || ih >= p.H) continue;
            for (int64_t s = 0; s < p.S; ++s) {
              int64_t iw = pw * p.stride_w - p.pad_w + s;
              if (iw < 0 || iw >= p.W) continue;
              const T* xrow = x + ((n * p.H + ih) * p.W + iw) * p.C;
              const T* wrow = w + (r * p.S + s) * p.C * p.K;
              for (int64_t c = 0; c < p.C; ++c) {
                T xv = xrow[c];
                const T* wk = wrow + c * p.K;
                for (int64_t k = 0; k < p.K; ++k) yrow[k] += xv * wk[k];
              }
            }
          }
        }
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_CPU_KERNEL_FLOATS("Conv2D", Conv2DOp)

template <typename T>
class Conv2DBackpropInputOp : public OpKernel {
 public:
  explicit Conv2DBackpropInputOp(OpKernelConstruction* ctx, bool side = false)
      : OpKernel(ctx), side_(side) {
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto in_sizes = IntVector(ctx->input(0));
    const Tensor& filter = ctx->input(1);
    const Tensor& dy = ctx->input(2);
    TensorShape in_shape(in_sizes);
    Conv2DParams p;
    OP_REQUIRES_OK(ctx, ComputeConvParams(in_shape, filter.shape(), strides_,
                                          padding_, &p));
    Tensor* out = ctx->allocate_output(0, in_shape);
    const T* w = filter.flat<T>();
    const T* g = dy.flat<T>();
    T* dx = out->flat<T>();
    if (side_) {
      // Conv2DBackpropInputAdd: accumulate on top of the side gradient
      // instead of zero (the add the gradient aggregator would have run).
      const Tensor& sd = ctx->input(3);
      OP_REQUIRES(ctx, sd.NumElements() == out->NumElements(),
                  errors::InvalidArgument(
                      "Conv2DBackpropInputAdd: side shape mismatch"));
      std::memcpy(dx, sd.flat<T>(), out->TotalBytes());
    } else {
      std::memset(dx, 0, out->TotalBytes());
    }
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw) {
          const T* grow = g + ((n * p.P + ph) * p.Q + pw) * p.K;
          for (int64_t r = 0; r < p.R; ++r) {
            int64_t ih = ph * p.stride_h - p.pad_h + r;
            if (ih < 0 || ih >= p.H) continue;
            for (int64_t s = 0; s < p.S; ++s) {
              int64_t iw = pw * p.stride_w - p.pad_w + s;
              if (iw < 0 || iw >= p.W) continue;
              T* xrow = dx + ((n * p.H + ih) * p.W + iw) * p.C;
              const T* wrow = w + (r * p.S + s) * p.C * p.K;
              for (int64_t c = 0; c < p.C; ++c) {
                const T* wk = wrow + c * p.K;
                T acc = 0;
                for (int64_t k = 0; k < p.K; ++k) acc += grow[k] * wk[k];
                xrow[c] += acc;
              }
            }
          }
        }
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
  bool side_;
};
REGISTER_CPU_KERNEL_FLOATS("Conv2DBackpropInput", Conv2DBackpropInputOp)

template <typename T>
class Conv2DBackpropInputAddOp : public Conv2DBackpropInputOp<T> {
 public:
  explicit Conv2DBackpropInputAddOp(OpKernelConstruction* ctx)
      : Conv2DBackpropInputOp<T>(ctx, true) {}
};
REGISTER_CPU_KERNEL_FLOATS("Conv2DBackpropInputAdd", Conv2DBackpropInputAddOp)

template <typename T>
class Conv2DBackpropFilterOp : public OpKernel {
 public:
  explicit Conv2DBackpropFilterOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    auto f_sizes = IntVector(ctx->input(1));
    const Tensor& dy = ctx->input(2);
    TensorShape f_shape(f_sizes);
    Conv2DParams p;
    OP_REQUIRES_OK(
        ctx, ComputeConvParams(in.shape(), f_shape, strides_, padding_, &p));
    Tensor* out = ctx->allocate_output(0, f_shape);
    const T* x = in.flat<T>();
    const T* g = dy.flat<T>();
    T* dw = out->flat<T>();
    std::memset(dw, 0, out->TotalBytes());
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw) {
          const T* grow = g + ((n * p.P + ph) * p.Q + pw) * p.K;
          for (int64_t r = 0; r < p.R; ++r) {
            int64_t ih = ph * p.stride_h - p.pad_h + r;
            if (ih < 0 || ih >= p.H) continue;
            for (int64_t s = 0; s < p.S; ++s) {
              int64_t iw = pw * p.stride_w - p.pad_w + s;
              if (iw < 0 || iw >= p.W) continue;
              const T* xrow = x + ((n * p.H + ih) * p.W + iw) * p.C;
              T* wrow = dw + (r * p.S + s) * p.C * p.K;
              for (int64_t c = 0; c < p.C; ++c) {
                T xv = xrow[c];
                T* wk = wrow + c * p.K;
                for (int64_t k = 0; k < p.K; ++k) wk[k] += xv * grow[k];
              }
            }
          }
        }
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_CPU_KERNEL_FLOATS("Conv2DBackpropFilter", Conv2DBackpropFilterOp)

// --------------------------------- pooling ---------------------------------
struct PoolParams {
  int64_t N, H, W, C, kh, kw, sh, sw, pad_h, pad_w, P, Q;
};
static PoolParams GetPoolParams(const TensorShape& input,
                                const std::vector<int64_t>& ksize,
                                const std::vector<int64_t>& strides,
                                const std::string& padding) {
  PoolParams p;
  p.N = input.dim_size(0);
  p.H = input.dim_size(1);
  p.W = input.dim_size(2);
  p.C = input.dim_size(3);
  p.kh = ksize[1];
  p.kw = ksize[2];
  p.sh = strides[1];
  p.sw = strides[2];
  if (padding == "SAME") {
    p.P = (p.H + p.sh - 1) / p.sh;
    p.Q = (p.W + p.sw - 1) / p.sw;
    p.pad_h = std::max<int64_t>(0, (p.P - 1) * p.sh + p.kh - p.H) / 2;
    p.pad_w = std::max<int64_t>(0, (p.Q - 1) * p.sw + p.kw - p.W) / 2;
  } else {
    p.P = (p.H - p.kh) / p.sh + 1;
    p.Q = (p.W - p.kw) / p.sw + 1;
    p.pad_h = p.pad_w = 0;
  }
  return p;
}

template <typename T, bool is_max>
class PoolOp : public OpKernel {
 public:
  explicit PoolOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("ksize", &ksize_);
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    PoolParams p = GetPoolParams(in.shape(), ksize_, strides_, padding_);
    Tensor* out = ctx->allocate_output(0, TensorShape({p.N, p.P, p.Q, p.C}));
    const T* x = in.flat<T>();
    T* y = out->flat<T>();
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw)
          for (int64_t c = 0; c < p.C; ++c) {
            T best = is_max ? std::numeric_limits<T>::lowest() : T(0);
            int64_t count = 0;
            for (int64_t kh = 0; kh < p.kh; ++kh) {
              int64_t ih = ph * p.sh - p.pad_h + kh;
              if (ih < 0 || ih >= p.H) continue;
              for (int64_t kw = 0; kw < p.kw; ++kw) {
                int64_t iw = pw * p.sw - p.pad_w + kw;
                if (iw < 0 || iw >= p.W) continue;
                T v = x[((n * p.H + ih) * p.W + iw) * p.C + c];
                if (is_max) best = v > best ? v : best;
                else best += v;
                ++count;
              }
            }
            y[((n * p.P + ph) * p.Q + pw) * p.C + c] =
                is_max ? best : best / (T)count;
          }
  }

 protected:
  std::vector<int64_t> ksize_, strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("MaxPool").Device(DEVICE_CPU).TypeConstraint<float>("T"), PoolOp<float, true>);
REGISTER_KERNEL_BUILDER(Name("MaxPool").Device(DEVICE_CPU).TypeConstraint<double>("T"), PoolOp<double, true>);
REGISTER_KERNEL_BUILDER(Name("AvgPool").Device(DEVICE_CPU).TypeConstraint<float>("T"), PoolOp<float, false>);
REGISTER_KERNEL_BUILDER(Name("AvgPool").Device(DEVICE_CPU).TypeConstraint<double>("T"), PoolOp<double, false>);

template <typename T>
class MaxPoolGradOp : public OpKernel {
 public:
  explicit MaxPoolGradOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("ksize", &ksize_);
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& dy = ctx->input(2);
    PoolParams p = GetPoolParams(in.shape(), ksize_, strides_, padding_);
    Tensor* out = ctx->allocate_output(0, in.shape());
    const T* x = in.flat<T>();
    const T* g = dy.flat<T>();
    T* dx = out->flat<T>();
    std::memset(dx, 0, out->TotalBytes());
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw)
          for (int64_t c = 0; c < p.C; ++c) {
            // find argmax
            T best = std::numeric_limits<T>::lowest();
            int64_t bi = -1;
            for (int64_t kh = 0; kh < p.kh; ++kh) {
              int64_t ih = ph * p.sh - p.pad_h + kh;
              if (ih < 0 || ih >= p.H) continue;
              for (int64_t kw = 0; kw < p.kw; ++kw) {
                int64_t iw = pw * p.sw - p.pad_w + kw;
                if (iw < 0 || iw >= p.W) continue;
                int64_t idx = ((n * p.H + ih) * p.W + iw) * p.C + c;
                if (x[idx] > best) {
                  best = x[idx];
                  bi = idx;
                }
              }
            }
            if (bi >= 0) dx[bi] += g[((n * p.P + ph) * p.Q + pw) * p.C + c];
          }
  }

 private:
  std::vector<int64_t> ksize_, strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("MaxPoolGrad").Device(DEVICE_CPU).TypeConstraint<float>("T"), MaxPoolGradOp<float>);
REGISTER_KERNEL_BUILDER(Name("MaxPoolGrad").Device(DEVICE_CPU).TypeConstraint<double>("T"), MaxPoolGradOp<double>);

template <typename T>
class AvgPoolGradOp : public OpKernel {
 public:
  explicit AvgPoolGradOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("ksize", &ksize_);
    ctx->GetAttr("strides", &strides_);
    ctx->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    auto in_sizes = IntVector(ctx->input(0));
    const Tensor& dy = ctx->input(1);
    TensorShape in_shape(in_sizes);
    PoolParams p = GetPoolParams(in_shape, ksize_, strides_, padding_);
    Tensor* out = ctx->allocate_output(0, in_shape);
    const T* g = dy.flat<T>();
    T* dx = out->flat<T>();
    std::memset(dx, 0, out->TotalBytes());
    for (int64_t n = 0; n < p.N; ++n)
      for (int64_t ph = 0; ph < p.P; ++ph)
        for (int64_t pw = 0; pw < p.Q; ++pw) {
          // count valid positions
          int64_t count = 0;
          for (int64_t kh = 0; kh < p.kh; ++kh) {
            int64_t ih = ph * p.sh - p.pad_h + kh;
            if (ih < 0 || ih >= p.H) continue;
            for (int64_t kw = 0; kw < p.kw; ++kw) {
              int64_t iw = pw * p.sw - p.pad_w + kw;
              if (iw >= 0 && iw < p.W) ++count;
            }
          }
          for (int64_t c = 0; c < p.C; ++c) {
            T gv = g[((n * p.P + ph) * p.Q + pw) * p.C + c] / (T)count;
            for (int64_t kh = 0; kh < p.kh; ++kh) {
              int64_t ih = ph * p.sh - p.pad_h + kh;
              if (ih < 0 || ih >= p.H) continue;
              for (int64_t kw = 0; kw < p.kw; ++kw) {
                int64_t iw = pw * p.sw - p.pad_w + kw;
                if (iw < 0 || iw >= p.W) continue;
                dx[((n * p.H + ih) * p.W + iw) * p.C + c] += gv;
              }
            }
          }
        }
  }

 private:
  std::vector<int64_t> ksize_, strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("AvgPoolGrad").Device(DEVICE_CPU).TypeConstraint<float>("T").HostMemory("orig_input_shape"), AvgPoolGradOp<float>);
REGISTER_KERNEL_BUILDER(Name("AvgPoolGrad").Device(DEVICE_CPU).TypeConstraint<double>("T").HostMemory("orig_input_shape"), AvgPoolGradOp<double>);

// ------------------------------ FusedBatchNorm -------------------------------
class FusedBatchNormOp : public OpKernel {
 public:
  explicit FusedBatchNormOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("epsilon", &eps_);
    ctx->GetAttr("is_training", &training_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& scale = ctx->input(1);
    const Tensor& offset = ctx->input(2);
    int64_t C = x.dim_size(3);
    int64_t rows = x.NumElements() / C;
    Tensor* y = ctx->allocate_output(0, x.shape());
    Tensor* mean_out = ctx->allocate_output(1, TensorShape({C}));
    Tensor* var_out = ctx->allocate_output(2, TensorShape({C}));
    Tensor* save_mean = ctx->allocate_output(3, TensorShape({C}));
    Tensor* save_inv = ctx->allocate_output(4, TensorShape({C}));
    const float* xp = x.flat<float>();
    float* yp = y->flat<float>();
    std::vector<double> mean(C, 0), var(C, 0);
    if (training_) {
      for (int64_t i = 0; i < rows; ++i)
        for (int64_t c = 0; c < C; ++c) mean[c] += xp[i * C + c];
      for (int64_t c = 0; c < C; ++c) mean[c] /= rows;
      for (int64_t i = 0; i < rows; ++i)
        for (int64_t c = 0; c < C; ++c) {
          double d = xp[i * C + c] - mean[c];
          var[c] += d * d;
        }
      for (int64_t c = 0; c < C; ++c) var[c] /= rows;
    } else {
      const float* m = ctx->input(3).flat<float>();
      const float* v = ctx->input(4).flat<float>();
      for (int64_t c = 0; c < C; ++c) {
        mean[c] = m[c];
        var[c] = v[c];
      }
    }
    const float* sc = scale.flat<float>();
    const float* of = offset.flat<float>();
    std::vector<float> inv(C);
    for (int64_t c = 0; c < C; ++c) {
      inv[c] = 1.0f / std::sqrt((float)var[c] + eps_);
      mean_out->flat<float>()[c] = (float)mean[c];
      var_out->flat<float>()[c] = (float)var[c];
      save_mean->flat<float>()[c] = (float)mean[c];
      save_inv->flat<float>()[c] = inv[c];
    }
    for (int64_t i = 0; i < rows; ++i)
      for (int64_t c = 0; c < C; ++c)
        yp[i * C + c] =
            ((xp[i * C + c] - (float)mean[c]) * inv[c]) * sc[c] + of[c];
  }

 private:
  float eps_ = 1e-4f;
  bool training_ = true;
};
REGISTER_KERNEL_BUILDER(Name("FusedBatchNorm").Device(DEVICE_CPU).TypeConstraint<float>("T"), FusedBatchNormOp);

class FusedBatchNormGradOp : public OpKernel {
 public:
  explicit FusedBatchNormGradOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("epsilon", &eps_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& dy = ctx->input(0);
    const Tensor& x = ctx->input(1);
    const Tensor& scale = ctx->input(2);
    const Tensor& saved_mean = ctx->input(3);
    const Tensor& saved_inv = ctx->input(4);  // 1/sqrt(var+eps)
    int64_t C = x.dim_size(3);
    int64_t rows = x.NumElements() / C;
    Tensor* dx = ctx->allocate_output(0, x.shape());
    Tensor* dscale = ctx->allocate_output(1, TensorShape({C}));
    Tensor* doffset = ctx->allocate_output(2, TensorShape({C}));
    ctx->allocate_output(3, TensorShape({0}));
    ctx->allocate_output(4, TensorShape({0}));
    const float* g = dy.flat<float>();
    const float* xp = x.flat<float>();
    const float* sc = scale.flat<float>();
    const float* mu = saved_mean.flat<float>();
    const float* inv = saved_inv.flat<float>();
    float* dxp = dx->flat<float>();
    std::vector<double> sum_dy(C, 0), sum_dy_xhat(C, 0);
    for (int64_t i = 0; i < rows; ++i)
      for (int64_t c = 0; c < C; ++c) {
        float xhat = (xp[i * C + c] - mu[c]) * inv[c];
        sum_dy[c] += g[i * C + c];
        sum_dy_xhat[c] += g[i * C + c] * xhat;
      }
    for (int64_t c = 0; c < C; ++c) {
      dscale->flat<float>()[c] = (float)sum_dy_xhat[c];
      doffset->flat<float>()[c] = (float)sum_dy[c];
    }
    for (int64_t i = 0; i < rows; ++i)
      for (int64_t c = 0; c < C; ++c) {
        float xhat = (xp[i * C + c] - mu[c]) * inv[c];
        dxp[i * C + c] =
            sc[c] * inv[c] *
            (g[i * C + c] - (float)sum_dy[c] / rows -
             xhat * (float)sum_dy_xhat[c] / rows);
      }
  }

 private:
  float eps_ = 1e-4f;
};
REGISTER_KERNEL_BUILDER(Name("FusedBatchNormGrad").Device(DEVICE_CPU).TypeConstraint<float>("T"), FusedBatchNormGradOp);


// ---------------------- depthwise conv (CPU reference) ----------------------
// (reference core/kernels/depthwise_conv_op.cc; plain loops, f32 accum)
namespace {

struct DwShape {
  int64_t N, H, W, C, R, S, sh, sw, ph, pw, P, Q, mult;
};

static Status DwShapeFrom(const TensorShape& x, const TensorShape& f,
                          const std::vector<int64_t>& strides,
                          const std::string& padding, DwShape* d) {
  d->N = x.dim_size(0); d->H = x.dim_size(1); d->W = x.dim_size(2);
  d->C = x.dim_size(3);
  d->R = f.dim_size(0); d->S = f.dim_size(1); d->mult = f.dim_size(3);
  d->sh = strides[1]; d->sw = strides[2];
  if (padding == "SAME") {
    d->P = (d->H + d->sh - 1) / d->sh;
    d->Q = (d->W + d->sw - 1) / d->sw;
    d->ph = std::max<int64_t>(0, (d->P - 1) * d->sh + d->R - d->H) / 2;
    d->pw = std::max<int64_t>(0, (d->Q - 1) * d->sw + d->S - d->W) / 2;
  } else {
    d->P = (d->H - d->R) / d->sh + 1;
    d->Q = (d->W - d->S) / d->sw + 1;
    d->ph = d->pw = 0;
  }
  return Status::OK();
}

template <typename T>
class CpuDepthwiseConvOp : public OpKernel {
 public:
  explicit CpuDepthwiseConvOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("strides", &strides_);
    c->GetAttr("padding", &padding_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& w = ctx->input(1);
    DwShape d;
    OP_REQUIRES_OK(ctx, DwShapeFrom(x.shape(), w.shape(), strides_,
                                    padding_, &d));
    Tensor* y = ctx->allocate_output(
        0, TensorShape({d.N, d.P, d.Q, d.C * d.mult}));
    const T* xp = x.flat<T>();
    const T* wp = w.flat<T>();
    T* yp = y->flat<T>();
    int64_t CM = d.C * d.mult;
    for (int64_t n = 0; n < d.N; ++n)
      for (int64_t p = 0; p < d.P; ++p)
        for (int64_t q = 0; q < d.Q; ++q)
          for (int64_t c = 0; c < d.C; ++c)
            for (int64_t m = 0; m < d.mult; ++m) {
              float acc = 0.f;
              for (int64_t r = 0; r < d.R; ++r) {
                int64_t ih = p * d.sh - d.ph + r;
                if (ih < 0 || ih >= d.H) continue;
                for (int64_t s2 = 0; s2 < d.S; ++s2) {
                  int64_t iw = q * d.sw - d.pw + s2;
                  if (iw < 0 || iw >= d.W) continue;
                  acc += (float)xp[((n * d.H + ih) * d.W + iw) * d.C + c] *
                         (float)wp[((r * d.S + s2) * d.C + c) * d.mult + m];
                }
              }
              yp[((n * d.P + p) * d.Q + q) * CM + c * d.mult + m] = (T)acc;
            }
  }

 private:
  std::vector<int64_t> strides_;
  std::string padding_;
};
REGISTER_KERNEL_BUILDER(Name("DepthwiseConv2dNative").Device(DEVICE_CPU).TypeConstraint<float>("T"), CpuDepthwiseConvOp<float>);
REGISTER_KERNEL_BUILDER(Name("DepthwiseConv2dNative").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), CpuDepthwiseConvOp<bfloat16>);

// Local response normalization over the channel (last) axis:
// y_i = x_i / (bias + alpha * sum_{|j-i|<=r} x_j^2)^beta
// (reference lrn_op.cc; the backward pass is a python composite in
// python/ops/nn_grad.py _lrn_grad).
class LRNOp : public OpKernel {
 public:
  explicit LRNOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("depth_radius", &radius_);
    ctx->GetAttr("bias", &bias_);
    ctx->GetAttr("alpha", &alpha_);
    ctx->GetAttr("beta", &beta_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    int64_t c = in.shape().dim_size(in.shape().dims() - 1);
    int64_t rows = in.NumElements() / c;
    const float* x = in.flat<float>();
    float* y = out->flat<float>();
    for (int64_t rI = 0; rI < rows; ++rI) {
      const float* xr = x + rI * c;
      float* yr = y + rI * c;
      for (int64_t k = 0; k < c; ++k) {
        int64_t lo = k - radius_ < 0 ? 0 : k - radius_;
        int64_t hi = k + radius_ + 1 > c ? c : k + radius_ + 1;
        float s = bias_;
        for (int64_t j = lo; j < hi; ++j) s += alpha_ * xr[j] * xr[j];
        yr[k] = xr[k] / std::pow(s, beta_);
      }
    }
  }

 private:
  int64_t radius_ = 5;
  float bias_ = 1.f, alpha_ = 1.f, beta_ = 0.5f;
};
REGISTER_KERNEL_BUILDER(Name("LRN").Device(DEVICE_CPU), LRNOp);

// ------------------------- fused LSTM cell pointwise ------------------------
// CPU reference of the GPU LstmGates kernels (nn_kernels.hip); gate order
// i, j, f, o as produced by BasicLSTMCell's single [B, 4H] GEMM.
template <typename T>
class LSTMGatesOp : public OpKernel {
 public:
  explicit LSTMGatesOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    ctx->GetAttr("forget_bias", &forget_bias_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& gates = ctx->input(0);
    const Tensor& c_prev = ctx->input(1);
    int64_t B = c_prev.shape().dim_size(0);
    int64_t H = c_prev.shape().dim_size(1);
    Tensor* outs[7];
    for (int k = 0; k < 7; ++k) outs[k] = ctx->allocate_output(k, c_prev.shape());
    const T* g = gates.flat<T>();
    const T* cp = c_prev.flat<T>();
    for (int64_t b = 0; b < B; ++b) {
      const T* gr = g + b * 4 * H;
      for (int64_t h = 0; h < H; ++h) {
        float i = 1.f / (1.f + std::exp(-(float)gr[h]));
        float ci = std::tanh((float)gr[H + h]);
        float f = 1.f / (1.f + std::exp(-((float)gr[2 * H + h] + forget_bias_)));
        float o = 1.f / (1.f + std::exp(-(float)gr[3 * H + h]));
        float cs = f * (float)cp[b * H + h] + i * ci;
        float co = std::tanh(cs);
        int64_t idx = b * H + h;
        outs[0]->flat<T>()[idx] = (T)i;
        outs[1]->flat<T>()[idx] = (T)f;
        outs[2]->flat<T>()[idx] = (T)o;
        outs[3]->flat<T>()[idx] = (T)ci;
        outs[4]->flat<T>()[idx] = (T)cs;
        outs[5]->flat<T>()[idx] = (T)co;
        outs[6]->flat<T>()[idx] = (T)(o * co);
      }
    }
  }

 private:
  float forget_bias_ = 1.f;
};
REGISTER_KERNEL_BUILDER(Name("LSTMGates").Device(DEVICE_CPU).TypeConstraint<float>("T"), LSTMGatesOp<float>);
REGISTER_KERNEL_BUILDER(Name("LSTMGates").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), LSTMGatesOp<bfloat16>);

template <typename T>
class LSTMGatesGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& c_prev = ctx->input(0);
    int64_t B = c_prev.shape().dim_size(0);
    int64_t H = c_prev.shape().dim_size(1);
    const T* cp = c_prev.flat<T>();
    const T* i_ = ctx->input(1).flat<T>();
    const T* f_ = ctx->input(2).flat<T>();
    const T* o_ = ctx->input(3).flat<T>();
    const T* ci_ = ctx->input(4).flat<T>();
    const T* co_ = ctx->input(5).flat<T>();
    const T* dh = ctx->input(6).flat<T>();
    const T* dcsn = ctx->input(7).flat<T>();
    Tensor* dgates = ctx->allocate_output(0, TensorShape({B, 4 * H}));
    Tensor* dc_prev = ctx->allocate_output(1, c_prev.shape());
    T* dg = dgates->flat<T>();
    T* dcp = dc_prev->flat<T>();
    for (int64_t idx = 0; idx < B * H; ++idx) {
      int64_t b = idx / H, h = idx % H;
      float i = (float)i_[idx], f = (float)f_[idx], o = (float)o_[idx];
      float ci = (float)ci_[idx], co = (float)co_[idx];
      float dhv = (float)dh[idx];
      float dcs = dhv * o * (1.f - co * co) + (float)dcsn[idx];
      T* row = dg + b * 4 * H;
      row[h] = (T)(dcs * ci * i * (1.f - i));
      row[H + h] = (T)(dcs * i * (1.f - ci * ci));
      row[2 * H + h] = (T)(dcs * (float)cp[idx] * f * (1.f - f));
      row[3 * H + h] = (T)(dhv * co * o * (1.f - o));
      dcp[idx] = (T)(dcs * f);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("LSTMGatesGrad").Device(DEVICE_CPU).TypeConstraint<float>("T"), LSTMGatesGradOp<float>);
REGISTER_KERNEL_BUILDER(Name("LSTMGatesGrad").Device(DEVICE_CPU).TypeConstraint<bfloat16>("T"), LSTMGatesGradOp<bfloat16>);

}  // namespace

}  // namespace stf
