// Quantized op kernels (capability analogs of reference quantize_op.cc,
// dequantize_op.cc, quantized_matmul_op.cc, quantization_utils.h):
// MIN_COMBINED affine mapping float <-> quint8 (carried as uint8), qint32
// accumulation for the quantized matmul with exact range propagation.
#include <algorithm>
#include <cmath>
#include <limits>

#include "framework/op_kernel.h"

namespace stf {
namespace {

// MIN_COMBINED: q = round((x - min) / range * 255); x = min + q * range/255.
class QuantizeV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float mn = ctx->input(1).flat<float>()[0];
    float mx = ctx->input(2).flat<float>()[0];
    if (mx <= mn) mx = mn + 1e-6f;
    Tensor* out = ctx->allocate_output(0, in.shape());
    float scale = 255.f / (mx - mn);
    const float* p = in.flat<float>();
    uint8_t* o = out->flat<uint8_t>();
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      float q = std::round((p[i] - mn) * scale);
      o[i] = (uint8_t)std::min(255.f, std::max(0.f, q));
    }
    ctx->allocate_output(1, TensorShape({}))->flat<float>()[0] = mn;
    ctx->allocate_output(2, TensorShape({}))->flat<float>()[0] = mx;
  }
};
REGISTER_KERNEL_BUILDER(Name("QuantizeV2").Device(DEVICE_CPU), QuantizeV2Op);

class DequantizeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float mn = ctx->input(1).flat<float>()[0];
    float mx = ctx->input(2).flat<float>()[0];
    Tensor* out = ctx->allocate_output(0, in.shape());
    float* o = out->flat<float>();
    if (in.dtype() == DT_UINT8) {
      float step = (mx - mn) / 255.f;
      const uint8_t* p = in.flat<uint8_t>();
      for (int64_t i = 0; i < in.NumElements(); ++i)
        o[i] = mn + p[i] * step;
    } else {  // qint32 carrier: min/max map the full int32 range
      const int32_t* p = in.flat<int32_t>();
      double scale = ((double)mx - mn) / 4294967295.0;
      for (int64_t i = 0; i < in.NumElements(); ++i)
        o[i] = (float)(((double)p[i] + 2147483648.0) * scale + mn);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Dequantize").Device(DEVICE_CPU), DequantizeOp);

// out_int32 = sum (a_q - a_zero) * (b_q - b_zero) in float-equivalent units:
// we accumulate raw products and propagate the float range exactly as the
// reference does (min/max of the int32 result in float units).
class QuantizedMatMulOp : public OpKernel {
 public:
  explicit QuantizedMatMulOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("transpose_a", &ta_);
    c->GetAttr("transpose_b", &tb_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& a = ctx->input(0);
    const Tensor& b = ctx->input(1);
    float min_a = ctx->input(2).flat<float>()[0];
    float max_a = ctx->input(3).flat<float>()[0];
    float min_b = ctx->input(4).flat<float>()[0];
    float max_b = ctx->input(5).flat<float>()[0];
    int64_t M = ta_ ? a.dim_size(1) : a.dim_size(0);
    int64_t K = ta_ ? a.dim_size(0) : a.dim_size(1);
    int64_t N = tb_ ? b.dim_size(0) : b.dim_size(1);
    Tensor* out = ctx->allocate_output(0, TensorShape({M, N}));
    const uint8_t* ap = a.flat<uint8_t>();
    const uint8_t* bp = b.flat<uint8_t>();
    int32_t* op = out->flat<int32_t>();
    // zero points for MIN_COMBINED quint8
    float sa = (max_a - min_a) / 255.f;
    float sb = (max_b - min_b) / 255.f;
    int32_t za = (int32_t)std::lround(-min_a / (sa != 0 ? sa : 1.f));
    int32_t zb = (int32_t)std::lround(-min_b / (sb != 0 ? sb : 1.f));
    for (int64_t m = 0; m < M; ++m)
      for (int64_t n = 0; n < N; ++n) {
        int64_t acc = 0;
        for (int64_t k = 0; k < K; ++k) {
          int32_t av = ta_ ? ap[k * M + m] : ap[m * K + k];
          int32_t bv = tb_ ? bp[n * K + k] : bp[k * N + n];
          acc += (int64_t)(av - za) * (bv - zb);
        }
        op[m * N + n] = (int32_t)acc;
      }
    // float value of one output unit = sa * sb; int32 range maps to:
    float unit = sa * sb;
    ctx->allocate_output(1, TensorShape({}))->flat<float>()[0] =
        (float)(-2147483648.0 * unit);
    ctx->allocate_output(2, TensorShape({}))->flat<float>()[0] =
        (float)(2147483647.0 * unit);
  }

 private:
  bool ta_ = false, tb_ = false;
};
REGISTER_KERNEL_BUILDER(Name("QuantizedMatMul").Device(DEVICE_CPU),
                        QuantizedMatMulOp);

class QuantizedReluOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float mn = ctx->input(1).flat<float>()[0];
    float mx = ctx->input(2).flat<float>()[0];
    Tensor* out = ctx->allocate_output(0, in.shape());
    float step = (mx - mn) / 255.f;
    // the quantized value representing 0.0
    int32_t zero_q = (int32_t)std::lround(-mn / (step != 0 ? step : 1.f));
    const uint8_t* p = in.flat<uint8_t>();
    uint8_t* o = out->flat<uint8_t>();
    for (int64_t i = 0; i < in.NumElements(); ++i)
      o[i] = (uint8_t)std::max<int32_t>(p[i], std::max(0, zero_q));
    ctx->allocate_output(1, TensorShape({}))->flat<float>()[0] = mn;
    ctx->allocate_output(2, TensorShape({}))->flat<float>()[0] = mx;
  }
};
REGISTER_KERNEL_BUILDER(Name("QuantizedRelu").Device(DEVICE_CPU),
                        QuantizedReluOp);

class RequantizationRangeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float mn = ctx->input(1).flat<float>()[0];
    float mx = ctx->input(2).flat<float>()[0];
    double unit = ((double)mx - mn) / 4294967295.0;
    const int32_t* p = in.flat<int32_t>();
    int32_t lo = 0, hi = 0;
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      lo = std::min(lo, p[i]);
      hi = std::max(hi, p[i]);
    }
    ctx->allocate_output(0, TensorShape({}))->flat<float>()[0] =
        (float)(lo * unit);
    ctx->allocate_output(1, TensorShape({}))->flat<float>()[0] =
        (float)(hi * unit);
  }
};
REGISTER_KERNEL_BUILDER(Name("RequantizationRange").Device(DEVICE_CPU),
                        RequantizationRangeOp);

class QuantizeDownAndShrinkRangeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float mn = ctx->input(1).flat<float>()[0];
    float mx = ctx->input(2).flat<float>()[0];
    double unit = ((double)mx - mn) / 4294967295.0;
    const int32_t* p = in.flat<int32_t>();
    int64_t n = in.NumElements();
    double lo = 0, hi = 0;
    for (int64_t i = 0; i < n; ++i) {
      double v = p[i] * unit;
      lo = std::min(lo, v);
      hi = std::max(hi, v);
    }
    if (hi <= lo) hi = lo + 1e-6;
    Tensor* out = ctx->allocate_output(0, in.shape());
    uint8_t* o = out->flat<uint8_t>();
    double scale = 255.0 / (hi - lo);
    for (int64_t i = 0; i < n; ++i) {
      double v = (p[i] * unit - lo) * scale;
      o[i] = (uint8_t)std::min(255.0, std::max(0.0, std::round(v)));
    }
    ctx->allocate_output(1, TensorShape({}))->flat<float>()[0] = (float)lo;
    ctx->allocate_output(2, TensorShape({}))->flat<float>()[0] = (float)hi;
  }
};
REGISTER_KERNEL_BUILDER(Name("QuantizeDownAndShrinkRange").Device(DEVICE_CPU),
                        QuantizeDownAndShrinkRangeOp);

}  // namespace
}  // namespace stf
