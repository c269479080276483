// Summary kernels: serialize Summary protos (analog of reference
// core/kernels/summary_op.cc + lib/histogram): ScalarSummary,
// HistogramSummary, MergeSummary -> DT_STRING scalar holding Summary bytes.
#include <cmath>

#include "core/pb.h"
#include "kernels/kernel_util.h"

namespace stf {

namespace {

// Summary { Value value = 1 { tag = 1; simple_value = 2; histo = 5 } }
void AppendScalarValue(pb::Writer* w, const std::string& tag, float v) {
  pb::Writer val;
  val.PutString(1, tag);
  val.PutTag(2, 5);
  char tmp[4];
  std::memcpy(tmp, &v, 4);
  val.buf().append(tmp, 4);
  w->PutMessage(1, val.buf());
}

void PutDouble(pb::Writer* w, int field, double v) {
  w->PutTag(field, 1);
  char tmp[8];
  std::memcpy(tmp, &v, 8);
  w->buf().append(tmp, 8);
}

}  // namespace

template <typename T>
class ScalarSummaryOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& tags = ctx->input(0);
    const Tensor& values = ctx->input(1);
    OP_REQUIRES(ctx, tags.NumElements() == values.NumElements(),
                errors::InvalidArgument("tags/values size mismatch"));
    pb::Writer w;
    for (int64_t i = 0; i < tags.NumElements(); ++i)
      AppendScalarValue(&w, tags.flat<std::string>()[i],
                        (float)values.flat<T>()[i]);
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = w.buf();
  }
};
REGISTER_KERNEL_BUILDER(Name("ScalarSummary").Device(DEVICE_CPU).TypeConstraint<float>("T"), ScalarSummaryOp<float>);
REGISTER_KERNEL_BUILDER(Name("ScalarSummary").Device(DEVICE_CPU).TypeConstraint<double>("T"), ScalarSummaryOp<double>);
REGISTER_KERNEL_BUILDER(Name("ScalarSummary").Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), ScalarSummaryOp<int32_t>);
REGISTER_KERNEL_BUILDER(Name("ScalarSummary").Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), ScalarSummaryOp<int64_t>);

template <typename T>
class HistogramSummaryOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& tag = ctx->input(0).flat<std::string>()[0];
    const Tensor& values = ctx->input(1);
    // histogram with exponential bucket limits (reference lib/histogram
    // default buckets: ... 1e-12 * 1.1^k ...)
    std::vector<double> limits;
    for (double v = 1e-12; v < 1e20; v *= 1.1) limits.push_back(v);
    double mn = INFINITY, mx = -INFINITY, sum = 0, sumsq = 0;
    std::vector<double> neg(limits.size(), 0), pos(limits.size(), 0);
    int64_t n = values.NumElements();
    for (int64_t i = 0; i < n; ++i) {
      double v = (double)values.flat<T>()[i];
      OP_REQUIRES(ctx, std::isfinite(v),
                  errors::InvalidArgument("non-finite value in histogram"));
      mn = std::min(mn, v);
      mx = std::max(mx, v);
      sum += v;
      sumsq += v * v;
      double a = std::abs(v);
      size_t b = std::lower_bound(limits.begin(), limits.end(), a) -
                 limits.begin();
      if (b >= limits.size()) b = limits.size() - 1;
      (v < 0 ? neg : pos)[b] += 1;
    }
    // HistogramProto: min=1,max=2,num=3,sum=4,sum_squares=5,
    // bucket_limit=6(packed double),bucket=7(packed double)
    pb::Writer h;
    PutDouble(&h, 1, mn);
    PutDouble(&h, 2, mx);
    PutDouble(&h, 3, (double)n);
    PutDouble(&h, 4, sum);
    PutDouble(&h, 5, sumsq);
    std::string lim_buf, cnt_buf;
    std::vector<double> all_limits;
    std::vector<double> all_counts;
    for (int i = (int)limits.size() - 1; i >= 0; --i)
      if (neg[i] > 0) {
        all_limits.push_back(-limits[i] * (1.0 / 1.1));
        all_counts.push_back(neg[i]);
      }
    for (size_t i = 0; i < limits.size(); ++i)
      if (pos[i] > 0) {
        all_limits.push_back(limits[i]);
        all_counts.push_back(pos[i]);
      }
    if (all_limits.empty()) {
      all_limits.push_back(1e-12);
      all_counts.push_back(0);
    }
    for (double v : all_limits) {
      char tmp[8];
      std::memcpy(tmp, &v, 8);
      lim_buf.append(tmp, 8);
    }
    for (double v : all_counts) {
      char tmp[8];
      std::memcpy(tmp, &v, 8);
      cnt_buf.append(tmp, 8);
    }
    h.PutMessage(6, lim_buf);
    h.PutMessage(7, cnt_buf);
    pb::Writer w;
    pb::Writer val;
    val.PutString(1, tag);
    val.PutMessage(5, h.buf());
    w.PutMessage(1, val.buf());
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = w.buf();
  }
};
REGISTER_KERNEL_BUILDER(Name("HistogramSummary").Device(DEVICE_CPU).TypeConstraint<float>("T"), HistogramSummaryOp<float>);
REGISTER_KERNEL_BUILDER(Name("HistogramSummary").Device(DEVICE_CPU).TypeConstraint<double>("T"), HistogramSummaryOp<double>);

class MergeSummaryOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    std::string merged;
    for (int i = 0; i < num_inputs(); ++i)
      merged += ctx->input(i).flat<std::string>()[0];
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = merged;
  }
};
REGISTER_KERNEL_BUILDER(Name("MergeSummary").Device(DEVICE_CPU), MergeSummaryOp);

}  // namespace stf
