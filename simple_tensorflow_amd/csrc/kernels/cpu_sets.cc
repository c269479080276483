// Set operation kernels (capability analog of reference
// core/kernels/set_kernels.cc): rows are sets over the last dimension;
// results come back as a sparse [rows, max_result_len] tensor.
#include <algorithm>
#include <set>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

template <typename T>
class DenseToDenseSetOperationOp : public OpKernel {
 public:
  explicit DenseToDenseSetOperationOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("set_operation", &op_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& s1 = ctx->input(0);
    const Tensor& s2 = ctx->input(1);
    int d1 = s1.shape().dims(), d2 = s2.shape().dims();
    if (d1 < 1 || d1 != d2) {
      ctx->SetStatus(errors::InvalidArgument("set op: rank mismatch"));
      return;
    }
    int64_t n1 = s1.shape().dim_size(d1 - 1);
    int64_t n2 = s2.shape().dim_size(d2 - 1);
    int64_t rows = s1.NumElements() / (n1 ? n1 : 1);
    if (rows != s2.NumElements() / (n2 ? n2 : 1)) {
      ctx->SetStatus(errors::InvalidArgument("set op: row count mismatch"));
      return;
    }
    std::vector<std::vector<T>> results(rows);
    const T* p1 = s1.flat<T>();
    const T* p2 = s2.flat<T>();
    int64_t total = 0, maxlen = 0;
    for (int64_t r = 0; r < rows; ++r) {
      std::set<T> a(p1 + r * n1, p1 + (r + 1) * n1);
      std::set<T> b(p2 + r * n2, p2 + (r + 1) * n2);
      std::vector<T>& out = results[r];
      if (op_ == "union") {
        std::set_union(a.begin(), a.end(), b.begin(), b.end(),
                       std::back_inserter(out));
      } else if (op_ == "intersection") {
        std::set_intersection(a.begin(), a.end(), b.begin(), b.end(),
                              std::back_inserter(out));
      } else if (op_ == "a-b") {
        std::set_difference(a.begin(), a.end(), b.begin(), b.end(),
                            std::back_inserter(out));
      } else if (op_ == "b-a") {
        std::set_difference(b.begin(), b.end(), a.begin(), a.end(),
                            std::back_inserter(out));
      } else {
        ctx->SetStatus(errors::InvalidArgument("set op: unknown operation ",
                                               op_));
        return;
      }
      total += (int64_t)out.size();
      maxlen = std::max<int64_t>(maxlen, (int64_t)out.size());
    }
    // output indices use the batch shape of set1 (all dims but last) + pos
    int nd = d1;  // batch dims + position dim
    Tensor* oi = ctx->allocate_output(0, TensorShape({total, nd}));
    Tensor* ov = ctx->allocate_output(1, TensorShape({total}));
    Tensor* os = ctx->allocate_output(2, TensorShape({nd}));
    std::vector<int64_t> bdims(nd - 1);
    for (int i = 0; i < nd - 1; ++i) bdims[i] = s1.shape().dim_size(i);
    int64_t k = 0;
    for (int64_t r = 0; r < rows; ++r) {
      // decompose r into batch coords
      std::vector<int64_t> coord(nd - 1);
      int64_t rem = r;
      for (int i = nd - 2; i >= 0; --i) {
        coord[i] = rem % bdims[i];
        rem /= bdims[i];
      }
      for (size_t j = 0; j < results[r].size(); ++j, ++k) {
        for (int i = 0; i < nd - 1; ++i)
          oi->flat<int64_t>()[k * nd + i] = coord[i];
        oi->flat<int64_t>()[k * nd + nd - 1] = (int64_t)j;
        ov->flat<T>()[k] = results[r][j];
      }
    }
    for (int i = 0; i < nd - 1; ++i) os->flat<int64_t>()[i] = bdims[i];
    os->flat<int64_t>()[nd - 1] = maxlen;
  }

 private:
  std::string op_;
};
REGISTER_KERNEL_BUILDER(Name("DenseToDenseSetOperation").Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), DenseToDenseSetOperationOp<int32_t>);
REGISTER_KERNEL_BUILDER(Name("DenseToDenseSetOperation").Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), DenseToDenseSetOperationOp<int64_t>);

template <typename T>
class SetSizeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& idx = ctx->input(0);
    const Tensor& vals = ctx->input(1);
    const Tensor& shape = ctx->input(2);
    int nd = (int)shape.NumElements();
    // rows = product of all dims but last
    std::vector<int64_t> bdims(nd - 1);
    int64_t rows = 1;
    for (int i = 0; i < nd - 1; ++i) {
      bdims[i] = shape.flat<int64_t>()[i];
      rows *= bdims[i];
    }
    TensorShape out_shape;
    for (int i = 0; i < nd - 1; ++i) out_shape.AddDim(bdims[i]);
    Tensor* out = ctx->allocate_output(0, out_shape);
    std::vector<std::set<T>> sets(rows);
    int64_t nnz = vals.NumElements();
    for (int64_t k = 0; k < nnz; ++k) {
      int64_t row = 0;
      for (int i = 0; i < nd - 1; ++i)
        row = row * bdims[i] + idx.flat<int64_t>()[k * nd + i];
      if (row >= 0 && row < rows)
        sets[row].insert(vals.flat<T>()[k]);
    }
    for (int64_t r = 0; r < rows; ++r)
      out->flat<int32_t>()[r] = (int32_t)sets[r].size();
  }
};
REGISTER_KERNEL_BUILDER(Name("SetSize").Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), SetSizeOp<int32_t>);
REGISTER_KERNEL_BUILDER(Name("SetSize").Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), SetSizeOp<int64_t>);

}  // namespace
}  // namespace stf
