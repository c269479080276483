// Sparse algebra kernels (capability analogs of the reference's
// core/kernels/sparse_add_op.cc, sparse_dense_binary_op_shared.cc,
// sparse_reorder_op.cc, sparse_reduce_op.cc, sparse_concat_op.cc):
// COO [nnz, ndims] int64 indices + values + dense shape.
#include <algorithm>
#include <cmath>
#include <map>
#include <numeric>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

// lexicographic compare of two index rows
struct RowLess {
  const int64_t* idx;
  int nd;
  bool operator()(int64_t a, int64_t b) const {
    for (int d = 0; d < nd; ++d) {
      if (idx[a * nd + d] != idx[b * nd + d])
        return idx[a * nd + d] < idx[b * nd + d];
    }
    return false;
  }
};

template <typename T>
class SparseAddOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& ai = ctx->input(0);
    const Tensor& av = ctx->input(1);
    const Tensor& as = ctx->input(2);
    const Tensor& bi = ctx->input(3);
    const Tensor& bv = ctx->input(4);
    double thresh = 0.0;
    if (ctx->num_inputs() > 6) {
      const Tensor& th = ctx->input(6);
      if (th.NumElements() == 1) {
        if (th.dtype() == DT_FLOAT) thresh = th.flat<float>()[0];
        else if (th.dtype() == DT_DOUBLE) thresh = th.flat<double>()[0];
      }
    }
    int nd = (int)as.NumElements();
    // merge rows into a map keyed by the index tuple
    std::map<std::vector<int64_t>, double> acc;
    auto fold = [&](const Tensor& it, const Tensor& vt) {
      int64_t nnz = vt.NumElements();
      for (int64_t k = 0; k < nnz; ++k) {
        std::vector<int64_t> key(nd);
        for (int d = 0; d < nd; ++d) key[d] = it.flat<int64_t>()[k * nd + d];
        acc[key] += (double)vt.flat<T>()[k];
      }
    };
    fold(ai, av);
    fold(bi, bv);
    std::vector<std::pair<std::vector<int64_t>, double>> rows;
    for (auto& kv : acc)
      if (std::fabs(kv.second) > thresh) rows.push_back(kv);
    int64_t n = (int64_t)rows.size();
    Tensor* oi = ctx->allocate_output(0, TensorShape({n, nd}));
    Tensor* ov = ctx->allocate_output(1, TensorShape({n}));
    Tensor* os = ctx->allocate_output(2, TensorShape({nd}));
    for (int64_t k = 0; k < n; ++k) {
      for (int d = 0; d < nd; ++d)
        oi->flat<int64_t>()[k * nd + d] = rows[k].first[d];
      ov->flat<T>()[k] = (T)rows[k].second;
    }
    for (int d = 0; d < nd; ++d)
      os->flat<int64_t>()[d] = as.flat<int64_t>()[d];
  }
};

template <typename T, typename TI>
class SparseTensorDenseAddOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& ai = ctx->input(0);
    const Tensor& av = ctx->input(1);
    const Tensor& b = ctx->input(3);
    Tensor* out = ctx->allocate_output(0, b.shape());
    std::memcpy(out->raw_data(), b.raw_data(), b.TotalBytes());
    int nd = b.shape().dims();
    int64_t nnz = av.NumElements();
    std::vector<int64_t> strides(nd, 1);
    for (int d = nd - 2; d >= 0; --d)
      strides[d] = strides[d + 1] * b.shape().dim_size(d + 1);
    for (int64_t k = 0; k < nnz; ++k) {
      int64_t off = 0;
      for (int d = 0; d < nd; ++d) {
        int64_t ix = (int64_t)ai.flat<TI>()[k * nd + d];
        if (ix < 0 || ix >= b.shape().dim_size(d)) {
          ctx->SetStatus(errors::InvalidArgument(
              "SparseTensorDenseAdd: index out of bounds"));
          return;
        }
        off += ix * strides[d];
      }
      out->flat<T>()[off] += av.flat<T>()[k];
    }
  }
};

template <typename T>
class SparseReorderOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& it = ctx->input(0);
    const Tensor& vt = ctx->input(1);
    int64_t nnz = vt.NumElements();
    int nd = it.shape().dims() > 1 ? (int)it.shape().dim_size(1) : 1;
    std::vector<int64_t> order(nnz);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              RowLess{it.flat<int64_t>(), nd});
    Tensor* oi = ctx->allocate_output(0, it.shape());
    Tensor* ov = ctx->allocate_output(1, vt.shape());
    for (int64_t k = 0; k < nnz; ++k) {
      for (int d = 0; d < nd; ++d)
        oi->flat<int64_t>()[k * nd + d] =
            it.flat<int64_t>()[order[k] * nd + d];
      ov->flat<T>()[k] = vt.flat<T>()[order[k]];
    }
  }
};

template <typename T>
class SparseReduceSumOp : public OpKernel {
 public:
  explicit SparseReduceSumOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("keep_dims", &keep_dims_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& it = ctx->input(0);
    const Tensor& vt = ctx->input(1);
    const Tensor& st = ctx->input(2);
    const Tensor& axes_t = ctx->input(3);
    int nd = (int)st.NumElements();
    std::vector<bool> reduce(nd, false);
    for (int64_t k = 0; k < axes_t.NumElements(); ++k) {
      int a = axes_t.flat<int32_t>()[k];
      if (a < 0) a += nd;
      if (a < 0 || a >= nd) {
        ctx->SetStatus(errors::InvalidArgument("SparseReduceSum: bad axis"));
        return;
      }
      reduce[a] = true;
    }
    TensorShape out_shape;
    std::vector<int64_t> out_dims;
    for (int d = 0; d < nd; ++d) {
      if (!reduce[d]) {
        out_shape.AddDim(st.flat<int64_t>()[d]);
        out_dims.push_back(st.flat<int64_t>()[d]);
      } else if (keep_dims_) {
        out_shape.AddDim(1);
      }
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    int64_t out_n = out->NumElements();
    for (int64_t i = 0; i < out_n; ++i) out->flat<T>()[i] = (T)0;
    int64_t nnz = vt.NumElements();
    for (int64_t k = 0; k < nnz; ++k) {
      int64_t off = 0;
      for (int d = 0; d < nd; ++d) {
        if (reduce[d]) continue;
        off = off * st.flat<int64_t>()[d] +
              it.flat<int64_t>()[k * nd + d];
      }
      out->flat<T>()[off] += vt.flat<T>()[k];
    }
  }

 private:
  bool keep_dims_ = false;
};

template <typename T>
class SparseConcatOp : public OpKernel {
 public:
  explicit SparseConcatOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("concat_dim", &concat_dim_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n3 = num_inputs();
    int n = n3 / 3;
    int nd = (int)ctx->input(2 * n).NumElements();
    int cd = (int)concat_dim_;
    if (cd < 0) cd += nd;
    // output shape: sum along cd
    std::vector<int64_t> oshape(nd);
    for (int d = 0; d < nd; ++d)
      oshape[d] = ctx->input(2 * n).flat<int64_t>()[d];
    int64_t total_nnz = 0;
    std::vector<int64_t> offsets(n, 0);
    int64_t run = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& sh = ctx->input(2 * n + i);
      offsets[i] = run;
      run += sh.flat<int64_t>()[cd];
      total_nnz += ctx->input(n + i).NumElements();
    }
    oshape[cd] = run;
    Tensor* oi = ctx->allocate_output(0, TensorShape({total_nnz, nd}));
    Tensor* ov = ctx->allocate_output(1, TensorShape({total_nnz}));
    Tensor* os = ctx->allocate_output(2, TensorShape({nd}));
    for (int d = 0; d < nd; ++d) os->flat<int64_t>()[d] = oshape[d];
    int64_t k = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& it = ctx->input(i);
      const Tensor& vt = ctx->input(n + i);
      int64_t nnz = vt.NumElements();
      for (int64_t e = 0; e < nnz; ++e, ++k) {
        for (int d = 0; d < nd; ++d) {
          int64_t v = it.flat<int64_t>()[e * nd + d];
          if (d == cd) v += offsets[i];
          oi->flat<int64_t>()[k * nd + d] = v;
        }
        ov->flat<T>()[k] = vt.flat<T>()[e];
      }
    }
    // canonical row-major order
    std::vector<int64_t> order(total_nnz);
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(),
              RowLess{oi->flat<int64_t>(), nd});
    Tensor si(DT_INT64, TensorShape({total_nnz, nd}));
    Tensor sv(vt_dtype(ctx), TensorShape({total_nnz}));
    for (int64_t e = 0; e < total_nnz; ++e) {
      for (int d = 0; d < nd; ++d)
        si.flat<int64_t>()[e * nd + d] =
            oi->flat<int64_t>()[order[e] * nd + d];
      sv.flat<T>()[e] = ov->flat<T>()[order[e]];
    }
    std::memcpy(oi->raw_data(), si.raw_data(), si.TotalBytes());
    std::memcpy(ov->raw_data(), sv.raw_data(), sv.TotalBytes());
  }

 private:
  DataType vt_dtype(OpKernelContext* ctx) {
    return ctx->input(num_inputs() / 3).dtype();
  }
  int64_t concat_dim_ = 0;
};

#define REG_SPARSE_T(NAME, OP)                                                \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<float>("T"), OP<float>);   \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<double>("T"), OP<double>); \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"),             \
      OP<int32_t>);                                                           \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), OP<int64_t>);

REG_SPARSE_T("SparseAdd", SparseAddOp)
REG_SPARSE_T("SparseReorder", SparseReorderOp)
REG_SPARSE_T("SparseReduceSum", SparseReduceSumOp)
REG_SPARSE_T("SparseConcat", SparseConcatOp)
#undef REG_SPARSE_T

REGISTER_KERNEL_BUILDER(Name("SparseTensorDenseAdd").Device(DEVICE_CPU).TypeConstraint<float>("T").TypeConstraint<int64_t>("Tindices"), SparseTensorDenseAddOp<float, int64_t>);
REGISTER_KERNEL_BUILDER(Name("SparseTensorDenseAdd").Device(DEVICE_CPU).TypeConstraint<float>("T").TypeConstraint<int32_t>("Tindices"), SparseTensorDenseAddOp<float, int32_t>);
REGISTER_KERNEL_BUILDER(Name("SparseTensorDenseAdd").Device(DEVICE_CPU).TypeConstraint<double>("T").TypeConstraint<int64_t>("Tindices"), SparseTensorDenseAddOp<double, int64_t>);

}  // namespace
}  // namespace stf
