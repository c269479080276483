// Breadth wave of CPU kernels (round 2): search/index ops (Where, Unique,
// TopK), sorted-segment reductions, scan (Cumprod), array restructuring
// (ReverseV2, ListDiff, DynamicPartition, GatherNd, ScatterNd, diag family,
// SpaceToDepth/DepthToSpace, MirrorPad, ReverseSequence, Bitcast) and
// variable scatter updates. Reference analogs live in
// tensorflow/core/kernels/{where_op,unique_op,topk_op,segment_reduction_ops,
// scan_ops,reverse_op,listdiff_op,dynamic_partition_op,gather_nd_op,
// scatter_nd_op,diag_op,matrix_diag_op,matrix_band_part_op,spacetodepth_op,
// mirror_pad_op,reverse_sequence_op,bitcast_op,scatter_op}.cc — these are
// plain-loop redesigns, not ports (the reference is Eigen expression code).
#include <algorithm>
#include <cmath>
#include <cstring>
#include <map>
#include <numeric>
#include <vector>

#include "kernels/kernel_util.h"

namespace stf {
namespace {

// ------------------------------- Where ------------------------------------
class WhereOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& c = ctx->input(0);
    int rank = c.dims() ? c.dims() : 1;
    std::vector<int64_t> dims(rank, 1);
    for (int i = 0; i < c.dims(); ++i) dims[i] = c.dim_size(i);
    const bool* p = c.flat<bool>();
    int64_t n = c.NumElements();
    int64_t hits = 0;
    for (int64_t i = 0; i < n; ++i) hits += p[i] ? 1 : 0;
    Tensor* out = ctx->allocate_output(0, TensorShape({hits, (int64_t)rank}));
    int64_t* o = out->flat<int64_t>();
    int64_t row = 0;
    for (int64_t i = 0; i < n; ++i) {
      if (!p[i]) continue;
      int64_t rem = i;
      for (int d = rank - 1; d >= 0; --d) {
        o[row * rank + d] = rem % dims[d];
        rem /= dims[d];
      }
      ++row;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Where").Device(DEVICE_CPU), WhereOp);

// ------------------------------- Unique ------------------------------------
template <typename T>
class UniqueOp : public OpKernel {
 public:
  explicit UniqueOp(OpKernelConstruction* c) : OpKernel(c) {
    with_counts_ = (num_outputs() == 3);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t n = x.NumElements();
    const T* p = x.flat<T>();
    std::map<T, int32_t> first;
    std::vector<T> uniq;
    std::vector<int32_t> idx(n), counts;
    for (int64_t i = 0; i < n; ++i) {
      auto it = first.find(p[i]);
      if (it == first.end()) {
        int32_t id = (int32_t)uniq.size();
        first.emplace(p[i], id);
        uniq.push_back(p[i]);
        counts.push_back(1);
        idx[i] = id;
      } else {
        idx[i] = it->second;
        counts[it->second]++;
      }
    }
    Tensor* y = ctx->allocate_output(0, TensorShape({(int64_t)uniq.size()}));
    std::copy(uniq.begin(), uniq.end(), y->flat<T>());
    Tensor* ix = ctx->allocate_output(1, TensorShape({n}));
    std::copy(idx.begin(), idx.end(), ix->flat<int32_t>());
    if (with_counts_) {
      Tensor* cn =
          ctx->allocate_output(2, TensorShape({(int64_t)uniq.size()}));
      std::copy(counts.begin(), counts.end(), cn->flat<int32_t>());
    }
  }

 private:
  bool with_counts_ = false;
};
REGISTER_CPU_KERNEL_TYPES("Unique", UniqueOp)
REGISTER_CPU_KERNEL_TYPES("UniqueWithCounts", UniqueOp)

// -------------------------------- TopK -------------------------------------
template <typename T>
class TopKOp : public OpKernel {
 public:
  explicit TopKOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("sorted", &sorted_);
    has_k_attr_ = c->GetAttr("k", &k_attr_).ok();
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t k = has_k_attr_ && num_inputs() == 1
                    ? k_attr_
                    : IntVector(ctx->input(1))[0];
    OP_REQUIRES(ctx, x.dims() >= 1,
                errors::InvalidArgument("TopK input must have rank >= 1"));
    int64_t cols = x.dim_size(x.dims() - 1);
    OP_REQUIRES(ctx, k >= 0 && k <= cols,
                errors::InvalidArgument("k out of range"));
    int64_t rows = x.NumElements() / (cols ? cols : 1);
    TensorShape out_shape = x.shape();
    out_shape.set_dim(x.dims() - 1, k);
    Tensor* vals = ctx->allocate_output(0, out_shape);
    Tensor* idxs = ctx->allocate_output(1, out_shape);
    const T* src = x.flat<T>();
    T* vp = vals->flat<T>();
    int32_t* ip = idxs->flat<int32_t>();
    std::vector<int32_t> ord(cols);
    for (int64_t r = 0; r < rows; ++r) {
      std::iota(ord.begin(), ord.end(), 0);
      const T* rowp = src + r * cols;
      // stable order on ties: by value desc, then index asc
      std::partial_sort(ord.begin(), ord.begin() + k, ord.end(),
                        [&](int32_t a, int32_t b) {
                          if (rowp[a] != rowp[b]) return rowp[a] > rowp[b];
                          return a < b;
                        });
      for (int64_t j = 0; j < k; ++j) {
        vp[r * k + j] = rowp[ord[j]];
        ip[r * k + j] = ord[j];
      }
    }
  }

 private:
  bool sorted_ = true;
  bool has_k_attr_ = false;
  int64_t k_attr_ = 0;
};
REGISTER_CPU_KERNEL_TYPES("TopKV2", TopKOp)
REGISTER_CPU_KERNEL_TYPES("TopK", TopKOp)

// --------------------------- Cumsum / Cumprod -------------------------------
template <typename T, bool PROD>
class ScanOp : public OpKernel {
 public:
  explicit ScanOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("exclusive", &exclusive_);
    c->GetAttr("reverse", &reverse_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t axis = IntVector(ctx->input(1))[0];
    if (axis < 0) axis += x.dims();
    Tensor* out = ctx->allocate_output(0, x.shape());
    int64_t outer = 1, inner = 1, len = x.dim_size(axis);
    for (int i = 0; i < axis; ++i) outer *= x.dim_size(i);
    for (int i = axis + 1; i < x.dims(); ++i) inner *= x.dim_size(i);
    const T* src = x.flat<T>();
    T* dst = out->flat<T>();
    for (int64_t o = 0; o < outer; ++o) {
      for (int64_t in = 0; in < inner; ++in) {
        T acc = PROD ? T(1) : T(0);
        for (int64_t j = 0; j < len; ++j) {
          int64_t jj = reverse_ ? len - 1 - j : j;
          int64_t off = (o * len + jj) * inner + in;
          if (exclusive_) {
            dst[off] = acc;
            acc = PROD ? acc * src[off] : acc + src[off];
          } else {
            acc = PROD ? acc * src[off] : acc + src[off];
            dst[off] = acc;
          }
        }
      }
    }
  }

 private:
  bool exclusive_ = false;
  bool reverse_ = false;
};
template <typename T>
using CumprodOp = ScanOp<T, true>;
template <typename T>
using CumsumOp = ScanOp<T, false>;
REGISTER_CPU_KERNEL_TYPES("Cumprod", CumprodOp)
REGISTER_CPU_KERNEL_TYPES("Cumsum", CumsumOp)

// ----------------------------- DynamicStitch --------------------------------
class DynamicStitchOp : public OpKernel {
 public:
  explicit DynamicStitchOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("N", &n_);
  }
  void Compute(OpKernelContext* ctx) override {
    // inputs: N index tensors then N data tensors
    int64_t max_idx = -1;
    for (int i = 0; i < n_; ++i)
      for (int64_t v : IntVector(ctx->input(i)))
        max_idx = std::max(max_idx, v);
    const Tensor& d0 = ctx->input(n_);
    int idx_dims = ctx->input(0).dims();
    int64_t row = 1;
    TensorShape shape({max_idx + 1});
    for (int i = idx_dims; i < d0.dims(); ++i) {
      shape.AddDim(d0.dim_size(i));
      row *= d0.dim_size(i);
    }
    Tensor* out = ctx->allocate_output(0, shape);
    size_t es = DataTypeSize(d0.dtype());
    std::memset(out->raw_data(), 0, out->TotalBytes());
    for (int i = 0; i < n_; ++i) {
      auto ids = IntVector(ctx->input(i));
      const Tensor& d = ctx->input(n_ + i);
      for (size_t r = 0; r < ids.size(); ++r)
        std::memcpy((char*)out->raw_data() + ids[r] * row * es,
                    (const char*)d.raw_data() + r * row * es, row * es);
    }
  }

 private:
  int64_t n_ = 1;
};
REGISTER_KERNEL_BUILDER(Name("DynamicStitch").Device(DEVICE_CPU),
                        DynamicStitchOp);

// --------------------------- segment reductions ----------------------------
// Sorted-segment ops: ids non-decreasing; output rows = ids.back()+1.
template <typename T, int RED>  // 0 sum, 1 mean, 2 max, 3 min, 4 prod
class SegmentReduceOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& data = ctx->input(0);
    auto ids = IntVector(ctx->input(1));
    OP_REQUIRES(ctx, (int64_t)ids.size() == data.dim_size(0),
                errors::InvalidArgument("segment_ids size mismatch"));
    for (size_t i = 1; i < ids.size(); ++i)
      OP_REQUIRES(ctx, ids[i] >= ids[i - 1],
                  errors::InvalidArgument("segment ids must be sorted"));
    int64_t nseg = ids.empty() ? 0 : ids.back() + 1;
    TensorShape out_shape = data.shape();
    out_shape.set_dim(0, nseg);
    Tensor* out = ctx->allocate_output(0, out_shape);
    int64_t row = data.NumElements() / std::max<int64_t>(data.dim_size(0), 1);
    const T* src = data.flat<T>();
    T* dst = out->flat<T>();
    std::vector<int64_t> count(nseg, 0);
    for (int64_t s = 0; s < nseg * row; ++s)
      dst[s] = RED == 2 ? std::numeric_limits<T>::lowest()
               : RED == 3 ? std::numeric_limits<T>::max()
               : RED == 4 ? T(1)
                          : T(0);
    for (size_t i = 0; i < ids.size(); ++i) {
      int64_t s = ids[i];
      count[s]++;
      for (int64_t j = 0; j < row; ++j) {
        T v = src[i * row + j];
        T& d = dst[s * row + j];
        if (RED == 0 || RED == 1) d += v;
        else if (RED == 2) d = std::max(d, v);
        else if (RED == 3) d = std::min(d, v);
        else d = d * v;
      }
    }
    if (RED == 1) {
      for (int64_t s = 0; s < nseg; ++s) {
        T div = T(count[s] ? count[s] : 1);
        for (int64_t j = 0; j < row; ++j) dst[s * row + j] /= div;
      }
    }
    // empty segments of max/min are 0 in the reference
    if (RED == 2 || RED == 3) {
      for (int64_t s = 0; s < nseg; ++s)
        if (!count[s])
          for (int64_t j = 0; j < row; ++j) dst[s * row + j] = T(0);
  }
  }
};
#define REG_SEGMENT(NAME, RED)                                               \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<float>("T"),              \
      (SegmentReduceOp<float, RED>));                                        \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<double>("T"),             \
      (SegmentReduceOp<double, RED>));                                       \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"),            \
      (SegmentReduceOp<int32_t, RED>));
REG_SEGMENT("SegmentSum", 0)
REG_SEGMENT("SegmentMean", 1)
REG_SEGMENT("SegmentMax", 2)
REG_SEGMENT("SegmentMin", 3)
REG_SEGMENT("SegmentProd", 4)
#undef REG_SEGMENT

// ------------------------------- ReverseV2 ---------------------------------
class ReverseV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto axes = IntVector(ctx->input(1));
    int rank = x.dims();
    std::vector<bool> rev(rank, false);
    for (auto a : axes) rev[(a % rank + rank) % rank] = true;
    Tensor* out = ctx->allocate_output(0, x.shape());
    size_t es = DataTypeSize(x.dtype());
    std::vector<int64_t> strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      strides[i] = strides[i + 1] * x.dim_size(i + 1);
    int64_t n = x.NumElements();
    const char* src = (const char*)x.raw_data();
    char* dst = (char*)out->raw_data();
    std::vector<int64_t> idx(rank, 0);
    for (int64_t i = 0; i < n; ++i) {
      int64_t off = 0;
      for (int d = 0; d < rank; ++d) {
        int64_t v = rev[d] ? x.dim_size(d) - 1 - idx[d] : idx[d];
        off += v * strides[d];
      }
      std::memcpy(dst + i * es, src + off * es, es);
      for (int d = rank - 1; d >= 0; --d) {
        if (++idx[d] < x.dim_size(d)) break;
        idx[d] = 0;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("ReverseV2").Device(DEVICE_CPU), ReverseV2Op);

// ------------------------------- ListDiff ----------------------------------
template <typename T>
class ListDiffOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& y = ctx->input(1);
    const T* xp = x.flat<T>();
    const T* yp = y.flat<T>();
    std::vector<T> yset(yp, yp + y.NumElements());
    std::sort(yset.begin(), yset.end());
    std::vector<T> out;
    std::vector<int32_t> idx;
    for (int64_t i = 0; i < x.NumElements(); ++i) {
      if (!std::binary_search(yset.begin(), yset.end(), xp[i])) {
        out.push_back(xp[i]);
        idx.push_back((int32_t)i);
      }
    }
    Tensor* o = ctx->allocate_output(0, TensorShape({(int64_t)out.size()}));
    std::copy(out.begin(), out.end(), o->flat<T>());
    Tensor* oi = ctx->allocate_output(1, TensorShape({(int64_t)idx.size()}));
    std::copy(idx.begin(), idx.end(), oi->flat<int32_t>());
  }
};
REGISTER_CPU_KERNEL_TYPES("ListDiff", ListDiffOp)

// --------------------------- DynamicPartition -------------------------------
class DynamicPartitionOp : public OpKernel {
 public:
  explicit DynamicPartitionOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("num_partitions", &nparts_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& data = ctx->input(0);
    const Tensor& parts = ctx->input(1);
    auto pv = IntVector(parts);
    int64_t prefix = parts.NumElements();
    OP_REQUIRES(ctx, data.dims() >= parts.dims(),
                errors::InvalidArgument("partitions rank too high"));
    int64_t row = 1;
    for (int i = parts.dims(); i < data.dims(); ++i) row *= data.dim_size(i);
    std::vector<std::vector<int64_t>> members(nparts_);
    for (int64_t i = 0; i < prefix; ++i) {
      OP_REQUIRES(ctx, pv[i] >= 0 && pv[i] < nparts_,
                  errors::InvalidArgument("partition id out of range"));
      members[pv[i]].push_back(i);
    }
    size_t es = DataTypeSize(data.dtype());
    for (int p = 0; p < nparts_; ++p) {
      TensorShape shape({(int64_t)members[p].size()});
      for (int i = parts.dims(); i < data.dims(); ++i)
        shape.AddDim(data.dim_size(i));
      Tensor* out = ctx->allocate_output(p, shape);
      char* dst = (char*)out->raw_data();
      for (size_t r = 0; r < members[p].size(); ++r)
        std::memcpy(dst + r * row * es,
                    (const char*)data.raw_data() + members[p][r] * row * es,
                    row * es);
    }
  }

 private:
  int64_t nparts_ = 1;
};
REGISTER_KERNEL_BUILDER(Name("DynamicPartition").Device(DEVICE_CPU),
                        DynamicPartitionOp);

// -------------------------------- GatherNd ----------------------------------
class GatherNdOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& params = ctx->input(0);
    const Tensor& indices = ctx->input(1);
    int64_t id = indices.dim_size(indices.dims() - 1);
    OP_REQUIRES(ctx, id <= params.dims(),
                errors::InvalidArgument("index depth > params rank"));
    int64_t nidx = indices.NumElements() / id;
    int64_t row = 1;
    for (int i = (int)id; i < params.dims(); ++i) row *= params.dim_size(i);
    TensorShape out_shape;
    for (int i = 0; i < indices.dims() - 1; ++i)
      out_shape.AddDim(indices.dim_size(i));
    for (int i = (int)id; i < params.dims(); ++i)
      out_shape.AddDim(params.dim_size(i));
    Tensor* out = ctx->allocate_output(0, out_shape);
    std::vector<int64_t> strides(id, 1);
    int64_t s = row;
    for (int d = (int)id - 1; d >= 0; --d) {
      strides[d] = s;
      s *= params.dim_size(d);
    }
    auto iv = IntVector(indices);
    size_t es = DataTypeSize(params.dtype());
    char* dst = (char*)out->raw_data();
    for (int64_t i = 0; i < nidx; ++i) {
      int64_t off = 0;
      bool oob = false;
      for (int64_t d = 0; d < id; ++d) {
        int64_t v = iv[i * id + d];
        if (v < 0 || v >= params.dim_size((int)d)) { oob = true; break; }
        off += v * strides[d];
      }
      OP_REQUIRES(ctx, !oob, errors::InvalidArgument("GatherNd index OOB"));
      std::memcpy(dst + i * row * es,
                  (const char*)params.raw_data() + off * es, row * es);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("GatherNd").Device(DEVICE_CPU), GatherNdOp);

// -------------------------------- ScatterNd ---------------------------------
template <typename T>
class ScatterNdOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& indices = ctx->input(0);
    const Tensor& updates = ctx->input(1);
    auto shp = IntVector(ctx->input(2));
    TensorShape out_shape(shp);
    Tensor* out = ctx->allocate_output(0, out_shape);
    std::memset(out->raw_data(), 0, out->TotalBytes());
    int64_t id = indices.dim_size(indices.dims() - 1);
    int64_t nidx = indices.NumElements() / std::max<int64_t>(id, 1);
    int64_t row = 1;
    for (size_t i = id; i < shp.size(); ++i) row *= shp[i];
    std::vector<int64_t> strides(id, 1);
    int64_t s = row;
    for (int64_t d = id - 1; d >= 0; --d) {
      strides[d] = s;
      s *= shp[d];
    }
    auto iv = IntVector(indices);
    const T* up = updates.flat<T>();
    T* dst = out->flat<T>();
    for (int64_t i = 0; i < nidx; ++i) {
      int64_t off = 0;
      for (int64_t d = 0; d < id; ++d) off += iv[i * id + d] * strides[d];
      for (int64_t j = 0; j < row; ++j) dst[off + j] += up[i * row + j];
    }
  }
};
REGISTER_CPU_KERNEL_TYPES("ScatterNd", ScatterNdOp)

// --------------------------------- diag family ------------------------------
class DiagOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t n = x.NumElements();
    TensorShape shape;
    for (int i = 0; i < x.dims(); ++i) shape.AddDim(x.dim_size(i));
    for (int i = 0; i < x.dims(); ++i) shape.AddDim(x.dim_size(i));
    Tensor* out = ctx->allocate_output(0, shape);
    size_t es = DataTypeSize(x.dtype());
    std::memset(out->raw_data(), 0, out->TotalBytes());
    for (int64_t i = 0; i < n; ++i)
      std::memcpy((char*)out->raw_data() + (i * n + i) * es,
                  (const char*)x.raw_data() + i * es, es);
  }
};
REGISTER_KERNEL_BUILDER(Name("Diag").Device(DEVICE_CPU), DiagOp);

class DiagPartOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int half = x.dims() / 2;
    TensorShape shape;
    int64_t n = 1;
    for (int i = 0; i < half; ++i) {
      shape.AddDim(x.dim_size(i));
      n *= x.dim_size(i);
    }
    Tensor* out = ctx->allocate_output(0, shape);
    size_t es = DataTypeSize(x.dtype());
    for (int64_t i = 0; i < n; ++i)
      std::memcpy((char*)out->raw_data() + i * es,
                  (const char*)x.raw_data() + (i * n + i) * es, es);
  }
};
REGISTER_KERNEL_BUILDER(Name("DiagPart").Device(DEVICE_CPU), DiagPartOp);

// batched matrix diag ops: operate on the innermost 1 or 2 dims.
class MatrixDiagOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t n = x.dim_size(x.dims() - 1);
    int64_t batch = x.NumElements() / n;
    TensorShape shape = x.shape();
    shape.AddDim(n);
    Tensor* out = ctx->allocate_output(0, shape);
    size_t es = DataTypeSize(x.dtype());
    std::memset(out->raw_data(), 0, out->TotalBytes());
    for (int64_t b = 0; b < batch; ++b)
      for (int64_t i = 0; i < n; ++i)
        std::memcpy((char*)out->raw_data() + ((b * n + i) * n + i) * es,
                    (const char*)x.raw_data() + (b * n + i) * es, es);
  }
};
REGISTER_KERNEL_BUILDER(Name("MatrixDiag").Device(DEVICE_CPU), MatrixDiagOp);
REGISTER_KERNEL_BUILDER(Name("BatchMatrixDiag").Device(DEVICE_CPU),
                        MatrixDiagOp);

class MatrixDiagPartOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t m = x.dim_size(x.dims() - 2), n = x.dim_size(x.dims() - 1);
    int64_t d = std::min(m, n);
    int64_t batch = x.NumElements() / (m * n);
    TensorShape shape;
    for (int i = 0; i < x.dims() - 2; ++i) shape.AddDim(x.dim_size(i));
    shape.AddDim(d);
    Tensor* out = ctx->allocate_output(0, shape);
    size_t es = DataTypeSize(x.dtype());
    for (int64_t b = 0; b < batch; ++b)
      for (int64_t i = 0; i < d; ++i)
        std::memcpy((char*)out->raw_data() + (b * d + i) * es,
                    (const char*)x.raw_data() + (b * m * n + i * n + i) * es,
                    es);
  }
};
REGISTER_KERNEL_BUILDER(Name("MatrixDiagPart").Device(DEVICE_CPU),
                        MatrixDiagPartOp);
REGISTER_KERNEL_BUILDER(Name("BatchMatrixDiagPart").Device(DEVICE_CPU),
                        MatrixDiagPartOp);

class MatrixSetDiagOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& diag = ctx->input(1);
    int64_t m = x.dim_size(x.dims() - 2), n = x.dim_size(x.dims() - 1);
    int64_t d = std::min(m, n);
    int64_t batch = x.NumElements() / (m * n);
    Tensor* out = ctx->allocate_output(0, x.shape());
    size_t es = DataTypeSize(x.dtype());
    std::memcpy(out->raw_data(), x.raw_data(), x.TotalBytes());
    for (int64_t b = 0; b < batch; ++b)
      for (int64_t i = 0; i < d; ++i)
        std::memcpy((char*)out->raw_data() + (b * m * n + i * n + i) * es,
                    (const char*)diag.raw_data() + (b * d + i) * es, es);
  }
};
REGISTER_KERNEL_BUILDER(Name("MatrixSetDiag").Device(DEVICE_CPU),
                        MatrixSetDiagOp);
REGISTER_KERNEL_BUILDER(Name("BatchMatrixSetDiag").Device(DEVICE_CPU),
                        MatrixSetDiagOp);

template <typename T>
class MatrixBandPartOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    int64_t lower = IntVector(ctx->input(1))[0];
    int64_t upper = IntVector(ctx->input(2))[0];
    int64_t m = x.dim_size(x.dims() - 2), n = x.dim_size(x.dims() - 1);
    int64_t batch = x.NumElements() / (m * n);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* src = x.flat<T>();
    T* dst = out->flat<T>();
    for (int64_t b = 0; b < batch; ++b)
      for (int64_t i = 0; i < m; ++i)
        for (int64_t j = 0; j < n; ++j) {
          bool keep = (lower < 0 || i - j <= lower) &&
                      (upper < 0 || j - i <= upper);
          dst[(b * m + i) * n + j] = keep ? src[(b * m + i) * n + j] : T(0);
        }
  }
};
REGISTER_CPU_KERNEL_TYPES("MatrixBandPart", MatrixBandPartOp)
REGISTER_CPU_KERNEL_TYPES("BatchMatrixBandPart", MatrixBandPartOp)

// --------------------------- SpaceToDepth / DepthToSpace --------------------
class SpaceToDepthOp : public OpKernel {
 public:
  explicit SpaceToDepthOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("block_size", &bs_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);  // NHWC
    int64_t n = x.dim_size(0), h = x.dim_size(1), w = x.dim_size(2),
            c = x.dim_size(3);
    OP_REQUIRES(ctx, h % bs_ == 0 && w % bs_ == 0,
                errors::InvalidArgument("dims not divisible by block_size"));
    Tensor* out = ctx->allocate_output(
        0, TensorShape({n, h / bs_, w / bs_, c * bs_ * bs_}));
    size_t es = DataTypeSize(x.dtype());
    const char* src = (const char*)x.raw_data();
    char* dst = (char*)out->raw_data();
    int64_t oh = h / bs_, ow = w / bs_;
    for (int64_t b = 0; b < n; ++b)
      for (int64_t i = 0; i < oh; ++i)
        for (int64_t j = 0; j < ow; ++j)
          for (int64_t bi = 0; bi < bs_; ++bi)
            for (int64_t bj = 0; bj < bs_; ++bj) {
              int64_t so = ((b * h + i * bs_ + bi) * w + j * bs_ + bj) * c;
              int64_t doff = ((b * oh + i) * ow + j) * c * bs_ * bs_ +
                             (bi * bs_ + bj) * c;
              std::memcpy(dst + doff * es, src + so * es, c * es);
            }
  }

 private:
  int64_t bs_ = 2;
};
REGISTER_KERNEL_BUILDER(Name("SpaceToDepth").Device(DEVICE_CPU),
                        SpaceToDepthOp);

class DepthToSpaceOp : public OpKernel {
 public:
  explicit DepthToSpaceOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("block_size", &bs_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);  // NHWC
    int64_t n = x.dim_size(0), h = x.dim_size(1), w = x.dim_size(2),
            c = x.dim_size(3);
    int64_t oc = c / (bs_ * bs_);
    OP_REQUIRES(ctx, oc * bs_ * bs_ == c,
                errors::InvalidArgument("depth not divisible"));
    Tensor* out =
        ctx->allocate_output(0, TensorShape({n, h * bs_, w * bs_, oc}));
    size_t es = DataTypeSize(x.dtype());
    const char* src = (const char*)x.raw_data();
    char* dst = (char*)out->raw_data();
    for (int64_t b = 0; b < n; ++b)
      for (int64_t i = 0; i < h; ++i)
        for (int64_t j = 0; j < w; ++j)
          for (int64_t bi = 0; bi < bs_; ++bi)
            for (int64_t bj = 0; bj < bs_; ++bj) {
              int64_t so = ((b * h + i) * w + j) * c + (bi * bs_ + bj) * oc;
              int64_t doff =
                  ((b * h * bs_ + i * bs_ + bi) * w * bs_ + j * bs_ + bj) *
                  oc;
              std::memcpy(dst + doff * es, src + so * es, oc * es);
            }
  }

 private:
  int64_t bs_ = 2;
};
REGISTER_KERNEL_BUILDER(Name("DepthToSpace").Device(DEVICE_CPU),
                        DepthToSpaceOp);

// -------------------------------- MirrorPad ---------------------------------
// mode REFLECT (no edge dup) / SYMMETRIC (edge dup).
class MirrorPadOp : public OpKernel {
 public:
  explicit MirrorPadOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("mode", &mode_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto pv = IntVector(ctx->input(1));
    int rank = x.dims();
    TensorShape out_shape;
    for (int i = 0; i < rank; ++i)
      out_shape.AddDim(x.dim_size(i) + pv[2 * i] + pv[2 * i + 1]);
    Tensor* out = ctx->allocate_output(0, out_shape);
    size_t es = DataTypeSize(x.dtype());
    std::vector<int64_t> xs(rank, 1);
    for (int i = rank - 2; i >= 0; --i) xs[i] = xs[i + 1] * x.dim_size(i + 1);
    int64_t n = out->NumElements();
    std::vector<int64_t> idx(rank, 0);
    const char* src = (const char*)x.raw_data();
    char* dst = (char*)out->raw_data();
    bool reflect = mode_ == "REFLECT";
    for (int64_t i = 0; i < n; ++i) {
      int64_t so = 0;
      for (int d = 0; d < rank; ++d) {
        int64_t v = idx[d] - pv[2 * d];
        int64_t dim = x.dim_size(d);
        // REFLECT mirrors about the edge element (no duplication);
        // SYMMETRIC mirrors about the edge boundary (edge duplicated).
        if (v < 0) v = reflect ? -v : -v - 1;
        else if (v >= dim) v = reflect ? 2 * dim - 2 - v : 2 * dim - 1 - v;
        so += v * xs[d];
      }
      std::memcpy(dst + i * es, src + so * es, es);
      for (int d = rank - 1; d >= 0; --d) {
        if (++idx[d] < out_shape.dim_size(d)) break;
        idx[d] = 0;
      }
    }
  }

 private:
  std::string mode_ = "REFLECT";
};
REGISTER_KERNEL_BUILDER(Name("MirrorPad").Device(DEVICE_CPU), MirrorPadOp);

// ----------------------------- ReverseSequence ------------------------------
template <typename T>
class ReverseSequenceOp : public OpKernel {
 public:
  explicit ReverseSequenceOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("seq_dim", &seq_dim_);
    c->GetAttr("batch_dim", &batch_dim_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    auto lens = IntVector(ctx->input(1));
    Tensor* out = ctx->allocate_output(0, x.shape());
    int rank = x.dims();
    std::vector<int64_t> strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      strides[i] = strides[i + 1] * x.dim_size(i + 1);
    const T* src = x.flat<T>();
    T* dst = out->flat<T>();
    int64_t n = x.NumElements();
    std::vector<int64_t> idx(rank, 0);
    for (int64_t i = 0; i < n; ++i) {
      int64_t b = idx[batch_dim_];
      int64_t s = idx[seq_dim_];
      int64_t s2 = (s < lens[b]) ? lens[b] - 1 - s : s;
      int64_t so = 0;
      for (int d = 0; d < rank; ++d)
        so += (d == seq_dim_ ? s2 : idx[d]) * strides[d];
      dst[i] = src[so];
      for (int d = rank - 1; d >= 0; --d) {
        if (++idx[d] < x.dim_size(d)) break;
        idx[d] = 0;
      }
    }
  }

 private:
  int64_t seq_dim_ = 0;
  int64_t batch_dim_ = 0;
};
REGISTER_CPU_KERNEL_TYPES("ReverseSequence", ReverseSequenceOp)

// --------------------------------- Bitcast ----------------------------------
class BitcastOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    DataType out_t = output_type(0);
    size_t in_es = DataTypeSize(x.dtype());
    size_t out_es = DataTypeSize(out_t);
    TensorShape shape = x.shape();
    if (in_es > out_es) {
      shape.AddDim(in_es / out_es);
    } else if (in_es < out_es) {
      OP_REQUIRES(ctx,
                  shape.dims() > 0 &&
                      shape.dim_size(shape.dims() - 1) ==
                          (int64_t)(out_es / in_es),
                  errors::InvalidArgument("bitcast: bad innermost dim"));
      shape.RemoveDim(shape.dims() - 1);
    }
    Tensor* out = ctx->allocate_output(0, shape);
    std::memcpy(out->raw_data(), x.raw_data(), x.TotalBytes());
  }
};
REGISTER_KERNEL_BUILDER(Name("Bitcast").Device(DEVICE_CPU), BitcastOp);

// -------------------------- variable scatter updates ------------------------
// op semantics: ref[indices[i], :] op= updates[i, :]; returns the ref.
template <typename T, int OPK>  // 0 update, 1 mul, 2 div
class ScatterUpdateOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    Tensor ref = ctx->input(0);
    auto ids = IntVector(ctx->input(1));
    const Tensor& up = ctx->input(2);
    int64_t row = ref.NumElements() / std::max<int64_t>(ref.dim_size(0), 1);
    T* dst = ref.flat<T>();
    const T* u = up.flat<T>();
    for (size_t i = 0; i < ids.size(); ++i) {
      OP_REQUIRES(ctx, ids[i] >= 0 && ids[i] < ref.dim_size(0),
                  errors::InvalidArgument("scatter index out of range"));
      for (int64_t j = 0; j < row; ++j) {
        T v = u[i * row + j];
        T& d = dst[ids[i] * row + j];
        if (OPK == 0) d = v;
        else if (OPK == 1) d = d * v;
        else d = d / v;
      }
    }
    ctx->set_output(0, ref);
  }
};
#define REG_SCATTER(NAME, OPK)                                               \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<float>("T"),              \
      (ScatterUpdateOp<float, OPK>));                                        \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<double>("T"),             \
      (ScatterUpdateOp<double, OPK>));                                       \
  REGISTER_KERNEL_BUILDER(                                                   \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"),            \
      (ScatterUpdateOp<int32_t, OPK>));
REG_SCATTER("ScatterUpdate", 0)
REG_SCATTER("ScatterMul", 1)
REG_SCATTER("ScatterDiv", 2)
#undef REG_SCATTER

// ------------------------------ StridedSlice --------------------------------
// Full reference semantics (core/util/strided_slice_op.cc redesigned):
// sparse spec (begin/end/strides + 5 masks) canonicalized to one dense
// (begin,end,stride) per input dim plus the final shape (new_axis dims
// inserted, shrink dims removed). The element ORDER of the processing
// output equals the final-shape output, so the kernel writes the output
// tensor directly.
namespace strided {

struct Spec {
  std::vector<int64_t> begin, end, stride;  // dense, one per input dim
  std::vector<int64_t> proc_dims;           // extent per input dim
  std::vector<int64_t> final_dims;          // output shape
  int64_t total = 1;
};

inline Status Build(const TensorShape& in, const std::vector<int64_t>& b,
                    const std::vector<int64_t>& e,
                    const std::vector<int64_t>& st, int64_t begin_mask,
                    int64_t end_mask, int64_t ellipsis_mask,
                    int64_t new_axis_mask, int64_t shrink_mask, Spec* out) {
  int n = (int)b.size();
  int rank = in.dims();
  // count spec entries that consume an input dim
  int consuming = 0;
  for (int i = 0; i < n; ++i)
    if (!((new_axis_mask >> i) & 1) && !((ellipsis_mask >> i) & 1))
      ++consuming;
  int ell_dims = rank - consuming;  // dims covered by the ellipsis
  if (ell_dims < 0) return errors::InvalidArgument("too many slice specs");
  bool has_ellipsis = ellipsis_mask != 0;
  int dim = 0;  // current input dim
  auto add_full = [&](int64_t) {
    out->begin.push_back(0);
    out->end.push_back(in.dim_size(dim));
    out->stride.push_back(1);
    int64_t ext = in.dim_size(dim);
    out->proc_dims.push_back(ext);
    out->final_dims.push_back(ext);
    ++dim;
  };
  for (int i = 0; i < n; ++i) {
    if ((ellipsis_mask >> i) & 1) {
      for (int k = 0; k < ell_dims; ++k) add_full(0);
      continue;
    }
    if ((new_axis_mask >> i) & 1) {
      out->final_dims.push_back(1);
      continue;
    }
    if (dim >= rank) return errors::InvalidArgument("slice spec OOB");
    int64_t d = in.dim_size(dim);
    int64_t stride = st[i] == 0 ? 1 : st[i];
    int64_t bi = b[i], ei = e[i];
    if ((shrink_mask >> i) & 1) {
      if (bi < 0) bi += d;
      if (bi < 0 || bi >= d)
        return errors::InvalidArgument("shrink index out of range");
      out->begin.push_back(bi);
      out->end.push_back(bi + 1);
      out->stride.push_back(1);
      out->proc_dims.push_back(1);
      ++dim;
      continue;  // no final dim
    }
    if ((begin_mask >> i) & 1) bi = stride > 0 ? 0 : d - 1;
    else if (bi < 0) bi += d;
    if ((end_mask >> i) & 1) ei = stride > 0 ? d : -d - 1;
    else if (ei < 0) ei += d;
    // clamp
    if (stride > 0) {
      bi = std::min(std::max<int64_t>(bi, 0), d);
      ei = std::min(std::max<int64_t>(ei, 0), d);
    } else {
      bi = std::min(std::max<int64_t>(bi, -1), d - 1);
      ei = std::min(std::max<int64_t>(ei, -d - 1), d - 1);
      if ((end_mask >> i) & 1) ei = -1;
    }
    int64_t ext = stride > 0 ? (ei - bi + stride - 1) / stride
                             : (bi - ei - stride - 1) / (-stride);
    if (ext < 0) ext = 0;
    out->begin.push_back(bi);
    out->end.push_back(ei);
    out->stride.push_back(stride);
    out->proc_dims.push_back(ext);
    out->final_dims.push_back(ext);
    ++dim;
  }
  // remaining input dims: full range
  while (dim < rank) add_full(0);
  out->total = 1;
  for (int64_t x : out->proc_dims) out->total *= x;
  return Status::OK();
}

}  // namespace strided

class StridedSliceOp : public OpKernel {
 public:
  explicit StridedSliceOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("begin_mask", &bm_);
    c->GetAttr("end_mask", &em_);
    c->GetAttr("ellipsis_mask", &elm_);
    c->GetAttr("new_axis_mask", &nam_);
    c->GetAttr("shrink_axis_mask", &sam_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    strided::Spec sp;
    OP_REQUIRES_OK(ctx, strided::Build(x.shape(), IntVector(ctx->input(1)),
                                       IntVector(ctx->input(2)),
                                       IntVector(ctx->input(3)), bm_, em_,
                                       elm_, nam_, sam_, &sp));
    Tensor* out = ctx->allocate_output(0, TensorShape(sp.final_dims));
    int rank = x.dims();
    std::vector<int64_t> in_strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * x.dim_size(i + 1);
    size_t es = DataTypeSize(x.dtype());
    const char* src = (const char*)x.raw_data();
    char* dst = (char*)out->raw_data();
    std::vector<int64_t> idx(rank, 0);
    for (int64_t i = 0; i < sp.total; ++i) {
      int64_t off = 0;
      for (int d = 0; d < rank; ++d)
        off += (sp.begin[d] + idx[d] * sp.stride[d]) * in_strides[d];
      std::memcpy(dst + i * es, src + off * es, es);
      for (int d = rank - 1; d >= 0; --d) {
        if (++idx[d] < sp.proc_dims[d]) break;
        idx[d] = 0;
      }
    }
  }

 private:
  int64_t bm_ = 0, em_ = 0, elm_ = 0, nam_ = 0, sam_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("StridedSlice").Device(DEVICE_CPU),
                        StridedSliceOp);

template <typename T>
class StridedSliceGradOp : public OpKernel {
 public:
  explicit StridedSliceGradOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("begin_mask", &bm_);
    c->GetAttr("end_mask", &em_);
    c->GetAttr("ellipsis_mask", &elm_);
    c->GetAttr("new_axis_mask", &nam_);
    c->GetAttr("shrink_axis_mask", &sam_);
  }
  void Compute(OpKernelContext* ctx) override {
    TensorShape in_shape(IntVector(ctx->input(0)));
    const Tensor& dy = ctx->input(4);
    strided::Spec sp;
    OP_REQUIRES_OK(ctx, strided::Build(in_shape, IntVector(ctx->input(1)),
                                       IntVector(ctx->input(2)),
                                       IntVector(ctx->input(3)), bm_, em_,
                                       elm_, nam_, sam_, &sp));
    OP_REQUIRES(ctx, dy.NumElements() == sp.total,
                errors::InvalidArgument("dy size mismatch"));
    Tensor* out = ctx->allocate_output(0, in_shape);
    std::memset(out->raw_data(), 0, out->TotalBytes());
    int rank = in_shape.dims();
    std::vector<int64_t> in_strides(rank, 1);
    for (int i = rank - 2; i >= 0; --i)
      in_strides[i] = in_strides[i + 1] * in_shape.dim_size(i + 1);
    const T* src = dy.flat<T>();
    T* dst = out->flat<T>();
    std::vector<int64_t> idx(rank, 0);
    for (int64_t i = 0; i < sp.total; ++i) {
      int64_t off = 0;
      for (int d = 0; d < rank; ++d)
        off += (sp.begin[d] + idx[d] * sp.stride[d]) * in_strides[d];
      dst[off] += src[i];
      for (int d = rank - 1; d >= 0; --d) {
        if (++idx[d] < sp.proc_dims[d]) break;
        idx[d] = 0;
      }
    }
  }

 private:
  int64_t bm_ = 0, em_ = 0, elm_ = 0, nam_ = 0, sam_ = 0;
};
REGISTER_CPU_KERNEL_TYPES("StridedSliceGrad", StridedSliceGradOp)
REGISTER_KERNEL_BUILDER(Name("StridedSliceGrad").Device(DEVICE_CPU)
                            .TypeConstraint<bfloat16>("T"),
                        StridedSliceGradOp<bfloat16>);

// ------------------------ math breadth (trig/special) -----------------------
template <typename T, typename F>
class UnaryXOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* xp = x.flat<T>();
    T* op = out->flat<T>();
    F f;
    for (int64_t i = 0; i < x.NumElements(); ++i) op[i] = f((double)xp[i]);
  }
};
#define XFN(NAME, EXPR)                                            \
  struct NAME {                                                    \
    double operator()(double a) const { return EXPR; }             \
  };
XFN(FTan, std::tan(a))
XFN(FAsin, std::asin(a))
XFN(FAcos, std::acos(a))
XFN(FAtan, std::atan(a))
XFN(FErf, std::erf(a))
XFN(FErfc, std::erfc(a))
XFN(FExpm1, std::expm1(a))
XFN(FLgamma, std::lgamma(a))
XFN(FRint2, std::rint(a))
XFN(FSoftsign, a / (1.0 + std::abs(a)))
XFN(FInvX, 1.0 / a)
#undef XFN
struct FDigamma {
  double operator()(double x) const {
    // standard asymptotic series with upward recurrence
    double r = 0.0;
    while (x < 6.0) { r -= 1.0 / x; x += 1.0; }
    double f = 1.0 / (x * x);
    return r + std::log(x) - 0.5 / x -
           f * (1.0 / 12 - f * (1.0 / 120 - f * (1.0 / 252 - f / 240)));
  }
};
#define REG_XUNARY(OP, F)                                                     \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"),                 \
      (UnaryXOp<float, F>));                                                  \
  REGISTER_KERNEL_BUILDER(                                                    \
      Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"),                \
      (UnaryXOp<double, F>));
REG_XUNARY("Tan", FTan)
REG_XUNARY("Asin", FAsin)
REG_XUNARY("Acos", FAcos)
REG_XUNARY("Atan", FAtan)
REG_XUNARY("Erf", FErf)
REG_XUNARY("Erfc", FErfc)
REG_XUNARY("Expm1", FExpm1)
REG_XUNARY("Lgamma", FLgamma)
REG_XUNARY("Digamma", FDigamma)
REG_XUNARY("Rint", FRint2)
REG_XUNARY("Softsign", FSoftsign)
REG_XUNARY("Inv", FInvX)
#undef REG_XUNARY

template <typename T>
class SoftsignGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& g = ctx->input(0);
    const Tensor& x = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* gp = g.flat<T>();
    const T* xp = x.flat<T>();
    T* op = out->flat<T>();
    for (int64_t i = 0; i < x.NumElements(); ++i) {
      T d = T(1) + (xp[i] < T(0) ? -xp[i] : xp[i]);
      op[i] = gp[i] / (d * d);
    }
  }
};
REGISTER_CPU_KERNEL_FLOATS("SoftsignGrad", SoftsignGradOp)

template <typename T>
class InvGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& y = ctx->input(0);
    const Tensor& dy = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, y.shape());
    const T* yp = y.flat<T>();
    const T* dp = dy.flat<T>();
    T* op = out->flat<T>();
    for (int64_t i = 0; i < y.NumElements(); ++i)
      op[i] = -dp[i] * yp[i] * yp[i];
  }
};
REGISTER_CPU_KERNEL_FLOATS("InvGrad", InvGradOp)

// truncated mod (C semantics), matching the reference Mod op
template <typename T>
class ModOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& y = ctx->input(1);
    BCast b(x.shape(), y.shape());
    OP_REQUIRES(ctx, b.valid, errors::InvalidArgument("Mod: bad broadcast"));
    Tensor* out = ctx->allocate_output(0, b.out_shape());
    const T* xp = x.flat<T>();
    const T* yp = y.flat<T>();
    T* op = out->flat<T>();
    for (int64_t i = 0; i < b.num_elements; ++i) {
      int64_t xi, yi;
      b.Map(i, &xi, &yi);
      if constexpr (std::is_integral<T>::value)
        op[i] = xp[xi] % yp[yi];
      else
        op[i] = std::fmod((double)xp[xi], (double)yp[yi]);
    }
  }
};
REGISTER_CPU_KERNEL_TYPES("Mod", ModOp)

template <typename T>
class ApproximateEqualOp : public OpKernel {
 public:
  explicit ApproximateEqualOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("tolerance", &tol_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    const Tensor& y = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, x.shape());
    const T* xp = x.flat<T>();
    const T* yp = y.flat<T>();
    bool* op = out->flat<bool>();
    for (int64_t i = 0; i < x.NumElements(); ++i)
      op[i] = std::abs((double)(xp[i] - yp[i])) < (double)tol_;
  }

 private:
  float tol_ = 1e-5f;
};
REGISTER_CPU_KERNEL_TYPES("ApproximateEqual", ApproximateEqualOp)

class AsStringOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& x = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, x.shape());
    for (int64_t i = 0; i < x.NumElements(); ++i) {
      std::string v;
      switch (x.dtype()) {
        case DT_FLOAT: v = std::to_string(x.flat<float>()[i]); break;
        case DT_DOUBLE: v = std::to_string(x.flat<double>()[i]); break;
        case DT_INT32: v = std::to_string(x.flat<int32_t>()[i]); break;
        case DT_INT64: v = std::to_string(x.flat<int64_t>()[i]); break;
        case DT_BOOL: v = x.flat<bool>()[i] ? "true" : "false"; break;
        default:
          ctx->SetStatus(errors::InvalidArgument("AsString dtype"));
          return;
      }
      out->flat<std::string>()[i] = v;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("AsString").Device(DEVICE_CPU), AsStringOp);

class DecodeRawOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    DataType out_t = output_type(0);
    size_t es = DataTypeSize(out_t);
    OP_REQUIRES(ctx, in.NumElements() > 0,
                errors::InvalidArgument("DecodeRaw: empty input"));
    const std::string& first = in.flat<std::string>()[0];
    int64_t elems = (int64_t)(first.size() / es);
    TensorShape shape = in.shape();
    shape.AddDim(elems);
    Tensor* out = ctx->allocate_output(0, shape);
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      const std::string& s = in.flat<std::string>()[i];
      OP_REQUIRES(ctx, (int64_t)(s.size() / es) == elems,
                  errors::InvalidArgument("DecodeRaw: ragged strings"));
      std::memcpy((char*)out->raw_data() + i * elems * es, s.data(),
                  elems * es);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("DecodeRaw").Device(DEVICE_CPU), DecodeRawOp);

}  // namespace
}  // namespace stf
