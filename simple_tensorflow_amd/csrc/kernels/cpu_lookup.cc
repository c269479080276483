// Lookup tables, stacks and barriers (reference core/kernels/
// lookup_table_op.cc, lookup_table_init_op.cc, stack_ops.cc,
// barrier_ops.cc): resource-manager-backed structures addressed by a
// string handle, following this framework's queue pattern
// (cpu_queue.cc). Keys are stored by their byte encoding so one table
// implementation serves int32/int64/string keys.
#include <condition_variable>
#include <cstring>
#include <deque>
#include <unordered_map>
#include <vector>

#include "kernels/kernel_util.h"
#include "kernels/resource_mgr.h"

namespace stf {
namespace {

std::string KeyBytes(const Tensor& t, int64_t i) {
  if (t.dtype() == DT_STRING) return t.flat<std::string>()[i];
  size_t es = DataTypeSize(t.dtype());
  return std::string((const char*)t.raw_data() + i * es, es);
}

class TableResource : public ResourceBase {
 public:
  TableResource(DataType key_t, DataType value_t, bool mutable_table)
      : key_t_(key_t), value_t_(value_t), mutable_(mutable_table) {}

  Status Insert(const Tensor& keys, const Tensor& values) {
    std::lock_guard<std::mutex> l(mu_);
    if (!mutable_ && initialized_)
      return errors::FailedPrecondition("table already initialized");
    if (keys.dtype() != key_t_ || values.dtype() != value_t_)
      return errors::InvalidArgument("table dtype mismatch on insert");
    for (int64_t i = 0; i < keys.NumElements(); ++i) {
      std::string k = KeyBytes(keys, i);
      auto it = index_.find(k);
      if (it == index_.end()) {
        index_[k] = (int64_t)rows_.size();
        rows_.push_back(ValueAt(values, i));
        key_rows_.push_back(k);
      } else {
        rows_[it->second] = ValueAt(values, i);
      }
    }
    initialized_ = true;
    return Status::OK();
  }

  void Find(const Tensor& keys, const Tensor& dflt, Tensor* out) {
    std::lock_guard<std::mutex> l(mu_);
    size_t es = DataTypeSize(value_t_);
    for (int64_t i = 0; i < keys.NumElements(); ++i) {
      auto it = index_.find(KeyBytes(keys, i));
      if (value_t_ == DT_STRING) {
        out->flat<std::string>()[i] =
            it == index_.end() ? dflt.flat<std::string>()[0]
                               : rows_[it->second];
      } else {
        const void* src = it == index_.end() ? dflt.raw_data()
                                             : rows_[it->second].data();
        std::memcpy((char*)out->raw_data() + i * es, src, es);
      }
    }
  }

  int64_t size() {
    std::lock_guard<std::mutex> l(mu_);
    return (int64_t)rows_.size();
  }

  void Export(Tensor* keys, Tensor* values) {
    std::lock_guard<std::mutex> l(mu_);
    size_t kes = DataTypeSize(key_t_);
    size_t ves = DataTypeSize(value_t_);
    for (size_t i = 0; i < rows_.size(); ++i) {
      if (key_t_ == DT_STRING)
        keys->flat<std::string>()[i] = key_rows_[i];
      else
        std::memcpy((char*)keys->raw_data() + i * kes, key_rows_[i].data(),
                    kes);
      if (value_t_ == DT_STRING)
        values->flat<std::string>()[i] = rows_[i];
      else
        std::memcpy((char*)values->raw_data() + i * ves, rows_[i].data(),
                    ves);
    }
  }

  DataType key_t() const { return key_t_; }
  DataType value_t() const { return value_t_; }
  bool initialized() {
    std::lock_guard<std::mutex> l(mu_);
    return initialized_;
  }

 private:
  static std::string ValueAt(const Tensor& t, int64_t i) {
    if (t.dtype() == DT_STRING) return t.flat<std::string>()[i];
    size_t es = DataTypeSize(t.dtype());
    return std::string((const char*)t.raw_data() + i * es, es);
  }

  DataType key_t_, value_t_;
  bool mutable_;
  std::mutex mu_;
  bool initialized_ = false;
  std::unordered_map<std::string, int64_t> index_;
  std::vector<std::string> rows_;      // value bytes per row
  std::vector<std::string> key_rows_;  // key bytes per row (export order)
};

TableResource* GetTable(OpKernelContext* ctx, const std::string& handle,
                        DataType kt = DT_INT64, DataType vt = DT_FLOAT,
                        bool mutable_table = true) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<TableResource>(handle, [&]() {
    return new TableResource(kt, vt, mutable_table);
  });
}

class HashTableOp : public OpKernel {
 public:
  explicit HashTableOp(OpKernelConstruction* c, bool mutable_table = false)
      : OpKernel(c), mutable_(mutable_table) {
    c->GetAttr("key_dtype", &key_t_);
    c->GetAttr("value_dtype", &value_t_);
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    OP_REQUIRES(ctx, GetTable(ctx, name(), key_t_, value_t_, mutable_),
                errors::Internal("no resource manager"));
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = name();
  }

 private:
  DataType key_t_, value_t_;
  bool mutable_;
};
class MutableHashTableOp : public HashTableOp {
 public:
  explicit MutableHashTableOp(OpKernelConstruction* c)
      : HashTableOp(c, true) {}
};
REGISTER_KERNEL_BUILDER(Name("HashTable").Device(DEVICE_CPU), HashTableOp);
REGISTER_KERNEL_BUILDER(Name("MutableHashTable").Device(DEVICE_CPU),
                        MutableHashTableOp);

class InitializeTableOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& h = ctx->input(0).flat<std::string>()[0];
    TableResource* t = GetTable(ctx, h, ctx->input(1).dtype(),
                                ctx->input(2).dtype(), false);
    OP_REQUIRES(ctx, t, errors::Internal("no resource manager"));
    OP_REQUIRES_OK(ctx, t->Insert(ctx->input(1), ctx->input(2)));
  }
};
REGISTER_KERNEL_BUILDER(Name("InitializeTable").Device(DEVICE_CPU),
                        InitializeTableOp);

class LookupTableInsertOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& h = ctx->input(0).flat<std::string>()[0];
    TableResource* t = GetTable(ctx, h, ctx->input(1).dtype(),
                                ctx->input(2).dtype(), true);
    OP_REQUIRES(ctx, t, errors::Internal("no resource manager"));
    OP_REQUIRES_OK(ctx, t->Insert(ctx->input(1), ctx->input(2)));
  }
};
REGISTER_KERNEL_BUILDER(Name("LookupTableInsert").Device(DEVICE_CPU),
                        LookupTableInsertOp);
REGISTER_KERNEL_BUILDER(Name("LookupTableImport").Device(DEVICE_CPU),
                        LookupTableInsertOp);

class LookupTableFindOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& h = ctx->input(0).flat<std::string>()[0];
    TableResource* t = GetTable(ctx, h);
    OP_REQUIRES(ctx, t, errors::Internal("no resource manager"));
    const Tensor& keys = ctx->input(1);
    const Tensor& dflt = ctx->input(2);
    Tensor* out = ctx->allocate_output(0, keys.shape());
    t->Find(keys, dflt, out);
  }
};
REGISTER_KERNEL_BUILDER(Name("LookupTableFind").Device(DEVICE_CPU),
                        LookupTableFindOp);

class LookupTableSizeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& h = ctx->input(0).flat<std::string>()[0];
    TableResource* t = GetTable(ctx, h);
    OP_REQUIRES(ctx, t, errors::Internal("no resource manager"));
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int64_t>()[0] = t->size();
  }
};
REGISTER_KERNEL_BUILDER(Name("LookupTableSize").Device(DEVICE_CPU),
                        LookupTableSizeOp);

class LookupTableExportOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& h = ctx->input(0).flat<std::string>()[0];
    TableResource* t = GetTable(ctx, h);
    OP_REQUIRES(ctx, t, errors::Internal("no resource manager"));
    int64_t n = t->size();
    Tensor* keys = ctx->allocate_output(0, TensorShape({n}));
    Tensor* values = ctx->allocate_output(1, TensorShape({n}));
    OP_REQUIRES(ctx, keys->dtype() == t->key_t() &&
                         values->dtype() == t->value_t(),
                errors::InvalidArgument("table dtype mismatch"));
    t->Export(keys, values);
  }
};
REGISTER_KERNEL_BUILDER(Name("LookupTableExport").Device(DEVICE_CPU),
                        LookupTableExportOp);

// --------------------------------- stacks ----------------------------------
class StackResource : public ResourceBase {
 public:
  std::mutex mu;
  std::vector<Tensor> items;
  bool closed = false;
};

StackResource* GetStack(OpKernelContext* ctx, const std::string& handle) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<StackResource>(handle,
                                            [] { return new StackResource(); });
}

class StackOp : public OpKernel {
 public:
  explicit StackOp(OpKernelConstruction* c) : OpKernel(c) {
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    OP_REQUIRES(ctx, GetStack(ctx, name()),
                errors::Internal("no resource manager"));
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    // per-step frame/iter uniqueness matters inside while loops
    out->flat<std::string>()[0] =
        name() + "/" + ctx->frame_name;
  }
};
REGISTER_KERNEL_BUILDER(Name("Stack").Device(DEVICE_CPU), StackOp);

class StackPushOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    StackResource* s = GetStack(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, s, errors::Internal("no resource manager"));
    {
      std::lock_guard<std::mutex> l(s->mu);
      OP_REQUIRES(ctx, !s->closed,
                  errors::FailedPrecondition("stack closed"));
      s->items.push_back(ctx->input(1));
    }
    ctx->set_output(0, ctx->input(1));
  }
};
REGISTER_KERNEL_BUILDER(Name("StackPush").Device(DEVICE_CPU), StackPushOp);

class StackPopOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    StackResource* s = GetStack(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, s, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(s->mu);
    OP_REQUIRES(ctx, !s->items.empty(),
                errors::InvalidArgument("pop on empty stack"));
    ctx->set_output(0, s->items.back());
    s->items.pop_back();
  }
};
REGISTER_KERNEL_BUILDER(Name("StackPop").Device(DEVICE_CPU), StackPopOp);

class StackCloseOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    StackResource* s = GetStack(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, s, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(s->mu);
    s->closed = true;
    s->items.clear();
  }
};
REGISTER_KERNEL_BUILDER(Name("StackClose").Device(DEVICE_CPU), StackCloseOp);

// --------------------------------- barrier ----------------------------------
// Reference barrier_ops.cc: keyed rendezvous of value components; TakeMany
// blocks until `num_elements` keys have ALL components set.
class BarrierResource : public ResourceBase {
 public:
  std::mutex mu;
  std::condition_variable cv;
  int num_components = 0;
  bool closed = false;
  // key -> per-component value (empty Tensor until inserted)
  std::map<std::string, std::vector<Tensor>> pending;
  std::deque<std::pair<std::string, std::vector<Tensor>>> ready;

  bool Complete(const std::vector<Tensor>& comps) {
    for (const Tensor& t : comps)
      if (!t.IsInitialized()) return false;
    return true;
  }
};

BarrierResource* GetBarrier(OpKernelContext* ctx, const std::string& h) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<BarrierResource>(
      h, [] { return new BarrierResource(); });
}

class BarrierOp : public OpKernel {
 public:
  explicit BarrierOp(OpKernelConstruction* c) : OpKernel(c) {
    auto it = c->def().attr.find("component_types");
    if (it != c->def().attr.end())
      ncomp_ = (int)it->second.list.type.size();
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    BarrierResource* b = GetBarrier(ctx, name());
    OP_REQUIRES(ctx, b, errors::Internal("no resource manager"));
    {
      std::lock_guard<std::mutex> l(b->mu);
      b->num_components = ncomp_;
    }
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = name();
  }

 private:
  int ncomp_ = 1;
};
REGISTER_KERNEL_BUILDER(Name("Barrier").Device(DEVICE_CPU), BarrierOp);

class BarrierInsertManyOp : public OpKernel {
 public:
  explicit BarrierInsertManyOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("component_index", &comp_);
  }
  void Compute(OpKernelContext* ctx) override {
    BarrierResource* b = GetBarrier(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, b, errors::Internal("no resource manager"));
    const Tensor& keys = ctx->input(1);
    const Tensor& vals = ctx->input(2);
    int64_t n = keys.NumElements();
    int64_t row = vals.NumElements() / std::max<int64_t>(n, 1);
    std::lock_guard<std::mutex> l(b->mu);
    OP_REQUIRES(ctx, !b->closed, errors::Aborted("barrier closed"));
    for (int64_t i = 0; i < n; ++i) {
      const std::string& k = keys.flat<std::string>()[i];
      auto& comps = b->pending[k];
      comps.resize(b->num_components);
      // slice row i out of vals
      TensorShape rshape;
      for (int d = 1; d < vals.dims(); ++d) rshape.AddDim(vals.dim_size(d));
      Tensor rv(vals.dtype(), rshape);
      std::memcpy(rv.raw_data(),
                  (const char*)vals.raw_data() +
                      i * row * DataTypeSize(vals.dtype()),
                  row * DataTypeSize(vals.dtype()));
      comps[comp_] = rv;
      if (b->Complete(comps)) {
        b->ready.emplace_back(k, comps);
        b->pending.erase(k);
      }
    }
    b->cv.notify_all();
  }

 private:
  int64_t comp_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("BarrierInsertMany").Device(DEVICE_CPU),
                        BarrierInsertManyOp);

class BarrierTakeManyOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    BarrierResource* b = GetBarrier(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, b, errors::Internal("no resource manager"));
    int64_t want = IntVector(ctx->input(1))[0];
    std::unique_lock<std::mutex> l(b->mu);
    b->cv.wait(l, [&] {
      return b->closed || (int64_t)b->ready.size() >= want ||
             (ctx->is_cancelled && ctx->is_cancelled());
    });
    if ((int64_t)b->ready.size() < want) {
      ctx->SetStatus(errors::OutOfRange("barrier closed with too few"));
      return;
    }
    int nc = b->num_components;
    Tensor* idx = ctx->allocate_output(0, TensorShape({want}));
    Tensor* keys = ctx->allocate_output(1, TensorShape({want}));
    std::vector<std::vector<Tensor>> taken;
    for (int64_t i = 0; i < want; ++i) {
      idx->flat<int64_t>()[i] = i;
      keys->flat<std::string>()[i] = b->ready.front().first;
      taken.push_back(b->ready.front().second);
      b->ready.pop_front();
    }
    for (int c = 0; c < nc; ++c) {
      TensorShape shape({want});
      for (int d = 0; d < taken[0][c].dims(); ++d)
        shape.AddDim(taken[0][c].dim_size(d));
      Tensor* out = ctx->allocate_output(2 + c, shape);
      size_t es = taken[0][c].TotalBytes();
      for (int64_t i = 0; i < want; ++i)
        std::memcpy((char*)out->raw_data() + i * es,
                    taken[i][c].raw_data(), es);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("BarrierTakeMany").Device(DEVICE_CPU),
                        BarrierTakeManyOp);

class BarrierCloseOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    BarrierResource* b = GetBarrier(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, b, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(b->mu);
    b->closed = true;
    b->cv.notify_all();
  }
};
REGISTER_KERNEL_BUILDER(Name("BarrierClose").Device(DEVICE_CPU),
                        BarrierCloseOp);

class BarrierSizeOp : public OpKernel {
 public:
  explicit BarrierSizeOp(OpKernelConstruction* c) : OpKernel(c) {
    incomplete_ = (def().op == "BarrierIncompleteSize");
  }
  void Compute(OpKernelContext* ctx) override {
    BarrierResource* b = GetBarrier(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, b, errors::Internal("no resource manager"));
    std::lock_guard<std::mutex> l(b->mu);
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int32_t>()[0] =
        incomplete_ ? (int32_t)b->pending.size() : (int32_t)b->ready.size();
  }

 private:
  bool incomplete_ = false;
};
REGISTER_KERNEL_BUILDER(Name("BarrierReadySize").Device(DEVICE_CPU),
                        BarrierSizeOp);
REGISTER_KERNEL_BUILDER(Name("BarrierIncompleteSize").Device(DEVICE_CPU),
                        BarrierSizeOp);

}  // namespace
}  // namespace stf
