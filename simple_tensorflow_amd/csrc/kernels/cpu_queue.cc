// Queue kernels: FIFOQueue / RandomShuffleQueue / PaddingFIFOQueue +
// enqueue/dequeue(+Many)/close/size — the reference's input-pipeline
// synchronization primitives (core/kernels/fifo_queue.cc, typed_queue.h,
// random_shuffle_queue_op.cc) rebuilt on callback-async kernels: a blocked
// dequeue parks a callback instead of a thread (matches our executor's
// AsyncOpKernel path, like the reference's async queue kernels).
#include <deque>
#include <random>

#include "kernels/kernel_util.h"
#include "kernels/resource_mgr.h"

namespace stf {

namespace {

using DoneCb = std::function<void()>;

class QueueResource : public ResourceBase {
 public:
  QueueResource(std::vector<DataType> dtypes, int64_t capacity, bool shuffle,
                int64_t min_after_dequeue, uint64_t seed)
      : dtypes_(std::move(dtypes)),
        capacity_(capacity < 0 ? (1ll << 60) : capacity), shuffle_(shuffle),
        min_after_(min_after_dequeue), rng_(seed ? seed : 0x2545F491) {}

  void TryEnqueue(std::vector<Tensor> item, OpKernelContext* ctx,
                  DoneCb done) {
    {
      std::unique_lock<std::mutex> l(mu_);
      if (closed_) {
        ctx->SetStatus(errors::Cancelled("queue is closed"));
      } else if ((int64_t)items_.size() < capacity_) {
        items_.push_back(std::move(item));
      } else {
        pending_enqueue_.push_back({std::move(item), ctx, std::move(done)});
        return;
      }
    }
    Progress();
    done();
  }

  void TryDequeueMany(int64_t n, OpKernelContext* ctx, DoneCb done) {
    {
      std::unique_lock<std::mutex> l(mu_);
      pending_dequeue_.push_back({n, ctx, std::move(done)});
    }
    Progress();
  }

  void Close(bool cancel_pending) {
    std::deque<PendingEq> eq;
    {
      std::unique_lock<std::mutex> l(mu_);
      closed_ = true;
      if (cancel_pending) {
        eq.swap(pending_enqueue_);
      }
    }
    for (auto& e : eq) {
      e.ctx->SetStatus(errors::Cancelled("enqueue cancelled"));
      e.done();
    }
    Progress();
  }

  int64_t size() {
    std::unique_lock<std::mutex> l(mu_);
    return (int64_t)items_.size();
  }

  const std::vector<DataType>& dtypes() const { return dtypes_; }

 private:
  struct PendingEq {
    std::vector<Tensor> item;
    OpKernelContext* ctx;
    DoneCb done;
  };
  struct PendingDq {
    int64_t n;  // -1 = single dequeue (no batch dim)
    OpKernelContext* ctx;
    DoneCb done;
  };

  // Move waiting work forward; fires completed callbacks outside the lock.
  void Progress() {
    for (;;) {
      std::vector<std::pair<PendingDq, std::vector<std::vector<Tensor>>>> fire;
      std::vector<PendingDq> oor;
      std::vector<PendingEq> eq_fire;
      std::vector<PendingDq> cancelled_dq;
      std::vector<PendingEq> cancelled_eq;
      {
        std::unique_lock<std::mutex> l(mu_);
        // Purge waiters whose step was cancelled (Run timeout / abort):
        // they must not be completed later against a dead step, and must not
        // consume items meant for live steps.
        for (auto it = pending_dequeue_.begin();
             it != pending_dequeue_.end();) {
          if (it->ctx->is_cancelled && it->ctx->is_cancelled()) {
            cancelled_dq.push_back(std::move(*it));
            it = pending_dequeue_.erase(it);
          } else {
            ++it;
          }
        }
        for (auto it = pending_enqueue_.begin();
             it != pending_enqueue_.end();) {
          if (it->ctx->is_cancelled && it->ctx->is_cancelled()) {
            cancelled_eq.push_back(std::move(*it));
            it = pending_enqueue_.erase(it);
          } else {
            ++it;
          }
        }
        // admit pending enqueues
        while (!pending_enqueue_.empty() &&
               (int64_t)items_.size() < capacity_ && !closed_) {
          items_.push_back(std::move(pending_enqueue_.front().item));
          eq_fire.push_back({{},
                             pending_enqueue_.front().ctx,
                             std::move(pending_enqueue_.front().done)});
          pending_enqueue_.pop_front();
        }
        // serve dequeues
        while (!pending_dequeue_.empty()) {
          PendingDq& d = pending_dequeue_.front();
          int64_t want = d.n < 0 ? 1 : d.n;
          int64_t avail = (int64_t)items_.size();
          int64_t reserve = (shuffle_ && !closed_) ? min_after_ : 0;
          if (avail - want >= reserve && avail >= want) {
            std::vector<std::vector<Tensor>> taken;
            for (int64_t i = 0; i < want; ++i) {
              size_t idx = 0;
              if (shuffle_) idx = rng_() % items_.size();
              taken.push_back(std::move(items_[idx]));
              items_.erase(items_.begin() + idx);
            }
            fire.emplace_back(std::move(d), std::move(taken));
            pending_dequeue_.pop_front();
          } else if (closed_ && pending_enqueue_.empty()) {
            oor.push_back(std::move(d));
            pending_dequeue_.pop_front();
          } else {
            break;
          }
        }
      }
      if (fire.empty() && oor.empty() && eq_fire.empty() &&
          cancelled_dq.empty() && cancelled_eq.empty())
        return;
      for (auto& d : cancelled_dq) {
        d.ctx->SetStatus(errors::Cancelled("dequeue cancelled (step aborted)"));
        d.done();
      }
      for (auto& e : cancelled_eq) {
        e.ctx->SetStatus(errors::Cancelled("enqueue cancelled (step aborted)"));
        e.done();
      }
      for (auto& e : eq_fire) e.done();
      for (auto& d : oor) {
        d.ctx->SetStatus(errors::OutOfRange(
            "queue is closed and has insufficient elements"));
        d.done();
      }
      for (auto& f : fire) {
        DeliverLocked(f.first, f.second);
        f.first.done();
      }
    }
  }

  void DeliverLocked(PendingDq& d,
                     std::vector<std::vector<Tensor>>& taken) {
    int k = (int)dtypes_.size();
    if (d.n < 0) {
      for (int c = 0; c < k; ++c) d.ctx->set_output(c, taken[0][c]);
      return;
    }
    // stack along new dim 0
    for (int c = 0; c < k; ++c) {
      TensorShape shape = taken[0][c].shape();
      TensorShape out_shape;
      out_shape.AddDim((int64_t)taken.size());
      for (auto dd : shape.dim_sizes()) out_shape.AddDim(dd);
      Tensor out(dtypes_[c], out_shape);
      size_t row = taken[0][c].TotalBytes();
      for (size_t i = 0; i < taken.size(); ++i)
        std::memcpy((char*)out.raw_data() + i * row, taken[i][c].raw_data(),
                    row);
      d.ctx->set_output(c, out);
    }
  }

  std::mutex mu_;
  std::vector<DataType> dtypes_;
  int64_t capacity_;
  bool shuffle_;
  int64_t min_after_;
  std::mt19937_64 rng_;
  bool closed_ = false;
  std::deque<std::vector<Tensor>> items_;
  std::deque<PendingEq> pending_enqueue_;
  std::deque<PendingDq> pending_dequeue_;
};

QueueResource* GetQueue(OpKernelContext* ctx, const std::string& handle) {
  auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
  if (!mgr) return nullptr;
  return mgr->LookupOrCreate<QueueResource>(handle, [&]() {
    return new QueueResource({}, 0, false, 0, 0);  // placeholder; see QueueOp
  });
}

// ----------------------------- queue creation -------------------------------
class QueueOp : public OpKernel {
 public:
  QueueOp(OpKernelConstruction* c, bool shuffle) : OpKernel(c) {
    auto it = c->def().attr.find("component_types");
    if (it != c->def().attr.end())
      for (int t : it->second.list.type) dtypes_.push_back((DataType)t);
    c->GetAttr("capacity", &capacity_);
    if (shuffle) {
      c->GetAttr("min_after_dequeue", &min_after_);
      c->GetAttr("seed", &seed_);
    }
    shuffle_ = shuffle;
    set_expensive(false);
  }
  void Compute(OpKernelContext* ctx) override {
    auto* mgr = static_cast<ResourceMgr*>(ctx->resource_mgr);
    OP_REQUIRES(ctx, mgr, errors::Internal("no resource manager"));
    mgr->LookupOrCreate<QueueResource>(name(), [&]() {
      return new QueueResource(dtypes_, capacity_, shuffle_, min_after_,
                               (uint64_t)seed_);
    });
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<std::string>()[0] = name();
  }

 private:
  std::vector<DataType> dtypes_;
  int64_t capacity_ = -1;
  int64_t min_after_ = 0;
  int64_t seed_ = 0;
  bool shuffle_ = false;
};
class FIFOQueueOp : public QueueOp {
 public:
  explicit FIFOQueueOp(OpKernelConstruction* c) : QueueOp(c, false) {}
};
class RandomShuffleQueueOp : public QueueOp {
 public:
  explicit RandomShuffleQueueOp(OpKernelConstruction* c) : QueueOp(c, true) {}
};
REGISTER_KERNEL_BUILDER(Name("FIFOQueue").Device(DEVICE_CPU), FIFOQueueOp);
REGISTER_KERNEL_BUILDER(Name("PaddingFIFOQueue").Device(DEVICE_CPU), FIFOQueueOp);
REGISTER_KERNEL_BUILDER(Name("RandomShuffleQueue").Device(DEVICE_CPU),
                        RandomShuffleQueueOp);

// ----------------------------- enqueue/dequeue ------------------------------
class QueueEnqueueOp : public AsyncOpKernel {
 public:
  using AsyncOpKernel::AsyncOpKernel;
  void ComputeAsync(OpKernelContext* ctx, DoneCallback done) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES_ASYNC(ctx, q, errors::NotFound("queue not found"), done);
    std::vector<Tensor> item;
    for (int i = 1; i < num_inputs(); ++i) item.push_back(ctx->input(i));
    q->TryEnqueue(std::move(item), ctx, std::move(done));
  }
};
REGISTER_KERNEL_BUILDER(Name("QueueEnqueue").Device(DEVICE_CPU),
                        QueueEnqueueOp);

class QueueEnqueueManyOp : public AsyncOpKernel {
 public:
  using AsyncOpKernel::AsyncOpKernel;
  void ComputeAsync(OpKernelContext* ctx, DoneCallback done) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES_ASYNC(ctx, q, errors::NotFound("queue not found"), done);
    int64_t n = ctx->input(1).dim_size(0);
    int k = num_inputs() - 1;
    // split rows into items; enqueue each (last one carries the done)
    struct Counter {
      std::atomic<int64_t> left;
      DoneCallback done;
    };
    auto* counter = new Counter{{n}, std::move(done)};
    for (int64_t r = 0; r < n; ++r) {
      std::vector<Tensor> item;
      for (int c = 0; c < k; ++c) {
        const Tensor& comp = ctx->input(1 + c);
        TensorShape row_shape = comp.shape();
        row_shape.RemoveDim(0);
        Tensor row(comp.dtype(), row_shape);
        size_t bytes = row.TotalBytes();
        if (comp.dtype() == DT_STRING) {
          for (int64_t e = 0; e < row.NumElements(); ++e)
            row.flat<std::string>()[e] =
                comp.flat<std::string>()[r * row.NumElements() + e];
        } else {
          std::memcpy(row.raw_data(),
                      (const char*)comp.raw_data() + r * bytes, bytes);
        }
        item.push_back(row);
      }
      q->TryEnqueue(std::move(item), ctx, [counter]() {
        if (counter->left.fetch_sub(1) == 1) {
          counter->done();
          delete counter;
        }
      });
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("QueueEnqueueMany").Device(DEVICE_CPU),
                        QueueEnqueueManyOp);

class QueueDequeueOp : public AsyncOpKernel {
 public:
  using AsyncOpKernel::AsyncOpKernel;
  void ComputeAsync(OpKernelContext* ctx, DoneCallback done) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES_ASYNC(ctx, q, errors::NotFound("queue not found"), done);
    q->TryDequeueMany(-1, ctx, std::move(done));
  }
};
REGISTER_KERNEL_BUILDER(Name("QueueDequeue").Device(DEVICE_CPU),
                        QueueDequeueOp);

class QueueDequeueManyOp : public AsyncOpKernel {
 public:
  using AsyncOpKernel::AsyncOpKernel;
  void ComputeAsync(OpKernelContext* ctx, DoneCallback done) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES_ASYNC(ctx, q, errors::NotFound("queue not found"), done);
    int64_t n = ctx->input(1).flat<int32_t>()[0];
    q->TryDequeueMany(n, ctx, std::move(done));
  }
};
REGISTER_KERNEL_BUILDER(Name("QueueDequeueMany").Device(DEVICE_CPU),
                        QueueDequeueManyOp);

class QueueCloseOp : public OpKernel {
 public:
  explicit QueueCloseOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("cancel_pending_enqueues", &cancel_);
  }
  void Compute(OpKernelContext* ctx) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, q, errors::NotFound("queue not found"));
    q->Close(cancel_);
  }

 private:
  bool cancel_ = false;
};
REGISTER_KERNEL_BUILDER(Name("QueueClose").Device(DEVICE_CPU), QueueCloseOp);

class QueueSizeOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    QueueResource* q = GetQueue(ctx, ctx->input(0).flat<std::string>()[0]);
    OP_REQUIRES(ctx, q, errors::NotFound("queue not found"));
    Tensor* out = ctx->allocate_output(0, TensorShape({}));
    out->flat<int32_t>()[0] = (int32_t)q->size();
  }
};
REGISTER_KERNEL_BUILDER(Name("QueueSize").Device(DEVICE_CPU), QueueSizeOp);

}  // namespace
}  // namespace stf
