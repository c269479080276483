#include "runtime/executor.h"

#include <condition_variable>
#include <chrono>
#include <deque>
#include <tuple>

namespace stf {

namespace {

struct Entry {
  Tensor val;
  bool has_value = false;
  bool is_dead = false;
};

struct TaggedNode;

}  // namespace

// ---------------------------------------------------------------------------
// Executor::Create / Initialize
// ---------------------------------------------------------------------------
Status Executor::Create(std::unique_ptr<Graph> graph, Device* device,
                        OpSegment* opseg, std::unique_ptr<Executor>* out) {
  std::unique_ptr<Executor> e(new Executor());
  e->graph_ = std::move(graph);
  e->device_ = device;
  e->opseg_ = opseg;
  STF_RETURN_IF_ERROR(e->Initialize());
  *out = std::move(e);
  return Status::OK();
}

Executor::~Executor() {
  for (auto& item : items_) {
    if (item.owned_kernel) delete item.kernel;
  }
}

Status Executor::Initialize() {
  items_.resize(graph_->num_node_ids());
  total_inputs_ = 0;
  for (Node* n : graph_->nodes()) {
    NodeItem& item = items_[n->id];
    item.node = n;
    item.input_start = total_inputs_;
    total_inputs_ += n->num_inputs();
    item.is_merge = n->IsMerge();
    item.is_switch = n->IsSwitch();
    item.is_enter = n->IsEnter();
    item.is_exit = n->IsExit();
    item.is_next_iter = n->IsNextIteration();
    item.is_noop = n->op() == "NoOp";
    if (item.is_enter) {
      bool c = false;
      GetAttrBool(n->def, "is_constant", &c);
      item.is_constant_enter = c;
    }
    // Control-flow routing nodes and Send/Recv are executed by the executor
    // itself; everything else gets a kernel.
    bool executor_routed = n->IsControlFlow() || n->IsSend() || n->IsRecv() ||
                           item.is_noop;
    if (!executor_routed) {
      if (n->op_def->is_stateful && opseg_) {
        OpKernel* k = nullptr;
        STF_RETURN_IF_ERROR(opseg_->FindOrCreate(
            n->name(),
            [&](std::unique_ptr<OpKernel>* out_k) {
              return CreateOpKernel(device_->device_type(), device_, n->def,
                                    out_k);
            },
            &k));
        item.kernel = k;
        item.owned_kernel = false;
      } else {
        std::unique_ptr<OpKernel> k;
        STF_RETURN_IF_ERROR(
            CreateOpKernel(device_->device_type(), device_, n->def, &k));
        item.kernel = k.release();
        item.owned_kernel = true;
      }
      item.is_async = item.kernel->IsAsync();
    }
    if (n->in_edges.empty()) roots_.push_back(n);
  }
  return Status::OK();
}

// ---------------------------------------------------------------------------
// ExecutorState
// ---------------------------------------------------------------------------
class ExecutorState {
 public:
  struct IterState {
    std::vector<Entry> entries;
    std::vector<int> pending;      // remaining arrivals per node (non-merge)
    std::vector<int> dead_count;   // dead arrivals per node
    // merge bookkeeping
    std::vector<int> merge_ctrl_remaining;
    std::vector<int> merge_dead;
    std::vector<int8_t> merge_live;
    std::vector<int8_t> merge_fired;
  };

  struct FrameState {
    std::string name;
    FrameState* parent = nullptr;
    int64_t parent_iter = 0;
    std::map<int64_t, std::unique_ptr<IterState>> iters;
    std::map<std::pair<int64_t, std::string>, std::unique_ptr<FrameState>>
        children;
    // Loop-invariant (constant Enter) values: (enter node, value, dead).
    std::vector<std::tuple<Node*, Tensor, bool>> invariants;
    // True when the frame was entered dead (a loop-var Merge fired dead at
    // iteration 0): only then do dead Exits propagate to the parent.
    bool is_dead = false;
  };

  struct TaggedNode {
    Node* node;
    FrameState* frame;
    int64_t iter;
  };

  ExecutorState(Executor* impl, const ExecutorArgs& args,
                std::function<void(Status)> done)
      : impl_(impl), args_(args), done_(std::move(done)) {}

  void Run() {
    root_.name = "";
    GetOrCreateIter(&root_, 0);
    if (impl_->roots_.empty()) {
      finished_ = true;
      done_(Status::OK());
      return;
    }
    outstanding_ = (int64_t)impl_->roots_.size();
    for (Node* n : impl_->roots_) Schedule({n, &root_, 0});
  }

 private:
  Executor* impl_;
  ExecutorArgs args_;
  std::function<void(Status)> done_;
  std::mutex mu_;
  FrameState root_;
  int64_t outstanding_ = 0;
  Status status_;
  bool finished_ = false;

  IterState* GetOrCreateIter(FrameState* f, int64_t iter) {
    auto it = f->iters.find(iter);
    if (it != f->iters.end()) return it->second.get();
    auto is = std::make_unique<IterState>();
    int num_nodes = impl_->graph_->num_node_ids();
    is->entries.resize(impl_->total_inputs_);
    is->pending.resize(num_nodes, 0);
    is->dead_count.resize(num_nodes, 0);
    is->merge_ctrl_remaining.resize(num_nodes, 0);
    is->merge_dead.resize(num_nodes, 0);
    is->merge_live.resize(num_nodes, 0);
    is->merge_fired.resize(num_nodes, 0);
    for (Node* n : impl_->graph_->nodes()) {
      int data = 0, ctrl = 0;
      for (auto* e : n->in_edges) (e->IsControl() ? ctrl : data)++;
      is->pending[n->id] = data + ctrl;
      if (impl_->items_[n->id].is_merge) {
        is->merge_ctrl_remaining[n->id] = ctrl;
        // Slots that cannot arrive in this iteration count as dead:
        // NextIteration-fed slots at iter 0; Enter-fed slots at iter > 0.
        for (auto* e : n->in_edges) {
          if (e->IsControl()) continue;
          if ((e->src->IsNextIteration() && iter == 0) ||
              (e->src->IsEnter() && iter > 0))
            is->merge_dead[n->id]++;
        }
      }
    }
    IterState* raw = is.get();
    f->iters[iter] = std::move(is);
    // Replay loop invariants into the new iteration (edges only — the Enter
    // transition itself must not re-run).
    std::vector<TaggedNode> ready;
    for (auto& inv : f->invariants) {
      Node* enter = std::get<0>(inv);
      DeliverEdgesLocked(enter, f, iter, raw, {std::get<1>(inv)},
                         {std::get<2>(inv)}, &ready);
    }
    for (auto& t : ready) {
      outstanding_++;
      ScheduleLocked(t);
    }
    return raw;
  }

  void Schedule(const TaggedNode& t) {
    args_.schedule([this, t]() { Process(t); });
  }
  void ScheduleLocked(const TaggedNode& t) { Schedule(t); }

  int NumDataInputs(Node* n) {
    int d = 0;
    for (auto* e : n->in_edges)
      if (!e->IsControl()) ++d;
    return d;
  }

  // Record one arrival at (dst, frame, iter). For data edges the entry has
  // already been stored. Appends dst to `ready` if it became runnable.
  void ArriveLocked(Node* dst, FrameState* f, int64_t iter, IterState* is,
                    bool is_dead, bool is_control,
                    std::vector<TaggedNode>* ready) {
    Executor::NodeItem& item = impl_->items_[dst->id];
    if (item.is_merge) {
      if (is_control) {
        if (is_dead) is->dead_count[dst->id]++;
        is->merge_ctrl_remaining[dst->id]--;
      } else if (is_dead) {
        is->merge_dead[dst->id]++;
      } else {
        is->merge_live[dst->id] = 1;
      }
      bool can_live = is->merge_live[dst->id] &&
                      is->merge_ctrl_remaining[dst->id] == 0;
      bool can_dead = is->merge_dead[dst->id] >= NumDataInputs(dst) &&
                      is->merge_ctrl_remaining[dst->id] == 0;
      if (!is->merge_fired[dst->id] && (can_live || can_dead)) {
        is->merge_fired[dst->id] = 1;
        ready->push_back({dst, f, iter});
      }
      return;
    }
    if (is_dead) is->dead_count[dst->id]++;
    if (--is->pending[dst->id] == 0) ready->push_back({dst, f, iter});
  }

  // Deliver the outputs of `src` executed at (f, iter) to its consumers.
  void DeliverFromLocked(Node* src, FrameState* f, int64_t iter, IterState* is,
                         const std::vector<Tensor>& outs,
                         const std::vector<bool>& dead,
                         std::vector<TaggedNode>* ready) {
    Executor::NodeItem& sitem = impl_->items_[src->id];
    // Determine destination frame/iter by the src node's type.
    FrameState* out_f = f;
    int64_t out_iter = iter;
    IterState* out_is = is;
    if (sitem.is_enter) {
      std::string fname;
      GetAttrString(src->def, "frame_name", &fname);
      auto key = std::make_pair(iter, fname);
      auto it = f->children.find(key);
      if (it == f->children.end()) {
        auto child = std::make_unique<FrameState>();
        child->name = fname;
        child->parent = f;
        child->parent_iter = iter;
        out_f = child.get();
        f->children[key] = std::move(child);
      } else {
        out_f = it->second.get();
      }
      out_iter = 0;
      out_is = GetOrCreateIter(out_f, 0);
      if (sitem.is_constant_enter) {
        out_f->invariants.emplace_back(src, outs.empty() ? Tensor() : outs[0],
                                       dead.empty() ? false : dead[0]);
        // Deliver into all existing iterations (the one just created gets it
        // below via the normal path for iter 0 only; replay covers others).
        for (auto& kv : out_f->iters) {
          if (kv.first == 0) continue;
          DeliverEdgesLocked(src, out_f, kv.first, kv.second.get(), outs, dead,
                             ready);
        }
      }
    } else if (sitem.is_exit) {
      // A dead Exit fires every non-final iteration; it must reach the parent
      // only when the whole frame is dead (reference: dead_exits deferral in
      // executor.cc). Otherwise the single live Exit is the one delivery.
      if (!dead.empty() && dead[0] && !f->is_dead) return;
      out_f = f->parent;
      out_iter = f->parent_iter;
      CHECK(out_f != nullptr) << "Exit outside a frame";
      out_is = GetOrCreateIter(out_f, out_iter);
    } else if (sitem.is_next_iter) {
      if (!dead.empty() && dead[0]) return;  // loop termination: drop
      out_iter = iter + 1;
      out_is = GetOrCreateIter(f, out_iter);
    }
    DeliverEdgesLocked(src, out_f, out_iter, out_is, outs, dead, ready);
  }

  void DeliverEdgesLocked(Node* src, FrameState* f, int64_t iter, IterState* is,
                          const std::vector<Tensor>& outs,
                          const std::vector<bool>& dead,
                          std::vector<TaggedNode>* ready) {
    bool all_dead = true;
    for (size_t i = 0; i < dead.size(); ++i) all_dead &= dead[i];
    if (dead.empty()) all_dead = false;
    for (auto* e : src->out_edges) {
      Node* dst = e->dst;
      Executor::NodeItem& ditem = impl_->items_[dst->id];
      if (e->IsControl()) {
        ArriveLocked(dst, f, iter, is, /*dead=*/!dead.empty() && all_dead,
                     /*control=*/true, ready);
      } else {
        Entry& ent = is->entries[ditem.input_start + e->dst_input];
        ent.val = outs[e->src_output];
        ent.has_value = true;
        ent.is_dead = dead[e->src_output];
        ArriveLocked(dst, f, iter, is, ent.is_dead, /*control=*/false, ready);
      }
    }
  }

  std::string FrameKey(FrameState* f, int64_t iter) {
    if (f == &root_) return "";
    return f->name + "@" + std::to_string(iter);
  }

  // ------------------------- node processing ------------------------------
  void Process(TaggedNode tagged) {
    std::deque<TaggedNode> inline_q;
    inline_q.push_back(tagged);
    while (!inline_q.empty()) {
      TaggedNode t = inline_q.front();
      inline_q.pop_front();
      ProcessOne(t, &inline_q);
    }
  }

  void ProcessOne(TaggedNode t, std::deque<TaggedNode>* inline_q) {
    Node* n = t.node;
    Executor::NodeItem& item = impl_->items_[n->id];

    // Gather inputs.
    std::vector<Tensor> inputs(n->num_inputs());
    std::vector<bool> in_dead(n->num_inputs(), false);
    bool any_dead = false;
    int merge_live_slot = -1;
    bool aborted = false;
    {
      std::lock_guard<std::mutex> l(mu_);
      aborted = !status_.ok();
    }
    if (aborted) {
      Done(t, {}, {}, inline_q);
      return;
    }
    {
      std::lock_guard<std::mutex> l(mu_);
      IterState* is = t.frame->iters[t.iter].get();
      for (auto* e : n->in_edges) {
        if (e->IsControl()) continue;
        Entry& ent = is->entries[item.input_start + e->dst_input];
        inputs[e->dst_input] = ent.val;
        in_dead[e->dst_input] = ent.is_dead || !ent.has_value;
        if (ent.has_value && !ent.is_dead && merge_live_slot < 0)
          merge_live_slot = e->dst_input;
        ent.val = Tensor();  // release ref
        ent.has_value = false;
      }
      any_dead = is->dead_count[n->id] > 0;
      for (bool d : in_dead) any_dead |= d;
    }

    std::vector<Tensor> outs(n->num_outputs());
    std::vector<bool> dead(n->num_outputs(), false);

    static bool debug = getenv("STF_EXEC_DEBUG") != nullptr;
    if (debug) {
      fprintf(stderr, "[exec] %s (%s) frame=%s iter=%lld dead=%d\n",
              n->name().c_str(), n->op().c_str(), t.frame->name.c_str(),
              (long long)t.iter, (int)any_dead);
    }

    // --- executor-routed ops ---
    if (item.is_merge) {
      if (merge_live_slot < 0 && t.iter == 0) {
        std::lock_guard<std::mutex> l(mu_);
        t.frame->is_dead = true;
      }
      if (merge_live_slot >= 0) {
        outs[0] = inputs[merge_live_slot];
        Tensor idx(DT_INT32, TensorShape({}));
        idx.flat<int32_t>()[0] = merge_live_slot;
        if (n->num_outputs() > 1) outs[1] = idx;
      } else {
        dead.assign(dead.size(), true);
      }
      Done(t, outs, dead, inline_q);
      return;
    }
    if (item.is_switch) {
      if (any_dead) {
        dead.assign(dead.size(), true);
      } else {
        bool pred = ReadBoolScalar(inputs[1]);
        int port = pred ? 1 : 0;
        outs[port] = inputs[0];
        dead[1 - port] = true;
      }
      Done(t, outs, dead, inline_q);
      return;
    }
    if (item.is_enter || item.is_exit || item.is_next_iter) {
      if (any_dead) {
        dead.assign(dead.size(), true);
      } else {
        outs[0] = inputs[0];
      }
      Done(t, outs, dead, inline_q);
      return;
    }
    if (item.is_noop) {
      Done(t, outs, any_dead ? std::vector<bool>(outs.size(), true) : dead,
           inline_q);
      return;
    }
    if (n->IsSend()) {
      std::string tensor_name;
      GetAttrString(n->def, "tensor_name", &tensor_name);
      Status s = args_.rendezvous->Send(
          SendRecvKey(n->def, tensor_name, t), any_dead ? Tensor() : inputs[0],
          any_dead);
      if (!s.ok()) {
        Fail(s);
      }
      Done(t, outs, dead, inline_q);
      return;
    }
    if (n->IsRecv()) {
      std::string tensor_name;
      GetAttrString(n->def, "tensor_name", &tensor_name);
      std::string key = SendRecvKey(n->def, tensor_name, t);
      args_.rendezvous->RecvAsync(
          key, [this, t](const Status& s, const Tensor& val, bool is_dead) {
            if (!s.ok()) Fail(s);
            std::vector<Tensor> outs = {val};
            std::vector<bool> dead = {is_dead};
            std::deque<TaggedNode> q;
            Done(t, outs, dead, &q);
            for (auto& nt : q) Process(nt);
          });
      return;
    }

    // --- dead regular node: propagate deadness without executing ---
    if (any_dead) {
      dead.assign(dead.size(), true);
      Done(t, outs, dead, inline_q);
      return;
    }

    // --- kernel execution ---
    // Convert input memory spaces to what the kernel expects.
    Device* dev = impl_->device_;
    for (int i = 0; i < n->num_inputs(); ++i) {
      MemSpace want = item.kernel->input_mem.empty()
                          ? MemSpace::HOST
                          : item.kernel->input_mem[i];
      if (!inputs[i].IsInitialized()) continue;
      if (want == MemSpace::HOST && inputs[i].mem_space() == MemSpace::DEVICE) {
        Tensor host;
        Status s = dev->CopyDeviceTensorToHost(inputs[i], &host);
        if (!s.ok()) {
          Fail(Status(s.code(), "input " + std::to_string(i) + " of node " +
                                    n->name() + " (" + n->op() +
                                    "): " + s.message()));
          Done(t, outs, dead, inline_q);
          return;
        }
        inputs[i] = host;
      } else if (want == MemSpace::DEVICE &&
                 inputs[i].mem_space() == MemSpace::HOST) {
        Tensor devt;
        Status s = dev->CopyHostTensorToDevice(inputs[i], &devt);
        if (!s.ok()) { Fail(s); Done(t, outs, dead, inline_q); return; }
        inputs[i] = devt;
      }
    }

    if (item.is_async) {
      auto* ctx = new OpKernelContext(item.kernel, dev, std::move(inputs));
      FillCtx(ctx, t);
      auto* akernel = static_cast<AsyncOpKernel*>(item.kernel);
      akernel->ComputeAsync(ctx, [this, ctx, t]() {
        std::vector<Tensor> outs = ctx->outputs();
        std::vector<bool> dead(outs.size(), false);
        if (!ctx->status().ok()) Fail(ctx->status());
        delete ctx;
        std::deque<TaggedNode> q;
        Done(t, outs, dead, &q);
        for (auto& nt : q) Process(nt);
      });
      return;
    }

    OpKernelContext ctx(item.kernel, dev, std::move(inputs));
    FillCtx(&ctx, t);
    int64_t t0 = 0;
    void* trace_tag = nullptr;
    if (args_.stats) {
      t0 = std::chrono::duration_cast<std::chrono::microseconds>(
               std::chrono::system_clock::now().time_since_epoch())
               .count();
      if (args_.stats->gpu_pre) trace_tag = args_.stats->gpu_pre(dev);
    }
    dev->Compute(item.kernel, &ctx);
    if (args_.stats) {
      if (trace_tag) args_.stats->gpu_post(dev, trace_tag, n->name(), n->op());
      int64_t t1 = std::chrono::duration_cast<std::chrono::microseconds>(
                       std::chrono::system_clock::now().time_since_epoch())
                       .count();
      args_.stats->Add(n->name(), n->op(), t0, t1);
    }
    if (getenv("STF_DEBUG_LAUNCH")) {
      std::string line = n->name() + " (" + n->op() + ") in:";
      char b[32];
      for (int i = 0; i < ctx.num_inputs(); ++i) {
        snprintf(b, sizeof(b), " %p", ctx.input(i).raw_data());
        line += b;
      }
      line += " out:";
      for (auto& o : ctx.outputs()) {
        snprintf(b, sizeof(b), " %p", o.raw_data());
        line += b;
      }
      fprintf(stderr, "[launch cap=%d] %s\n", dev->capturing() ? 1 : 0,
              line.c_str());
    }
    if (!ctx.status().ok()) {
      Fail(Status(ctx.status().code(),
                  "node " + n->name() + " (" + n->op() + "): " +
                      ctx.status().message()));
      Done(t, outs, dead, inline_q);
      return;
    }
    Done(t, ctx.outputs(), dead, inline_q);
  }

  void FillCtx(OpKernelContext* ctx, const TaggedNode& t) {
    ctx->rendezvous = args_.rendezvous;
    ctx->step_id = args_.step_id;
    ctx->frame_name = t.frame->name;
    ctx->iter_id = t.iter;
    ctx->resource_mgr = args_.resource_mgr;
    ctx->is_cancelled = args_.is_cancelled;
  }

  std::string SendRecvKey(const NodeDef& def, const std::string& tensor_name,
                          const TaggedNode& t) {
    std::string send_dev, recv_dev;
    GetAttrString(def, "send_device", &send_dev);
    GetAttrString(def, "recv_device", &recv_dev);
    return RendezvousKey(send_dev, recv_dev, tensor_name, FrameKey(t.frame, 0),
                         t.iter);
  }

  bool ReadBoolScalar(const Tensor& pred) {
    if (pred.mem_space() == MemSpace::DEVICE) {
      Tensor host;
      Status s = impl_->device_->CopyDeviceTensorToHost(pred, &host);
      if (!s.ok()) {
        Fail(s);
        return false;
      }
      return host.flat<bool>()[0];
    }
    return pred.flat<bool>()[0];
  }

  void Fail(const Status& s) {
    std::lock_guard<std::mutex> l(mu_);
    if (status_.ok()) {
      status_ = s;
      if (args_.rendezvous) args_.rendezvous->StartAbort(s);
    }
  }

  // Propagate outputs, update bookkeeping, maybe finish. Pushes newly ready
  // nodes: first to inline_q, rest scheduled.
  void Done(const TaggedNode& t, const std::vector<Tensor>& outs,
            const std::vector<bool>& dead, std::deque<TaggedNode>* inline_q) {
    std::vector<TaggedNode> ready;
    bool finish = false;
    Status st;
    {
      std::lock_guard<std::mutex> l(mu_);
      if (status_.ok()) {
        IterState* is = t.frame->iters[t.iter].get();
        DeliverFromLocked(t.node, t.frame, t.iter, is, outs, dead, &ready);
      }
      outstanding_ += (int64_t)ready.size() - 1;
      if (outstanding_ == 0 && !finished_) {
        finished_ = true;
        finish = true;
        st = status_;
      }
    }
    for (size_t i = 0; i < ready.size(); ++i) {
      if (i == 0 && inline_q) inline_q->push_back(ready[i]);
      else Schedule(ready[i]);
    }
    if (finish) done_(st);
  }

 public:
  friend class Executor;
};

void Executor::RunAsync(const ExecutorArgs& args,
                        std::function<void(Status)> done) {
  // Heap cell so the completion callback can delete the state after it runs.
  auto** cell = new ExecutorState*;
  *cell = new ExecutorState(this, args, [done, cell](Status s) {
    ExecutorState* self = *cell;
    delete cell;
    done(s);
    delete self;
  });
  (*cell)->Run();
}

Status Executor::Run(const ExecutorArgs& args) {
  std::mutex mu;
  std::condition_variable cv;
  bool done_flag = false;
  Status result;
  auto* state = new ExecutorState(this, args, [&](Status s) {
    std::lock_guard<std::mutex> l(mu);
    result = s;
    done_flag = true;
    cv.notify_one();
  });
  state->Run();
  {
    std::unique_lock<std::mutex> l(mu);
    cv.wait(l, [&]() { return done_flag; });
  }
  delete state;
  return result;
}

}  // namespace stf
