// Dataflow executor with dead-token control flow and while-loop frames.
//
// Capability analog of the reference's common_runtime/executor.cc design
// (GraphView + PendingCounts + FrameState; reference executor.cc:332,782),
// rebuilt compactly: Switch/Merge/Enter/Exit/NextIteration are routed by the
// executor itself (pure data movement — no kernel launch), merges use
// explicit arrival-event logic with "cannot-arrive" slots counted dead
// (Enter-fed slots in iter>0, NextIteration-fed slots in iter 0), and GPU
// kernels enqueue onto the device's single compute hipStream.
#pragma once

#include <atomic>
#include <functional>
#include <map>
#include <memory>
#include <mutex>

#include "core/threadpool.h"
#include "framework/device.h"
#include "framework/op_kernel.h"
#include "graph/graph.h"
#include "runtime/rendezvous.h"

namespace stf {

// Session-lifetime cache of stateful kernels (variables, queues, RNG state)
// keyed by node name — the analog of the reference's OpSegment, so every
// executor built by one session shares variable storage.
class OpSegment {
 public:
  Status FindOrCreate(const std::string& node_name,
                      std::function<Status(std::unique_ptr<OpKernel>*)> create,
                      OpKernel** kernel) {
    std::lock_guard<std::mutex> l(mu_);
    auto it = kernels_.find(node_name);
    if (it != kernels_.end()) {
      *kernel = it->second.get();
      return Status::OK();
    }
    std::unique_ptr<OpKernel> k;
    STF_RETURN_IF_ERROR(create(&k));
    *kernel = k.get();
    kernels_[node_name] = std::move(k);
    return Status::OK();
  }

  // Drops every cached stateful kernel (variables, queues' op instances) —
  // Session.reset support.
  void Clear() {
    std::lock_guard<std::mutex> l(mu_);
    kernels_.clear();
  }

 private:
  std::mutex mu_;
  std::map<std::string, std::unique_ptr<OpKernel>> kernels_;
};

class ResourceMgr;  // defined in kernels (queues etc.)

// Per-node timing record (StepStats; reference step_stats.proto NodeExecStats
// — host-side enqueue times, the analog of the reference's
// StepStatsCollector in common_runtime/step_stats_collector.cc).
struct NodeStats {
  std::string node;
  std::string op;
  int64_t start_us = 0;
  int64_t end_us = 0;
  // "" = host enqueue lane; the GpuTracer appends device-lane entries
  // ("/device:GPU:n/stream:compute") with hardware kernel intervals.
  std::string device;
};

struct StatsCollector {
  std::mutex mu;
  std::vector<NodeStats> stats;
  void Add(const std::string& node, const std::string& op, int64_t start_us,
           int64_t end_us) {
    std::lock_guard<std::mutex> l(mu);
    stats.push_back({node, op, start_us, end_us, ""});
  }
  // Hardware-trace hooks (gpu/gpu_tracer.h): bracket a GPU node's kernel
  // enqueue with stream events. Unset when tracing is off or no GPU.
  std::function<void*(Device*)> gpu_pre;
  std::function<void(Device*, void*, const std::string&, const std::string&)>
      gpu_post;
};

struct ExecutorArgs {
  int64_t step_id = 0;
  Rendezvous* rendezvous = nullptr;
  // Work scheduler (normally ThreadPool::Schedule; during hipGraph capture a
  // single-threaded trampoline so every kernel enqueue happens on the
  // capturing thread).
  std::function<void(std::function<void()>)> schedule;
  void* resource_mgr = nullptr;
  std::function<bool()> is_cancelled;
  StatsCollector* stats = nullptr;  // non-null => collect per-node timings
};

class Executor {
 public:
  // Takes ownership of `graph`. Kernels for stateful ops come from `opseg`.
  static Status Create(std::unique_ptr<Graph> graph, Device* device,
                       OpSegment* opseg, std::unique_ptr<Executor>* out);
  ~Executor();

  void RunAsync(const ExecutorArgs& args, std::function<void(Status)> done);
  Status Run(const ExecutorArgs& args);

 private:
  friend class ExecutorState;
  Executor() {}
  Status Initialize();

  struct NodeItem {
    Node* node = nullptr;
    OpKernel* kernel = nullptr;  // null for executor-routed control flow
    bool owned_kernel = false;
    bool is_merge = false, is_switch = false, is_enter = false,
         is_exit = false, is_next_iter = false, is_async = false,
         is_constant_enter = false, is_noop = false;
    int input_start = 0;  // offset into per-iteration entry array
  };

  std::unique_ptr<Graph> graph_;
  Device* device_ = nullptr;
  OpSegment* opseg_ = nullptr;
  std::vector<NodeItem> items_;  // indexed by node id
  std::vector<Node*> roots_;     // zero-input nodes
  int total_inputs_ = 0;
};

}  // namespace stf
