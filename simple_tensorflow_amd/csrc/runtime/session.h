// DirectSession: the single-process session runtime.
// Capability analog of the reference's DirectSession
// (common_runtime/direct_session.cc) + SimpleGraphExecutionState +
// SimplePlacer + graph partitioner: owns the full graph, rewrites
// feeds/fetches to client _Recv/_Send through a per-step rendezvous, places
// nodes on CPU/GPU, partitions by device inserting _Send/_Recv pairs, builds
// one dataflow executor per device partition and caches executors by the
// (feeds, fetches, targets) signature.
#pragma once

#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "core/threadpool.h"
#include "framework/device.h"
#include "runtime/executor.h"

namespace stf {

class DirectSession {
 public:
  // Creates the session with all visible devices (CPU + any HIP GPUs).
  // `force_cpu_only` ignores GPUs (used in tests).
  explicit DirectSession(bool force_cpu_only = false, int num_threads = 0);
  ~DirectSession();

  Status Create(const GraphDef& def);
  Status Extend(const GraphDef& def);

  Status Run(const std::vector<std::pair<std::string, Tensor>>& feeds,
             const std::vector<std::string>& fetches,
             const std::vector<std::string>& targets,
             std::vector<Tensor>* outputs, StatsCollector* stats = nullptr,
             int64_t timeout_ms = 0);

  // Session.reset analog: drops all stateful kernels (variables, queues,
  // tables) and cached executors; the graph itself is retained.
  void Reset();

  // Partial-run support (reference DirectSession::PRunSetup/PRun): launch
  // the executors for the union of feeds/fetches now, feed and fetch
  // incrementally across multiple PartialRun calls.
  Status PartialRunSetup(const std::vector<std::string>& feeds,
                         const std::vector<std::string>& fetches,
                         const std::vector<std::string>& targets,
                         std::string* handle);
  Status PartialRun(const std::string& handle,
                    const std::vector<std::pair<std::string, Tensor>>& feeds,
                    const std::vector<std::string>& fetches,
                    std::vector<Tensor>* outputs);

  DeviceMgr* device_mgr() { return &devices_; }
  // Blocks until all device work is complete (bench timing bracket).
  Status SyncAllDevices() {
    for (auto& d : devices_.devices()) STF_RETURN_IF_ERROR(d->Sync());
    return Status::OK();
  }
  void* resource_mgr() { return resource_mgr_; }

 private:
  struct PartialRunState;
  struct ExecutorsAndKeys {
    struct Item {
      Device* device;
      std::unique_ptr<Executor> executor;
    };
    std::vector<Item> items;
    // feed key -> (recv_device name); fetch key -> send_device name
    std::map<std::string, std::string> feed_devices;
    std::map<std::string, std::string> fetch_devices;
    // hipGraph capture of the steady-state step (no feeds/fetches, one GPU
    // partition, host side trivial): after `kCaptureAfter` plain runs the
    // next run is recorded and subsequent runs replay the instantiated graph.
    bool capture_eligible = false;
    bool capture_broken = false;
    int plain_runs = 0;
    Device* capture_device = nullptr;
    void* graph_exec = nullptr;
  };

  Status GetOrCreateExecutors(
      const std::vector<std::string>& feeds,
      const std::vector<std::string>& fetches,
      const std::vector<std::string>& targets, ExecutorsAndKeys** out);

  Status BuildExecutors(const std::vector<std::string>& feeds,
                        const std::vector<std::string>& fetches,
                        const std::vector<std::string>& targets,
                        std::unique_ptr<ExecutorsAndKeys>* out);

  std::mutex mu_;
  GraphDef graph_def_;
  DeviceMgr devices_;
  OpSegment opseg_;
  std::unique_ptr<ThreadPool> pool_;
  std::map<std::string, std::unique_ptr<ExecutorsAndKeys>> executors_;
  std::map<std::string, std::shared_ptr<PartialRunState>> partial_runs_;
  int64_t partial_run_counter_ = 0;
  int64_t step_counter_ = 0;
  void* resource_mgr_ = nullptr;  // owned; see kernels/resource_mgr
};

}  // namespace stf
