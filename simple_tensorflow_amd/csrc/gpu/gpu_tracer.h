// Hardware kernel-interval tracing for RunOptions(FULL_TRACE): hipEvent
// pairs bracket every GPU node's enqueue on the device's compute stream and
// resolve to wall-clock micros after the step.
//
// Capability analog of the reference's GPUTracer / CUPTIManager
// (common_runtime/gpu/gpu_tracer.cc:104): the reference pulls CUPTI activity
// records out-of-band; the MI355X-native redesign records stream-ordered
// events inline — no external profiler dependency, correct under the
// single-compute-stream execution model, zero cost when tracing is off.
#pragma once

#include <cstdint>
#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "core/base.h"

namespace stf {

class Device;
struct StatsCollector;

class GpuTracer {
 public:
  ~GpuTracer();

  // Record a reference event + host timestamp on every GPU device so device
  // elapsed times can be mapped onto the host clock.
  Status Start(const std::vector<Device*>& gpus);

  // Called before/after a GPU node's Compute (kernel enqueue). Pre returns
  // an opaque tag (nullptr for non-GPU devices → Post is skipped).
  void* Pre(Device* dev);
  void Post(Device* dev, void* tag, const std::string& node,
            const std::string& op);

  // Sync the devices, resolve all event pairs and append them to `out` under
  // a "/device:GPU:n/stream:compute" lane.
  Status Collect(StatsCollector* out);

 private:
  struct Rec;
  std::mutex mu_;
  std::map<Device*, std::pair<void*, int64_t>> ref_;  // hipEvent_t + host us
  std::vector<Rec*> recs_;
};

}  // namespace stf
