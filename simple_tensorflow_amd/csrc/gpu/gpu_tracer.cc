#include "gpu/gpu_tracer.h"

#include <hip/hip_runtime.h>

#include <chrono>

#include "framework/device.h"
#include "runtime/executor.h"

namespace stf {

struct GpuTracer::Rec {
  hipEvent_t start = nullptr;
  hipEvent_t stop = nullptr;
  std::string node, op;
  Device* dev = nullptr;
  bool closed = false;
};

static int64_t NowUs() {
  return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

GpuTracer::~GpuTracer() {
  for (Rec* r : recs_) {
    if (r->start) (void)hipEventDestroy(r->start);
    if (r->stop) (void)hipEventDestroy(r->stop);
    delete r;
  }
  for (auto& kv : ref_)
    if (kv.second.first) (void)hipEventDestroy((hipEvent_t)kv.second.first);
}

Status GpuTracer::Start(const std::vector<Device*>& gpus) {
  for (Device* d : gpus) {
    hipSetDevice(d->gpu_ordinal());
    hipEvent_t ev = nullptr;
    if (hipEventCreate(&ev) != hipSuccess)
      return errors::Internal("GpuTracer: hipEventCreate failed");
    hipStream_t s = (hipStream_t)d->compute_stream();
    if (hipEventRecord(ev, s) != hipSuccess ||
        hipEventSynchronize(ev) != hipSuccess)
      return errors::Internal("GpuTracer: reference event failed");
    // The reference event has completed: NOW is its host-clock position.
    ref_[d] = {(void*)ev, NowUs()};
  }
  return Status::OK();
}

void* GpuTracer::Pre(Device* dev) {
  if (!dev->is_gpu()) return nullptr;
  std::lock_guard<std::mutex> l(mu_);
  if (!ref_.count(dev)) return nullptr;
  Rec* r = new Rec();
  r->dev = dev;
  hipSetDevice(dev->gpu_ordinal());
  if (hipEventCreate(&r->start) != hipSuccess ||
      hipEventCreate(&r->stop) != hipSuccess) {
    delete r;
    return nullptr;
  }
  (void)hipEventRecord(r->start, (hipStream_t)dev->compute_stream());
  recs_.push_back(r);
  return r;
}

void GpuTracer::Post(Device* dev, void* tag, const std::string& node,
                     const std::string& op) {
  if (!tag) return;
  Rec* r = (Rec*)tag;
  std::lock_guard<std::mutex> l(mu_);
  hipSetDevice(dev->gpu_ordinal());
  (void)hipEventRecord(r->stop, (hipStream_t)dev->compute_stream());
  r->node = node;
  r->op = op;
  r->closed = true;
}

Status GpuTracer::Collect(StatsCollector* out) {
  std::lock_guard<std::mutex> l(mu_);
  for (auto& kv : ref_) (void)kv.first->Sync();
  for (Rec* r : recs_) {
    if (!r->closed) continue;
    auto it = ref_.find(r->dev);
    if (it == ref_.end()) continue;
    hipEvent_t ref_ev = (hipEvent_t)it->second.first;
    float ms0 = 0.f, ms1 = 0.f;
    if (hipEventElapsedTime(&ms0, ref_ev, r->start) != hipSuccess ||
        hipEventElapsedTime(&ms1, ref_ev, r->stop) != hipSuccess)
      continue;
    int64_t base = it->second.second;
    NodeStats ns;
    ns.node = r->node;
    ns.op = r->op;
    ns.start_us = base + (int64_t)(ms0 * 1000.f);
    ns.end_us = base + (int64_t)(ms1 * 1000.f);
    ns.device = "/device:GPU:" + std::to_string(r->dev->gpu_ordinal()) +
                "/stream:compute";
    std::lock_guard<std::mutex> ol(out->mu);
    out->stats.push_back(std::move(ns));
  }
  return Status::OK();
}

}  // namespace stf
