// MI355X HIP device: stream group (compute/h2d/d2h), BFC allocator over
// hipMalloc sized for 288 GB HBM3E, pinned-host allocator.
//
// Capability analog of the reference's BaseGPUDevice + GPUBFCAllocator
// (common_runtime/gpu/gpu_device.cc, gpu_bfc_allocator.cc, bfc_allocator.h:44)
// without StreamExecutor: there is exactly one GPU backend (HIP/gfx950), so
// kernels enqueue directly on the device's hipStream_t. All compute runs on a
// single compute stream, which makes BFC reuse stream-ordered by
// construction; H2D/D2H copies run on side streams fenced by hipEvents.
#include <hip/hip_runtime.h>

#include <chrono>
#include <condition_variable>
#include <cstring>
#include <memory>
#include <map>
#include <mutex>
#include <set>
#include <vector>

#include "framework/device.h"
#include "framework/op_kernel.h"

namespace stf {

#define HIP_CHECK_STATUS(expr)                                             \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      return errors::Internal("HIP error: ", hipGetErrorString(_e), " at ", \
                              __FILE__, ":", __LINE__);                    \
  } while (0)

// ---------------------------------------------------------------------------
// BFC allocator over hipMalloc: power-of-two bins, best-fit, split+coalesce,
// on-demand region growth (allow_growth style — the MI355X has 288 GB, we
// grab 1 GiB regions as needed and reuse aggressively).
// ---------------------------------------------------------------------------
class GpuBfcAllocator : public Allocator {
 public:
  explicit GpuBfcAllocator(int ordinal) : ordinal_(ordinal) {}

  void* Allocate(size_t bytes) override {
    if (bytes == 0) bytes = 256;
    size_t size = RoundUp(bytes);
    std::unique_lock<std::mutex> l(mu_);
    Chunk* c = FindFree(size);
    if (!c) {
      if (!Grow(size)) {
        // Out of device memory right now. Another thread (a concurrent
        // session / the EventMgr draining d2h copies) may free soon: wait for
        // a Deallocate and retry, up to ~2 s total (reference
        // allocator_retry.cc behavior), then dump state and fail hard.
        auto deadline =
            std::chrono::steady_clock::now() + std::chrono::seconds(2);
        while (!c && std::chrono::steady_clock::now() < deadline) {
          if (free_cv_.wait_until(l, deadline) == std::cv_status::timeout)
            break;
          c = FindFree(size);
          if (!c && Grow(size)) c = FindFree(size);
        }
        if (!c) {
          LOG(ERROR) << "GPU" << ordinal_ << " BFC: out of memory allocating "
                     << bytes << " bytes after retry\n"
                     << DumpState();
          return nullptr;
        }
      } else {
        c = FindFree(size);
        if (!c) return nullptr;
      }
    }
    RemoveFromFree(c);
    // Split if the remainder is useful.
    if (c->size >= size + 512) {
      Chunk* rest = new Chunk();
      rest->ptr = (char*)c->ptr + size;
      rest->size = c->size - size;
      rest->prev = c;
      rest->next = c->next;
      if (rest->next) rest->next->prev = rest;
      c->next = rest;
      c->size = size;
      InsertFree(rest);
    }
    c->in_use = true;
    in_use_bytes_ += c->size;
    by_ptr_[c->ptr] = c;
    return c->ptr;
  }

  void Deallocate(void* ptr, size_t) override {
    if (!ptr) return;
    std::lock_guard<std::mutex> l(mu_);
    auto it = by_ptr_.find(ptr);
    if (it == by_ptr_.end()) {
      LOG(ERROR) << "BFC: free of unknown pointer";
      return;
    }
    Chunk* c = it->second;
    by_ptr_.erase(it);
    c->in_use = false;
    in_use_bytes_ -= c->size;
    // Coalesce with neighbors inside the same region.
    if (c->next && !c->next->in_use) {
      Chunk* n = c->next;
      RemoveFromFree(n);
      c->size += n->size;
      c->next = n->next;
      if (c->next) c->next->prev = c;
      delete n;
    }
    if (c->prev && !c->prev->in_use) {
      Chunk* p = c->prev;
      RemoveFromFree(p);
      p->size += c->size;
      p->next = c->next;
      if (p->next) p->next->prev = p;
      delete c;
      c = p;
    }
    InsertFree(c);
    free_cv_.notify_all();
  }

  MemSpace space() const override { return MemSpace::DEVICE; }
  int device_ordinal() const override { return ordinal_; }
  const char* name() const override { return "gpu_bfc"; }

 private:
  struct Chunk {
    void* ptr = nullptr;
    size_t size = 0;
    bool in_use = false;
    Chunk* prev = nullptr;  // adjacent in region
    Chunk* next = nullptr;
  };
  struct BySize {
    bool operator()(const Chunk* a, const Chunk* b) const {
      if (a->size != b->size) return a->size < b->size;
      return a->ptr < b->ptr;
    }
  };

  static size_t RoundUp(size_t b) { return (b + 255) & ~size_t(255); }

  Chunk* FindFree(size_t size) {
    Chunk probe;
    probe.size = size;
    probe.ptr = nullptr;
    auto it = free_.lower_bound(&probe);
    return it == free_.end() ? nullptr : *it;
  }
  void InsertFree(Chunk* c) { free_.insert(c); }
  void RemoveFromFree(Chunk* c) { free_.erase(c); }

  // Bin/fragmentation summary for the OOM log (gpu_bfc_allocator DumpMemoryLog
  // analog). Caller holds mu_.
  std::string DumpState() const {
    size_t free_bytes = 0, largest_free = 0;
    std::map<int, std::pair<int, size_t>> bins;  // log2 -> {count, bytes}
    for (const Chunk* c : free_) {
      free_bytes += c->size;
      largest_free = std::max(largest_free, c->size);
      int b = 0;
      while ((1ull << (b + 1)) <= c->size) ++b;
      bins[b].first++;
      bins[b].second += c->size;
    }
    std::string s = "BFC state: total=" + std::to_string(total_bytes_) +
                    " in_use=" + std::to_string(in_use_bytes_) +
                    " free=" + std::to_string(free_bytes) +
                    " largest_free_chunk=" + std::to_string(largest_free) +
                    " live_allocs=" + std::to_string(by_ptr_.size()) + "\n";
    for (auto& kv : bins)
      s += "  free bin 2^" + std::to_string(kv.first) + ": " +
           std::to_string(kv.second.first) + " chunks, " +
           std::to_string(kv.second.second) + " bytes\n";
    return s;
  }

  bool Grow(size_t min_bytes) {
    size_t region = 1ull << 30;  // 1 GiB
    while (region < min_bytes) region <<= 1;
    void* p = nullptr;
    hipError_t e = hipSuccess;
    for (;;) {
      e = hipMalloc(&p, region);
      if (e == hipSuccess) break;
      if (region <= min_bytes || region <= (1ull << 26)) return false;
      region >>= 1;
      if (region < min_bytes) region = RoundUp(min_bytes);
    }
    Chunk* c = new Chunk();
    c->ptr = p;
    c->size = region;
    InsertFree(c);
    total_bytes_ += region;
    return true;
  }

  int ordinal_;
  std::mutex mu_;
  std::condition_variable free_cv_;
  std::set<Chunk*, BySize> free_;
  std::map<void*, Chunk*> by_ptr_;
  size_t total_bytes_ = 0;
  size_t in_use_bytes_ = 0;
};

// Pinned host memory for fast DMA.
class PinnedAllocator : public Allocator {
 public:
  void* Allocate(size_t bytes) override {
    void* p = nullptr;
    if (hipHostMalloc(&p, bytes ? bytes : 1) != hipSuccess) return nullptr;
    return p;
  }
  void Deallocate(void* ptr, size_t) override { hipHostFree(ptr); }
  const char* name() const override { return "pinned"; }
};

// ---------------------------------------------------------------------------
// Debug allocators (reference gpu_debug_allocator.h:33,63 capability analogs),
// selected with STF_GPU_ALLOC_DEBUG=guard|nan|guard,nan. They wrap the BFC
// allocator; guard mode brackets every allocation with 64-byte patterns
// verified on free (catches out-of-bounds kernel writes at the faulting
// allocation), nan mode fills new/freed memory with f32 quiet-NaNs (catches
// reads of uninitialized or dangling buffers in any fp32 consumer).
// ---------------------------------------------------------------------------
class GpuGuardAllocator : public Allocator {
 public:
  static constexpr size_t kGuard = 64;  // bytes each side, 256-aligned base
  GpuGuardAllocator(Allocator* base, bool guard, bool nan)
      : base_(base), guard_(guard), nan_(nan) {
    for (size_t i = 0; i < kGuard / 8; ++i) pattern_[i] = 0x5A5A5A5A5A5A5A5Aull;
  }

  void* Allocate(size_t bytes) override {
    if (bytes == 0) bytes = 256;
    size_t padded = guard_ ? bytes + 2 * 256 : bytes;
    char* base = (char*)base_->Allocate(padded);
    if (!base) return nullptr;
    char* user = base;
    if (guard_) {
      user = base + 256;
      // 256-byte pads keep user alignment; the pattern occupies the 64 bytes
      // adjacent to the user range on each side.
      hipMemcpy(user - kGuard, pattern_, kGuard, hipMemcpyHostToDevice);
      hipMemcpy(user + bytes, pattern_, kGuard, hipMemcpyHostToDevice);
      std::lock_guard<std::mutex> l(mu_);
      live_[user] = {base, bytes};
    }
    if (nan_)
      hipMemsetD32((hipDeviceptr_t)user, 0x7fc00000u, bytes / 4);
    return user;
  }

  void Deallocate(void* ptr, size_t bytes) override {
    if (!ptr) return;
    char* user = (char*)ptr;
    char* base = user;
    size_t n = bytes;
    if (guard_) {
      std::lock_guard<std::mutex> l(mu_);
      auto it = live_.find(user);
      if (it == live_.end()) {
        LOG(ERROR) << "guard allocator: free of unknown pointer " << ptr;
        return;
      }
      base = it->second.first;
      n = it->second.second;
      live_.erase(it);
    }
    if (guard_ && !CheckGuards(user, n))
      LOG(FATAL) << "guard allocator: out-of-bounds GPU write detected on a "
                 << n << "-byte allocation at " << (void*)user;
    if (nan_ && n >= 4)
      hipMemsetD32((hipDeviceptr_t)user, 0x7fc00000u, n / 4);
    base_->Deallocate(base, 0);
  }

  MemSpace space() const override { return MemSpace::DEVICE; }
  int device_ordinal() const override { return base_->device_ordinal(); }
  const char* name() const override { return "gpu_debug"; }

 private:
  bool CheckGuards(char* user, size_t bytes) {
    uint64_t got[kGuard / 8];
    hipDeviceSynchronize();  // settle pending kernels before reading
    hipMemcpy(got, user - kGuard, kGuard, hipMemcpyDeviceToHost);
    if (memcmp(got, pattern_, kGuard) != 0) return false;
    hipMemcpy(got, user + bytes, kGuard, hipMemcpyDeviceToHost);
    return memcmp(got, pattern_, kGuard) == 0;
  }

  Allocator* base_;
  bool guard_, nan_;
  uint64_t pattern_[kGuard / 8];
  std::mutex mu_;
  std::map<void*, std::pair<char*, size_t>> live_;  // user -> {base, bytes}
};

// ---------------------------------------------------------------------------
// GpuDevice
// ---------------------------------------------------------------------------
class GpuDevice : public Device {
 public:
  GpuDevice(int ordinal, const std::string& name)
      : Device(name, "GPU"), ordinal_(ordinal), bfc_(ordinal) {
    hipSetDevice(ordinal_);
    hipStreamCreateWithFlags(&compute_, hipStreamNonBlocking);
    hipStreamCreateWithFlags(&h2d_, hipStreamNonBlocking);
    hipStreamCreateWithFlags(&d2h_, hipStreamNonBlocking);
  }
  ~GpuDevice() override {
    hipStreamDestroy(compute_);
    hipStreamDestroy(h2d_);
    hipStreamDestroy(d2h_);
  }

  Allocator* allocator() override {
    static const char* dbg = getenv("STF_GPU_ALLOC_DEBUG");
    if (dbg && *dbg) {
      if (!debug_alloc_) {
        std::string mode(dbg);
        debug_alloc_.reset(new GpuGuardAllocator(
            &bfc_, mode.find("guard") != std::string::npos,
            mode.find("nan") != std::string::npos));
      }
      return debug_alloc_.get();
    }
    return &bfc_;
  }
  Allocator* host_allocator() override {
    static PinnedAllocator* pinned = new PinnedAllocator();
    return pinned;
  }
  void* compute_stream() override { return (void*)compute_; }
  int gpu_ordinal() const override { return ordinal_; }

  void Compute(OpKernel* kernel, OpKernelContext* ctx) override {
    hipSetDevice(ordinal_);
    kernel->Compute(ctx);
  }

  Status Sync() override {
    hipSetDevice(ordinal_);
    HIP_CHECK_STATUS(hipDeviceSynchronize());
    return Status::OK();
  }

  Status BeginGraphCapture() override {
    hipSetDevice(ordinal_);
    HIP_CHECK_STATUS(hipStreamBeginCapture(compute_,
                                           hipStreamCaptureModeRelaxed));
    capturing_ = true;
    return Status::OK();
  }
  Status EndGraphCapture(void** graph_exec) override {
    hipSetDevice(ordinal_);
    capturing_ = false;
    hipGraph_t graph = nullptr;
    HIP_CHECK_STATUS(hipStreamEndCapture(compute_, &graph));
    if (const char* dot = getenv("STF_GRAPH_DOT")) {
      size_t n_nodes = 0;
      hipGraphGetNodes(graph, nullptr, &n_nodes);
      LOG(INFO) << "captured hipGraph with " << n_nodes << " nodes -> " << dot;
      hipGraphDebugDotPrint(graph, dot, 1 /*verbose*/);
    }
    hipGraphExec_t exec = nullptr;
    hipError_t e = hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0);
    hipGraphDestroy(graph);
    if (e != hipSuccess)
      return errors::Internal("hipGraphInstantiate: ", hipGetErrorString(e));
    *graph_exec = (void*)exec;
    return Status::OK();
  }
  Status LaunchCapturedGraph(void* graph_exec) override {
    hipSetDevice(ordinal_);
    HIP_CHECK_STATUS(hipGraphLaunch((hipGraphExec_t)graph_exec, compute_));
    return Status::OK();
  }
  bool capturing() const override { return capturing_; }

  Status CopyDeviceTensorToHost(const Tensor& src, Tensor* dst) override {
    if (capturing_)
      return errors::FailedPrecondition(
          "d2h copy during hipGraph capture (host-dependent op in the "
          "captured step)");
    hipSetDevice(ordinal_);
    Tensor host(host_allocator(), src.dtype(), src.shape());
    // Fence: wait for pending compute that may produce src.
    hipEvent_t ev;
    HIP_CHECK_STATUS(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    HIP_CHECK_STATUS(hipEventRecord(ev, compute_));
    HIP_CHECK_STATUS(hipStreamWaitEvent(d2h_, ev, 0));
    HIP_CHECK_STATUS(hipMemcpyAsync(host.raw_data(), src.raw_data(),
                                    src.TotalBytes(), hipMemcpyDeviceToHost,
                                    d2h_));
    HIP_CHECK_STATUS(hipStreamSynchronize(d2h_));
    HIP_CHECK_STATUS(hipEventDestroy(ev));
    *dst = host;
    return Status::OK();
  }

  Status CopyHostTensorToDevice(const Tensor& src, Tensor* dst) override {
    hipSetDevice(ordinal_);
    Tensor dev(&bfc_, src.dtype(), src.shape());
    // BFC reuse is stream-ordered ONLY on the compute stream: `dev` may be
    // a just-freed block that pending compute kernels (enqueued before the
    // free) still reference. Writing it on the h2d stream "now" would race
    // those kernels — order the copy after all compute work enqueued so
    // far. (Symmetric to the d2h path above. Skipped while capturing:
    // recording an event on a capturing compute stream would splice the
    // h2d stream into the capture; nothing executes during capture, so
    // there is no pending work to race with.)
    hipEvent_t ev = nullptr;
    if (!capturing_) {
      HIP_CHECK_STATUS(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
      HIP_CHECK_STATUS(hipEventRecord(ev, compute_));
      HIP_CHECK_STATUS(hipStreamWaitEvent(h2d_, ev, 0));
    }
    HIP_CHECK_STATUS(hipMemcpyAsync(dev.raw_data(), src.raw_data(),
                                    src.TotalBytes(), hipMemcpyHostToDevice,
                                    h2d_));
    // Sync the copy: the host source buffer may be released by the caller,
    // and completion here also orders the data before any compute-stream
    // consumer enqueued after this call.
    HIP_CHECK_STATUS(hipStreamSynchronize(h2d_));
    if (ev) HIP_CHECK_STATUS(hipEventDestroy(ev));
    if (capturing_) {
      // The h2d stream is not part of the capture, so this copy is NOT a
      // node of the hipGraph: replayed kernels will re-read `dev` directly.
      // The value is static (capture-eligible steps only move shape-derived
      // host scalars), but the buffer must outlive every replay — pin it.
      std::lock_guard<std::mutex> l(keepalive_mu_);
      capture_keepalive_.push_back(dev);
    }
    *dst = dev;
    return Status::OK();
  }

 private:
  int ordinal_;
  GpuBfcAllocator bfc_;
  std::unique_ptr<GpuGuardAllocator> debug_alloc_;
  hipStream_t compute_, h2d_, d2h_;
  bool capturing_ = false;
  // Device buffers referenced by captured hipGraphs but not produced inside
  // them (host->device shape scalars copied at capture time). A few bytes
  // per captured graph; lives as long as the device.
  std::mutex keepalive_mu_;
  std::vector<Tensor> capture_keepalive_;
};

void AddGpuDevices(DeviceMgr* mgr) {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess) return;
  for (int i = 0; i < count; ++i) {
    mgr->AddDevice(std::make_unique<GpuDevice>(i, "/gpu:" + std::to_string(i)));
  }
}

}  // namespace stf
