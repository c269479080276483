// Device abstraction: exactly two device types — CPU (ThreadPoolDevice) and
// the MI355X HIP device. Capability analog of the reference's
// common_runtime/device.h + gpu/gpu_device.h, without StreamExecutor's
// multi-platform plugin generality (there is one GPU backend: HIP/gfx950).
#pragma once

#include <memory>
#include <string>
#include <vector>

#include "core/base.h"
#include "core/tensor.h"

namespace stf {

class OpKernel;
class OpKernelContext;

class Device {
 public:
  Device(std::string name, std::string type)
      : name_(std::move(name)), type_(std::move(type)) {}
  virtual ~Device() {}

  const std::string& name() const { return name_; }
  const std::string& device_type() const { return type_; }
  bool is_gpu() const { return type_ == "GPU"; }

  // Device-memory allocator (BFC arena for GPU; plain aligned malloc for CPU).
  virtual Allocator* allocator() = 0;
  // Host-memory allocator used for host-resident args of this device's
  // kernels (pinned memory for the GPU device).
  virtual Allocator* host_allocator() { return cpu_allocator(); }

  // Run the kernel. For GPU this activates the device and enqueues on the
  // compute stream; the call returns once enqueued.
  virtual void Compute(OpKernel* kernel, OpKernelContext* ctx);

  // Block until all pending device work completes.
  virtual Status Sync() { return Status::OK(); }

  // hipStream_t of the compute stream (GPU only; nullptr on CPU).
  virtual void* compute_stream() { return nullptr; }
  virtual int gpu_ordinal() const { return -1; }

  // hipGraph capture of the steady-state step (north-star: hipGraph-captured
  // step loops). Begin/End bracket one executor run on the compute stream;
  // Launch replays the instantiated graph.
  virtual Status BeginGraphCapture() {
    return errors::Unimplemented("not a GPU device");
  }
  virtual Status EndGraphCapture(void** graph_exec) {
    return errors::Unimplemented("not a GPU device");
  }
  virtual Status LaunchCapturedGraph(void* graph_exec) {
    return errors::Unimplemented("not a GPU device");
  }
  virtual bool capturing() const { return false; }

  // Tensor movement. `done` is invoked when the copy is complete.
  virtual Status CopyDeviceTensorToHost(const Tensor& src, Tensor* dst);
  virtual Status CopyHostTensorToDevice(const Tensor& src, Tensor* dst);

 private:
  std::string name_;
  std::string type_;
};

class ThreadPoolDevice : public Device {
 public:
  explicit ThreadPoolDevice(const std::string& name)
      : Device(name, "CPU") {}
  Allocator* allocator() override { return cpu_allocator(); }
};

class DeviceMgr {
 public:
  void AddDevice(std::unique_ptr<Device> d) { devices_.push_back(std::move(d)); }
  // Accepts full ("/job:localhost/replica:0/task:0/device:GPU:0") or short
  // ("/gpu:0", "/device:GPU:0", "gpu:0") names; empty → default device.
  Device* LookUp(const std::string& name) const;
  Device* Default() const {
    return devices_.empty() ? nullptr : devices_[0].get();
  }
  const std::vector<std::unique_ptr<Device>>& devices() const {
    return devices_;
  }

 private:
  std::vector<std::unique_ptr<Device>> devices_;
};

// Canonicalize a device string to "TYPE:index" (e.g. "GPU:0"); empty string if
// unparseable. Understands "/gpu:0", "/device:GPU:0", full job/replica/task
// names, and bare "cpu:0".
std::string CanonicalDevice(const std::string& name);

}  // namespace stf
