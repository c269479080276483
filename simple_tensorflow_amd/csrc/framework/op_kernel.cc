#include "framework/op_kernel.h"

#include <algorithm>
#include <mutex>

#include "framework/device.h"

namespace stf {

Tensor* OpKernelContext::allocate_output(int i, const TensorShape& shape) {
  CHECK(i >= 0 && i < num_outputs());
  DataType dt = kernel_->output_type(i);
  Allocator* alloc =
      (kernel_->output_mem.empty() || kernel_->output_mem[i] == MemSpace::HOST)
          ? device_->host_allocator()
          : device_->allocator();
  if (!device_->is_gpu()) alloc = cpu_allocator();
  outputs_[i] = Tensor(alloc, dt, shape);
  return &outputs_[i];
}

Tensor OpKernelContext::allocate_temp(DataType dtype, const TensorShape& shape) {
  Allocator* alloc = device_->is_gpu() ? device_->allocator() : cpu_allocator();
  return Tensor(alloc, dtype, shape);
}

KernelRegistry* KernelRegistry::Global() {
  static KernelRegistry* r = new KernelRegistry();
  return r;
}

static std::mutex& kreg_mu() {
  static std::mutex mu;
  return mu;
}

void KernelRegistry::Register(const KernelDef& def) {
  std::lock_guard<std::mutex> l(kreg_mu());
  kernels_.emplace(def.op + "|" + def.device_type, def);
}

const KernelDef* KernelRegistry::Find(const NodeDef& node,
                                      const std::string& device_type) const {
  std::lock_guard<std::mutex> l(kreg_mu());
  auto range = kernels_.equal_range(node.op + "|" + device_type);
  for (auto it = range.first; it != range.second; ++it) {
    const KernelDef& kd = it->second;
    bool ok = true;
    for (auto& c : kd.constraints) {
      DataType dt;
      if (!GetAttrType(node, c.first, &dt)) {
        ok = false;
        break;
      }
      if (std::find(c.second.begin(), c.second.end(), dt) == c.second.end()) {
        ok = false;
        break;
      }
    }
    if (ok) return &kd;
  }
  return nullptr;
}

Status CreateOpKernel(const std::string& device_type, Device* device,
                      const NodeDef& node, std::unique_ptr<OpKernel>* kernel) {
  const OpDef* op_def = OpRegistry::Global()->LookUp(node.op);
  if (!op_def) return errors::NotFound("Op not registered: ", node.op);
  const KernelDef* kd = KernelRegistry::Global()->Find(node, device_type);
  if (!kd)
    return errors::NotFound("No ", device_type, " kernel for op ", node.op,
                            " (node ", node.name, ")");
  std::vector<DataType> in_types, out_types;
  STF_RETURN_IF_ERROR(InOutTypesForNode(node, *op_def, &in_types, &out_types));
  OpKernelConstruction ctx(&node, op_def, in_types, out_types, device);
  std::unique_ptr<OpKernel> k(kd->factory(&ctx));
  if (!ctx.status().ok()) return ctx.status();

  // Resolve per-arg memory spaces from the KernelDef's HostMemory set by
  // walking the OpDef args (expanding number_attr repeats).
  auto resolve = [&](const std::vector<OpDef::ArgDef>& args,
                     std::vector<MemSpace>* out) {
    for (auto& arg : args) {
      int64_t n = 1;
      if (!arg.number_attr.empty()) GetAttrInt(node, arg.number_attr, &n);
      if (!arg.type_list_attr.empty()) {
        auto it = node.attr.find(arg.type_list_attr);
        n = (it != node.attr.end()) ? (int64_t)it->second.list.type.size() : 0;
      }
      MemSpace ms = (device_type == DEVICE_GPU &&
                     kd->host_memory.count(arg.name) == 0)
                        ? MemSpace::DEVICE
                        : MemSpace::HOST;
      for (int64_t i = 0; i < n; ++i) out->push_back(ms);
    }
  };
  resolve(op_def->input_arg, &k->input_mem);
  resolve(op_def->output_arg, &k->output_mem);
  *kernel = std::move(k);
  return Status::OK();
}

// ------------------------------ Device -------------------------------------
void Device::Compute(OpKernel* kernel, OpKernelContext* ctx) {
  kernel->Compute(ctx);
}

Status Device::CopyDeviceTensorToHost(const Tensor& src, Tensor* dst) {
  *dst = src;  // CPU device: same memory space
  return Status::OK();
}

Status Device::CopyHostTensorToDevice(const Tensor& src, Tensor* dst) {
  *dst = src;
  return Status::OK();
}

Device* DeviceMgr::LookUp(const std::string& name) const {
  if (name.empty()) return Default();
  std::string want = CanonicalDevice(name);
  if (want.empty()) return nullptr;
  for (auto& d : devices_) {
    if (CanonicalDevice(d->name()) == want) return d.get();
  }
  // Allow "CPU:*"→ the CPU device, "GPU:*" unmatched returns nullptr.
  return nullptr;
}

std::string CanonicalDevice(const std::string& name) {
  // Find the last component mentioning cpu/gpu.
  std::string s = name;
  std::string out;
  auto parts = StrSplit(s, '/');
  for (auto& p : parts) {
    std::string q = p;
    if (StrStartsWith(q, "device:")) q = q.substr(7);
    std::string lower;
    for (char c : q) lower += (char)tolower(c);
    if (StrStartsWith(lower, "cpu") || StrStartsWith(lower, "gpu")) {
      std::string type = lower.substr(0, 3);
      std::string idx = "0";
      auto colon = q.find(':');
      if (colon != std::string::npos) idx = q.substr(colon + 1);
      if (idx.empty() || idx == "*") idx = "0";
      out = (type == "cpu" ? "CPU:" : "GPU:") + idx;
    }
  }
  return out;
}

}  // namespace stf
