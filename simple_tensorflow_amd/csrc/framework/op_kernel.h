// OpKernel / OpKernelContext / kernel registry.
//
// Capability analog of the reference's core/framework/op_kernel.h and
// REGISTER_KERNEL_BUILDER dispatch. MI355X-first differences: exactly two
// device types (CPU, GPU=HIP gfx950); memory placement of each arg is explicit
// in the kernel registration (HostMemory) with no implicit int32 rule; GPU
// kernels enqueue onto the device's compute hipStream and return.
#pragma once

#include <functional>
#include <map>
#include <memory>
#include <set>
#include <string>
#include <vector>

#include "core/base.h"
#include "core/tensor.h"
#include "framework/op.h"

namespace stf {

class Device;

constexpr const char* DEVICE_CPU = "CPU";
constexpr const char* DEVICE_GPU = "GPU";

class OpKernel;
class OpKernelContext;

// ---------------------------------------------------------------------------
// Construction-time context: node def + resolved types.
// ---------------------------------------------------------------------------
class OpKernelConstruction {
 public:
  OpKernelConstruction(const NodeDef* def, const OpDef* op_def,
                       std::vector<DataType> in_types,
                       std::vector<DataType> out_types, Device* device)
      : def_(def), op_def_(op_def), in_types_(std::move(in_types)),
        out_types_(std::move(out_types)), device_(device) {}

  const NodeDef& def() const { return *def_; }
  const OpDef& op_def() const { return *op_def_; }
  Device* device() const { return device_; }
  const std::vector<DataType>& input_types() const { return in_types_; }
  const std::vector<DataType>& output_types() const { return out_types_; }

  template <typename T>
  Status GetAttr(const std::string& name, T* out) const;

  void SetStatus(const Status& s) { status_ = s; }
  const Status& status() const { return status_; }

 private:
  const NodeDef* def_;
  const OpDef* op_def_;
  std::vector<DataType> in_types_;
  std::vector<DataType> out_types_;
  Device* device_;
  Status status_;
};

// ---------------------------------------------------------------------------
// OpKernel
// ---------------------------------------------------------------------------
class OpKernel {
 public:
  explicit OpKernel(OpKernelConstruction* ctx)
      : def_(ctx->def()), in_types_(ctx->input_types()),
        out_types_(ctx->output_types()) {}
  virtual ~OpKernel() {}

  virtual void Compute(OpKernelContext* ctx) = 0;
  virtual bool IsAsync() const { return false; }

  const NodeDef& def() const { return def_; }
  const std::string& name() const { return def_.name; }
  const std::string& type_string() const { return def_.op; }
  int num_inputs() const { return (int)in_types_.size(); }
  int num_outputs() const { return (int)out_types_.size(); }
  DataType input_type(int i) const { return in_types_[i]; }
  DataType output_type(int i) const { return out_types_[i]; }

  // Memory space per arg, filled by CreateOpKernel from the KernelDef.
  std::vector<MemSpace> input_mem;
  std::vector<MemSpace> output_mem;
  // True if this kernel is cheap enough to run inline in the executor loop.
  virtual bool IsExpensive() const { return expensive_; }
  void set_expensive(bool e) { expensive_ = e; }

 private:
  NodeDef def_;
  std::vector<DataType> in_types_;
  std::vector<DataType> out_types_;
  bool expensive_ = true;
};

class AsyncOpKernel : public OpKernel {
 public:
  using OpKernel::OpKernel;
  using DoneCallback = std::function<void()>;
  virtual void ComputeAsync(OpKernelContext* ctx, DoneCallback done) = 0;
  void Compute(OpKernelContext*) override { LOG(FATAL) << "use ComputeAsync"; }
  bool IsAsync() const override { return true; }
};

// ---------------------------------------------------------------------------
// Runtime context for one kernel invocation.
// ---------------------------------------------------------------------------
class OpKernelContext {
 public:
  OpKernelContext(OpKernel* kernel, Device* device, std::vector<Tensor> inputs)
      : kernel_(kernel), device_(device), inputs_(std::move(inputs)) {
    outputs_.resize(kernel->num_outputs());
  }

  OpKernel* kernel() const { return kernel_; }
  Device* device() const { return device_; }

  int num_inputs() const { return (int)inputs_.size(); }
  const Tensor& input(int i) const { return inputs_[i]; }
  Tensor* mutable_input(int i) { return &inputs_[i]; }

  int num_outputs() const { return (int)outputs_.size(); }

  // Allocates output i with the kernel's registered memory space.
  Tensor* allocate_output(int i, const TensorShape& shape);
  // Allocate a scratch tensor in device memory (or host if on CPU).
  Tensor allocate_temp(DataType dtype, const TensorShape& shape);
  void set_output(int i, const Tensor& t) { outputs_[i] = t; }
  const Tensor& output(int i) const { return outputs_[i]; }

  void SetStatus(const Status& s) { status_ = s; }
  const Status& status() const { return status_; }

  // Per-step rendezvous handle (used by _Send/_Recv); opaque to most kernels.
  void* rendezvous = nullptr;
  int64_t step_id = 0;
  // Frame/iteration of the executing node (control flow).
  std::string frame_name;
  int64_t iter_id = 0;
  // Session-level resource map (queues, variables keyed by name).
  void* resource_mgr = nullptr;
  // Cancellation/termination signal shared by the step.
  std::function<bool()> is_cancelled;

  std::vector<Tensor>& outputs() { return outputs_; }

 private:
  OpKernel* kernel_;
  Device* device_;
  std::vector<Tensor> inputs_;
  std::vector<Tensor> outputs_;
  Status status_;
};

#define OP_REQUIRES(ctx, cond, status)  \
  if (!(cond)) {                        \
    (ctx)->SetStatus(status);           \
    return;                             \
  }
#define OP_REQUIRES_OK(ctx, ...)        \
  {                                     \
    ::stf::Status _s = (__VA_ARGS__);   \
    if (!_s.ok()) {                     \
      (ctx)->SetStatus(_s);             \
      return;                           \
    }                                   \
  }
#define OP_REQUIRES_ASYNC(ctx, cond, status, done) \
  if (!(cond)) {                                   \
    (ctx)->SetStatus(status);                      \
    done();                                        \
    return;                                        \
  }

// ---------------------------------------------------------------------------
// Kernel registry
// ---------------------------------------------------------------------------
struct KernelDef {
  std::string op;
  std::string device_type;
  std::map<std::string, std::vector<DataType>> constraints;
  std::set<std::string> host_memory;  // arg names forced to host memory
  std::function<OpKernel*(OpKernelConstruction*)> factory;
};

class KernelRegistry {
 public:
  static KernelRegistry* Global();
  void Register(const KernelDef& def);
  // Find a kernel def for (op, device) matching node attrs; nullptr if none.
  const KernelDef* Find(const NodeDef& node, const std::string& device_type) const;
  bool HasKernel(const NodeDef& node, const std::string& device_type) const {
    return Find(node, device_type) != nullptr;
  }

 private:
  std::multimap<std::string, KernelDef> kernels_;
};

class KernelDefBuilder {
 public:
  explicit KernelDefBuilder(const std::string& op) { def_.op = op; }
  KernelDefBuilder& Device(const std::string& d) {
    def_.device_type = d;
    return *this;
  }
  template <typename T>
  KernelDefBuilder& TypeConstraint(const std::string& attr) {
    def_.constraints[attr].push_back(DataTypeToEnum<T>::v);
    return *this;
  }
  KernelDefBuilder& TypeConstraintList(const std::string& attr,
                                       std::vector<DataType> dts) {
    auto& v = def_.constraints[attr];
    v.insert(v.end(), dts.begin(), dts.end());
    return *this;
  }
  KernelDefBuilder& HostMemory(const std::string& arg) {
    def_.host_memory.insert(arg);
    return *this;
  }
  KernelDef def_;
};

inline KernelDefBuilder Name(const std::string& op) {
  return KernelDefBuilder(op);
}

namespace register_kernel {
struct Registrar {
  Registrar(KernelDefBuilder& b,
            std::function<OpKernel*(OpKernelConstruction*)> factory) {
    b.def_.factory = std::move(factory);
    KernelRegistry::Global()->Register(b.def_);
  }
};
}  // namespace register_kernel

#define REGISTER_KERNEL_BUILDER_UNIQ(ctr, builder, ...)                       \
  static ::stf::register_kernel::Registrar registrar_kernel_##ctr(            \
      builder, [](::stf::OpKernelConstruction* ctx) -> ::stf::OpKernel* {     \
        return new __VA_ARGS__(ctx);                                          \
      });
#define REGISTER_KERNEL_BUILDER_UNIQ_HELPER(ctr, builder, ...) \
  REGISTER_KERNEL_BUILDER_UNIQ(ctr, builder, __VA_ARGS__)
#define REGISTER_KERNEL_BUILDER(builder, ...) \
  REGISTER_KERNEL_BUILDER_UNIQ_HELPER(__COUNTER__, builder, __VA_ARGS__)

// Instantiate the kernel for `node` on `device_type`, resolving memory spaces.
Status CreateOpKernel(const std::string& device_type, Device* device,
                      const NodeDef& node, std::unique_ptr<OpKernel>* kernel);

// ---------------------------------------------------------------------------
// GetAttr impls
// ---------------------------------------------------------------------------
template <>
inline Status OpKernelConstruction::GetAttr<int64_t>(const std::string& name,
                                                     int64_t* out) const {
  if (GetAttrInt(def(), name, out)) return Status::OK();
  return errors::NotFound("attr ", name);
}
template <>
inline Status OpKernelConstruction::GetAttr<int>(const std::string& name,
                                                 int* out) const {
  int64_t v;
  STF_RETURN_IF_ERROR(GetAttr<int64_t>(name, &v));
  *out = (int)v;
  return Status::OK();
}
template <>
inline Status OpKernelConstruction::GetAttr<bool>(const std::string& name,
                                                  bool* out) const {
  if (GetAttrBool(def(), name, out)) return Status::OK();
  return errors::NotFound("attr ", name);
}
template <>
inline Status OpKernelConstruction::GetAttr<float>(const std::string& name,
                                                   float* out) const {
  auto it = def().attr.find(name);
  if (it == def().attr.end() || it->second.kind != 'f')
    return errors::NotFound("attr ", name);
  *out = it->second.f;
  return Status::OK();
}
template <>
inline Status OpKernelConstruction::GetAttr<std::string>(
    const std::string& name, std::string* out) const {
  if (GetAttrString(def(), name, out)) return Status::OK();
  return errors::NotFound("attr ", name);
}
template <>
inline Status OpKernelConstruction::GetAttr<DataType>(const std::string& name,
                                                      DataType* out) const {
  if (GetAttrType(def(), name, out)) return Status::OK();
  return errors::NotFound("attr ", name);
}
template <>
inline Status OpKernelConstruction::GetAttr<std::vector<int64_t>>(
    const std::string& name, std::vector<int64_t>* out) const {
  auto it = def().attr.find(name);
  if (it == def().attr.end() || it->second.kind != 'l')
    return errors::NotFound("attr ", name);
  *out = it->second.list.i;
  return Status::OK();
}
template <>
inline Status OpKernelConstruction::GetAttr<std::vector<float>>(
    const std::string& name, std::vector<float>* out) const {
  auto it = def().attr.find(name);
  if (it == def().attr.end() || it->second.kind != 'l')
    return errors::NotFound("attr ", name);
  *out = it->second.list.f;
  return Status::OK();
}
template <>
inline Status OpKernelConstruction::GetAttr<std::vector<std::string>>(
    const std::string& name, std::vector<std::string>* out) const {
  auto it = def().attr.find(name);
  if (it == def().attr.end() || it->second.kind != 'l')
    return errors::NotFound("attr ", name);
  *out = it->second.list.s;
  return Status::OK();
}
template <>
inline Status OpKernelConstruction::GetAttr<TensorShape>(
    const std::string& name, TensorShape* out) const {
  auto it = def().attr.find(name);
  if (it == def().attr.end() || it->second.kind != 'h')
    return errors::NotFound("attr ", name);
  *out = it->second.shape.AsShape();
  return Status::OK();
}

}  // namespace stf
