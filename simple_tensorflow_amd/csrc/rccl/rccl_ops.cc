// RCCL collectives over xGMI as first-class graph ops — the MI355X-native
// replacement for the reference's PS/AddN gradient aggregation (SURVEY.md
// §2.3: the reference trim has no collectives at all; RCCL over the 7 xGMI
// links per GPU is the designed multi-GPU path, BASELINE.json config 3).
//
// One process per GPU, a single global communicator bootstrapped from Python
// (unique id exchanged over a gloo rendezvous).
//
// Two tiers of ops:
//  - RcclAllReduce / RcclBroadcast: one collective per tensor, enqueued on
//    the COMPUTE stream (strictly ordered, used for variable broadcast and
//    ad-hoc reductions).
//  - RcclBucketAllReduce + RcclCommSync: the gradient hot path. A bucket of
//    up to ~120 gradient tensors is packed into one flat f32 staging buffer,
//    reduced with a single ncclAllReduce, and unpacked (scaled by 1/world)
//    — all on a dedicated COMM stream fenced by hipEvents so the collective
//    overlaps with the rest of backprop on the compute stream. RcclCommSync
//    makes the compute stream wait on every outstanding bucket before the
//    optimizer-apply ops consume the reduced gradients (PyTorch-DDP-shaped
//    schedule, built as graph ops).
//
// Buffer-lifetime rule: BFC reuse is stream-ordered on the COMPUTE stream
// only, so every tensor the comm stream touches (inputs, staging, outputs)
// is ref-held in RcclState::keepalive until RcclCommSync has enqueued the
// compute-stream waits — after that, any reuse of those blocks is enqueued
// behind the comm work and cannot race it.
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <mutex>
#include <vector>

#include "kernels/kernel_util.h"

extern "C" {
int stf_comm_max_segments();
hipError_t stf_comm_pack(int nseg, const void* const* ptrs,
                         const int64_t* ends, const unsigned char* is_bf16,
                         float* dst, int64_t total, hipStream_t stream);
hipError_t stf_comm_unpack(int nseg, void* const* ptrs, const int64_t* ends,
                           const unsigned char* is_bf16, const float* src,
                           float scale, int64_t total, hipStream_t stream);
}

namespace stf {

namespace {

struct RcclState {
  ncclComm_t comm = nullptr;
  int nranks = 0;
  int rank = -1;
  hipStream_t comm_stream = nullptr;
  std::mutex mu;
  std::vector<hipEvent_t> pending;   // bucket done-events awaiting a sync
  std::vector<Tensor> keepalive;     // tensors the comm stream still reads
};

RcclState* GlobalRccl() {
  static RcclState* s = new RcclState();
  return s;
}

#define OP_NCCL_OK(ctx, expr)                                              \
  {                                                                        \
    ncclResult_t _r = (expr);                                              \
    if (_r != ncclSuccess) {                                               \
      (ctx)->SetStatus(errors::Internal("RCCL failure: ",                  \
                                        ncclGetErrorString(_r)));          \
      return;                                                              \
    }                                                                      \
  }

#define OP_HIP_OK(ctx, expr)                                               \
  {                                                                        \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      (ctx)->SetStatus(errors::Internal("HIP failure: ",                   \
                                        hipGetErrorString(_e)));           \
      return;                                                              \
    }                                                                      \
  }

ncclDataType_t ToNccl(DataType dt) {
  switch (dt) {
    case DT_FLOAT: return ncclFloat32;
    case DT_BFLOAT16: return ncclBfloat16;
    case DT_HALF: return ncclFloat16;
    default: return ncclFloat32;
  }
}

class RcclAllReduceOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    RcclState* st = GlobalRccl();
    OP_REQUIRES(ctx, st->comm != nullptr,
                errors::FailedPrecondition(
                    "RCCL communicator not initialized (call "
                    "parallel.dist.init first)"));
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    hipStream_t s = (hipStream_t)ctx->device()->compute_stream();
    OP_NCCL_OK(ctx, ncclAllReduce(in.raw_data(), out->raw_data(),
                                  in.NumElements(), ToNccl(in.dtype()),
                                  ncclSum, st->comm, s));
  }
};
REGISTER_KERNEL_BUILDER(Name("RcclAllReduce").Device(DEVICE_GPU), RcclAllReduceOp);

class RcclBroadcastOp : public OpKernel {
 public:
  explicit RcclBroadcastOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("root", &root_);
  }
  void Compute(OpKernelContext* ctx) override {
    RcclState* st = GlobalRccl();
    OP_REQUIRES(ctx, st->comm != nullptr,
                errors::FailedPrecondition("RCCL not initialized"));
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    hipStream_t s = (hipStream_t)ctx->device()->compute_stream();
    OP_NCCL_OK(ctx, ncclBroadcast(in.raw_data(), out->raw_data(),
                                  in.NumElements(), ToNccl(in.dtype()),
                                  (int)root_, st->comm, s));
  }

 private:
  int64_t root_ = 0;
};
REGISTER_KERNEL_BUILDER(Name("RcclBroadcast").Device(DEVICE_GPU), RcclBroadcastOp);

// Fused-bucket all-reduce on the comm stream. Outputs are only valid for
// compute-stream consumers scheduled after a RcclCommSync (the
// DistributedOptimizer wires that control edge).
class RcclBucketAllReduceOp : public OpKernel {
 public:
  explicit RcclBucketAllReduceOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("scale", &scale_);
  }
  ~RcclBucketAllReduceOp() override {
    if (ev_ready_) hipEventDestroy(ev_ready_);
    if (ev_done_) hipEventDestroy(ev_done_);
  }

  void Compute(OpKernelContext* ctx) override {
    RcclState* st = GlobalRccl();
    OP_REQUIRES(ctx, st->comm != nullptr,
                errors::FailedPrecondition(
                    "RCCL communicator not initialized (call "
                    "parallel.dist.init first)"));
    OP_REQUIRES(ctx, !ctx->device()->capturing(),
                errors::FailedPrecondition(
                    "RcclBucketAllReduce inside hipGraph capture is not "
                    "supported yet; run multi-GPU with STF_NO_HIPGRAPH=1"));
    int n = ctx->num_inputs();
    OP_REQUIRES(ctx, n <= stf_comm_max_segments(),
                errors::InvalidArgument("bucket has ", n, " tensors, max ",
                                        stf_comm_max_segments()));
    std::vector<const void*> in_ptrs(n);
    std::vector<void*> out_ptrs(n);
    std::vector<int64_t> ends(n);
    std::vector<unsigned char> bf16(n);
    int64_t total = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& in = ctx->input(i);
      OP_REQUIRES(ctx, in.dtype() == DT_FLOAT || in.dtype() == DT_BFLOAT16,
                  errors::InvalidArgument("bucket dtype must be f32/bf16"));
      Tensor* out = ctx->allocate_output(i, in.shape());
      in_ptrs[i] = in.raw_data();
      out_ptrs[i] = out->raw_data();
      total += in.NumElements();
      ends[i] = total;
      bf16[i] = in.dtype() == DT_BFLOAT16 ? 1 : 0;
    }
    Tensor flat(ctx->device()->allocator(), DT_FLOAT, TensorShape({total}));
    OP_REQUIRES(ctx, flat.raw_data() != nullptr,
                errors::ResourceExhausted("bucket staging alloc failed"));

    if (!ev_ready_) {
      OP_HIP_OK(ctx, hipEventCreateWithFlags(&ev_ready_,
                                             hipEventDisableTiming));
      OP_HIP_OK(ctx, hipEventCreateWithFlags(&ev_done_,
                                             hipEventDisableTiming));
    }
    hipStream_t compute = (hipStream_t)ctx->device()->compute_stream();
    hipStream_t comm = st->comm_stream;
    // Inputs (and any freed block the staging buffer may reuse) are complete
    // once all compute work enqueued so far completes.
    OP_HIP_OK(ctx, hipEventRecord(ev_ready_, compute));
    OP_HIP_OK(ctx, hipStreamWaitEvent(comm, ev_ready_, 0));
    OP_HIP_OK(ctx, stf_comm_pack(n, in_ptrs.data(), ends.data(), bf16.data(),
                                 (float*)flat.raw_data(), total, comm));
    OP_NCCL_OK(ctx, ncclAllReduce(flat.raw_data(), flat.raw_data(), total,
                                  ncclFloat32, ncclSum, st->comm, comm));
    OP_HIP_OK(ctx, stf_comm_unpack(n, out_ptrs.data(), ends.data(),
                                   bf16.data(), (const float*)flat.raw_data(),
                                   scale_, total, comm));
    OP_HIP_OK(ctx, hipEventRecord(ev_done_, comm));
    {
      std::lock_guard<std::mutex> l(st->mu);
      st->pending.push_back(ev_done_);
      for (int i = 0; i < n; ++i) st->keepalive.push_back(ctx->input(i));
      st->keepalive.push_back(flat);
      for (int i = 0; i < n; ++i) st->keepalive.push_back(ctx->output(i));
    }
  }

 private:
  float scale_ = 1.0f;
  hipEvent_t ev_ready_ = nullptr;
  hipEvent_t ev_done_ = nullptr;
};
REGISTER_KERNEL_BUILDER(Name("RcclBucketAllReduce").Device(DEVICE_GPU),
                        RcclBucketAllReduceOp);

// Joins the comm stream back into the compute stream: every op downstream
// of this one (by control edge) sees completed bucket all-reduces.
class RcclCommSyncOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    RcclState* st = GlobalRccl();
    hipStream_t compute = (hipStream_t)ctx->device()->compute_stream();
    std::vector<hipEvent_t> evs;
    std::vector<Tensor> dead;
    {
      std::lock_guard<std::mutex> l(st->mu);
      evs.swap(st->pending);
      dead.swap(st->keepalive);
    }
    for (hipEvent_t ev : evs)
      OP_HIP_OK(ctx, hipStreamWaitEvent(compute, ev, 0));
    // `dead` now drops the refs: any BFC reuse of those blocks is enqueued
    // after the waits above, hence ordered behind the comm-stream work.
  }
};
REGISTER_KERNEL_BUILDER(Name("RcclCommSync").Device(DEVICE_GPU),
                        RcclCommSyncOp);

}  // namespace

// ---- python bootstrap hooks (called from pybind/module.cc) ----
std::string RcclGetUniqueId() {
  ncclUniqueId id;
  if (ncclGetUniqueId(&id) != ncclSuccess) return "";
  return std::string(id.internal, NCCL_UNIQUE_ID_BYTES);
}

Status RcclInit(int nranks, int rank, const std::string& id_bytes) {
  RcclState* st = GlobalRccl();
  if (st->comm) return Status::OK();
  if (id_bytes.size() != NCCL_UNIQUE_ID_BYTES)
    return errors::InvalidArgument("bad rccl unique id size ",
                                   id_bytes.size());
  ncclUniqueId id;
  memcpy(id.internal, id_bytes.data(), NCCL_UNIQUE_ID_BYTES);
  hipSetDevice(0);  // one visible GPU per process (HIP_VISIBLE_DEVICES)
  ncclResult_t r = ncclCommInitRank(&st->comm, nranks, id, rank);
  if (r != ncclSuccess)
    return errors::Internal("ncclCommInitRank: ", ncclGetErrorString(r));
  if (hipStreamCreateWithFlags(&st->comm_stream, hipStreamNonBlocking) !=
      hipSuccess)
    return errors::Internal("comm stream creation failed");
  st->nranks = nranks;
  st->rank = rank;
  return Status::OK();
}

}  // namespace stf
