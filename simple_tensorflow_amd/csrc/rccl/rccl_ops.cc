// RCCL collectives over xGMI as first-class graph ops (RcclAllReduce,
// RcclBroadcast) — the MI355X-native replacement for the reference's
// PS/AddN gradient aggregation (the reference trim has no
// collectives at all; RCCL over the 7 xGMI links per GPU is the designed
// multi-GPU path, BASELINE.json config 3).
//
// One process per GPU: a single global communicator, initialized from Python
// (bootstrap id exchanged out-of-band), collectives enqueued on the device's
// compute stream so the per-rank enqueue ORDER (fixed by grad-bucket control
// edges) matches across ranks.
#include <rccl/rccl.h>

#include <mutex>

#include "kernels/kernel_util.h"

namespace stf {

namespace {

struct RcclState {
  ncclComm_t comm = nullptr;
  int nranks = 0;
  int rank = -1;
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
  st->nranks = nranks;
  st->rank = rank;
  return Status::OK();
}

}  // namespace stf
