// CPU random ops on the shared Philox engine (analog of reference
// core/kernels/random_op.cc).
#include <atomic>
#include <cmath>

#include "kernels/kernel_util.h"
#include "kernels/philox.h"

namespace stf {

static uint64_t PickSeed(int64_t seed, int64_t seed2) {
  if (seed == 0 && seed2 == 0) {
    static std::atomic<uint64_t> counter{0x9E3779B97F4A7C15ull};
    return counter.fetch_add(0x12345);
  }
  return ((uint64_t)seed << 32) | (uint32_t)seed2;
}

enum class Dist { UNIFORM, NORMAL, TRUNCATED };

template <Dist D>
class RandomOp : public OpKernel {
 public:
  explicit RandomOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    int64_t seed = 0, seed2 = 0;
    ctx->GetAttr("seed", &seed);
    ctx->GetAttr("seed2", &seed2);
    seed_ = PickSeed(seed, seed2);
  }
  void Compute(OpKernelContext* ctx) override {
    auto dims = IntVector(ctx->input(0));
    Tensor* out = ctx->allocate_output(0, TensorShape(dims));
    OP_REQUIRES(ctx, out->dtype() == DT_FLOAT,
                errors::Unimplemented("CPU random supports float32"));
    float* p = out->flat<float>();
    int64_t n = out->NumElements();
    uint64_t offset = offset_.fetch_add((n + 3) / 4 + 4);
    random::Philox4x32 rng(seed_, offset);
    uint32_t r[4];
    if (D == Dist::UNIFORM) {
      for (int64_t i = 0; i < n; i += 4) {
        rng.Next(r);
        for (int j = 0; j < 4 && i + j < n; ++j)
          p[i + j] = random::Uint32ToFloat01(r[j]);
      }
    } else if (D == Dist::NORMAL) {
      for (int64_t i = 0; i < n; i += 4) {
        rng.Next(r);
        float z0, z1, z2, z3;
        random::BoxMuller(r[0], r[1], &z0, &z1);
        random::BoxMuller(r[2], r[3], &z2, &z3);
        float z[4] = {z0, z1, z2, z3};
        for (int j = 0; j < 4 && i + j < n; ++j) p[i + j] = z[j];
      }
    } else {  // truncated: redraw until |z| < 2
      for (int64_t i = 0; i < n; ++i) {
        float z = 3.0f;
        while (std::abs(z) >= 2.0f) {
          rng.Next(r);
          float z1;
          random::BoxMuller(r[0], r[1], &z, &z1);
        }
        p[i] = z;
      }
    }
  }

 private:
  uint64_t seed_;
  std::atomic<uint64_t> offset_{0};
};
REGISTER_KERNEL_BUILDER(Name("RandomUniform").Device(DEVICE_CPU), RandomOp<Dist::UNIFORM>);
REGISTER_KERNEL_BUILDER(Name("RandomStandardNormal").Device(DEVICE_CPU), RandomOp<Dist::NORMAL>);
REGISTER_KERNEL_BUILDER(Name("TruncatedNormal").Device(DEVICE_CPU), RandomOp<Dist::TRUNCATED>);

class RandomUniformIntOp : public OpKernel {
 public:
  explicit RandomUniformIntOp(OpKernelConstruction* ctx) : OpKernel(ctx) {
    int64_t seed = 0, seed2 = 0;
    ctx->GetAttr("seed", &seed);
    ctx->GetAttr("seed2", &seed2);
    seed_ = PickSeed(seed, seed2);
  }
  void Compute(OpKernelContext* ctx) override {
    auto dims = IntVector(ctx->input(0));
    int64_t lo = IntVector(ctx->input(1))[0];
    int64_t hi = IntVector(ctx->input(2))[0];
    Tensor* out = ctx->allocate_output(0, TensorShape(dims));
    int64_t n = out->NumElements();
    uint64_t offset = offset_.fetch_add((n + 3) / 4 + 4);
    random::Philox4x32 rng(seed_, offset);
    uint32_t r[4];
    uint64_t range = (uint64_t)(hi - lo);
    for (int64_t i = 0; i < n; i += 4) {
      rng.Next(r);
      for (int j = 0; j < 4 && i + j < n; ++j) {
        int64_t v = lo + (int64_t)(r[j] % range);
        if (out->dtype() == DT_INT32) out->flat<int32_t>()[i + j] = (int32_t)v;
        else out->flat<int64_t>()[i + j] = v;
      }
    }
  }

 private:
  uint64_t seed_;
  std::atomic<uint64_t> offset_{0};
};
REGISTER_KERNEL_BUILDER(Name("RandomUniformInt").Device(DEVICE_CPU), RandomUniformIntOp);

}  // namespace stf
