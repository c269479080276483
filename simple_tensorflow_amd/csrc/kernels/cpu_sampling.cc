// Candidate samplers + accidental-hit detection (capability analogs of
// reference core/kernels/candidate_sampler_ops.cc + range_sampler.cc):
// uniform / log-uniform (Zipfian) / learned-unigram samplers with exact
// expected-count math, and ComputeAccidentalHits for sampled-softmax
// correction.
#include <cmath>
#include <random>
#include <set>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

// Expected count of `id` after `trials` draws with replacement given
// per-draw probability p: 1 - (1-p)^trials (== p*trials for unique=false in
// the reference's convention; the reference uses this same formula for
// unique sampling and p*trials otherwise).
float ExpectedCount(double p, int64_t trials, bool unique) {
  if (unique) return (float)(-std::expm1(trials * std::log1p(-p)));
  return (float)(p * trials);
}

class BaseSamplerOp : public OpKernel {
 public:
  // mode: 0 uniform, 1 log-uniform, 2 learned-unigram (uniform prior — no
  // Update() path feeds it in-graph, matching reference defaults at init).
  BaseSamplerOp(OpKernelConstruction* c, int mode) : OpKernel(c), mode_(mode) {
    c->GetAttr("num_true", &num_true_);
    c->GetAttr("num_sampled", &num_sampled_);
    c->GetAttr("unique", &unique_);
    c->GetAttr("range_max", &range_max_);
    int64_t seed = 0, seed2 = 0;
    c->GetAttr("seed", &seed);
    c->GetAttr("seed2", &seed2);
    rng_.seed(seed || seed2 ? (uint64_t)(seed * 0x9E3779B97F4A7C15ull + seed2)
                            : std::random_device{}());
  }

  double Prob(int64_t id) const {
    if (mode_ == 1) {
      // log-uniform: P(id) = log((id+2)/(id+1)) / log(range_max+1)
      return (std::log((double)(id + 2)) - std::log((double)(id + 1))) /
             std::log((double)(range_max_ + 1));
    }
    return 1.0 / (double)range_max_;
  }

  int64_t SampleOne() {
    if (mode_ == 1) {
      // inverse transform for the Zipfian CDF
      double u = dist_(rng_);
      int64_t v = (int64_t)(std::exp(u * std::log((double)(range_max_ + 1)))) - 1;
      return std::min<int64_t>(std::max<int64_t>(v, 0), range_max_ - 1);
    }
    return (int64_t)(dist_(rng_) * range_max_) % range_max_;
  }

  void Compute(OpKernelContext* ctx) override {
    const Tensor& true_classes = ctx->input(0);
    int64_t batch = true_classes.shape().dims() > 0
                        ? true_classes.shape().dim_size(0)
                        : 1;
    Tensor* sampled = ctx->allocate_output(0, TensorShape({num_sampled_}));
    Tensor* true_ec =
        ctx->allocate_output(1, TensorShape({batch, num_true_}));
    Tensor* sampled_ec = ctx->allocate_output(2, TensorShape({num_sampled_}));
    std::vector<int64_t> picks;
    int64_t trials = 0;
    if (unique_) {
      std::set<int64_t> seen;
      // rejection-sample until num_sampled unique ids; count trials for the
      // expected-count formula
      while ((int64_t)picks.size() < num_sampled_) {
        ++trials;
        int64_t id = SampleOne();
        if (seen.insert(id).second) picks.push_back(id);
        if (trials > num_sampled_ * 1000 + 10000) {
          ctx->SetStatus(errors::InvalidArgument(
              "sampler: cannot draw ", num_sampled_, " unique from range ",
              range_max_));
          return;
        }
      }
    } else {
      trials = num_sampled_;
      for (int64_t i = 0; i < num_sampled_; ++i) picks.push_back(SampleOne());
    }
    for (int64_t i = 0; i < num_sampled_; ++i) {
      sampled->flat<int64_t>()[i] = picks[i];
      sampled_ec->flat<float>()[i] =
          ExpectedCount(Prob(picks[i]), trials, unique_);
    }
    const int64_t* tc = true_classes.flat<int64_t>();
    for (int64_t i = 0; i < batch * num_true_; ++i)
      true_ec->flat<float>()[i] = ExpectedCount(Prob(tc[i]), trials, unique_);
  }

 private:
  int mode_;
  int64_t num_true_ = 1, num_sampled_ = 1, range_max_ = 1;
  bool unique_ = true;
  std::mt19937_64 rng_;
  std::uniform_real_distribution<double> dist_{0.0, 1.0};
};

#define REG_SAMPLER(NAME, MODE)                                          \
  class NAME##Op : public BaseSamplerOp {                                \
   public:                                                               \
    explicit NAME##Op(OpKernelConstruction* c) : BaseSamplerOp(c, MODE) {} \
  };                                                                     \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_CPU), NAME##Op);
REG_SAMPLER(UniformCandidateSampler, 0)
REG_SAMPLER(LogUniformCandidateSampler, 1)
REG_SAMPLER(LearnedUnigramCandidateSampler, 2)
#undef REG_SAMPLER

class ComputeAccidentalHitsOp : public OpKernel {
 public:
  explicit ComputeAccidentalHitsOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("num_true", &num_true_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& true_classes = ctx->input(0);
    const Tensor& sampled = ctx->input(1);
    int64_t batch = true_classes.shape().dim_size(0);
    int64_t ns = sampled.NumElements();
    std::vector<int32_t> idx;
    std::vector<int64_t> ids;
    const int64_t* tc = true_classes.flat<int64_t>();
    const int64_t* sp = sampled.flat<int64_t>();
    for (int64_t b = 0; b < batch; ++b)
      for (int64_t t = 0; t < num_true_; ++t)
        for (int64_t s = 0; s < ns; ++s)
          if (tc[b * num_true_ + t] == sp[s]) {
            idx.push_back((int32_t)b);
            ids.push_back(s);
          }
    int64_t n = (int64_t)idx.size();
    Tensor* oi = ctx->allocate_output(0, TensorShape({n}));
    Tensor* od = ctx->allocate_output(1, TensorShape({n}));
    Tensor* ow = ctx->allocate_output(2, TensorShape({n}));
    for (int64_t i = 0; i < n; ++i) {
      oi->flat<int32_t>()[i] = idx[i];
      od->flat<int64_t>()[i] = ids[i];
      ow->flat<float>()[i] = -3.4e38f;  // -FLT_MAX: mask the logit
    }
  }

 private:
  int64_t num_true_ = 1;
};
REGISTER_KERNEL_BUILDER(Name("ComputeAccidentalHits").Device(DEVICE_CPU),
                        ComputeAccidentalHitsOp);

}  // namespace
}  // namespace stf
