// CTC loss + greedy decoder (capability analogs of the reference's
// core/kernels/ctc_loss_op.cc and ctc_decoder_ops.cc, which wrap
// util/ctc/*): log-space forward-backward over the blank-extended label
// sequence, gradient w.r.t. the raw (pre-softmax) logits. Blank label is
// num_classes - 1 (TF convention). Inputs are time-major
// [max_time, batch, num_classes]; labels arrive as a sparse [batch, time]
// matrix (indices/values).
#include <algorithm>
#include <cmath>
#include <limits>
#include <map>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

constexpr double kLogZero = -std::numeric_limits<double>::infinity();

double LogAdd(double a, double b) {
  if (a == kLogZero) return b;
  if (b == kLogZero) return a;
  double mx = std::max(a, b);
  return mx + std::log1p(std::exp(std::min(a, b) - mx));
}

class CTCLossOp : public OpKernel {
 public:
  explicit CTCLossOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("preprocess_collapse_repeated", &collapse_);
    c->GetAttr("ctc_merge_repeated", &merge_);
  }

  void Compute(OpKernelContext* ctx) override {
    const Tensor& inputs = ctx->input(0);   // [T, B, C]
    const Tensor& lidx = ctx->input(1);     // [nnz, 2]
    const Tensor& lval = ctx->input(2);     // [nnz]
    const Tensor& seq_len = ctx->input(3);  // [B]
    if (inputs.shape().dims() != 3) {
      ctx->SetStatus(errors::InvalidArgument("CTCLoss: inputs must be 3-D"));
      return;
    }
    int64_t T = inputs.shape().dim_size(0);
    int64_t B = inputs.shape().dim_size(1);
    int64_t C = inputs.shape().dim_size(2);
    int blank = (int)C - 1;
    Tensor* loss_t = ctx->allocate_output(0, TensorShape({B}));
    Tensor* grad_t = ctx->allocate_output(1, inputs.shape());
    float* loss = loss_t->flat<float>();
    float* grad = grad_t->flat<float>();
    std::fill(grad, grad + inputs.NumElements(), 0.f);

    // gather labels per batch
    std::vector<std::vector<int>> labels(B);
    const int64_t* ip = lidx.flat<int64_t>();
    const int32_t* vp = lval.flat<int32_t>();
    int64_t nnz = lval.NumElements();
    std::vector<std::vector<std::pair<int64_t, int>>> per_b(B);
    for (int64_t k = 0; k < nnz; ++k) {
      int64_t b = ip[k * 2], t = ip[k * 2 + 1];
      if (b < 0 || b >= B) continue;
      per_b[b].push_back({t, vp[k]});
    }
    for (int64_t b = 0; b < B; ++b) {
      std::sort(per_b[b].begin(), per_b[b].end());
      for (auto& kv : per_b[b]) {
        if (collapse_ && !labels[b].empty() && labels[b].back() == kv.second)
          continue;
        labels[b].push_back(kv.second);
      }
    }

    const float* x = inputs.flat<float>();
    std::vector<double> logp((size_t)T * C);  // log softmax for one batch
    for (int64_t b = 0; b < B; ++b) {
      int64_t Tb = seq_len.flat<int32_t>()[b];
      if (Tb > T) Tb = T;
      const std::vector<int>& l = labels[b];
      int64_t L = (int64_t)l.size();
      int64_t S = 2 * L + 1;  // blank-extended
      if (Tb < L) {
        ctx->SetStatus(errors::InvalidArgument(
            "CTCLoss: not enough time steps (", Tb, ") for label length ", L,
            " at batch ", b));
        return;
      }
      // log-softmax per frame
      for (int64_t t = 0; t < Tb; ++t) {
        const float* row = x + (t * B + b) * C;
        double mx = row[0];
        for (int64_t c = 1; c < C; ++c) mx = std::max(mx, (double)row[c]);
        double sum = 0;
        for (int64_t c = 0; c < C; ++c) sum += std::exp((double)row[c] - mx);
        double lse = mx + std::log(sum);
        for (int64_t c = 0; c < C; ++c) logp[t * C + c] = (double)row[c] - lse;
      }
      auto lab = [&](int64_t s) -> int {
        return (s & 1) ? l[s >> 1] : blank;
      };
      // forward
      std::vector<double> alpha((size_t)Tb * S, kLogZero);
      alpha[0] = logp[blank];
      if (S > 1) alpha[1] = logp[lab(1)];
      for (int64_t t = 1; t < Tb; ++t) {
        for (int64_t s = 0; s < S; ++s) {
          double a = alpha[(t - 1) * S + s];
          if (s > 0) a = LogAdd(a, alpha[(t - 1) * S + s - 1]);
          if (s > 1 && lab(s) != blank &&
              (!merge_ || lab(s) != lab(s - 2)))
            a = LogAdd(a, alpha[(t - 1) * S + s - 2]);
          alpha[t * S + s] =
              a == kLogZero ? kLogZero : a + logp[t * C + lab(s)];
        }
      }
      double ll = alpha[(Tb - 1) * S + S - 1];
      if (S > 1) ll = LogAdd(ll, alpha[(Tb - 1) * S + S - 2]);
      loss[b] = (float)(-ll);
      if (ll == kLogZero) continue;  // no valid path; zero grad
      // backward
      std::vector<double> beta((size_t)Tb * S, kLogZero);
      beta[(Tb - 1) * S + S - 1] = 0.0;
      if (S > 1) beta[(Tb - 1) * S + S - 2] = 0.0;
      for (int64_t t = Tb - 2; t >= 0; --t) {
        for (int64_t s = 0; s < S; ++s) {
          double v = beta[(t + 1) * S + s] + logp[(t + 1) * C + lab(s)];
          if (s + 1 < S)
            v = LogAdd(v, beta[(t + 1) * S + s + 1] +
                              logp[(t + 1) * C + lab(s + 1)]);
          if (s + 2 < S && lab(s + 2) != blank &&
              (!merge_ || lab(s + 2) != lab(s)))
            v = LogAdd(v, beta[(t + 1) * S + s + 2] +
                              logp[(t + 1) * C + lab(s + 2)]);
          beta[t * S + s] = v;
        }
      }
      // gradient wrt logits: softmax(t,c) - sum_{s: lab(s)=c} gamma(t,s)
      for (int64_t t = 0; t < Tb; ++t) {
        std::vector<double> lab_sum(C, kLogZero);
        for (int64_t s = 0; s < S; ++s) {
          double g = alpha[t * S + s] + beta[t * S + s];
          if (g == kLogZero) continue;
          int c = lab(s);
          lab_sum[c] = LogAdd(lab_sum[c], g);
        }
        float* grow = grad + (t * B + b) * C;
        for (int64_t c = 0; c < C; ++c) {
          double sm = std::exp(logp[t * C + c]);
          double target =
              lab_sum[c] == kLogZero ? 0.0 : std::exp(lab_sum[c] - ll);
          grow[c] = (float)(sm - target);
        }
      }
    }
  }

 private:
  bool collapse_ = false, merge_ = true;
};
REGISTER_KERNEL_BUILDER(Name("CTCLoss").Device(DEVICE_CPU), CTCLossOp);

class CTCGreedyDecoderOp : public OpKernel {
 public:
  explicit CTCGreedyDecoderOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("merge_repeated", &merge_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& inputs = ctx->input(0);  // [T, B, C]
    const Tensor& seq_len = ctx->input(1);
    int64_t T = inputs.shape().dim_size(0);
    int64_t B = inputs.shape().dim_size(1);
    int64_t C = inputs.shape().dim_size(2);
    int blank = (int)C - 1;
    const float* x = inputs.flat<float>();
    std::vector<std::vector<int64_t>> decoded(B);
    Tensor* logp_t = ctx->allocate_output(3, TensorShape({B, 1}));
    for (int64_t b = 0; b < B; ++b) {
      int64_t Tb = std::min<int64_t>(seq_len.flat<int32_t>()[b], T);
      double lp = 0.0;
      int prev = -1;
      for (int64_t t = 0; t < Tb; ++t) {
        const float* row = x + (t * B + b) * C;
        int arg = 0;
        for (int64_t c = 1; c < C; ++c)
          if (row[c] > row[arg]) arg = (int)c;
        // log prob of the greedy path (normalized)
        double mx = row[0];
        for (int64_t c = 1; c < C; ++c) mx = std::max(mx, (double)row[c]);
        double sum = 0;
        for (int64_t c = 0; c < C; ++c) sum += std::exp((double)row[c] - mx);
        lp += (double)row[arg] - (mx + std::log(sum));
        if (arg != blank && (!merge_ || arg != prev))
          decoded[b].push_back(arg);
        prev = arg;
      }
      logp_t->flat<float>()[b] = (float)(-lp);
    }
    int64_t total = 0, maxlen = 0;
    for (auto& d : decoded) {
      total += (int64_t)d.size();
      maxlen = std::max<int64_t>(maxlen, (int64_t)d.size());
    }
    Tensor* idx = ctx->allocate_output(0, TensorShape({total, 2}));
    Tensor* val = ctx->allocate_output(1, TensorShape({total}));
    Tensor* shp = ctx->allocate_output(2, TensorShape({2}));
    int64_t k = 0;
    for (int64_t b = 0; b < B; ++b)
      for (size_t t = 0; t < decoded[b].size(); ++t, ++k) {
        idx->flat<int64_t>()[k * 2] = b;
        idx->flat<int64_t>()[k * 2 + 1] = (int64_t)t;
        val->flat<int64_t>()[k] = decoded[b][t];
      }
    shp->flat<int64_t>()[0] = B;
    shp->flat<int64_t>()[1] = maxlen;
  }

 private:
  bool merge_ = false;
};
REGISTER_KERNEL_BUILDER(Name("CTCGreedyDecoder").Device(DEVICE_CPU),
                        CTCGreedyDecoderOp);

}  // namespace
}  // namespace stf
