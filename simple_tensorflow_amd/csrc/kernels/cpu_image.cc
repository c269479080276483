// Image kernels: resize bilinear / nearest (+grads) — capability analog of
// reference core/kernels/resize_bilinear_op.cc, resize_nearest_neighbor_op.cc
// (input-pipeline ops; they run on the host like the reference's CPU path).
#include <algorithm>
#include <numeric>
#include <random>
#include <cmath>
#include <cstring>

#include "kernels/kernel_util.h"

namespace stf {
namespace {

struct ResizeArgs {
  int64_t batch, in_h, in_w, channels, out_h, out_w;
  float h_scale, w_scale;
};

Status GetResizeArgs(OpKernelContext* ctx, bool align_corners,
                     ResizeArgs* a) {
  const Tensor& images = ctx->input(0);
  if (images.dims() != 4)
    return errors::InvalidArgument("resize expects 4-D NHWC input");
  const Tensor& size = ctx->input(1);
  if (size.NumElements() != 2)
    return errors::InvalidArgument("size must have 2 elements");
  a->batch = images.dim_size(0);
  a->in_h = images.dim_size(1);
  a->in_w = images.dim_size(2);
  a->channels = images.dim_size(3);
  a->out_h = size.flat<int32_t>()[0];
  a->out_w = size.flat<int32_t>()[1];
  auto scale = [&](int64_t in, int64_t out) {
    if (align_corners && out > 1)
      return (float)(in - 1) / (float)(out - 1);
    return (float)in / (float)out;
  };
  a->h_scale = scale(a->in_h, a->out_h);
  a->w_scale = scale(a->in_w, a->out_w);
  return Status::OK();
}

class ResizeBilinearOp : public OpKernel {
 public:
  explicit ResizeBilinearOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("align_corners", &align_);
  }
  void Compute(OpKernelContext* ctx) override {
    ResizeArgs a;
    OP_REQUIRES_OK(ctx, GetResizeArgs(ctx, align_, &a));
    const float* in = ctx->input(0).flat<float>();
    Tensor* out = ctx->allocate_output(
        0, TensorShape({a.batch, a.out_h, a.out_w, a.channels}));
    float* o = out->flat<float>();
    for (int64_t b = 0; b < a.batch; ++b) {
      const float* base = in + b * a.in_h * a.in_w * a.channels;
      for (int64_t y = 0; y < a.out_h; ++y) {
        float fy = y * a.h_scale;
        int64_t y0 = (int64_t)fy;
        int64_t y1 = std::min(y0 + 1, a.in_h - 1);
        float ly = fy - y0;
        for (int64_t x = 0; x < a.out_w; ++x) {
          float fx = x * a.w_scale;
          int64_t x0 = (int64_t)fx;
          int64_t x1 = std::min(x0 + 1, a.in_w - 1);
          float lx = fx - x0;
          for (int64_t c = 0; c < a.channels; ++c) {
            float tl = base[(y0 * a.in_w + x0) * a.channels + c];
            float tr = base[(y0 * a.in_w + x1) * a.channels + c];
            float bl = base[(y1 * a.in_w + x0) * a.channels + c];
            float br = base[(y1 * a.in_w + x1) * a.channels + c];
            float top = tl + (tr - tl) * lx;
            float bot = bl + (br - bl) * lx;
            *o++ = top + (bot - top) * ly;
          }
        }
      }
    }
  }

 private:
  bool align_ = false;
};

class ResizeBilinearGradOp : public OpKernel {
 public:
  explicit ResizeBilinearGradOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("align_corners", &align_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& grads = ctx->input(0);   // [b, out_h, out_w, c]
    const Tensor& orig = ctx->input(1);    // original input (for shape)
    int64_t batch = orig.dim_size(0), in_h = orig.dim_size(1),
            in_w = orig.dim_size(2), channels = orig.dim_size(3);
    int64_t out_h = grads.dim_size(1), out_w = grads.dim_size(2);
    auto scale = [&](int64_t in, int64_t out) {
      if (align_ && out > 1) return (float)(in - 1) / (float)(out - 1);
      return (float)in / (float)out;
    };
    float h_scale = scale(in_h, out_h), w_scale = scale(in_w, out_w);
    Tensor* out = ctx->allocate_output(0, orig.shape());
    float* o = out->flat<float>();
    std::memset(o, 0, out->TotalBytes());
    const float* g = grads.flat<float>();
    for (int64_t b = 0; b < batch; ++b) {
      float* base = o + b * in_h * in_w * channels;
      for (int64_t y = 0; y < out_h; ++y) {
        float fy = y * h_scale;
        int64_t y0 = (int64_t)fy;
        int64_t y1 = std::min(y0 + 1, in_h - 1);
        float ly = fy - y0;
        for (int64_t x = 0; x < out_w; ++x) {
          float fx = x * w_scale;
          int64_t x0 = (int64_t)fx;
          int64_t x1 = std::min(x0 + 1, in_w - 1);
          float lx = fx - x0;
          for (int64_t c = 0; c < channels; ++c) {
            float gv = *g++;
            base[(y0 * in_w + x0) * channels + c] +=
                gv * (1 - ly) * (1 - lx);
            base[(y0 * in_w + x1) * channels + c] += gv * (1 - ly) * lx;
            base[(y1 * in_w + x0) * channels + c] += gv * ly * (1 - lx);
            base[(y1 * in_w + x1) * channels + c] += gv * ly * lx;
          }
        }
      }
    }
  }

 private:
  bool align_ = false;
};

class ResizeNearestNeighborOp : public OpKernel {
 public:
  explicit ResizeNearestNeighborOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("align_corners", &align_);
  }
  void Compute(OpKernelContext* ctx) override {
    ResizeArgs a;
    OP_REQUIRES_OK(ctx, GetResizeArgs(ctx, align_, &a));
    const float* in = ctx->input(0).flat<float>();
    Tensor* out = ctx->allocate_output(
        0, TensorShape({a.batch, a.out_h, a.out_w, a.channels}));
    float* o = out->flat<float>();
    for (int64_t b = 0; b < a.batch; ++b) {
      const float* base = in + b * a.in_h * a.in_w * a.channels;
      for (int64_t y = 0; y < a.out_h; ++y) {
        int64_t sy = std::min((int64_t)(align_ ? std::lround(y * a.h_scale)
                                               : std::floor(y * a.h_scale)),
                              a.in_h - 1);
        for (int64_t x = 0; x < a.out_w; ++x) {
          int64_t sx = std::min(
              (int64_t)(align_ ? std::lround(x * a.w_scale)
                               : std::floor(x * a.w_scale)),
              a.in_w - 1);
          std::memcpy(o, base + (sy * a.in_w + sx) * a.channels,
                      a.channels * sizeof(float));
          o += a.channels;
        }
      }
    }
  }

 private:
  bool align_ = false;
};

REGISTER_KERNEL_BUILDER(Name("ResizeBilinear").Device(DEVICE_CPU),
                        ResizeBilinearOp);
REGISTER_KERNEL_BUILDER(Name("ResizeBilinearGrad").Device(DEVICE_CPU),
                        ResizeBilinearGradOp);
REGISTER_KERNEL_BUILDER(Name("ResizeNearestNeighbor").Device(DEVICE_CPU),
                        ResizeNearestNeighborOp);


// ---- color space (reference core/kernels/colorspace_op.cc analog) ----
class RGBToHSVOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    int64_t n = in.NumElements() / 3;
    const float* p = in.flat<float>();
    float* o = out->flat<float>();
    for (int64_t i = 0; i < n; ++i) {
      float r = p[i * 3], g = p[i * 3 + 1], b = p[i * 3 + 2];
      float mx = std::max(r, std::max(g, b));
      float mn = std::min(r, std::min(g, b));
      float v = mx, d = mx - mn;
      float s = mx > 0 ? d / mx : 0.f;
      float h = 0.f;
      if (d > 0) {
        if (mx == r) h = (g - b) / d;
        else if (mx == g) h = 2.f + (b - r) / d;
        else h = 4.f + (r - g) / d;
        h /= 6.f;
        if (h < 0) h += 1.f;
      }
      o[i * 3] = h;
      o[i * 3 + 1] = s;
      o[i * 3 + 2] = v;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("RGBToHSV").Device(DEVICE_CPU), RGBToHSVOp);

class HSVToRGBOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    int64_t n = in.NumElements() / 3;
    const float* p = in.flat<float>();
    float* o = out->flat<float>();
    for (int64_t i = 0; i < n; ++i) {
      float h = p[i * 3], s = p[i * 3 + 1], v = p[i * 3 + 2];
      float c = v * s;
      float hh = h * 6.f;
      float x = c * (1.f - std::fabs(std::fmod(hh, 2.f) - 1.f));
      float r = 0, g = 0, b = 0;
      int seg = (int)hh % 6;
      switch (seg) {
        case 0: r = c; g = x; break;
        case 1: r = x; g = c; break;
        case 2: g = c; b = x; break;
        case 3: g = x; b = c; break;
        case 4: r = x; b = c; break;
        default: r = c; b = x; break;
      }
      float m = v - c;
      o[i * 3] = r + m;
      o[i * 3 + 1] = g + m;
      o[i * 3 + 2] = b + m;
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("HSVToRGB").Device(DEVICE_CPU), HSVToRGBOp);

// ---- contrast (reference adjust_contrast_op.cc v2 semantics) ----
class AdjustContrastV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    float factor = ctx->input(1).flat<float>()[0];
    Tensor* out = ctx->allocate_output(0, in.shape());
    int dims = in.shape().dims();
    int64_t c = in.shape().dim_size(dims - 1);
    int64_t hw = in.shape().dim_size(dims - 3) * in.shape().dim_size(dims - 2);
    int64_t batch = in.NumElements() / (hw * c);
    const float* p = in.flat<float>();
    float* o = out->flat<float>();
    std::vector<double> mean(c);
    for (int64_t b = 0; b < batch; ++b) {
      const float* img = p + b * hw * c;
      float* oimg = o + b * hw * c;
      std::fill(mean.begin(), mean.end(), 0.0);
      for (int64_t i = 0; i < hw; ++i)
        for (int64_t ch = 0; ch < c; ++ch) mean[ch] += img[i * c + ch];
      for (int64_t ch = 0; ch < c; ++ch) mean[ch] /= hw;
      for (int64_t i = 0; i < hw; ++i)
        for (int64_t ch = 0; ch < c; ++ch)
          oimg[i * c + ch] =
              (float)((img[i * c + ch] - mean[ch]) * factor + mean[ch]);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("AdjustContrastv2").Device(DEVICE_CPU),
                        AdjustContrastV2Op);

// ---- NMS (reference non_max_suppression_op.cc) ----
inline float BoxIoU(const float* a, const float* b) {
  float ay1 = std::min(a[0], a[2]), ay2 = std::max(a[0], a[2]);
  float ax1 = std::min(a[1], a[3]), ax2 = std::max(a[1], a[3]);
  float by1 = std::min(b[0], b[2]), by2 = std::max(b[0], b[2]);
  float bx1 = std::min(b[1], b[3]), bx2 = std::max(b[1], b[3]);
  float iy = std::max(0.f, std::min(ay2, by2) - std::max(ay1, by1));
  float ix = std::max(0.f, std::min(ax2, bx2) - std::max(ax1, bx1));
  float inter = iy * ix;
  float ua = (ay2 - ay1) * (ax2 - ax1) + (by2 - by1) * (bx2 - bx1) - inter;
  return ua > 0 ? inter / ua : 0.f;
}

class NonMaxSuppressionOp : public OpKernel {
 public:
  explicit NonMaxSuppressionOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("iou_threshold", &thresh_attr_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& boxes = ctx->input(0);
    const Tensor& scores = ctx->input(1);
    int64_t max_out = ctx->input(2).flat<int32_t>()[0];
    float thresh = thresh_attr_;
    if (ctx->num_inputs() > 3 && ctx->input(3).NumElements() == 1)
      thresh = ctx->input(3).flat<float>()[0];
    int64_t n = scores.NumElements();
    std::vector<int> order(n);
    std::iota(order.begin(), order.end(), 0);
    const float* sp = scores.flat<float>();
    std::sort(order.begin(), order.end(),
              [&](int a, int b) { return sp[a] > sp[b]; });
    const float* bp = boxes.flat<float>();
    std::vector<int32_t> keep;
    for (int i : order) {
      if ((int64_t)keep.size() >= max_out) break;
      bool ok = true;
      for (int k : keep)
        if (BoxIoU(bp + i * 4, bp + k * 4) > thresh) {
          ok = false;
          break;
        }
      if (ok) keep.push_back(i);
    }
    Tensor* out = ctx->allocate_output(0, TensorShape({(int64_t)keep.size()}));
    for (size_t i = 0; i < keep.size(); ++i)
      out->flat<int32_t>()[i] = keep[i];
  }

 private:
  float thresh_attr_ = 0.5f;
};
REGISTER_KERNEL_BUILDER(Name("NonMaxSuppression").Device(DEVICE_CPU),
                        NonMaxSuppressionOp);
REGISTER_KERNEL_BUILDER(Name("NonMaxSuppressionV2").Device(DEVICE_CPU),
                        NonMaxSuppressionOp);

// ---- sample_distorted_bounding_box (inception-style random crop;
// reference sample_distorted_bounding_box_op.cc) ----
class SampleDistortedBBoxOp : public OpKernel {
 public:
  explicit SampleDistortedBBoxOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("min_object_covered", &min_covered_);
    c->GetAttr("aspect_ratio_range", &ar_range_);
    c->GetAttr("area_range", &area_range_);
    c->GetAttr("max_attempts", &max_attempts_);
    int64_t seed = 0, seed2 = 0;
    c->GetAttr("seed", &seed);
    c->GetAttr("seed2", &seed2);
    rng_.seed(seed || seed2 ? (uint64_t)(seed * 7919 + seed2)
                            : std::random_device{}());
    if (ar_range_.size() < 2) ar_range_ = {0.75f, 1.33f};
    if (area_range_.size() < 2) area_range_ = {0.05f, 1.0f};
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& size_in = ctx->input(0);
    int64_t H = size_in.flat<int32_t>()[0];
    int64_t W = size_in.flat<int32_t>()[1];
    std::uniform_real_distribution<float> uni(0.f, 1.f);
    int64_t ch = H, cw = W, cy = 0, cx = 0;
    for (int attempt = 0; attempt < max_attempts_; ++attempt) {
      float area_frac = area_range_[0] +
                        uni(rng_) * (area_range_[1] - area_range_[0]);
      float ar = ar_range_[0] + uni(rng_) * (ar_range_[1] - ar_range_[0]);
      double target_area = (double)H * W * area_frac;
      int64_t w = (int64_t)std::lround(std::sqrt(target_area * ar));
      int64_t h = (int64_t)std::lround(std::sqrt(target_area / ar));
      if (w <= 0 || h <= 0 || w > W || h > H) continue;
      ch = h;
      cw = w;
      cy = (int64_t)(uni(rng_) * (H - h + 1));
      cx = (int64_t)(uni(rng_) * (W - w + 1));
      break;
    }
    Tensor* begin = ctx->allocate_output(0, TensorShape({3}));
    Tensor* size = ctx->allocate_output(1, TensorShape({3}));
    Tensor* bboxes = ctx->allocate_output(2, TensorShape({1, 1, 4}));
    begin->flat<int32_t>()[0] = (int32_t)cy;
    begin->flat<int32_t>()[1] = (int32_t)cx;
    begin->flat<int32_t>()[2] = 0;
    size->flat<int32_t>()[0] = (int32_t)ch;
    size->flat<int32_t>()[1] = (int32_t)cw;
    size->flat<int32_t>()[2] = -1;
    bboxes->flat<float>()[0] = (float)cy / H;
    bboxes->flat<float>()[1] = (float)cx / W;
    bboxes->flat<float>()[2] = (float)(cy + ch) / H;
    bboxes->flat<float>()[3] = (float)(cx + cw) / W;
  }

 private:
  float min_covered_ = 0.1f;
  std::vector<float> ar_range_, area_range_;
  int64_t max_attempts_ = 100;
  std::mt19937_64 rng_;
};
REGISTER_KERNEL_BUILDER(Name("SampleDistortedBoundingBox").Device(DEVICE_CPU),
                        SampleDistortedBBoxOp);

}  // namespace
}  // namespace stf
