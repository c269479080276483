// Image kernels: resize bilinear / nearest (+grads) — capability analog of
// reference core/kernels/resize_bilinear_op.cc, resize_nearest_neighbor_op.cc
// (input-pipeline ops; they run on the host like the reference's CPU path).
#include <algorithm>
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

}  // namespace
}  // namespace stf
