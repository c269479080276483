// Spectral ops: FFT/IFFT (1D/2D/3D, complex64), RFFT/IRFFT (+2D) and the
// complex construction/accessor ops.
//
// Capability analog of the reference's core/kernels/fft_ops.cc (which
// delegates to Eigen TensorFFT / cuFFT): self-contained iterative radix-2
// Cooley-Tukey for power-of-two lengths and Bluestein's chirp-z transform
// for everything else, accumulating in double precision.
#include <cmath>
#include <complex>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

using cd = std::complex<double>;
using cf = std::complex<float>;
constexpr double kPi = 3.141592653589793238462643383279502884;

bool IsPow2(int64_t n) { return n > 0 && (n & (n - 1)) == 0; }

// In-place iterative radix-2 FFT; n must be a power of two.
void FftPow2(cd* a, int64_t n, bool inverse) {
  // bit reversal
  for (int64_t i = 1, j = 0; i < n; ++i) {
    int64_t bit = n >> 1;
    for (; j & bit; bit >>= 1) j ^= bit;
    j ^= bit;
    if (i < j) std::swap(a[i], a[j]);
  }
  for (int64_t len = 2; len <= n; len <<= 1) {
    double ang = 2 * kPi / len * (inverse ? 1 : -1);
    cd wl(std::cos(ang), std::sin(ang));
    for (int64_t i = 0; i < n; i += len) {
      cd w(1);
      for (int64_t j = 0; j < len / 2; ++j) {
        cd u = a[i + j], v = a[i + j + len / 2] * w;
        a[i + j] = u + v;
        a[i + j + len / 2] = u - v;
        w *= wl;
      }
    }
  }
}

// General-length FFT via Bluestein: x_k -> chirp multiply, convolve with
// the conjugate chirp through a pow2 FFT of size >= 2n-1.
void FftAny(cd* a, int64_t n, bool inverse) {
  if (n <= 1) return;
  if (IsPow2(n)) {
    FftPow2(a, n, inverse);
    return;
  }
  int64_t m = 1;
  while (m < 2 * n - 1) m <<= 1;
  std::vector<cd> w(n), fa(m, cd(0)), fb(m, cd(0));
  double sign = inverse ? 1.0 : -1.0;
  for (int64_t k = 0; k < n; ++k) {
    // w_k = exp(sign * i*pi*k^2/n); k^2 mod 2n keeps the angle exact
    int64_t k2 = (int64_t)((__int128)k * k % (2 * n));
    double ang = sign * kPi * (double)k2 / (double)n;
    w[k] = cd(std::cos(ang), std::sin(ang));
    fa[k] = a[k] * w[k];
  }
  fb[0] = cd(1);
  for (int64_t k = 1; k < n; ++k)
    fb[k] = fb[m - k] = std::conj(w[k]);
  FftPow2(fa.data(), m, false);
  FftPow2(fb.data(), m, false);
  for (int64_t i = 0; i < m; ++i) fa[i] *= fb[i];
  FftPow2(fa.data(), m, true);
  for (int64_t k = 0; k < n; ++k) a[k] = fa[k] / (double)m * w[k];
}

// FFT over the trailing `rank` dims of a [outer, d0, d1, ...] block.
// dims = sizes of the trailing dims. Data in row-major complex.
void FftNd(cd* data, const std::vector<int64_t>& dims, bool inverse,
           bool normalize) {
  int rank = (int)dims.size();
  int64_t total = 1;
  for (auto d : dims) total *= d;
  // transform along each axis
  for (int ax = rank - 1; ax >= 0; --ax) {
    int64_t len = dims[ax];
    int64_t stride = 1;
    for (int i = ax + 1; i < rank; ++i) stride *= dims[i];
    int64_t nvec = total / len;
    std::vector<cd> buf(len);
    for (int64_t v = 0; v < nvec; ++v) {
      // compute the base offset of this 1-D line
      int64_t rem = v, base = 0, mul = 1;
      // lines: iterate all dims except ax
      int64_t coords[8] = {0};
      for (int i = rank - 1; i >= 0; --i) {
        if (i == ax) continue;
        coords[i] = rem % dims[i];
        rem /= dims[i];
      }
      (void)mul;
      base = 0;
      int64_t acc = 1;
      for (int i = rank - 1; i >= 0; --i) {
        base += coords[i] * acc;
        acc *= dims[i];
      }
      for (int64_t k = 0; k < len; ++k) buf[k] = data[base + k * stride];
      FftAny(buf.data(), len, inverse);
      for (int64_t k = 0; k < len; ++k) data[base + k * stride] = buf[k];
    }
  }
  if (inverse && normalize) {
    for (int64_t i = 0; i < total; ++i) data[i] /= (double)total;
  }
}

// ---- complex-to-complex ops ----

class FftOp : public OpKernel {
 public:
  FftOp(OpKernelConstruction* c, int rank, bool inverse)
      : OpKernel(c), rank_(rank), inverse_(inverse) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    if (in.shape().dims() < rank_) {
      ctx->SetStatus(errors::InvalidArgument("FFT rank > input rank"));
      return;
    }
    Tensor* out = ctx->allocate_output(0, in.shape());
    std::vector<int64_t> dims(rank_);
    int64_t inner = 1;
    for (int i = 0; i < rank_; ++i) {
      dims[i] = in.shape().dim_size(in.shape().dims() - rank_ + i);
      inner *= dims[i];
    }
    int64_t outer = in.NumElements() / (inner ? inner : 1);
    const cf* src = in.flat<cf>();
    cf* dst = out->flat<cf>();
    std::vector<cd> work(inner);
    for (int64_t b = 0; b < outer; ++b) {
      for (int64_t i = 0; i < inner; ++i) work[i] = cd(src[b * inner + i]);
      FftNd(work.data(), dims, inverse_, true);
      for (int64_t i = 0; i < inner; ++i) dst[b * inner + i] = cf(work[i]);
    }
  }

 private:
  int rank_;
  bool inverse_;
};

#define REG_FFT(NAME, RANK, INV)                                      \
  class NAME##Op : public FftOp {                                     \
   public:                                                            \
    explicit NAME##Op(OpKernelConstruction* c) : FftOp(c, RANK, INV) {} \
  };                                                                  \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_CPU), NAME##Op);
REG_FFT(FFT, 1, false)
REG_FFT(IFFT, 1, true)
REG_FFT(FFT2D, 2, false)
REG_FFT(IFFT2D, 2, true)
REG_FFT(FFT3D, 3, false)
REG_FFT(IFFT3D, 3, true)
#undef REG_FFT

// ---- real transforms ----

// RFFT: real [..., n] (padded/clipped to fft_length) -> complex
// [..., fft_length/2 + 1].
class RfftOp : public OpKernel {
 public:
  RfftOp(OpKernelConstruction* c, int rank) : OpKernel(c), rank_(rank) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& flt = ctx->input(1);
    if (flt.NumElements() != rank_) {
      ctx->SetStatus(errors::InvalidArgument("fft_length must have rank elements"));
      return;
    }
    std::vector<int64_t> fft(rank_);
    for (int i = 0; i < rank_; ++i) fft[i] = flt.flat<int32_t>()[i];
    int dims = in.shape().dims();
    std::vector<int64_t> in_dims(rank_);
    int64_t inner_in = 1;
    for (int i = 0; i < rank_; ++i) {
      in_dims[i] = in.shape().dim_size(dims - rank_ + i);
      inner_in *= in_dims[i];
    }
    std::vector<int64_t> out_dims = fft;
    out_dims[rank_ - 1] = fft[rank_ - 1] / 2 + 1;
    TensorShape os;
    for (int i = 0; i < dims - rank_; ++i) os.AddDim(in.shape().dim_size(i));
    for (int i = 0; i < rank_; ++i) os.AddDim(out_dims[i]);
    Tensor* out = ctx->allocate_output(0, os);
    int64_t inner_fft = 1, inner_out = 1;
    for (int i = 0; i < rank_; ++i) {
      inner_fft *= fft[i];
      inner_out *= out_dims[i];
    }
    int64_t outer = in.NumElements() / (inner_in ? inner_in : 1);
    const float* src = in.flat<float>();
    cf* dst = out->flat<cf>();
    std::vector<cd> work(inner_fft);
    for (int64_t b = 0; b < outer; ++b) {
      // gather with pad/clip per dim (rank <= 2 here)
      std::fill(work.begin(), work.end(), cd(0));
      if (rank_ == 1) {
        int64_t n = std::min(in_dims[0], fft[0]);
        for (int64_t i = 0; i < n; ++i)
          work[i] = cd(src[b * inner_in + i], 0.0);
      } else {
        int64_t r = std::min(in_dims[0], fft[0]);
        int64_t c = std::min(in_dims[1], fft[1]);
        for (int64_t i = 0; i < r; ++i)
          for (int64_t j = 0; j < c; ++j)
            work[i * fft[1] + j] =
                cd(src[b * inner_in + i * in_dims[1] + j], 0.0);
      }
      FftNd(work.data(), fft, false, true);
      // keep the non-redundant half along the last dim
      int64_t keep = out_dims[rank_ - 1];
      int64_t rows = inner_fft / fft[rank_ - 1];
      for (int64_t r = 0; r < rows; ++r)
        for (int64_t k = 0; k < keep; ++k)
          dst[b * inner_out + r * keep + k] =
              cf(work[r * fft[rank_ - 1] + k]);
    }
  }

 private:
  int rank_;
};

// IRFFT: complex [..., n/2+1] -> real [..., fft_length], hermitian
// reconstruction of the dropped half.
class IrfftOp : public OpKernel {
 public:
  IrfftOp(OpKernelConstruction* c, int rank) : OpKernel(c), rank_(rank) {}
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& flt = ctx->input(1);
    std::vector<int64_t> fft(rank_);
    for (int i = 0; i < rank_; ++i) fft[i] = flt.flat<int32_t>()[i];
    int dims = in.shape().dims();
    std::vector<int64_t> in_dims(rank_);
    int64_t inner_in = 1;
    for (int i = 0; i < rank_; ++i) {
      in_dims[i] = in.shape().dim_size(dims - rank_ + i);
      inner_in *= in_dims[i];
    }
    TensorShape os;
    for (int i = 0; i < dims - rank_; ++i) os.AddDim(in.shape().dim_size(i));
    for (int i = 0; i < rank_; ++i) os.AddDim(fft[i]);
    Tensor* out = ctx->allocate_output(0, os);
    int64_t inner_fft = 1;
    for (int i = 0; i < rank_; ++i) inner_fft *= fft[i];
    int64_t outer = in.NumElements() / (inner_in ? inner_in : 1);
    const cf* src = in.flat<cf>();
    float* dst = out->flat<float>();
    std::vector<cd> work(inner_fft);
    int64_t half = fft[rank_ - 1] / 2 + 1;
    for (int64_t b = 0; b < outer; ++b) {
      std::fill(work.begin(), work.end(), cd(0));
      int64_t rows_in = inner_in / in_dims[rank_ - 1];
      int64_t rows_fft = inner_fft / fft[rank_ - 1];
      // place the stored half (pad/clip rows for 2D)
      int64_t rcount = std::min(rows_in, rows_fft);
      int64_t ccount = std::min(in_dims[rank_ - 1], half);
      for (int64_t r = 0; r < rcount; ++r)
        for (int64_t k = 0; k < ccount; ++k)
          work[r * fft[rank_ - 1] + k] =
              cd(src[b * inner_in + r * in_dims[rank_ - 1] + k]);
      // hermitian completion along the last axis:
      // X[..., n-k] = conj(X[..., k])  (with row index negated for 2D)
      int64_t n = fft[rank_ - 1];
      if (rank_ == 1) {
        for (int64_t k = 1; k < n - half + 1; ++k)
          work[n - k] = std::conj(work[k]);
      } else {
        int64_t nr = fft[0];
        for (int64_t r = 0; r < nr; ++r)
          for (int64_t k = 1; k < n - half + 1; ++k)
            work[r * n + (n - k)] =
                std::conj(work[((nr - r) % nr) * n + k]);
      }
      FftNd(work.data(), fft, true, true);
      for (int64_t i = 0; i < inner_fft; ++i)
        dst[b * inner_fft + i] = (float)work[i].real();
    }
  }

 private:
  int rank_;
};

#define REG_RFFT(NAME, OP, RANK)                                        \
  class NAME##Op : public OP {                                          \
   public:                                                              \
    explicit NAME##Op(OpKernelConstruction* c) : OP(c, RANK) {}         \
  };                                                                    \
  REGISTER_KERNEL_BUILDER(Name(#NAME).Device(DEVICE_CPU), NAME##Op);
REG_RFFT(RFFT, RfftOp, 1)
REG_RFFT(IRFFT, IrfftOp, 1)
REG_RFFT(RFFT2D, RfftOp, 2)
REG_RFFT(IRFFT2D, IrfftOp, 2)
#undef REG_RFFT

// ---- complex construction / accessors ----

class ComplexOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& re = ctx->input(0);
    const Tensor& im = ctx->input(1);
    Tensor* out = ctx->allocate_output(0, re.shape());
    cf* o = out->flat<cf>();
    const float* r = re.flat<float>();
    const float* i = im.flat<float>();
    for (int64_t k = 0; k < re.NumElements(); ++k) o[k] = cf(r[k], i[k]);
  }
};
REGISTER_KERNEL_BUILDER(Name("Complex").Device(DEVICE_CPU), ComplexOp);

template <int MODE>  // 0 real, 1 imag, 2 abs
class ComplexPartOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    const cf* s = in.flat<cf>();
    float* o = out->flat<float>();
    for (int64_t k = 0; k < in.NumElements(); ++k)
      o[k] = MODE == 0 ? s[k].real()
                       : (MODE == 1 ? s[k].imag() : std::abs(s[k]));
  }
};
REGISTER_KERNEL_BUILDER(Name("Real").Device(DEVICE_CPU), ComplexPartOp<0>);
REGISTER_KERNEL_BUILDER(Name("Imag").Device(DEVICE_CPU), ComplexPartOp<1>);
REGISTER_KERNEL_BUILDER(Name("ComplexAbs").Device(DEVICE_CPU), ComplexPartOp<2>);

class ConjOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    const cf* s = in.flat<cf>();
    cf* o = out->flat<cf>();
    for (int64_t k = 0; k < in.NumElements(); ++k) o[k] = std::conj(s[k]);
  }
};
REGISTER_KERNEL_BUILDER(Name("Conj").Device(DEVICE_CPU), ConjOp);

}  // namespace
}  // namespace stf
