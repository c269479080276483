// String op kernels (capability analogs of the reference's
// core/kernels/string_join_op.cc, string_split_op.cc, substr_op.cc,
// string_to_hash_bucket_op.cc, string_to_number_op.cc, reduce_join_op.cc,
// encode/decode_base64 in core/lib/strings). Hashing uses a MurmurHash64A-
// style 64-bit mix (public algorithm) rather than the reference's FarmHash.
#include <algorithm>
#include <cstring>
#include <string>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

uint64_t Hash64(const char* data, size_t n, uint64_t seed = 0xc3a5c85c97cb3127ull) {
  const uint64_t m = 0xc6a4a7935bd1e995ull;
  const int r = 47;
  uint64_t h = seed ^ (n * m);
  const unsigned char* p = (const unsigned char*)data;
  const unsigned char* end = p + (n & ~size_t(7));
  while (p != end) {
    uint64_t k;
    std::memcpy(&k, p, 8);
    p += 8;
    k *= m;
    k ^= k >> r;
    k *= m;
    h ^= k;
    h *= m;
  }
  switch (n & 7) {
    case 7: h ^= (uint64_t)p[6] << 48; [[fallthrough]];
    case 6: h ^= (uint64_t)p[5] << 40; [[fallthrough]];
    case 5: h ^= (uint64_t)p[4] << 32; [[fallthrough]];
    case 4: h ^= (uint64_t)p[3] << 24; [[fallthrough]];
    case 3: h ^= (uint64_t)p[2] << 16; [[fallthrough]];
    case 2: h ^= (uint64_t)p[1] << 8; [[fallthrough]];
    case 1: h ^= (uint64_t)p[0]; h *= m;
  }
  h ^= h >> r;
  h *= m;
  h ^= h >> r;
  return h;
}

class StringJoinOp : public OpKernel {
 public:
  explicit StringJoinOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("separator", &sep_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n = num_inputs();
    // output shape: broadcast of scalars against the common non-scalar shape
    TensorShape shape;  // scalar by default
    for (int i = 0; i < n; ++i)
      if (ctx->input(i).shape().dims() > 0) shape = ctx->input(i).shape();
    for (int i = 0; i < n; ++i) {
      const TensorShape& s = ctx->input(i).shape();
      if (s.dims() > 0 && !(s == shape)) {
        ctx->SetStatus(errors::InvalidArgument(
            "StringJoin: inputs must be scalars or share one shape"));
        return;
      }
    }
    Tensor* out = ctx->allocate_output(0, shape);
    int64_t count = out->NumElements();
    for (int64_t e = 0; e < count; ++e) {
      std::string acc;
      for (int i = 0; i < n; ++i) {
        if (i) acc += sep_;
        const Tensor& t = ctx->input(i);
        acc += t.flat<std::string>()[t.shape().dims() == 0 ? 0 : e];
      }
      out->flat<std::string>()[e] = std::move(acc);
    }
  }

 private:
  std::string sep_;
};
REGISTER_KERNEL_BUILDER(Name("StringJoin").Device(DEVICE_CPU), StringJoinOp);

class StringSplitOp : public OpKernel {
 public:
  explicit StringSplitOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("skip_empty", &skip_empty_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& delim_t = ctx->input(1);
    std::string delim = delim_t.flat<std::string>()[0];
    int64_t n = in.NumElements();
    std::vector<std::vector<std::string>> rows(n);
    for (int64_t i = 0; i < n; ++i) {
      const std::string& s = in.flat<std::string>()[i];
      if (delim.empty()) {
        // split on whitespace
        size_t p = 0;
        while (p < s.size()) {
          while (p < s.size() && std::isspace((unsigned char)s[p])) ++p;
          size_t q = p;
          while (q < s.size() && !std::isspace((unsigned char)s[q])) ++q;
          if (q > p) rows[i].push_back(s.substr(p, q - p));
          p = q;
        }
      } else {
        size_t p = 0;
        while (true) {
          size_t q = s.find(delim, p);
          std::string tok = s.substr(p, q == std::string::npos
                                            ? std::string::npos
                                            : q - p);
          if (!tok.empty() || !skip_empty_) rows[i].push_back(tok);
          if (q == std::string::npos) break;
          p = q + delim.size();
        }
      }
    }
    int64_t total = 0, maxlen = 0;
    for (auto& r : rows) {
      total += (int64_t)r.size();
      maxlen = std::max<int64_t>(maxlen, (int64_t)r.size());
    }
    Tensor* idx = ctx->allocate_output(0, TensorShape({total, 2}));
    Tensor* val = ctx->allocate_output(1, TensorShape({total}));
    Tensor* shp = ctx->allocate_output(2, TensorShape({2}));
    int64_t k = 0;
    for (int64_t i = 0; i < n; ++i)
      for (size_t j = 0; j < rows[i].size(); ++j, ++k) {
        idx->flat<int64_t>()[k * 2] = i;
        idx->flat<int64_t>()[k * 2 + 1] = (int64_t)j;
        val->flat<std::string>()[k] = std::move(rows[i][j]);
      }
    shp->flat<int64_t>()[0] = n;
    shp->flat<int64_t>()[1] = maxlen;
  }

 private:
  bool skip_empty_ = true;
};
REGISTER_KERNEL_BUILDER(Name("StringSplit").Device(DEVICE_CPU), StringSplitOp);

template <typename T>
class SubstrOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& pos = ctx->input(1);
    const Tensor& len = ctx->input(2);
    Tensor* out = ctx->allocate_output(0, in.shape());
    int64_t n = in.NumElements();
    bool scalar_pos = pos.NumElements() == 1;
    for (int64_t i = 0; i < n; ++i) {
      const std::string& s = in.flat<std::string>()[i];
      int64_t p = (int64_t)pos.flat<T>()[scalar_pos ? 0 : i];
      int64_t l = (int64_t)len.flat<T>()[scalar_pos ? 0 : i];
      if (p < 0 || p > (int64_t)s.size()) {
        ctx->SetStatus(errors::InvalidArgument("Substr: pos out of range"));
        return;
      }
      out->flat<std::string>()[i] = s.substr(p, l);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("Substr").Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), SubstrOp<int32_t>);
REGISTER_KERNEL_BUILDER(Name("Substr").Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), SubstrOp<int64_t>);

class StringToHashBucketOp : public OpKernel {
 public:
  StringToHashBucketOp(OpKernelConstruction* c, uint64_t seed)
      : OpKernel(c), seed_(seed) {
    c->GetAttr("num_buckets", &num_buckets_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      const std::string& s = in.flat<std::string>()[i];
      out->flat<int64_t>()[i] =
          (int64_t)(Hash64(s.data(), s.size(), seed_) %
                    (uint64_t)num_buckets_);
    }
  }

 protected:
  int64_t num_buckets_ = 1;
  uint64_t seed_;
};
class HashBucketFastOp : public StringToHashBucketOp {
 public:
  explicit HashBucketFastOp(OpKernelConstruction* c)
      : StringToHashBucketOp(c, 0xc3a5c85c97cb3127ull) {}
};
class HashBucketStrongOp : public StringToHashBucketOp {
 public:
  explicit HashBucketStrongOp(OpKernelConstruction* c)
      : StringToHashBucketOp(c, 0x9ae16a3b2f90404full) {
    std::vector<int64_t> key;
    c->GetAttr("key", &key);
    if (key.size() >= 2)
      seed_ = (uint64_t)key[0] * 0x9ddfea08eb382d69ull + (uint64_t)key[1];
  }
};
REGISTER_KERNEL_BUILDER(Name("StringToHashBucket").Device(DEVICE_CPU), HashBucketFastOp);
REGISTER_KERNEL_BUILDER(Name("StringToHashBucketFast").Device(DEVICE_CPU), HashBucketFastOp);
REGISTER_KERNEL_BUILDER(Name("StringToHashBucketStrong").Device(DEVICE_CPU), HashBucketStrongOp);

template <typename T>
class StringToNumberOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      const std::string& s = in.flat<std::string>()[i];
      try {
        if (std::is_integral<T>::value)
          out->flat<T>()[i] = (T)std::stoll(s);
        else
          out->flat<T>()[i] = (T)std::stod(s);
      } catch (...) {
        ctx->SetStatus(errors::InvalidArgument(
            "StringToNumber: could not parse '", s, "'"));
        return;
      }
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("StringToNumber").Device(DEVICE_CPU).TypeConstraint<float>("out_type"), StringToNumberOp<float>);
REGISTER_KERNEL_BUILDER(Name("StringToNumber").Device(DEVICE_CPU).TypeConstraint<double>("out_type"), StringToNumberOp<double>);
REGISTER_KERNEL_BUILDER(Name("StringToNumber").Device(DEVICE_CPU).TypeConstraint<int32_t>("out_type"), StringToNumberOp<int32_t>);
REGISTER_KERNEL_BUILDER(Name("StringToNumber").Device(DEVICE_CPU).TypeConstraint<int64_t>("out_type"), StringToNumberOp<int64_t>);

class ReduceJoinOp : public OpKernel {
 public:
  explicit ReduceJoinOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("keep_dims", &keep_dims_);
    c->GetAttr("separator", &sep_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    const Tensor& axes_t = ctx->input(1);
    int dims = in.shape().dims();
    std::vector<bool> reduce(dims, false);
    if (axes_t.NumElements() == 0) {
      for (int i = 0; i < dims; ++i) reduce[i] = true;
    } else {
      for (int64_t k = 0; k < axes_t.NumElements(); ++k) {
        int a = axes_t.flat<int32_t>()[k];
        if (a < 0) a += dims;
        if (a < 0 || a >= dims) {
          ctx->SetStatus(errors::InvalidArgument("ReduceJoin: bad axis"));
          return;
        }
        reduce[a] = true;
      }
    }
    TensorShape out_shape;
    for (int i = 0; i < dims; ++i) {
      if (!reduce[i]) out_shape.AddDim(in.shape().dim_size(i));
      else if (keep_dims_) out_shape.AddDim(1);
    }
    Tensor* out = ctx->allocate_output(0, out_shape);
    int64_t out_n = out->NumElements();
    std::vector<std::string> acc(out_n);
    std::vector<bool> started(out_n, false);
    // iterate input in order; reduced axes vary fastest within a group when
    // they are innermost — to join in index order along reduced dims we walk
    // the full index space in row-major order (which IS index order).
    std::vector<int64_t> in_dims(dims);
    for (int i = 0; i < dims; ++i) in_dims[i] = in.shape().dim_size(i);
    std::vector<int64_t> coord(dims, 0);
    for (int64_t flat = 0; flat < in.NumElements(); ++flat) {
      // output index from non-reduced coords
      int64_t oidx = 0;
      for (int i = 0; i < dims; ++i) {
        if (reduce[i]) continue;
        oidx = oidx * in_dims[i] + coord[i];
      }
      if (started[oidx]) acc[oidx] += sep_;
      acc[oidx] += in.flat<std::string>()[flat];
      started[oidx] = true;
      for (int i = dims - 1; i >= 0; --i) {
        if (++coord[i] < in_dims[i]) break;
        coord[i] = 0;
      }
    }
    for (int64_t i = 0; i < out_n; ++i)
      out->flat<std::string>()[i] = std::move(acc[i]);
  }

 private:
  bool keep_dims_ = false;
  std::string sep_;
};
REGISTER_KERNEL_BUILDER(Name("ReduceJoin").Device(DEVICE_CPU), ReduceJoinOp);

const char kB64[] =
    "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789-_";

class EncodeBase64Op : public OpKernel {
 public:
  explicit EncodeBase64Op(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("pad", &pad_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      const std::string& s = in.flat<std::string>()[i];
      std::string e;
      e.reserve((s.size() + 2) / 3 * 4);
      size_t j = 0;
      for (; j + 3 <= s.size(); j += 3) {
        uint32_t v = ((unsigned char)s[j] << 16) |
                     ((unsigned char)s[j + 1] << 8) | (unsigned char)s[j + 2];
        e += kB64[v >> 18];
        e += kB64[(v >> 12) & 63];
        e += kB64[(v >> 6) & 63];
        e += kB64[v & 63];
      }
      size_t rem = s.size() - j;
      if (rem == 1) {
        uint32_t v = (unsigned char)s[j] << 16;
        e += kB64[v >> 18];
        e += kB64[(v >> 12) & 63];
        if (pad_) e += "==";
      } else if (rem == 2) {
        uint32_t v = ((unsigned char)s[j] << 16) |
                     ((unsigned char)s[j + 1] << 8);
        e += kB64[v >> 18];
        e += kB64[(v >> 12) & 63];
        e += kB64[(v >> 6) & 63];
        if (pad_) e += "=";
      }
      out->flat<std::string>()[i] = std::move(e);
    }
  }

 private:
  bool pad_ = false;
};
REGISTER_KERNEL_BUILDER(Name("EncodeBase64").Device(DEVICE_CPU), EncodeBase64Op);

class DecodeBase64Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    int8_t rev[256];
    std::memset(rev, -1, sizeof(rev));
    for (int i = 0; i < 64; ++i) rev[(unsigned char)kB64[i]] = (int8_t)i;
    // accept the standard alphabet too
    rev[(unsigned char)'+'] = 62;
    rev[(unsigned char)'/'] = 63;
    const Tensor& in = ctx->input(0);
    Tensor* out = ctx->allocate_output(0, in.shape());
    for (int64_t i = 0; i < in.NumElements(); ++i) {
      const std::string& s = in.flat<std::string>()[i];
      std::string d;
      uint32_t v = 0;
      int bits = 0;
      for (char ch : s) {
        if (ch == '=') break;
        int8_t x = rev[(unsigned char)ch];
        if (x < 0) {
          ctx->SetStatus(errors::InvalidArgument("DecodeBase64: bad char"));
          return;
        }
        v = (v << 6) | (uint32_t)x;
        bits += 6;
        if (bits >= 8) {
          bits -= 8;
          d += (char)((v >> bits) & 0xff);
        }
      }
      out->flat<std::string>()[i] = std::move(d);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("DecodeBase64").Device(DEVICE_CPU), DecodeBase64Op);

}  // namespace
}  // namespace stf
