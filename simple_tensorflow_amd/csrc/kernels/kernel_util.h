// Shared helpers for CPU kernels: type dispatch, numpy-style broadcasting.
#pragma once

#include <algorithm>
#include <cmath>
#include <vector>

#include "framework/device.h"
#include "framework/op_kernel.h"

namespace stf {

// Numpy-style broadcast of two shapes; element strides are 0 on broadcast
// dims (capability analog of the reference's util/bcast.cc).
struct BCast {
  bool valid = false;
  std::vector<int64_t> out;       // output shape
  std::vector<int64_t> sx, sy;    // per-dim element strides into x and y
  int64_t num_elements = 1;

  BCast(const TensorShape& x, const TensorShape& y) {
    int rank = std::max(x.dims(), y.dims());
    std::vector<int64_t> xd(rank, 1), yd(rank, 1);
    for (int i = 0; i < x.dims(); ++i) xd[rank - x.dims() + i] = x.dim_size(i);
    for (int i = 0; i < y.dims(); ++i) yd[rank - y.dims() + i] = y.dim_size(i);
    out.resize(rank);
    valid = true;
    for (int i = 0; i < rank; ++i) {
      if (xd[i] == yd[i]) out[i] = xd[i];
      else if (xd[i] == 1) out[i] = yd[i];
      else if (yd[i] == 1) out[i] = xd[i];
      else { valid = false; return; }
      num_elements *= out[i];
    }
    sx.assign(rank, 0);
    sy.assign(rank, 0);
    int64_t s = 1;
    for (int i = rank - 1; i >= 0; --i) {
      if (xd[i] != 1) sx[i] = s;
      s *= xd[i];
    }
    s = 1;
    for (int i = rank - 1; i >= 0; --i) {
      if (yd[i] != 1) sy[i] = s;
      s *= yd[i];
    }
  }

  TensorShape out_shape() const { return TensorShape(out); }
  bool trivial() const {
    for (size_t i = 0; i < out.size(); ++i)
      if (sx[i] == 0 || sy[i] == 0) {
        // only trivial when both strides dense everywhere
        return false;
      }
    return true;
  }

  // Map output linear index -> (x index, y index).
  inline void Map(int64_t idx, int64_t* xi, int64_t* yi) const {
    int64_t x = 0, y = 0;
    for (int i = (int)out.size() - 1; i >= 0; --i) {
      int64_t c = idx % out[i];
      idx /= out[i];
      x += c * sx[i];
      y += c * sy[i];
    }
    *xi = x;
    *yi = y;
  }
};

// Reads an int32/int64 host tensor into a vector<int64>.
inline std::vector<int64_t> IntVector(const Tensor& t) {
  std::vector<int64_t> out;
  if (t.dtype() == DT_INT32) {
    const int32_t* p = t.flat<int32_t>();
    for (int64_t i = 0; i < t.NumElements(); ++i) out.push_back(p[i]);
  } else {
    const int64_t* p = t.flat<int64_t>();
    for (int64_t i = 0; i < t.NumElements(); ++i) out.push_back(p[i]);
  }
  return out;
}

// Registration helper macros for the standard CPU numeric types.
#define REGISTER_CPU_KERNEL_TYPES(OP, KERNEL)                                  \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), KERNEL<float>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), KERNEL<double>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int32_t>("T"), KERNEL<int32_t>); \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<int64_t>("T"), KERNEL<int64_t>);

#define REGISTER_CPU_KERNEL_FLOATS(OP, KERNEL)                                 \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<float>("T"), KERNEL<float>);   \
  REGISTER_KERNEL_BUILDER(Name(OP).Device(DEVICE_CPU).TypeConstraint<double>("T"), KERNEL<double>);

// Dispatch over all POD dtypes by size (for movement ops that don't compute).
template <typename Fn>
inline void DispatchBySize(size_t elem_size, Fn fn) {
  switch (elem_size) {
    case 1: fn(uint8_t{}); break;
    case 2: fn(uint16_t{}); break;
    case 4: fn(uint32_t{}); break;
    case 8: fn(uint64_t{}); break;
    default: LOG(FATAL) << "bad elem size " << elem_size;
  }
}

}  // namespace stf
