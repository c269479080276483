// Tensor + Allocator.
//
// Capability parity with the reference's Tensor/TensorBuffer/Allocator
// (reference: tensorflow/core/framework/tensor.h:43,480, allocator.h), built
// MI355X-first: a buffer is tagged with the memory space it lives in (host or
// a GPU's HBM) so the runtime can route copies; allocation is pluggable so the
// GPU device can install a BFC arena over hipMalloc.
#pragma once

#include <atomic>
#include <memory>

#include "core/base.h"
#include "core/protos.h"

namespace stf {

enum class MemSpace : int { HOST = 0, DEVICE = 1 };

class Allocator {
 public:
  virtual ~Allocator() {}
  virtual void* Allocate(size_t bytes) = 0;
  virtual void Deallocate(void* ptr, size_t bytes) = 0;
  virtual MemSpace space() const { return MemSpace::HOST; }
  // GPU ordinal for DEVICE space (-1 for host).
  virtual int device_ordinal() const { return -1; }
  virtual const char* name() const { return "cpu"; }
};

// 64-byte-aligned host allocator.
Allocator* cpu_allocator();

// Refcounted storage. For DT_STRING the buffer holds constructed std::string
// objects (host only); for POD dtypes raw bytes.
class TensorBuffer {
 public:
  TensorBuffer(Allocator* alloc, void* data, size_t size, bool is_string,
               int64_t n_strings)
      : alloc_(alloc), data_(data), size_(size), is_string_(is_string),
        n_strings_(n_strings) {}
  ~TensorBuffer() {
    if (data_ == nullptr) return;
    if (is_string_) {
      std::string* p = static_cast<std::string*>(data_);
      for (int64_t i = 0; i < n_strings_; ++i) p[i].~basic_string();
    }
    alloc_->Deallocate(data_, size_);
  }
  void* data() const { return data_; }
  size_t size() const { return size_; }
  Allocator* allocator() const { return alloc_; }

 private:
  Allocator* alloc_;
  void* data_;
  size_t size_;
  bool is_string_;
  int64_t n_strings_;
};

class Tensor {
 public:
  Tensor() : dtype_(DT_INVALID) {}
  Tensor(DataType dtype, const TensorShape& shape)
      : Tensor(cpu_allocator(), dtype, shape) {}
  Tensor(Allocator* alloc, DataType dtype, const TensorShape& shape)
      : dtype_(dtype), shape_(shape) {
    int64_t n = shape.num_elements();
    size_t bytes = (size_t)n * DataTypeSize(dtype);
    if (dtype == DT_STRING) {
      CHECK(alloc->space() == MemSpace::HOST) << "string tensors are host-only";
      void* mem = alloc->Allocate(bytes ? bytes : 1);
      std::string* p = static_cast<std::string*>(mem);
      for (int64_t i = 0; i < n; ++i) new (p + i) std::string();
      buf_ = std::make_shared<TensorBuffer>(alloc, mem, bytes ? bytes : 1, true, n);
    } else {
      void* mem = bytes ? alloc->Allocate(bytes) : nullptr;
      buf_ = bytes ? std::make_shared<TensorBuffer>(alloc, mem, bytes, false, 0)
                   : nullptr;
    }
  }

  DataType dtype() const { return dtype_; }
  const TensorShape& shape() const { return shape_; }
  int dims() const { return shape_.dims(); }
  int64_t dim_size(int i) const { return shape_.dim_size(i); }
  int64_t NumElements() const { return shape_.num_elements(); }
  size_t TotalBytes() const {
    return (size_t)NumElements() * DataTypeSize(dtype_);
  }
  bool IsInitialized() const { return dtype_ != DT_INVALID; }
  MemSpace mem_space() const {
    return buf_ ? buf_->allocator()->space() : MemSpace::HOST;
  }
  int device_ordinal() const {
    return buf_ ? buf_->allocator()->device_ordinal() : -1;
  }

  void* raw_data() const { return buf_ ? buf_->data() : nullptr; }
  template <typename T>
  T* flat() const {
    return static_cast<T*>(raw_data());
  }
  template <typename T>
  T scalar() const {
    CHECK_EQ(NumElements(), 1);
    return flat<T>()[0];
  }

  // Same buffer, different shape (no copy).
  Tensor Reshaped(const TensorShape& s) const {
    CHECK_EQ(s.num_elements(), NumElements());
    Tensor t = *this;
    t.shape_ = s;
    return t;
  }

  bool SharesBufferWith(const Tensor& o) const { return buf_ == o.buf_; }

  // Host-only (de)serialization to TensorProto.
  void AsProto(TensorProto* proto) const;
  static Status FromProto(const TensorProto& proto, Tensor* out);

  std::string DebugString() const {
    return std::string(DataTypeString(dtype_)) + shape_.DebugString();
  }

 private:
  DataType dtype_;
  TensorShape shape_;
  std::shared_ptr<TensorBuffer> buf_;
};

}  // namespace stf
