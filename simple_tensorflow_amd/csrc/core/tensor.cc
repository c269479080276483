#include "core/tensor.h"

#include <cstdlib>
#include <mutex>

namespace stf {

namespace {
class CPUAllocator : public Allocator {
 public:
  void* Allocate(size_t bytes) override {
    void* p = nullptr;
    if (posix_memalign(&p, 64, bytes) != 0) return nullptr;
    return p;
  }
  void Deallocate(void* ptr, size_t) override { free(ptr); }
  const char* name() const override { return "cpu"; }
};
}  // namespace

Allocator* cpu_allocator() {
  static CPUAllocator* a = new CPUAllocator();
  return a;
}

void Tensor::AsProto(TensorProto* proto) const {
  CHECK(mem_space() == MemSpace::HOST) << "AsProto requires a host tensor";
  proto->dtype = dtype_;
  proto->tensor_shape = TensorShapeProto::From(shape_);
  proto->has_shape = true;
  if (dtype_ == DT_STRING) {
    const std::string* p = flat<std::string>();
    for (int64_t i = 0; i < NumElements(); ++i) proto->string_val.push_back(p[i]);
  } else {
    proto->tensor_content.assign((const char*)raw_data(), TotalBytes());
  }
}

Status Tensor::FromProto(const TensorProto& proto, Tensor* out) {
  TensorShape shape = proto.tensor_shape.AsShape();
  Tensor t(proto.dtype, shape);
  int64_t n = shape.num_elements();
  if (proto.dtype == DT_STRING) {
    std::string* p = t.flat<std::string>();
    for (int64_t i = 0; i < n && i < (int64_t)proto.string_val.size(); ++i)
      p[i] = proto.string_val[i];
  } else if (!proto.tensor_content.empty()) {
    if (proto.tensor_content.size() != t.TotalBytes())
      return errors::InvalidArgument("tensor_content size mismatch: ",
                                     proto.tensor_content.size(), " vs ",
                                     t.TotalBytes());
    std::memcpy(t.raw_data(), proto.tensor_content.data(), t.TotalBytes());
  } else {
    // Typed value fields, with scalar splat semantics (repeat last value).
    auto fill = [&](auto* dst, const auto& src) {
      using T = typename std::remove_reference<decltype(dst[0])>::type;
      if (src.empty()) {
        std::memset((void*)dst, 0, t.TotalBytes());
        return;
      }
      for (int64_t i = 0; i < n; ++i)
        dst[i] = (T)src[i < (int64_t)src.size() ? i : src.size() - 1];
    };
    switch (proto.dtype) {
      case DT_FLOAT:
        fill(t.flat<float>(), proto.float_val);
        break;
      case DT_DOUBLE:
        fill(t.flat<double>(), proto.double_val);
        break;
      case DT_INT32:
        fill(t.flat<int32_t>(), proto.int_val);
        break;
      case DT_INT64:
        fill(t.flat<int64_t>(), proto.int64_val);
        break;
      case DT_BOOL: {
        bool* p = t.flat<bool>();
        if (proto.bool_val.empty()) {
          std::memset(p, 0, n);
        } else {
          for (int64_t i = 0; i < n; ++i)
            p[i] = proto.bool_val[i < (int64_t)proto.bool_val.size()
                                      ? i
                                      : proto.bool_val.size() - 1] != 0;
        }
        break;
      }
      case DT_BFLOAT16:
      case DT_HALF: {
        uint16_t* p = t.flat<uint16_t>();
        if (proto.half_val.empty()) {
          std::memset(p, 0, n * 2);
        } else {
          for (int64_t i = 0; i < n; ++i)
            p[i] = (uint16_t)proto.half_val[i < (int64_t)proto.half_val.size()
                                                ? i
                                                : proto.half_val.size() - 1];
        }
        break;
      }
      case DT_UINT8:
        fill(t.flat<uint8_t>(), proto.int_val);
        break;
      case DT_INT8:
        fill(t.flat<int8_t>(), proto.int_val);
        break;
      case DT_INT16:
        fill(t.flat<int16_t>(), proto.int_val);
        break;
      default:
        return errors::Unimplemented("FromProto: dtype ",
                                     DataTypeString(proto.dtype));
    }
  }
  *out = t;
  return Status::OK();
}

}  // namespace stf
