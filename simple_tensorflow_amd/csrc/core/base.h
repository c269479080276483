// Core base types: Status, logging, DataType, bfloat16/half, TensorShape.
//
// MI355X-native re-implementation of the capability surface described by the
// reference's core/lib + core/platform + core/framework basic types
// (reference: tensorflow/core/lib/core/status.h, core/framework/types.h,
// core/framework/tensor_shape.h, core/framework/bfloat16.h). Brand-new code,
// designed for a single-binary hipcc build (no Bazel, no Eigen).
#pragma once

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <cmath>
#include <memory>
#include <complex>
#include <string>
#include <vector>
#include <sstream>
#include <functional>

namespace stf {

// ---------------------------------------------------------------------------
// Logging
// ---------------------------------------------------------------------------
namespace internal {
class LogMessage {
 public:
  LogMessage(const char* file, int line, int sev) : sev_(sev) {
    stream_ << "[" << "IWEF"[sev] << "] " << file << ":" << line << " ";
  }
  ~LogMessage() {
    stream_ << "\n";
    fputs(stream_.str().c_str(), stderr);
    if (sev_ >= 3) abort();
  }
  std::ostringstream& stream() { return stream_; }

 private:
  std::ostringstream stream_;
  int sev_;
};
struct Voidify {
  void operator&(std::ostream&) {}
};
}  // namespace internal

#define STF_LOG_INFO ::stf::internal::LogMessage(__FILE__, __LINE__, 0).stream()
#define STF_LOG_WARN ::stf::internal::LogMessage(__FILE__, __LINE__, 1).stream()
#define STF_LOG_ERROR ::stf::internal::LogMessage(__FILE__, __LINE__, 2).stream()
#define STF_LOG_FATAL ::stf::internal::LogMessage(__FILE__, __LINE__, 3).stream()
#define LOG(severity) STF_LOG_##severity
#define CHECK(cond)                                        \
  if (!(cond)) STF_LOG_FATAL << "Check failed: " #cond " "
#define CHECK_EQ(a, b) CHECK((a) == (b)) << " (" << (a) << " vs " << (b) << ") "
#define CHECK_NE(a, b) CHECK((a) != (b))
#define CHECK_GE(a, b) CHECK((a) >= (b))
#define CHECK_GT(a, b) CHECK((a) > (b))
#define CHECK_LE(a, b) CHECK((a) <= (b))
#define CHECK_LT(a, b) CHECK((a) < (b))
#define DCHECK(cond) CHECK(cond)

// ---------------------------------------------------------------------------
// Status
// ---------------------------------------------------------------------------
enum class Code : int {
  OK = 0,
  CANCELLED = 1,
  UNKNOWN = 2,
  INVALID_ARGUMENT = 3,
  DEADLINE_EXCEEDED = 4,
  NOT_FOUND = 5,
  ALREADY_EXISTS = 6,
  PERMISSION_DENIED = 7,
  RESOURCE_EXHAUSTED = 8,
  FAILED_PRECONDITION = 9,
  ABORTED = 10,
  OUT_OF_RANGE = 11,
  UNIMPLEMENTED = 12,
  INTERNAL = 13,
  UNAVAILABLE = 14,
  DATA_LOSS = 15,
};

class Status {
 public:
  Status() {}
  Status(Code code, std::string msg)
      : state_(code == Code::OK ? nullptr : new State{code, std::move(msg)}) {}
  bool ok() const { return state_ == nullptr; }
  Code code() const { return ok() ? Code::OK : state_->code; }
  const std::string& message() const {
    static std::string empty;
    return ok() ? empty : state_->msg;
  }
  std::string ToString() const {
    if (ok()) return "OK";
    return "Error(" + std::to_string((int)code()) + "): " + message();
  }
  static Status OK() { return Status(); }

 private:
  struct State {
    Code code;
    std::string msg;
  };
  std::shared_ptr<State> state_;
};

namespace errors {
#define STF_DECLARE_ERROR(FN, CODE)                        \
  template <typename... Args>                              \
  inline Status FN(Args... args) {                         \
    std::ostringstream oss;                                \
    (void)std::initializer_list<int>{((oss << args), 0)...}; \
    return Status(Code::CODE, oss.str());                  \
  }
STF_DECLARE_ERROR(InvalidArgument, INVALID_ARGUMENT)
STF_DECLARE_ERROR(NotFound, NOT_FOUND)
STF_DECLARE_ERROR(AlreadyExists, ALREADY_EXISTS)
STF_DECLARE_ERROR(Internal, INTERNAL)
STF_DECLARE_ERROR(Unimplemented, UNIMPLEMENTED)
STF_DECLARE_ERROR(FailedPrecondition, FAILED_PRECONDITION)
STF_DECLARE_ERROR(ResourceExhausted, RESOURCE_EXHAUSTED)
STF_DECLARE_ERROR(OutOfRange, OUT_OF_RANGE)
STF_DECLARE_ERROR(Aborted, ABORTED)
STF_DECLARE_ERROR(Cancelled, CANCELLED)
STF_DECLARE_ERROR(Unavailable, UNAVAILABLE)
STF_DECLARE_ERROR(DataLoss, DATA_LOSS)
STF_DECLARE_ERROR(DeadlineExceeded, DEADLINE_EXCEEDED)
#undef STF_DECLARE_ERROR
}  // namespace errors

#define STF_RETURN_IF_ERROR(...)              \
  do {                                        \
    ::stf::Status _s = (__VA_ARGS__);         \
    if (!_s.ok()) return _s;                  \
  } while (0)

#define STF_CHECK_OK(...)                                      \
  do {                                                         \
    ::stf::Status _s = (__VA_ARGS__);                          \
    if (!_s.ok()) LOG(FATAL) << "Non-OK status: " << _s.ToString(); \
  } while (0)

// ---------------------------------------------------------------------------
// DataType — numeric values wire-compatible with the reference's
// core/framework/types.proto.
// ---------------------------------------------------------------------------
enum DataType : int {
  DT_INVALID = 0,
  DT_FLOAT = 1,
  DT_DOUBLE = 2,
  DT_INT32 = 3,
  DT_UINT8 = 4,
  DT_INT16 = 5,
  DT_INT8 = 6,
  DT_STRING = 7,
  DT_COMPLEX64 = 8,
  DT_INT64 = 9,
  DT_BOOL = 10,
  DT_QINT8 = 11,
  DT_QUINT8 = 12,
  DT_QINT32 = 13,
  DT_BFLOAT16 = 14,
  DT_QINT16 = 15,
  DT_QUINT16 = 16,
  DT_UINT16 = 17,
  DT_COMPLEX128 = 18,
  DT_HALF = 19,
  DT_RESOURCE = 20,
};

// bfloat16: fp32 with 16-bit mantissa truncation (round-to-nearest-even on
// conversion from float).
struct bfloat16 {
  uint16_t value = 0;
  bfloat16() = default;
  bfloat16(float f) {  // NOLINT: implicit by design (compute in float)
    uint32_t bits;
    std::memcpy(&bits, &f, 4);
    // round to nearest even
    uint32_t lsb = (bits >> 16) & 1;
    bits += 0x7fff + lsb;
    value = (uint16_t)(bits >> 16);
  }
  operator float() const {
    uint32_t bits = ((uint32_t)value) << 16;
    float f;
    std::memcpy(&f, &bits, 4);
    return f;
  }
  bfloat16& operator=(float f) {
    *this = bfloat16(f);
    return *this;
  }
  bfloat16& operator+=(float f) {
    *this = bfloat16((float)*this + f);
    return *this;
  }
  bfloat16& operator*=(float f) {
    *this = bfloat16((float)*this * f);
    return *this;
  }
  bfloat16& operator-=(float f) {
    *this = bfloat16((float)*this - f);
    return *this;
  }
  bfloat16& operator/=(float f) {
    *this = bfloat16((float)*this / f);
    return *this;
  }
};

// IEEE fp16 (storage-only on CPU paths).
struct half16 {
  uint16_t value = 0;
};

inline size_t DataTypeSize(DataType dt) {
  switch (dt) {
    case DT_FLOAT: return 4;
    case DT_DOUBLE: return 8;
    case DT_INT32: return 4;
    case DT_UINT8: return 1;
    case DT_INT16: return 2;
    case DT_INT8: return 1;
    case DT_INT64: return 8;
    case DT_BOOL: return 1;
    case DT_BFLOAT16: return 2;
    case DT_UINT16: return 2;
    case DT_HALF: return 2;
    case DT_COMPLEX64: return 8;
    case DT_COMPLEX128: return 16;
    case DT_STRING: return sizeof(std::string);
    default: return 0;
  }
}

inline const char* DataTypeString(DataType dt) {
  switch (dt) {
    case DT_FLOAT: return "float";
    case DT_DOUBLE: return "double";
    case DT_INT32: return "int32";
    case DT_UINT8: return "uint8";
    case DT_INT16: return "int16";
    case DT_INT8: return "int8";
    case DT_STRING: return "string";
    case DT_INT64: return "int64";
    case DT_BOOL: return "bool";
    case DT_BFLOAT16: return "bfloat16";
    case DT_UINT16: return "uint16";
    case DT_HALF: return "half";
    case DT_RESOURCE: return "resource";
    case DT_COMPLEX64: return "complex64";
    case DT_COMPLEX128: return "complex128";
    default: return "invalid";
  }
}

// Parse a type name ("float", "int32", ...) to DataType; DT_INVALID if unknown.
inline DataType DataTypeFromString(const std::string& s) {
  static const struct { const char* n; DataType d; } kNames[] = {
      {"float", DT_FLOAT},     {"float32", DT_FLOAT}, {"double", DT_DOUBLE},
      {"float64", DT_DOUBLE},  {"int32", DT_INT32},   {"uint8", DT_UINT8},
      {"int16", DT_INT16},     {"int8", DT_INT8},     {"string", DT_STRING},
      {"int64", DT_INT64},     {"bool", DT_BOOL},     {"bfloat16", DT_BFLOAT16},
      {"uint16", DT_UINT16},   {"half", DT_HALF},     {"float16", DT_HALF},
      {"resource", DT_RESOURCE}, {"complex64", DT_COMPLEX64},
  };
  for (auto& kv : kNames)
    if (s == kv.n) return kv.d;
  return DT_INVALID;
}

template <typename T>
struct DataTypeToEnum;
template <> struct DataTypeToEnum<float> { static constexpr DataType v = DT_FLOAT; };
template <> struct DataTypeToEnum<double> { static constexpr DataType v = DT_DOUBLE; };
template <> struct DataTypeToEnum<int32_t> { static constexpr DataType v = DT_INT32; };
template <> struct DataTypeToEnum<int64_t> { static constexpr DataType v = DT_INT64; };
template <> struct DataTypeToEnum<uint8_t> { static constexpr DataType v = DT_UINT8; };
template <> struct DataTypeToEnum<int8_t> { static constexpr DataType v = DT_INT8; };
template <> struct DataTypeToEnum<int16_t> { static constexpr DataType v = DT_INT16; };
template <> struct DataTypeToEnum<uint16_t> { static constexpr DataType v = DT_UINT16; };
template <> struct DataTypeToEnum<bool> { static constexpr DataType v = DT_BOOL; };
template <> struct DataTypeToEnum<bfloat16> { static constexpr DataType v = DT_BFLOAT16; };
template <> struct DataTypeToEnum<half16> { static constexpr DataType v = DT_HALF; };
template <> struct DataTypeToEnum<std::string> { static constexpr DataType v = DT_STRING; };
template <> struct DataTypeToEnum<std::complex<float>> { static constexpr DataType v = DT_COMPLEX64; };
template <> struct DataTypeToEnum<std::complex<double>> { static constexpr DataType v = DT_COMPLEX128; };

// ---------------------------------------------------------------------------
// TensorShape
// ---------------------------------------------------------------------------
class TensorShape {
 public:
  TensorShape() {}
  TensorShape(std::initializer_list<int64_t> dims) : dims_(dims) { Recompute(); }
  explicit TensorShape(const std::vector<int64_t>& dims) : dims_(dims) { Recompute(); }

  int dims() const { return (int)dims_.size(); }
  int64_t dim_size(int i) const {
    CHECK(i >= 0 && i < dims()) << "dim " << i << " out of range";
    return dims_[i];
  }
  const std::vector<int64_t>& dim_sizes() const { return dims_; }
  int64_t num_elements() const { return num_elements_; }
  void AddDim(int64_t d) {
    dims_.push_back(d);
    Recompute();
  }
  void InsertDim(int i, int64_t d) {
    dims_.insert(dims_.begin() + i, d);
    Recompute();
  }
  void RemoveDim(int i) {
    dims_.erase(dims_.begin() + i);
    Recompute();
  }
  void set_dim(int i, int64_t d) {
    dims_[i] = d;
    Recompute();
  }
  bool operator==(const TensorShape& o) const { return dims_ == o.dims_; }
  bool operator!=(const TensorShape& o) const { return !(*this == o); }
  bool IsSameSize(const TensorShape& o) const { return *this == o; }
  std::string DebugString() const {
    std::string s = "[";
    for (int i = 0; i < dims(); ++i) {
      if (i) s += ",";
      s += std::to_string(dims_[i]);
    }
    return s + "]";
  }

 private:
  void Recompute() {
    num_elements_ = 1;
    for (int64_t d : dims_) num_elements_ *= d;
  }
  std::vector<int64_t> dims_;
  int64_t num_elements_ = 1;
};

// String helpers.
inline std::vector<std::string> StrSplit(const std::string& s, char sep) {
  std::vector<std::string> out;
  size_t start = 0;
  for (size_t i = 0; i <= s.size(); ++i) {
    if (i == s.size() || s[i] == sep) {
      out.push_back(s.substr(start, i - start));
      start = i + 1;
    }
  }
  return out;
}
inline std::string StrStrip(const std::string& s) {
  size_t a = s.find_first_not_of(" \t\n");
  if (a == std::string::npos) return "";
  size_t b = s.find_last_not_of(" \t\n");
  return s.substr(a, b - a + 1);
}
inline bool StrStartsWith(const std::string& s, const std::string& p) {
  return s.size() >= p.size() && s.compare(0, p.size(), p) == 0;
}

}  // namespace stf
