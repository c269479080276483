// Minimal protobuf wire-format reader/writer (proto3 subset).
//
// We keep the on-wire bytes compatible with the reference's .proto schemas
// (tensorflow/core/framework/*.proto) without depending on libprotobuf:
// messages are hand-defined structs in protos.h with explicit field numbers.
// Wire types used: 0 = varint, 1 = 64-bit, 2 = length-delimited, 5 = 32-bit.
#pragma once

#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

#include "core/base.h"

namespace stf {
namespace pb {

// ------------------------------- Writer ------------------------------------
class Writer {
 public:
  std::string& buf() { return buf_; }

  void PutVarint(uint64_t v) {
    while (v >= 0x80) {
      buf_.push_back((char)(v | 0x80));
      v >>= 7;
    }
    buf_.push_back((char)v);
  }
  void PutTag(int field, int wire) { PutVarint(((uint64_t)field << 3) | wire); }

  void PutInt64(int field, int64_t v) {
    if (v == 0) return;
    PutTag(field, 0);
    PutVarint((uint64_t)v);
  }
  void PutUInt64(int field, uint64_t v) {
    if (v == 0) return;
    PutTag(field, 0);
    PutVarint(v);
  }
  void PutBool(int field, bool v) {
    if (!v) return;
    PutTag(field, 0);
    PutVarint(1);
  }
  void PutFloat(int field, float v) {
    if (v == 0.0f && !std::signbit(v)) return;
    PutTag(field, 5);
    char tmp[4];
    std::memcpy(tmp, &v, 4);
    buf_.append(tmp, 4);
  }
  void PutDouble(int field, double v) {
    if (v == 0.0 && !std::signbit(v)) return;
    PutTag(field, 1);
    char tmp[8];
    std::memcpy(tmp, &v, 8);
    buf_.append(tmp, 8);
  }
  void PutString(int field, const std::string& s) {
    if (s.empty()) return;
    PutTag(field, 2);
    PutVarint(s.size());
    buf_.append(s);
  }
  // Always emits, even when empty (for repeated-field elements).
  void PutStringAlways(int field, const std::string& s) {
    PutTag(field, 2);
    PutVarint(s.size());
    buf_.append(s);
  }
  void PutMessage(int field, const std::string& sub) {
    PutTag(field, 2);
    PutVarint(sub.size());
    buf_.append(sub);
  }
  // Packed repeated varints.
  void PutPackedVarints(int field, const std::vector<int64_t>& vs) {
    if (vs.empty()) return;
    Writer sub;
    for (int64_t v : vs) sub.PutVarint((uint64_t)v);
    PutMessage(field, sub.buf());
  }
  void PutPackedFloats(int field, const std::vector<float>& vs) {
    if (vs.empty()) return;
    Writer sub;
    for (float v : vs) {
      char tmp[4];
      std::memcpy(tmp, &v, 4);
      sub.buf_.append(tmp, 4);
    }
    PutMessage(field, sub.buf());
  }

 private:
  std::string buf_;
};

// ------------------------------- Reader ------------------------------------
class Reader {
 public:
  Reader(const char* data, size_t size) : p_(data), end_(data + size) {}
  explicit Reader(const std::string& s) : Reader(s.data(), s.size()) {}

  bool done() const { return p_ >= end_; }

  // Reads next tag; returns false at end. field/wire are outputs.
  bool ReadTag(int* field, int* wire) {
    if (done()) return false;
    uint64_t tag;
    if (!ReadVarint(&tag)) return false;
    *field = (int)(tag >> 3);
    *wire = (int)(tag & 7);
    return true;
  }

  bool ReadVarint(uint64_t* out) {
    uint64_t v = 0;
    int shift = 0;
    while (p_ < end_) {
      uint8_t b = (uint8_t)*p_++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) {
        *out = v;
        return true;
      }
      shift += 7;
      if (shift >= 64) return false;
    }
    return false;
  }
  bool ReadFixed32(uint32_t* out) {
    if (end_ - p_ < 4) return false;
    std::memcpy(out, p_, 4);
    p_ += 4;
    return true;
  }
  bool ReadFixed64(uint64_t* out) {
    if (end_ - p_ < 8) return false;
    std::memcpy(out, p_, 8);
    p_ += 8;
    return true;
  }
  bool ReadBytes(std::string* out) {
    uint64_t len;
    if (!ReadVarint(&len) || (uint64_t)(end_ - p_) < len) return false;
    out->assign(p_, len);
    p_ += len;
    return true;
  }
  // Returns a view (ptr,len) for a length-delimited field.
  bool ReadView(const char** data, size_t* len) {
    uint64_t l;
    if (!ReadVarint(&l) || (uint64_t)(end_ - p_) < l) return false;
    *data = p_;
    *len = l;
    p_ += l;
    return true;
  }
  bool SkipField(int wire) {
    switch (wire) {
      case 0: {
        uint64_t v;
        return ReadVarint(&v);
      }
      case 1: {
        uint64_t v;
        return ReadFixed64(&v);
      }
      case 2: {
        const char* d;
        size_t l;
        return ReadView(&d, &l);
      }
      case 5: {
        uint32_t v;
        return ReadFixed32(&v);
      }
      default:
        return false;
    }
  }

 private:
  const char* p_;
  const char* end_;
};

}  // namespace pb
}  // namespace stf
