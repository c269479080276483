#include "util/table.h"

#include <cstring>

namespace stf {
namespace table {

namespace {

// varint32/64 helpers (LEB128, same as protobuf)
void PutVarint(std::string* s, uint64_t v) {
  while (v >= 0x80) {
    s->push_back((char)(v | 0x80));
    v >>= 7;
  }
  s->push_back((char)v);
}
bool GetVarint(const char** p, const char* end, uint64_t* out) {
  uint64_t v = 0;
  int shift = 0;
  while (*p < end) {
    uint8_t b = (uint8_t)**p;
    ++*p;
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
void PutFixed32(std::string* s, uint32_t v) {
  char buf[4];
  std::memcpy(buf, &v, 4);
  s->append(buf, 4);
}

uint32_t crc_table[8][256];
bool crc_init_done = false;
void InitCrc() {
  if (crc_init_done) return;
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = i;
    for (int k = 0; k < 8; ++k)
      c = (c & 1) ? (0x82f63b78u ^ (c >> 1)) : (c >> 1);
    crc_table[0][i] = c;
  }
  for (uint32_t i = 0; i < 256; ++i) {
    uint32_t c = crc_table[0][i];
    for (int t = 1; t < 8; ++t) {
      c = crc_table[0][c & 0xff] ^ (c >> 8);
      crc_table[t][i] = c;
    }
  }
  crc_init_done = true;
}

}  // namespace

uint32_t Crc32c(const char* data, size_t n) {
  InitCrc();
  uint32_t crc = 0xffffffffu;
  const uint8_t* p = (const uint8_t*)data;
  while (n >= 8) {
    uint64_t v;
    std::memcpy(&v, p, 8);
    v ^= crc;
    crc = crc_table[7][v & 0xff] ^ crc_table[6][(v >> 8) & 0xff] ^
          crc_table[5][(v >> 16) & 0xff] ^ crc_table[4][(v >> 24) & 0xff] ^
          crc_table[3][(v >> 32) & 0xff] ^ crc_table[2][(v >> 40) & 0xff] ^
          crc_table[1][(v >> 48) & 0xff] ^ crc_table[0][(v >> 56) & 0xff];
    p += 8;
    n -= 8;
  }
  while (n--) crc = crc_table[0][(crc ^ *p++) & 0xff] ^ (crc >> 8);
  return crc ^ 0xffffffffu;
}

namespace {

// One block: prefix-compressed entries + restart array.
class BlockBuilder {
 public:
  void Add(const std::string& key, const std::string& value) {
    size_t shared = 0;
    if (counter_ < 16 && !last_key_.empty()) {
      size_t m = std::min(last_key_.size(), key.size());
      while (shared < m && last_key_[shared] == key[shared]) ++shared;
    } else {
      restarts_.push_back((uint32_t)buf_.size());
      counter_ = 0;
      shared = 0;
    }
    PutVarint(&buf_, shared);
    PutVarint(&buf_, key.size() - shared);
    PutVarint(&buf_, value.size());
    buf_.append(key.data() + shared, key.size() - shared);
    buf_.append(value);
    last_key_ = key;
    ++counter_;
  }
  std::string Finish() {
    if (restarts_.empty()) restarts_.push_back(0);
    for (uint32_t r : restarts_) PutFixed32(&buf_, r);
    PutFixed32(&buf_, (uint32_t)restarts_.size());
    return buf_;
  }
  size_t CurrentSize() const {
    return buf_.size() + restarts_.size() * 4 + 4;
  }
  bool empty() const { return buf_.empty(); }
  void Reset() {
    buf_.clear();
    restarts_.clear();
    last_key_.clear();
    counter_ = 0;
  }

 private:
  std::string buf_;
  std::vector<uint32_t> restarts_;
  std::string last_key_;
  int counter_ = 0;
};

struct BlockHandle {
  uint64_t offset = 0, size = 0;
  void Encode(std::string* s) const {
    PutVarint(s, offset);
    PutVarint(s, size);
  }
  bool Decode(const char** p, const char* end) {
    return GetVarint(p, end, &offset) && GetVarint(p, end, &size);
  }
};

// Write one block + trailer; returns its handle.
BlockHandle EmitBlock(std::string* out, const std::string& contents) {
  BlockHandle h;
  h.offset = out->size();
  h.size = contents.size();
  out->append(contents);
  char type = 0;  // kNoCompression
  out->push_back(type);
  // crc over contents + type byte, masked
  uint32_t crc = Crc32c(contents.data(), contents.size());
  // extend over the type byte
  std::string tail(1, type);
  std::string both = contents + tail;
  crc = Crc32c(both.data(), both.size());
  PutFixed32(out, MaskCrc(crc));
  return h;
}

Status ParseBlock(const std::string& data, const BlockHandle& h,
                  std::map<std::string, std::string>* out) {
  if (h.offset + h.size + 5 > data.size())
    return errors::InvalidArgument("table: block out of range");
  const char* base = data.data() + h.offset;
  if (h.size < 4) return errors::InvalidArgument("table: short block");
  uint32_t num_restarts;
  std::memcpy(&num_restarts, base + h.size - 4, 4);
  if ((uint64_t)num_restarts * 4 + 4 > h.size)
    return errors::InvalidArgument("table: bad restarts");
  const char* end = base + h.size - 4 - num_restarts * 4;
  const char* p = base;
  std::string key;
  while (p < end) {
    uint64_t shared, non_shared, vlen;
    if (!GetVarint(&p, end, &shared) || !GetVarint(&p, end, &non_shared) ||
        !GetVarint(&p, end, &vlen))
      return errors::InvalidArgument("table: bad entry");
    if (p + non_shared + vlen > end)
      return errors::InvalidArgument("table: entry overrun");
    key.resize(shared);
    key.append(p, non_shared);
    p += non_shared;
    (*out)[key] = std::string(p, vlen);
    p += vlen;
  }
  return Status::OK();
}

}  // namespace

Status BuildTable(const std::map<std::string, std::string>& entries,
                  std::string* out) {
  out->clear();
  BlockBuilder data_block;
  BlockBuilder index_block;
  std::string last_key;
  std::vector<std::pair<std::string, BlockHandle>> index_entries;
  auto flush = [&](const std::string& last) {
    if (data_block.empty()) return;
    BlockHandle h = EmitBlock(out, data_block.Finish());
    index_entries.emplace_back(last, h);
    data_block.Reset();
  };
  for (auto& kv : entries) {
    data_block.Add(kv.first, kv.second);
    last_key = kv.first;
    if (data_block.CurrentSize() >= 4096) flush(last_key);
  }
  flush(last_key);
  // metaindex (empty)
  BlockBuilder meta;
  BlockHandle meta_handle = EmitBlock(out, meta.Finish());
  // index block
  for (auto& e : index_entries) {
    std::string hv;
    e.second.Encode(&hv);
    index_block.Add(e.first, hv);
  }
  BlockHandle index_handle = EmitBlock(out, index_block.Finish());
  // footer: metaindex handle + index handle, padded to 40, + magic
  std::string footer;
  meta_handle.Encode(&footer);
  index_handle.Encode(&footer);
  footer.resize(40);
  PutFixed32(&footer, (uint32_t)(0xdb4775248b80fb57ull & 0xffffffffu));
  PutFixed32(&footer, (uint32_t)(0xdb4775248b80fb57ull >> 32));
  out->append(footer);
  return Status::OK();
}

Status ReadTable(const std::string& data,
                 std::map<std::string, std::string>* out) {
  if (data.size() < 48) return errors::InvalidArgument("table too short");
  const char* footer = data.data() + data.size() - 48;
  uint32_t magic_lo, magic_hi;
  std::memcpy(&magic_lo, footer + 40, 4);
  std::memcpy(&magic_hi, footer + 44, 4);
  uint64_t magic = ((uint64_t)magic_hi << 32) | magic_lo;
  if (magic != 0xdb4775248b80fb57ull)
    return errors::InvalidArgument("bad table magic");
  const char* p = footer;
  const char* end = footer + 40;
  BlockHandle meta_handle, index_handle;
  if (!meta_handle.Decode(&p, end) || !index_handle.Decode(&p, end))
    return errors::InvalidArgument("bad table footer");
  std::map<std::string, std::string> index;
  STF_RETURN_IF_ERROR(ParseBlock(data, index_handle, &index));
  for (auto& kv : index) {
    const char* hp = kv.second.data();
    BlockHandle h;
    if (!h.Decode(&hp, hp + kv.second.size()))
      return errors::InvalidArgument("bad index entry");
    STF_RETURN_IF_ERROR(ParseBlock(data, h, out));
  }
  return Status::OK();
}

}  // namespace table
}  // namespace stf
