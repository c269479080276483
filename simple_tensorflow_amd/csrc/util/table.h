// LevelDB-format sorted string table writer/reader — the checkpoint index
// format (capability analog of the reference's core/lib/io/{table,block,
// format}; byte-compatible: block-based layout, restart arrays, 5-byte block
// trailer (type + masked crc32c), 48-byte footer with magic
// 0xdb4775248b80fb57). Writer emits uncompressed blocks with restart interval
// 16 (prefix compression like the reference); reader handles any compliant
// uncompressed table.
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "core/base.h"

namespace stf {
namespace table {

// ---- crc32c (Castagnoli) + the LevelDB mask ----
uint32_t Crc32c(const char* data, size_t n);
inline uint32_t Crc32cExtend(uint32_t crc, const char* data, size_t n);
inline uint32_t MaskCrc(uint32_t crc) {
  return ((crc >> 15) | (crc << 17)) + 0xa282ead8u;
}
inline uint32_t UnmaskCrc(uint32_t masked) {
  uint32_t rot = masked - 0xa282ead8u;
  return (rot >> 17) | (rot << 15);
}

// Appends all (key, value) pairs (sorted by key) as a table to `out`.
Status BuildTable(const std::map<std::string, std::string>& entries,
                  std::string* out);

// Parses a table file's bytes back into a key->value map.
Status ReadTable(const std::string& data,
                 std::map<std::string, std::string>* out);

}  // namespace table
}  // namespace stf
