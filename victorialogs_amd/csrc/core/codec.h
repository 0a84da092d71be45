// Block codecs restating lib/logstorage/encoding.go (strings / uint64 / bytes
// blocks) and the VictoriaMetrics int64 timestamps codec
// (vendor/.../lib/encoding/{encoding,nearest_delta,nearest_delta2}.go).
#pragma once

#include <cstdint>
#include <vector>

#include "vl_base.h"

namespace vl {

// ---- bytes block (encoding.go:338-426) ----
// type 0 = plain (<128 bytes, 1-byte length); type 1 = zstd (varuint complen).
void marshal_bytes_block(bytes& dst, const uint8_t* src, size_t n);
// Appends decoded bytes to dst; returns bytes consumed from src.
size_t unmarshal_bytes_block(bytes& dst, const uint8_t* src, size_t n);

// ---- uint64 items / block (encoding.go:148-336) ----
void marshal_uint64_items(bytes& dst, const uint64_t* a, size_t n);
void marshal_uint64_block(bytes& dst, const uint64_t* a, size_t n);
void unmarshal_uint64_items(std::vector<uint64_t>& dst, const uint8_t* src, size_t n,
                            uint64_t items_count);
size_t unmarshal_uint64_block(std::vector<uint64_t>& dst, const uint8_t* src, size_t n,
                              uint64_t items_count);

// ---- strings block (encoding.go:16-133) ----
void marshal_strings_block(bytes& dst, const std::vector<strview>& a);

// Decoded strings block: concatenated row bytes + offsets (n+1 entries).
// For the const-string special case (encoding.go:113-120) is_const is set and
// data holds the single value with offsets {0, len}.
struct StringsBlockDec {
  bytes data;
  std::vector<uint32_t> offsets;
  bool is_const = false;
  uint64_t rows = 0;

  strview row(uint64_t i) const {
    if (is_const) return strview((const char*)data.data(), data.size());
    return strview((const char*)data.data() + offsets[i], offsets[i + 1] - offsets[i]);
  }
};
void unmarshal_strings_block(StringsBlockDec& dst, const uint8_t* src, size_t n,
                             uint64_t items_count);

// ---- VM int64 array codec (vendor/.../lib/encoding/encoding.go:119-250) ----
// marshal types (encoding.go:20-43)
enum class MarshalType : uint8_t {
  ZSTDNearestDelta2 = 1,
  DeltaConst = 2,
  Const = 3,
  ZSTDNearestDelta = 4,
  NearestDelta2 = 5,
  NearestDelta = 6,
};

// marshalInt64Array with precisionBits=64 (lossless; block.go:682 passes 64).
// Returns the marshal type and first value.
MarshalType marshal_int64_array(bytes& dst, const int64_t* a, size_t n,
                                int64_t* first_value);
void unmarshal_int64_array(std::vector<int64_t>& dst, const uint8_t* src, size_t n,
                           MarshalType mt, int64_t first_value, uint64_t items_count);

// MarshalVarInt64s / UnmarshalVarInt64s (vendor/.../lib/encoding/int.go:107-284)
void marshal_varint64s(bytes& dst, const int64_t* a, size_t n);
void unmarshal_varint64s(std::vector<int64_t>& dst, const uint8_t* src, size_t n,
                         size_t items);

}  // namespace vl
