// Base helpers shared by the victorialogs_amd host library and the CPU oracle.
//
// Byte-order and varint conventions restate the reference wire format:
//  - fixed-width ints are big-endian
//    (vendor/github.com/VictoriaMetrics/VictoriaMetrics/lib/encoding/int.go:12-84)
//  - int64 fixed-width values are zig-zag encoded then big-endian (int.go:69-84)
//  - varints are LEB128 little-endian base-128 (int.go:287-302, binary.Uvarint)
//  - var-int64s are zig-zag encoded varints (int.go:87-118)
#pragma once

#include <cstdint>
#include <cstring>
#include <cstdio>
#include <string>
#include <vector>
#include <stdexcept>

namespace vl {

using bytes = std::vector<uint8_t>;

struct Error : std::runtime_error {
  explicit Error(const std::string& msg) : std::runtime_error(msg) {}
};

[[noreturn]] inline void fail(const std::string& msg) { throw Error(msg); }

// ---- big-endian fixed-width (int.go:12-48) ----
inline void put_u16be(bytes& dst, uint16_t v) {
  dst.push_back(uint8_t(v >> 8));
  dst.push_back(uint8_t(v));
}
inline void put_u32be(bytes& dst, uint32_t v) {
  dst.push_back(uint8_t(v >> 24));
  dst.push_back(uint8_t(v >> 16));
  dst.push_back(uint8_t(v >> 8));
  dst.push_back(uint8_t(v));
}
inline void put_u64be(bytes& dst, uint64_t v) {
  for (int s = 56; s >= 0; s -= 8) dst.push_back(uint8_t(v >> s));
}
inline uint16_t get_u16be(const uint8_t* p) { return uint16_t(p[0]) << 8 | p[1]; }
inline uint32_t get_u32be(const uint8_t* p) {
  return uint32_t(p[0]) << 24 | uint32_t(p[1]) << 16 | uint32_t(p[2]) << 8 | p[3];
}
inline uint64_t get_u64be(const uint8_t* p) {
  uint64_t v = 0;
  for (int i = 0; i < 8; i++) v = v << 8 | p[i];
  return v;
}

// ---- zig-zag (int.go:69-84) ----
inline uint64_t zigzag_enc64(int64_t v) { return (uint64_t(v) << 1) ^ uint64_t(v >> 63); }
inline int64_t zigzag_dec64(uint64_t u) {
  return int64_t(u >> 1) ^ (int64_t(u << 63) >> 63);
}
inline void put_i64be_zigzag(bytes& dst, int64_t v) { put_u64be(dst, zigzag_enc64(v)); }
inline int64_t get_i64be_zigzag(const uint8_t* p) { return zigzag_dec64(get_u64be(p)); }

// ---- varints (int.go:287-302; Go binary.Uvarint semantics, max 10 bytes) ----
inline void put_uvarint(bytes& dst, uint64_t u) {
  while (u >= 0x80) {
    dst.push_back(uint8_t(u) | 0x80);
    u >>= 7;
  }
  dst.push_back(uint8_t(u));
}
// Returns number of bytes consumed; 0 means failure (like Go's nSize<=0).
inline int get_uvarint(const uint8_t* p, size_t n, uint64_t* out) {
  uint64_t u = 0;
  int shift = 0;
  for (size_t i = 0; i < n && i < 10; i++) {
    uint8_t c = p[i];
    if (c < 0x80) {
      if (i == 9 && c > 1) return 0;  // overflow (10th byte may only be 0 or 1)
      u |= uint64_t(c) << shift;
      *out = u;
      return int(i) + 1;
    }
    u |= uint64_t(c & 0x7f) << shift;
    shift += 7;
  }
  return 0;
}
inline void put_varint64(bytes& dst, int64_t v) { put_uvarint(dst, zigzag_enc64(v)); }
inline int get_varint64(const uint8_t* p, size_t n, int64_t* out) {
  uint64_t u;
  int sz = get_uvarint(p, n, &u);
  if (sz > 0) *out = zigzag_dec64(u);
  return sz;
}

// MarshalBytes = varuint length + bytes (int.go:506-510)
inline void put_len_prefixed(bytes& dst, const uint8_t* p, size_t n) {
  put_uvarint(dst, n);
  dst.insert(dst.end(), p, p + n);
}
inline void put_len_prefixed(bytes& dst, const std::string& s) {
  put_len_prefixed(dst, (const uint8_t*)s.data(), s.size());
}

// A cheap string view over decoded block data (rows are not NUL-terminated).
struct strview {
  const char* p = nullptr;
  size_t n = 0;
  strview() = default;
  strview(const char* p_, size_t n_) : p(p_), n(n_) {}
  explicit strview(const std::string& s) : p(s.data()), n(s.size()) {}
  bool operator==(const strview& o) const {
    return n == o.n && (n == 0 || memcmp(p, o.p, n) == 0);
  }
  bool operator==(const std::string& o) const {
    return n == o.size() && (n == 0 || memcmp(p, o.data(), n) == 0);
  }
  std::string str() const { return std::string(p, n); }
};

}  // namespace vl
