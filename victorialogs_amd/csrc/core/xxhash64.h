// Canonical XXH64 (seed 0), as used by the reference through
// github.com/cespare/xxhash/v2 v2.3.0 (vendored at
// vendor/github.com/cespare/xxhash/v2/xxhash.go).  Call sites on the hot path:
// lib/logstorage/bloomfilter.go:136,138,164 and hash_tokenizer.go:146.
//
// This is an independent implementation of the public XXH64 algorithm
// (https://cyan4973.github.io/xxHash/), pinned by the known-answer vectors in
// tests/test_hash_bloom.py and, transitively, by the reference's bloom-filter
// hex golden test (lib/logstorage/bloomfilter_test.go:105-119).
#pragma once

#include <cstdint>
#include <cstring>

// usable from HIP device code too (GPU ingest-side bloom build)
#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define VL_XX_HD __host__ __device__
#else
#define VL_XX_HD
#endif

namespace vl {

namespace xx {
constexpr uint64_t P1 = 11400714785074694791ULL;
constexpr uint64_t P2 = 14029467366897019727ULL;
constexpr uint64_t P3 = 1609587929392839161ULL;
constexpr uint64_t P4 = 9650029242287828579ULL;
constexpr uint64_t P5 = 2870177450012600261ULL;

VL_XX_HD inline uint64_t rotl(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }
VL_XX_HD inline uint64_t rd64(const uint8_t* p) {
  uint64_t v;
  memcpy(&v, p, 8);  // little-endian host
  return v;
}
VL_XX_HD inline uint32_t rd32(const uint8_t* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}
VL_XX_HD inline uint64_t round_(uint64_t acc, uint64_t input) {
  acc += input * P2;
  acc = rotl(acc, 31);
  acc *= P1;
  return acc;
}
VL_XX_HD inline uint64_t merge_round(uint64_t acc, uint64_t val) {
  val = round_(0, val);
  acc ^= val;
  acc = acc * P1 + P4;
  return acc;
}
}  // namespace xx

VL_XX_HD inline uint64_t xxhash64(const void* data, size_t len) {
  using namespace xx;
  const uint8_t* p = (const uint8_t*)data;
  const uint8_t* end = p + len;
  uint64_t h;
  if (len >= 32) {
    uint64_t v1 = P1 + P2, v2 = P2, v3 = 0, v4 = 0 - P1;
    do {
      v1 = round_(v1, rd64(p));
      v2 = round_(v2, rd64(p + 8));
      v3 = round_(v3, rd64(p + 16));
      v4 = round_(v4, rd64(p + 24));
      p += 32;
    } while (p <= end - 32);
    h = rotl(v1, 1) + rotl(v2, 7) + rotl(v3, 12) + rotl(v4, 18);
    h = merge_round(h, v1);
    h = merge_round(h, v2);
    h = merge_round(h, v3);
    h = merge_round(h, v4);
  } else {
    h = P5;
  }
  h += uint64_t(len);
  while (p + 8 <= end) {
    h ^= round_(0, rd64(p));
    h = rotl(h, 27) * P1 + P4;
    p += 8;
  }
  if (p + 4 <= end) {
    h ^= uint64_t(rd32(p)) * P1;
    h = rotl(h, 23) * P2 + P3;
    p += 4;
  }
  while (p < end) {
    h ^= uint64_t(*p) * P5;
    h = rotl(h, 11) * P1;
    p++;
  }
  h ^= h >> 33;
  h *= P2;
  h ^= h >> 29;
  h *= P3;
  h ^= h >> 32;
  return h;
}

}  // namespace vl
