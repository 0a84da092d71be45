#include "codec.h"

#include "zstd_wrap.h"

namespace vl {

// getCompressLevel for bytes blocks (lib/logstorage/encoding.go:362-370),
// keyed by data length in bytes.
static int bytes_block_compress_level(size_t data_len) {
  if (data_len <= 512) return 1;
  if (data_len <= 4 * 1024) return 2;
  return 3;
}

void marshal_bytes_block(bytes& dst, const uint8_t* src, size_t n) {
  // encoding.go:343-360
  if (n < 128) {
    dst.push_back(0);  // marshalBytesTypePlain
    dst.push_back(uint8_t(n));
    dst.insert(dst.end(), src, src + n);
    return;
  }
  dst.push_back(1);  // marshalBytesTypeZSTD
  bytes comp;
  zstd_compress(comp, src, n, bytes_block_compress_level(n));
  put_uvarint(dst, comp.size());
  dst.insert(dst.end(), comp.begin(), comp.end());
}

size_t unmarshal_bytes_block(bytes& dst, const uint8_t* src, size_t n) {
  // encoding.go:372-426
  if (n < 1) fail("bytes block: empty src");
  uint8_t block_type = src[0];
  size_t pos = 1;
  if (block_type == 0) {
    if (pos >= n) fail("bytes block: no plain size");
    size_t block_len = src[pos++];
    if (n - pos < block_len) fail("bytes block: truncated plain block");
    dst.insert(dst.end(), src + pos, src + pos + block_len);
    return pos + block_len;
  }
  if (block_type == 1) {
    uint64_t block_len;
    int sz = get_uvarint(src + pos, n - pos, &block_len);
    if (sz <= 0) fail("bytes block: bad compressed size");
    pos += sz;
    if (n - pos < block_len) fail("bytes block: truncated compressed block");
    zstd_decompress(dst, src + pos, size_t(block_len));
    return pos + size_t(block_len);
  }
  fail("bytes block: unexpected block type");
}

// encoding.go:177-187
enum : uint8_t {
  kU8 = 0, kU16 = 1, kU32 = 2, kU64 = 3,
  kConst8 = 4, kConst16 = 5, kConst32 = 6, kConst64 = 7,
};

static bool are_const_u64(const uint64_t* a, size_t n) {
  if (n == 0) return false;
  for (size_t i = 1; i < n; i++) {
    if (a[i] != a[0]) return false;
  }
  return true;
}

void marshal_uint64_items(bytes& dst, const uint64_t* a, size_t n) {
  // encoding.go:190-243
  uint64_t nmax = 0;
  for (size_t i = 0; i < n; i++) nmax = a[i] > nmax ? a[i] : nmax;
  bool consts = n >= 2 && are_const_u64(a, n);
  if (nmax < (1 << 8)) {
    if (consts) {
      dst.push_back(kConst8);
      dst.push_back(uint8_t(a[0]));
    } else {
      dst.push_back(kU8);
      for (size_t i = 0; i < n; i++) dst.push_back(uint8_t(a[i]));
    }
  } else if (nmax < (1 << 16)) {
    if (consts) {
      dst.push_back(kConst16);
      put_u16be(dst, uint16_t(a[0]));
    } else {
      dst.push_back(kU16);
      for (size_t i = 0; i < n; i++) put_u16be(dst, uint16_t(a[i]));
    }
  } else if (nmax < (uint64_t(1) << 32)) {
    if (consts) {
      dst.push_back(kConst32);
      put_u32be(dst, uint32_t(a[0]));
    } else {
      dst.push_back(kU32);
      for (size_t i = 0; i < n; i++) put_u32be(dst, uint32_t(a[i]));
    }
  } else {
    if (consts) {
      dst.push_back(kConst64);
      put_u64be(dst, a[0]);
    } else {
      dst.push_back(kU64);
      for (size_t i = 0; i < n; i++) put_u64be(dst, a[i]);
    }
  }
}

void unmarshal_uint64_items(std::vector<uint64_t>& dst, const uint8_t* src, size_t n,
                            uint64_t items_count) {
  // encoding.go:246-336
  if (n < 1) fail("uint64 items: empty src");
  uint8_t bt = src[0];
  src++;
  n--;
  size_t base = dst.size();
  dst.resize(base + items_count);
  switch (bt) {
    case kU8:
      if (n != items_count) fail("uint64 items: bad u8 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = src[i];
      break;
    case kU16:
      if (n != 2 * items_count) fail("uint64 items: bad u16 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u16be(src + 2 * i);
      break;
    case kU32:
      if (n != 4 * items_count) fail("uint64 items: bad u32 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u32be(src + 4 * i);
      break;
    case kU64:
      if (n != 8 * items_count) fail("uint64 items: bad u64 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u64be(src + 8 * i);
      break;
    case kConst8:
      if (n != 1) fail("uint64 items: bad const8 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = src[0];
      break;
    case kConst16:
      if (n != 2) fail("uint64 items: bad const16 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u16be(src);
      break;
    case kConst32:
      if (n != 4) fail("uint64 items: bad const32 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u32be(src);
      break;
    case kConst64:
      if (n != 8) fail("uint64 items: bad const64 block length");
      for (uint64_t i = 0; i < items_count; i++) dst[base + i] = get_u64be(src);
      break;
    default:
      fail("uint64 items: unexpected block type");
  }
}

void marshal_uint64_block(bytes& dst, const uint64_t* a, size_t n) {
  // encoding.go:149-155
  bytes tmp;
  marshal_uint64_items(tmp, a, n);
  marshal_bytes_block(dst, tmp.data(), tmp.size());
}

size_t unmarshal_uint64_block(std::vector<uint64_t>& dst, const uint8_t* src, size_t n,
                              uint64_t items_count) {
  // encoding.go:158-175
  bytes tmp;
  size_t consumed = unmarshal_bytes_block(tmp, src, n);
  unmarshal_uint64_items(dst, tmp.data(), tmp.size(), items_count);
  return consumed;
}

void marshal_strings_block(bytes& dst, const std::vector<strview>& a) {
  // encoding.go:16-50
  std::vector<uint64_t> lens(a.size());
  size_t total = 0;
  for (size_t i = 0; i < a.size(); i++) {
    lens[i] = a[i].n;
    total += a[i].n;
  }
  marshal_uint64_block(dst, lens.data(), lens.size());

  bool consts = !a.empty();
  for (size_t i = 1; i < a.size() && consts; i++) consts = a[i] == a[0];
  if (consts) {
    // areConstValues special case (encoding.go:29-32)
    marshal_bytes_block(dst, (const uint8_t*)a[0].p, a[0].n);
    return;
  }
  bytes concat;
  concat.reserve(total);
  for (const auto& s : a) concat.insert(concat.end(), s.p, s.p + s.n);
  marshal_bytes_block(dst, concat.data(), concat.size());
}

void unmarshal_strings_block(StringsBlockDec& dst, const uint8_t* src, size_t n,
                             uint64_t items_count) {
  // stringsBlockUnmarshaler.unmarshal (encoding.go:83-133)
  dst.data.clear();
  dst.offsets.clear();
  dst.is_const = false;
  dst.rows = items_count;

  std::vector<uint64_t> lens;
  size_t consumed = unmarshal_uint64_block(lens, src, n, items_count);
  src += consumed;
  n -= consumed;

  size_t consumed2 = unmarshal_bytes_block(dst.data, src, n);
  if (consumed2 != n) fail("strings block: unexpected tail after bytes block");

  // const-string special case (encoding.go:113-120)
  if (lens.size() >= 2 && are_const_u64(lens.data(), lens.size()) &&
      uint64_t(dst.data.size()) == lens[0]) {
    dst.is_const = true;
    dst.offsets = {0, uint32_t(dst.data.size())};
    return;
  }

  dst.offsets.resize(items_count + 1);
  uint64_t off = 0;
  for (uint64_t i = 0; i < items_count; i++) {
    dst.offsets[i] = uint32_t(off);
    off += lens[i];
    if (off > dst.data.size()) fail("strings block: row length exceeds data");
  }
  dst.offsets[items_count] = uint32_t(off);
}

// ---- VM int64 array codec ----

static bool is_const_i64(const int64_t* a, size_t n) {
  // vendor/.../lib/encoding/encoding.go:289-308
  if (n == 0) return false;
  for (size_t i = 1; i < n; i++) {
    if (a[i] != a[0]) return false;
  }
  return true;
}

static bool is_delta_const_i64(const int64_t* a, size_t n) {
  // encoding.go:311-324
  if (n < 2) return false;
  int64_t d1 = a[1] - a[0];
  for (size_t i = 2; i < n; i++) {
    if (a[i] - a[i - 1] != d1) return false;
  }
  return true;
}

static bool is_gauge_i64(const int64_t* a, size_t n) {
  // encoding.go:331-369
  if (n < 2) return false;
  int resets = 0;
  int64_t prev = a[0];
  if (prev < 0) return true;
  for (size_t i = 1; i < n; i++) {
    int64_t v = a[i];
    if (v < prev) {
      if (v < 0) return true;
      if (v > (prev >> 3)) return true;
      resets++;
    }
    prev = v;
  }
  if (resets <= 2) return false;
  return resets > int(n >> 3);
}

static int int64_array_compress_level(size_t items_count) {
  // encoding.go:371-385 (keyed by items count)
  if (items_count <= (1u << 6)) return 1;
  if (items_count <= (1u << 8)) return 2;
  if (items_count <= (1u << 10)) return 3;
  if (items_count <= (1u << 12)) return 4;
  return 5;
}

void marshal_varint64s(bytes& dst, const int64_t* a, size_t n) {
  for (size_t i = 0; i < n; i++) put_varint64(dst, a[i]);
}

void unmarshal_varint64s(std::vector<int64_t>& dst, const uint8_t* src, size_t n,
                         size_t items) {
  size_t base = dst.size();
  dst.resize(base + items);
  size_t pos = 0;
  for (size_t i = 0; i < items; i++) {
    int64_t v;
    int sz = get_varint64(src + pos, n - pos, &v);
    if (sz <= 0) fail("varint64s: truncated");
    pos += sz;
    dst[base + i] = v;
  }
  if (pos != n) fail("varint64s: unexpected tail");
}

MarshalType marshal_int64_array(bytes& dst, const int64_t* a, size_t n,
                                int64_t* first_value) {
  // marshalInt64Array (encoding.go:119-171) at precisionBits=64 (block.go:682)
  if (n == 0) fail("marshal_int64_array: empty input");
  if (is_const_i64(a, n)) {
    *first_value = a[0];
    return MarshalType::Const;
  }
  if (is_delta_const_i64(a, n)) {
    *first_value = a[0];
    put_varint64(dst, a[1] - a[0]);
    return MarshalType::DeltaConst;
  }

  MarshalType mt;
  bytes plain;
  if (is_gauge_i64(a, n)) {
    // marshalInt64NearestDelta, precisionBits=64 fast path (nearest_delta.go:26-36)
    mt = MarshalType::ZSTDNearestDelta;
    *first_value = a[0];
    std::vector<int64_t> deltas(n - 1);
    for (size_t i = 1; i < n; i++) deltas[i - 1] = a[i] - a[i - 1];
    marshal_varint64s(plain, deltas.data(), deltas.size());
  } else {
    // marshalInt64NearestDelta2, precisionBits=64 fast path (nearest_delta2.go:24-37)
    mt = MarshalType::ZSTDNearestDelta2;
    *first_value = a[0];
    int64_t d1 = a[1] - a[0];
    put_varint64(plain, d1);
    std::vector<int64_t> d2s(n - 2);
    int64_t v = a[1];
    for (size_t i = 2; i < n; i++) {
      int64_t d2 = a[i] - v - d1;
      d1 += d2;
      v += d1;
      d2s[i - 2] = d2;
    }
    marshal_varint64s(plain, d2s.data(), d2s.size());
  }

  // Try compressing (encoding.go:150-167); minCompressibleBlockSize=128.
  if (plain.size() >= 128) {
    size_t old = dst.size();
    zstd_compress(dst, plain.data(), plain.size(), int64_array_compress_level(n));
    if (double(dst.size() - old) <= 0.9 * double(plain.size())) {
      return mt;
    }
    dst.resize(old);
  }
  // Ineffective compression: store plain.
  dst.insert(dst.end(), plain.begin(), plain.end());
  return mt == MarshalType::ZSTDNearestDelta2 ? MarshalType::NearestDelta2
                                              : MarshalType::NearestDelta;
}

void unmarshal_int64_array(std::vector<int64_t>& dst, const uint8_t* src, size_t n,
                           MarshalType mt, int64_t first_value, uint64_t items_count) {
  // unmarshalInt64Array (encoding.go:173-250)
  switch (mt) {
    case MarshalType::Const: {
      if (n > 0) fail("int64 array: unexpected data for const");
      dst.insert(dst.end(), items_count, first_value);
      return;
    }
    case MarshalType::DeltaConst: {
      int64_t d;
      int sz = get_varint64(src, n, &d);
      if (sz <= 0 || size_t(sz) != n) fail("int64 array: bad delta const");
      int64_t v = first_value;
      for (uint64_t i = 0; i < items_count; i++) {
        dst.push_back(v);
        v += d;
      }
      return;
    }
    case MarshalType::ZSTDNearestDelta:
    case MarshalType::ZSTDNearestDelta2: {
      bytes plain;
      zstd_decompress(plain, src, n);
      MarshalType inner = mt == MarshalType::ZSTDNearestDelta
                              ? MarshalType::NearestDelta
                              : MarshalType::NearestDelta2;
      unmarshal_int64_array(dst, plain.data(), plain.size(), inner, first_value,
                            items_count);
      return;
    }
    case MarshalType::NearestDelta: {
      // unmarshalInt64NearestDelta (nearest_delta.go:56-80)
      if (items_count < 1) fail("int64 array: itemsCount must be >= 1");
      std::vector<int64_t> deltas;
      unmarshal_varint64s(deltas, src, n, items_count - 1);
      int64_t v = first_value;
      dst.push_back(v);
      for (int64_t d : deltas) {
        v += d;
        dst.push_back(v);
      }
      return;
    }
    case MarshalType::NearestDelta2: {
      // unmarshalInt64NearestDelta2 (nearest_delta2.go:57-92)
      if (items_count < 2) fail("int64 array: itemsCount must be >= 2");
      std::vector<int64_t> is;
      unmarshal_varint64s(is, src, n, items_count - 1);
      int64_t v = first_value;
      int64_t d1 = is[0];
      dst.push_back(v);
      v += d1;
      dst.push_back(v);
      for (size_t i = 1; i < is.size(); i++) {
        d1 += is[i];
        v += d1;
        dst.push_back(v);
      }
      return;
    }
    default:
      fail("int64 array: unknown marshal type");
  }
}

}  // namespace vl
