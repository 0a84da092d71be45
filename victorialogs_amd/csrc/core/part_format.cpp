#include "part_format.h"

#include <cinttypes>
#include <cstdio>

#include "zstd_wrap.h"

namespace vl {

void TimestampsHeader::marshal(bytes& dst) const {
  // block_header.go:990-997
  put_u64be(dst, block_offset);
  put_u64be(dst, block_size);
  put_u64be(dst, uint64_t(min_timestamp));
  put_u64be(dst, uint64_t(max_timestamp));
  dst.push_back(uint8_t(marshal_type));
}

size_t TimestampsHeader::unmarshal(const uint8_t* src, size_t n) {
  // block_header.go:1000-1014
  if (n < 33) fail("timestampsHeader: need 33 bytes");
  block_offset = get_u64be(src);
  block_size = get_u64be(src + 8);
  min_timestamp = int64_t(get_u64be(src + 16));
  max_timestamp = int64_t(get_u64be(src + 24));
  marshal_type = MarshalType(src[32]);
  return 33;
}

void ColumnHeader::marshal(bytes& dst) const {
  // block_header.go:634-712 (name goes to columnsHeaderIndex, not here)
  dst.push_back(uint8_t(type));
  switch (type) {
    case ValueType::String:
      break;
    case ValueType::Dict: {
      // valuesDict.marshal (values_encoder.go:1289-1297)
      dst.push_back(uint8_t(dict.size()));
      for (const auto& v : dict) put_len_prefixed(dst, v);
      break;
    }
    case ValueType::Uint8:
      dst.push_back(uint8_t(min_value));
      dst.push_back(uint8_t(max_value));
      break;
    case ValueType::Uint16:
      put_u16be(dst, uint16_t(min_value));
      put_u16be(dst, uint16_t(max_value));
      break;
    case ValueType::Uint32:
    case ValueType::IPv4:
      put_u32be(dst, uint32_t(min_value));
      put_u32be(dst, uint32_t(max_value));
      break;
    case ValueType::Uint64:
    case ValueType::Float64:
    case ValueType::TimestampISO8601:
      put_u64be(dst, min_value);
      put_u64be(dst, max_value);
      break;
    case ValueType::Int64:
      put_i64be_zigzag(dst, int64_t(min_value));
      put_i64be_zigzag(dst, int64_t(max_value));
      break;
    default:
      fail("columnHeader.marshal: unknown valueType");
  }
  put_uvarint(dst, values_offset);
  put_uvarint(dst, values_size);
  if (type != ValueType::String && type != ValueType::Dict) {
    put_uvarint(dst, bloom_offset);
    put_uvarint(dst, bloom_size);
  } else if (type == ValueType::String) {
    put_uvarint(dst, bloom_offset);
    put_uvarint(dst, bloom_size);
  }
}

size_t ColumnHeader::unmarshal(const uint8_t* src, size_t n) {
  // block_header.go:735-888, partFormatVersion >= 1 (no inline name)
  const uint8_t* p = src;
  size_t left = n;
  auto take = [&](size_t k) {
    if (left < k) fail("columnHeader: truncated");
    const uint8_t* q = p;
    p += k;
    left -= k;
    return q;
  };
  auto take_uvarint = [&]() {
    uint64_t v;
    int sz = get_uvarint(p, left, &v);
    if (sz <= 0) fail("columnHeader: bad varint");
    p += sz;
    left -= sz;
    return v;
  };

  dict.clear();
  type = ValueType(*take(1));
  switch (type) {
    case ValueType::String:
      break;
    case ValueType::Dict: {
      // valuesDict.unmarshalInplace (values_encoder.go:1302-1322)
      uint8_t cnt = *take(1);
      for (int i = 0; i < cnt; i++) {
        uint64_t len = take_uvarint();
        const uint8_t* q = take(size_t(len));
        dict.emplace_back((const char*)q, size_t(len));
      }
      break;
    }
    case ValueType::Uint8: {
      const uint8_t* q = take(2);
      min_value = q[0];
      max_value = q[1];
      break;
    }
    case ValueType::Uint16: {
      const uint8_t* q = take(4);
      min_value = get_u16be(q);
      max_value = get_u16be(q + 2);
      break;
    }
    case ValueType::Uint32:
    case ValueType::IPv4: {
      const uint8_t* q = take(8);
      min_value = get_u32be(q);
      max_value = get_u32be(q + 4);
      break;
    }
    case ValueType::Uint64:
    case ValueType::Float64:
    case ValueType::TimestampISO8601: {
      const uint8_t* q = take(16);
      min_value = get_u64be(q);
      max_value = get_u64be(q + 8);
      break;
    }
    case ValueType::Int64: {
      const uint8_t* q = take(16);
      min_value = uint64_t(get_i64be_zigzag(q));
      max_value = uint64_t(get_i64be_zigzag(q + 8));
      break;
    }
    default:
      fail("columnHeader: unexpected valueType");
  }
  values_offset = take_uvarint();
  values_size = take_uvarint();
  if (type != ValueType::Dict) {
    bloom_offset = take_uvarint();
    bloom_size = take_uvarint();
  } else {
    bloom_offset = 0;
    bloom_size = 0;
  }
  return n - left;
}

static void marshal_refs(bytes& dst, const std::vector<ColumnHeaderRef>& refs) {
  // block_header.go:306-313
  put_uvarint(dst, refs.size());
  for (const auto& r : refs) {
    put_uvarint(dst, r.column_name_id);
    put_uvarint(dst, r.offset);
  }
}

static size_t unmarshal_refs(std::vector<ColumnHeaderRef>& dst, const uint8_t* src,
                             size_t n) {
  const uint8_t* p = src;
  size_t left = n;
  uint64_t cnt;
  int sz = get_uvarint(p, left, &cnt);
  if (sz <= 0) fail("columnHeaderRefs: bad count");
  p += sz;
  left -= sz;
  for (uint64_t i = 0; i < cnt; i++) {
    ColumnHeaderRef r;
    sz = get_uvarint(p, left, &r.column_name_id);
    if (sz <= 0) fail("columnHeaderRefs: bad nameID");
    p += sz;
    left -= sz;
    sz = get_uvarint(p, left, &r.offset);
    if (sz <= 0) fail("columnHeaderRefs: bad offset");
    p += sz;
    left -= sz;
    dst.push_back(r);
  }
  return n - left;
}

void ColumnsHeaderIndex::marshal(bytes& dst) const {
  // block_header.go:275-279
  marshal_refs(dst, column_headers_refs);
  marshal_refs(dst, const_columns_refs);
}

void ColumnsHeaderIndex::unmarshal(const uint8_t* src, size_t n) {
  // block_header.go:284-304
  column_headers_refs.clear();
  const_columns_refs.clear();
  size_t used = unmarshal_refs(column_headers_refs, src, n);
  size_t used2 = unmarshal_refs(const_columns_refs, src + used, n - used);
  if (used + used2 != n) fail("columnsHeaderIndex: unexpected tail");
}

void BlockHeader::marshal(bytes& dst) const {
  // block_header.go:69-80
  stream_id.marshal(dst);
  put_uvarint(dst, uncompressed_size_bytes);
  put_uvarint(dst, rows_count);
  timestamps_header.marshal(dst);
  put_uvarint(dst, columns_header_index_offset);
  put_uvarint(dst, columns_header_index_size);
  put_uvarint(dst, columns_header_offset);
  put_uvarint(dst, columns_header_size);
}

size_t BlockHeader::unmarshal(const uint8_t* src, size_t n) {
  // block_header.go:83-159, partFormatVersion >= 1
  const uint8_t* p = src;
  size_t left = n;
  size_t used = stream_id.unmarshal(p, left);
  p += used;
  left -= used;
  auto take_uvarint = [&]() {
    uint64_t v;
    int sz = get_uvarint(p, left, &v);
    if (sz <= 0) fail("blockHeader: bad varint");
    p += sz;
    left -= sz;
    return v;
  };
  uncompressed_size_bytes = take_uvarint();
  rows_count = take_uvarint();
  used = timestamps_header.unmarshal(p, left);
  p += used;
  left -= used;
  columns_header_index_offset = take_uvarint();
  columns_header_index_size = take_uvarint();
  columns_header_offset = take_uvarint();
  columns_header_size = take_uvarint();
  return n - left;
}

void IndexBlockHeader::marshal(bytes& dst) const {
  // index_block_header.go:81-88
  stream_id.marshal(dst);
  put_u64be(dst, uint64_t(min_timestamp));
  put_u64be(dst, uint64_t(max_timestamp));
  put_u64be(dst, index_block_offset);
  put_u64be(dst, index_block_size);
}

size_t IndexBlockHeader::unmarshal(const uint8_t* src, size_t n) {
  // index_block_header.go:91-111
  size_t used = stream_id.unmarshal(src, n);
  if (n - used < 32) fail("indexBlockHeader: need 32 more bytes");
  min_timestamp = int64_t(get_u64be(src + used));
  max_timestamp = int64_t(get_u64be(src + used + 8));
  index_block_offset = get_u64be(src + used + 16);
  index_block_size = get_u64be(src + used + 24);
  return used + 32;
}

std::string PartHeader::to_json() const {
  // partHeader JSON fields (part_header.go:15-40); Go json.Marshal field names
  char buf[512];
  snprintf(buf, sizeof(buf),
           "{\"FormatVersion\":%" PRIu64 ",\"CompressedSizeBytes\":%" PRIu64
           ",\"UncompressedSizeBytes\":%" PRIu64 ",\"RowsCount\":%" PRIu64
           ",\"BlocksCount\":%" PRIu64 ",\"MinTimestamp\":%" PRId64
           ",\"MaxTimestamp\":%" PRId64 ",\"BloomValuesShardsCount\":%" PRIu64 "}",
           format_version, compressed_size_bytes, uncompressed_size_bytes, rows_count,
           blocks_count, min_timestamp, max_timestamp, bloom_values_shards_count);
  return buf;
}

static bool json_field_u64(const std::string& s, const char* name, uint64_t* out) {
  std::string pat = std::string("\"") + name + "\":";
  size_t at = s.find(pat);
  if (at == std::string::npos) return false;
  *out = strtoull(s.c_str() + at + pat.size(), nullptr, 10);
  return true;
}
static bool json_field_i64(const std::string& s, const char* name, int64_t* out) {
  std::string pat = std::string("\"") + name + "\":";
  size_t at = s.find(pat);
  if (at == std::string::npos) return false;
  *out = strtoll(s.c_str() + at + pat.size(), nullptr, 10);
  return true;
}

void PartHeader::from_json(const std::string& s) {
  format_version = 0;
  bloom_values_shards_count = 0;
  json_field_u64(s, "FormatVersion", &format_version);
  json_field_u64(s, "CompressedSizeBytes", &compressed_size_bytes);
  json_field_u64(s, "UncompressedSizeBytes", &uncompressed_size_bytes);
  json_field_u64(s, "RowsCount", &rows_count);
  json_field_u64(s, "BlocksCount", &blocks_count);
  json_field_i64(s, "MinTimestamp", &min_timestamp);
  json_field_i64(s, "MaxTimestamp", &max_timestamp);
  json_field_u64(s, "BloomValuesShardsCount", &bloom_values_shards_count);
  // part_header.go:74-81: v1 implies 8 shards; v<=1 must have 0 in JSON
  if (format_version == 1) bloom_values_shards_count = 8;
  if (format_version > 3) fail("unsupported part FormatVersion");
}

bytes marshal_column_names(const std::vector<std::string>& names) {
  // column_names.go:101-109
  bytes data;
  put_uvarint(data, names.size());
  for (const auto& nm : names) put_len_prefixed(data, nm);
  bytes dst;
  zstd_compress(dst, data.data(), data.size(), 1);
  return dst;
}

std::vector<std::string> unmarshal_column_names(const uint8_t* src, size_t n) {
  // column_names.go:111-134
  bytes data;
  zstd_decompress(data, src, n);
  const uint8_t* p = data.data();
  size_t left = data.size();
  uint64_t cnt;
  int sz = get_uvarint(p, left, &cnt);
  if (sz <= 0) fail("column_names: bad count");
  p += sz;
  left -= sz;
  std::vector<std::string> names;
  for (uint64_t i = 0; i < cnt; i++) {
    uint64_t len;
    sz = get_uvarint(p, left, &len);
    if (sz <= 0 || left - sz < len) fail("column_names: bad name");
    p += sz;
    left -= sz;
    names.emplace_back((const char*)p, size_t(len));
    p += len;
    left -= len;
  }
  return names;
}

bytes marshal_column_idxs(const std::vector<std::pair<uint64_t, uint64_t>>& idxs) {
  // column_names.go:33-40
  bytes dst;
  put_uvarint(dst, idxs.size());
  for (const auto& kv : idxs) {
    put_uvarint(dst, kv.first);
    put_uvarint(dst, kv.second);
  }
  return dst;
}

std::map<uint64_t, uint64_t> unmarshal_column_idxs(const uint8_t* src, size_t n) {
  // column_names.go:42-81
  const uint8_t* p = src;
  size_t left = n;
  uint64_t cnt;
  int sz = get_uvarint(p, left, &cnt);
  if (sz <= 0) fail("column_idxs: bad count");
  p += sz;
  left -= sz;
  std::map<uint64_t, uint64_t> m;
  for (uint64_t i = 0; i < cnt; i++) {
    uint64_t id, shard;
    sz = get_uvarint(p, left, &id);
    if (sz <= 0) fail("column_idxs: bad id");
    p += sz;
    left -= sz;
    sz = get_uvarint(p, left, &shard);
    if (sz <= 0) fail("column_idxs: bad shard");
    p += sz;
    left -= sz;
    m[id] = shard;
  }
  if (left != 0) fail("column_idxs: unexpected tail");
  return m;
}

}  // namespace vl
