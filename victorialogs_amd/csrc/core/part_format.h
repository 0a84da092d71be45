// On-disk part format restating lib/logstorage (FormatVersion 3):
//   metadata.json        partHeader as JSON (part_header.go:15-80)
//   metaindex.bin        zstd frame of indexBlockHeaders (index_block_header.go:81-88)
//   index.bin            zstd frames of blockHeaders (block_header.go:69-80)
//   columns_header_index.bin  raw columnHeaderRefs (block_header.go:275-313)
//   columns_header.bin   raw columnHeaders / const Fields (block_header.go:634-730)
//   column_names.bin     zstd frame of names (column_names.go:101-134)
//   column_idxs.bin      raw columnID->shard map (column_names.go:33-40, v3)
//   timestamps.bin       VM int64 codec blocks (block.go:676-692)
//   message_values.bin / message_bloom.bin      the "" (_msg) column
//   values.bin<N> / bloom.bin<N>                other columns, sharded
// See SURVEY.md Appendix A for the verified layouts.
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "codec.h"
#include "values.h"
#include "vl_base.h"

namespace vl {

struct StreamID {
  uint32_t account_id = 0;  // tenant_id.go:54-58 (BE u32)
  uint32_t project_id = 0;
  uint64_t id_hi = 0;  // u128.go:60-64 (BE u64)
  uint64_t id_lo = 0;

  void marshal(bytes& dst) const {
    put_u32be(dst, account_id);
    put_u32be(dst, project_id);
    put_u64be(dst, id_hi);
    put_u64be(dst, id_lo);
  }
  size_t unmarshal(const uint8_t* src, size_t n) {
    if (n < 24) fail("streamID: need 24 bytes");
    account_id = get_u32be(src);
    project_id = get_u32be(src + 4);
    id_hi = get_u64be(src + 8);
    id_lo = get_u64be(src + 16);
    return 24;
  }
};

struct TimestampsHeader {
  // block_header.go:954-1014
  uint64_t block_offset = 0;
  uint64_t block_size = 0;
  int64_t min_timestamp = 0;
  int64_t max_timestamp = 0;
  MarshalType marshal_type = MarshalType::Const;

  void marshal(bytes& dst) const;
  size_t unmarshal(const uint8_t* src, size_t n);
};

struct ColumnHeader {
  // block_header.go:584-615
  std::string name;
  ValueType type = ValueType::String;
  uint64_t min_value = 0;
  uint64_t max_value = 0;
  std::vector<std::string> dict;  // valueTypeDict only
  uint64_t values_offset = 0;
  uint64_t values_size = 0;
  uint64_t bloom_offset = 0;  // absent for dict
  uint64_t bloom_size = 0;

  void marshal(bytes& dst) const;                     // block_header.go:634-712
  size_t unmarshal(const uint8_t* src, size_t n);     // block_header.go:735-888 (v1+)
};

struct ColumnHeaderRef {
  uint64_t column_name_id = 0;
  uint64_t offset = 0;
};

struct ColumnsHeaderIndex {
  // block_header.go:233-240
  std::vector<ColumnHeaderRef> column_headers_refs;
  std::vector<ColumnHeaderRef> const_columns_refs;

  void marshal(bytes& dst) const;
  void unmarshal(const uint8_t* src, size_t n);
};

struct BlockHeader {
  // block_header.go:17-41
  StreamID stream_id;
  uint64_t uncompressed_size_bytes = 0;
  uint64_t rows_count = 0;
  TimestampsHeader timestamps_header;
  uint64_t columns_header_index_offset = 0;
  uint64_t columns_header_index_size = 0;
  uint64_t columns_header_offset = 0;
  uint64_t columns_header_size = 0;

  void marshal(bytes& dst) const;
  size_t unmarshal(const uint8_t* src, size_t n);  // v1+ layout
};

struct IndexBlockHeader {
  // index_block_header.go
  StreamID stream_id;
  int64_t min_timestamp = 0;
  int64_t max_timestamp = 0;
  uint64_t index_block_offset = 0;
  uint64_t index_block_size = 0;

  void marshal(bytes& dst) const;
  size_t unmarshal(const uint8_t* src, size_t n);
};

struct PartHeader {
  // part_header.go:15-40 (JSON)
  uint64_t format_version = 3;
  uint64_t compressed_size_bytes = 0;
  uint64_t uncompressed_size_bytes = 0;
  uint64_t rows_count = 0;
  uint64_t blocks_count = 0;
  int64_t min_timestamp = 0;
  int64_t max_timestamp = 0;
  uint64_t bloom_values_shards_count = 1;

  std::string to_json() const;
  void from_json(const std::string& s);
};

// column_names.bin (column_names.go:101-134)
bytes marshal_column_names(const std::vector<std::string>& names);
std::vector<std::string> unmarshal_column_names(const uint8_t* src, size_t n);

// column_idxs.bin (column_names.go:33-40)
bytes marshal_column_idxs(const std::vector<std::pair<uint64_t, uint64_t>>& idxs);
std::map<uint64_t, uint64_t> unmarshal_column_idxs(const uint8_t* src, size_t n);

// const column Field record, value-only for v1+ (rows.go:35-66 marshal(dst,false))
struct ConstColumn {
  std::string name;
  std::string value;
};

}  // namespace vl
