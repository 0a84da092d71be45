// Part writer: creates reference-format (FormatVersion 3) parts on disk.
//
// Restates the writer subset of lib/logstorage: block.mustWriteTo
// (block.go:457-480), column.mustWriteTo (block.go:131-175),
// columnsHeader.mustWriteTo (block_header.go:425-452), timestamps write
// (block.go:676-692), index/metaindex assembly (block_stream_writer.go,
// index_block_header.go:113-121).  Used for data generation (SURVEY.md §2
// marks the writer subset in-scope as a data-gen/oracle dependency) — the
// query hot path only reads parts.
#pragma once

#include <cstdio>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "part_format.h"

namespace vl {

// One input column for a block: name ("" = _msg) and one value per row.
struct InputColumn {
  std::string name;
  std::vector<std::string> values;
};

class PartWriter {
 public:
  // shards: BloomValuesShardsCount for non-message columns (we write 1).
  explicit PartWriter(const std::string& dir, uint64_t shards = 1);
  ~PartWriter();

  // Adds one block.  timestamps must be sorted (block.go:343-350); all
  // columns must have timestamps.size() values.  Blocks must be added in
  // (streamID, minTimestamp) order (block_header.go:198-215).
  void add_block(const StreamID& sid, const std::vector<int64_t>& timestamps,
                 std::vector<InputColumn>& columns);

  void finish();  // writes metaindex/column_names/column_idxs/metadata.json

 private:
  struct FileW {
    FILE* f = nullptr;
    uint64_t bytes_written = 0;
    void open(const std::string& path);
    void write(const uint8_t* p, size_t n);
    void close();
  };

  uint64_t shard_for_column(const std::string& name);
  uint64_t column_name_id(const std::string& name);
  void flush_index_block();

  std::string dir_;
  uint64_t shards_;
  FileW index_, columns_header_index_, columns_header_, timestamps_;
  FileW message_values_, message_bloom_;
  std::vector<std::unique_ptr<FileW>> shard_values_, shard_bloom_;

  std::map<std::string, uint64_t> name_ids_;
  std::vector<std::string> names_;
  std::map<uint64_t, uint64_t> column_shards_;  // nameID -> shard
  uint64_t next_shard_ = 0;

  bytes index_block_buf_;  // marshaled blockHeaders pending compression
  StreamID cur_stream_id_;
  int64_t ib_min_ts_ = 0, ib_max_ts_ = 0;
  bool ib_has_blocks_ = false;
  std::vector<IndexBlockHeader> metaindex_;

  PartHeader ph_;
  bool finished_ = false;
};

}  // namespace vl
