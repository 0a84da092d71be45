// Part reader: opens reference-format parts (FormatVersion 1..3) and decodes
// blocks for staging.  Restates the read side of lib/logstorage:
// mustOpenFilePart (part.go:105-173), mustReadBlockHeaders
// (block_search.go:508-533), the lazy per-block column fetches
// (block_search.go:232-506) and getBloomValuesFileForColumnName
// (part.go:195-218).
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "codec.h"
#include "part_format.h"

namespace vl {

class PartReader {
 public:
  explicit PartReader(const std::string& dir);
  ~PartReader();

  const PartHeader& header() const { return ph_; }
  const std::vector<IndexBlockHeader>& index_block_headers() const {
    return metaindex_;
  }
  const std::vector<std::string>& column_names() const { return names_; }

  // Reads + decompresses one index block and unmarshals its blockHeaders.
  void read_block_headers(const IndexBlockHeader& ih, std::vector<BlockHeader>& dst) const;
  // All block headers of the part, in order.
  std::vector<BlockHeader> read_all_block_headers() const;

  // Per-block lazy reads (block_search.go:232-506).  name is canonical
  // ("" for _msg; see log_rows.go:508-513).
  struct BlockColumns {
    ColumnsHeaderIndex index;
    bytes header_data;  // raw columns_header bytes for this block
    // parallel to index.*_refs
  };
  void read_block_columns(const BlockHeader& bh, BlockColumns& dst) const;

  // Returns true + fills ch if the block has a regular column `name`.
  bool get_column_header(const BlockColumns& bc, const std::string& name,
                         ColumnHeader* ch) const;
  // Returns true + fills value if the block has a const column `name`.
  bool get_const_column(const BlockColumns& bc, const std::string& name,
                        std::string* value) const;
  // Enumerates all (name, header) pairs for a block.
  void get_all_column_headers(const BlockColumns& bc,
                              std::vector<ColumnHeader>* chs,
                              std::vector<ConstColumn>* ccs) const;

  // Raw reads + decode
  void read_values(const ColumnHeader& ch, uint64_t rows_count,
                   StringsBlockDec& dst) const;
  void read_bloom(const ColumnHeader& ch, std::vector<uint64_t>& words) const;
  void read_timestamps(const BlockHeader& bh, std::vector<int64_t>& dst) const;

 private:
  struct FileR {
    int fd = -1;
    FileR() = default;
    FileR(const FileR&) = delete;
    FileR& operator=(const FileR&) = delete;
    FileR(FileR&& o) noexcept : fd(o.fd) { o.fd = -1; }
    ~FileR() { close(); }  // a long-lived process opens many parts
    void open(const std::string& path, bool required);
    void pread_full(uint8_t* dst, size_t n, uint64_t off) const;
    bool ok() const { return fd >= 0; }
    void close();
  };

  const FileR& values_file(const std::string& name) const;
  const FileR& bloom_file(const std::string& name) const;

  std::string dir_;
  PartHeader ph_;
  std::vector<std::string> names_;
  std::map<std::string, uint64_t> name_ids_;
  std::map<uint64_t, uint64_t> column_shards_;  // nameID -> shard (v3)
  std::vector<IndexBlockHeader> metaindex_;

  FileR index_, columns_header_index_, columns_header_, timestamps_;
  FileR message_values_, message_bloom_;
  std::vector<std::unique_ptr<FileR>> shard_values_, shard_bloom_;
};

}  // namespace vl
