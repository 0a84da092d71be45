#include "part_writer.h"

#include <sys/stat.h>

#include <algorithm>

#include "bloom.h"
#include "tokenizer.h"
#include "zstd_wrap.h"

namespace vl {

void PartWriter::FileW::open(const std::string& path) {
  f = fopen(path.c_str(), "wb");
  if (!f) fail("cannot create " + path);
}
void PartWriter::FileW::write(const uint8_t* p, size_t n) {
  if (n && fwrite(p, 1, n, f) != n) fail("short write");
  bytes_written += n;
}
void PartWriter::FileW::close() {
  if (f) {
    fclose(f);
    f = nullptr;
  }
}

PartWriter::PartWriter(const std::string& dir, uint64_t shards)
    : dir_(dir), shards_(shards) {
  mkdir(dir.c_str(), 0755);
  index_.open(dir + "/index.bin");
  columns_header_index_.open(dir + "/columns_header_index.bin");
  columns_header_.open(dir + "/columns_header.bin");
  timestamps_.open(dir + "/timestamps.bin");
  message_values_.open(dir + "/message_values.bin");
  message_bloom_.open(dir + "/message_bloom.bin");
  for (uint64_t i = 0; i < shards_; i++) {
    auto v = std::make_unique<FileW>();
    v->open(dir + "/values.bin" + std::to_string(i));
    shard_values_.push_back(std::move(v));
    auto b = std::make_unique<FileW>();
    b->open(dir + "/bloom.bin" + std::to_string(i));
    shard_bloom_.push_back(std::move(b));
  }
  ph_.format_version = 3;
  ph_.bloom_values_shards_count = shards_;
}

PartWriter::~PartWriter() {
  if (!finished_) {
    try {
      finish();
    } catch (...) {
    }
  }
}

uint64_t PartWriter::column_name_id(const std::string& name) {
  auto it = name_ids_.find(name);
  if (it != name_ids_.end()) return it->second;
  uint64_t id = names_.size();
  names_.push_back(name);
  name_ids_[name] = id;
  return id;
}

uint64_t PartWriter::shard_for_column(const std::string& name) {
  // column -> (bloom,values) shard assignment, recorded in column_idxs.bin
  // (part.go:195-218 for the read side).  Round-robin like the reference's
  // v3 writer; with shards_=1 everything lands in shard 0.
  uint64_t id = column_name_id(name);
  auto it = column_shards_.find(id);
  if (it != column_shards_.end()) return it->second;
  uint64_t shard = next_shard_ % shards_;
  next_shard_++;
  column_shards_[id] = shard;
  return shard;
}

// estimatedJSONFieldLen analog for uncompressedSizeBytes metadata
// (block.go:48-91); approximate — nothing on the scan hot path reads it.
static uint64_t est_row_size(const std::vector<InputColumn>& cols, size_t row) {
  uint64_t n = 3 + 10 + 35;  // {}\n + "_time":"" + RFC3339Nano
  for (const auto& c : cols) {
    const std::string& v = c.values[row];
    if (v.empty()) continue;
    n += 7 + (c.name.empty() ? 4 : c.name.size()) + v.size();
  }
  return n;
}

void PartWriter::add_block(const StreamID& sid, const std::vector<int64_t>& timestamps,
                           std::vector<InputColumn>& columns) {
  if (timestamps.empty()) return;

  BlockHeader bh;
  bh.stream_id = sid;
  bh.rows_count = timestamps.size();
  for (size_t r = 0; r < timestamps.size(); r++) {
    bh.uncompressed_size_bytes += est_row_size(columns, r);
  }

  // ---- timestamps (block.go:676-692, precisionBits=64) ----
  {
    bytes ts;
    int64_t first;
    MarshalType mt = marshal_int64_array(ts, timestamps.data(), timestamps.size(), &first);
    TimestampsHeader& th = bh.timestamps_header;
    th.marshal_type = mt;
    th.min_timestamp = first;
    th.max_timestamp = timestamps.back();
    th.block_offset = timestamps_.bytes_written;
    th.block_size = ts.size();
    timestamps_.write(ts.data(), ts.size());
  }

  // ---- split into const and regular columns; sort by name (block.go:304-320,379-397)
  std::sort(columns.begin(), columns.end(),
            [](const InputColumn& a, const InputColumn& b) { return a.name < b.name; });
  std::vector<ConstColumn> const_cols;
  std::vector<InputColumn*> reg_cols;
  for (auto& c : columns) {
    bool is_const = true;
    for (size_t i = 1; i < c.values.size() && is_const; i++) {
      is_const = c.values[i] == c.values[0];
    }
    if (is_const && c.values[0].size() <= 256 /* maxConstColumnValueSize */) {
      const_cols.push_back({c.name, c.values[0]});
    } else {
      reg_cols.push_back(&c);
    }
  }

  // ---- per-column values + bloom (block.go:131-175) ----
  std::vector<ColumnHeader> chs;
  for (InputColumn* c : reg_cols) {
    ColumnHeader ch;
    ch.name = c->name;
    FileW& values_f = c->name.empty() ? message_values_
                                      : *shard_values_[shard_for_column(c->name)];
    FileW& bloom_f = c->name.empty() ? message_bloom_
                                     : *shard_bloom_[shard_for_column(c->name)];

    EncodedColumn ec;
    encode_values(ec, c->values);
    ch.type = ec.type;
    ch.min_value = ec.min_value;
    ch.max_value = ec.max_value;
    ch.dict = ec.dict;

    bytes vblock;
    marshal_strings_block(vblock, ec.values);
    ch.values_offset = values_f.bytes_written;
    ch.values_size = vblock.size();
    values_f.write(vblock.data(), vblock.size());

    if (ch.type != ValueType::Dict) {
      // tokenizeHashes over the ORIGINAL values (block.go:160-162)
      std::vector<strview> vs;
      vs.reserve(c->values.size());
      for (const auto& v : c->values) vs.push_back(strview(v));
      bytes bf = bloom_marshal_hashes(tokenize_hashes(vs));
      ch.bloom_offset = bloom_f.bytes_written;
      ch.bloom_size = bf.size();
      bloom_f.write(bf.data(), bf.size());
    }
    chs.push_back(std::move(ch));
  }

  // ---- columnsHeader + index (block_header.go:425-484) ----
  bytes csh_data;
  ColumnsHeaderIndex csh_index;
  put_uvarint(csh_data, chs.size());
  for (const auto& ch : chs) {
    ColumnHeaderRef ref;
    ref.column_name_id = column_name_id(ch.name);
    ref.offset = csh_data.size();
    ch.marshal(csh_data);
    csh_index.column_headers_refs.push_back(ref);
  }
  put_uvarint(csh_data, const_cols.size());
  for (const auto& cc : const_cols) {
    ColumnHeaderRef ref;
    ref.column_name_id = column_name_id(cc.name);
    ref.offset = csh_data.size();
    put_len_prefixed(csh_data, cc.value);  // Field.marshal(dst, false), rows.go:35-41
    csh_index.const_columns_refs.push_back(ref);
  }
  bytes csh_index_data;
  csh_index.marshal(csh_index_data);

  bh.columns_header_index_offset = columns_header_index_.bytes_written;
  bh.columns_header_index_size = csh_index_data.size();
  columns_header_index_.write(csh_index_data.data(), csh_index_data.size());

  bh.columns_header_offset = columns_header_.bytes_written;
  bh.columns_header_size = csh_data.size();
  columns_header_.write(csh_data.data(), csh_data.size());

  // ---- append to index block ----
  if (!ib_has_blocks_) {
    cur_stream_id_ = sid;
    ib_min_ts_ = bh.timestamps_header.min_timestamp;
    ib_max_ts_ = bh.timestamps_header.max_timestamp;
    ib_has_blocks_ = true;
  } else {
    ib_min_ts_ = std::min(ib_min_ts_, bh.timestamps_header.min_timestamp);
    ib_max_ts_ = std::max(ib_max_ts_, bh.timestamps_header.max_timestamp);
  }
  bh.marshal(index_block_buf_);
  ph_.rows_count += bh.rows_count;
  ph_.blocks_count++;
  ph_.uncompressed_size_bytes += bh.uncompressed_size_bytes;
  if (ph_.blocks_count == 1 || bh.timestamps_header.min_timestamp < ph_.min_timestamp) {
    ph_.min_timestamp = bh.timestamps_header.min_timestamp;
  }
  if (ph_.blocks_count == 1 || bh.timestamps_header.max_timestamp > ph_.max_timestamp) {
    ph_.max_timestamp = bh.timestamps_header.max_timestamp;
  }
  if (index_block_buf_.size() >= 128 * 1024 /* maxUncompressedIndexBlockSize */) {
    flush_index_block();
  }
}

void PartWriter::flush_index_block() {
  if (index_block_buf_.empty()) return;
  IndexBlockHeader ih;
  ih.stream_id = cur_stream_id_;
  ih.min_timestamp = ib_min_ts_;
  ih.max_timestamp = ib_max_ts_;
  ih.index_block_offset = index_.bytes_written;
  bytes comp;
  zstd_compress(comp, index_block_buf_.data(), index_block_buf_.size(), 1);
  ih.index_block_size = comp.size();
  index_.write(comp.data(), comp.size());
  metaindex_.push_back(ih);
  index_block_buf_.clear();
  ib_has_blocks_ = false;
}

void PartWriter::finish() {
  if (finished_) return;
  finished_ = true;
  flush_index_block();

  // metaindex.bin: one zstd frame of indexBlockHeaders (index_block_header.go:113-121)
  {
    bytes data;
    for (const auto& ih : metaindex_) ih.marshal(data);
    bytes comp;
    zstd_compress(comp, data.data(), data.size(), 1);
    FileW f;
    f.open(dir_ + "/metaindex.bin");
    f.write(comp.data(), comp.size());
    f.close();
  }
  // column_names.bin
  {
    bytes data = marshal_column_names(names_);
    FileW f;
    f.open(dir_ + "/column_names.bin");
    f.write(data.data(), data.size());
    f.close();
  }
  // column_idxs.bin (only non-message columns appear in the map)
  {
    std::vector<std::pair<uint64_t, uint64_t>> idxs(column_shards_.begin(),
                                                    column_shards_.end());
    bytes data = marshal_column_idxs(idxs);
    FileW f;
    f.open(dir_ + "/column_idxs.bin");
    f.write(data.data(), data.size());
    f.close();
  }

  index_.close();
  columns_header_index_.close();
  columns_header_.close();
  timestamps_.close();
  message_values_.close();
  message_bloom_.close();
  uint64_t compressed = 0;
  for (auto& v : shard_values_) {
    compressed += v->bytes_written;
    v->close();
  }
  for (auto& b : shard_bloom_) {
    compressed += b->bytes_written;
    b->close();
  }
  compressed += index_.bytes_written + columns_header_index_.bytes_written +
                columns_header_.bytes_written + timestamps_.bytes_written +
                message_values_.bytes_written + message_bloom_.bytes_written;
  ph_.compressed_size_bytes = compressed;

  std::string js = ph_.to_json();
  FileW f;
  f.open(dir_ + "/metadata.json");
  f.write((const uint8_t*)js.data(), js.size());
  f.close();
}

}  // namespace vl
