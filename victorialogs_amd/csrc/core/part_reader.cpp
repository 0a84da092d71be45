#include "part_reader.h"

#include <fcntl.h>
#include <unistd.h>

#include "bloom.h"
#include "xxhash64.h"
#include "zstd_wrap.h"

namespace vl {

void PartReader::FileR::open(const std::string& path, bool required) {
  fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0 && required) fail("cannot open " + path);
}
void PartReader::FileR::pread_full(uint8_t* dst, size_t n, uint64_t off) const {
  size_t done = 0;
  while (done < n) {
    ssize_t r = ::pread(fd, dst + done, n - done, off + done);
    if (r <= 0) fail("pread failed/short");
    done += size_t(r);
  }
}
void PartReader::FileR::close() {
  if (fd >= 0) {
    ::close(fd);
    fd = -1;
  }
}

static bytes read_whole(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  if (!f) fail("cannot open " + path);
  fseek(f, 0, SEEK_END);
  long sz = ftell(f);
  fseek(f, 0, SEEK_SET);
  bytes data((size_t(sz)));
  if (sz > 0 && fread(data.data(), 1, size_t(sz), f) != size_t(sz)) {
    fclose(f);
    fail("short read " + path);
  }
  fclose(f);
  return data;
}

PartReader::PartReader(const std::string& dir) : dir_(dir) {
  // mustOpenFilePart (part.go:105-173)
  bytes meta = read_whole(dir + "/metadata.json");
  ph_.from_json(std::string((const char*)meta.data(), meta.size()));
  if (ph_.format_version < 1) {
    fail("FormatVersion 0 parts are not supported by this reader (v1+ only)");
  }

  {
    bytes cn = read_whole(dir + "/column_names.bin");
    names_ = unmarshal_column_names(cn.data(), cn.size());
    for (size_t i = 0; i < names_.size(); i++) name_ids_[names_[i]] = i;
  }
  if (ph_.format_version >= 3) {
    bytes ci = read_whole(dir + "/column_idxs.bin");
    column_shards_ = unmarshal_column_idxs(ci.data(), ci.size());
  }
  {
    // metaindex.bin: one zstd frame of indexBlockHeaders
    bytes mi = read_whole(dir + "/metaindex.bin");
    bytes data;
    zstd_decompress(data, mi.data(), mi.size());
    size_t pos = 0;
    while (pos < data.size()) {
      IndexBlockHeader ih;
      pos += ih.unmarshal(data.data() + pos, data.size() - pos);
      metaindex_.push_back(ih);
    }
  }

  index_.open(dir + "/index.bin", true);
  columns_header_index_.open(dir + "/columns_header_index.bin", true);
  columns_header_.open(dir + "/columns_header.bin", true);
  timestamps_.open(dir + "/timestamps.bin", true);
  message_values_.open(dir + "/message_values.bin", true);
  message_bloom_.open(dir + "/message_bloom.bin", true);
  for (uint64_t i = 0; i < ph_.bloom_values_shards_count; i++) {
    auto v = std::make_unique<FileR>();
    v->open(dir + "/values.bin" + std::to_string(i), true);
    shard_values_.push_back(std::move(v));
    auto b = std::make_unique<FileR>();
    b->open(dir + "/bloom.bin" + std::to_string(i), true);
    shard_bloom_.push_back(std::move(b));
  }
}

PartReader::~PartReader() = default;

static uint64_t shard_index(const std::map<uint64_t, uint64_t>& column_shards,
                            const std::map<std::string, uint64_t>& name_ids,
                            uint64_t shards, uint64_t format_version,
                            const std::string& name) {
  // part.go:195-218
  if (format_version < 3) {
    if (shards <= 1) return 0;
    uint64_t h = xxhash64(name.data(), name.size());
    return h % shards;
  }
  auto it = name_ids.find(name);
  if (it == name_ids.end()) fail("unknown column name for shard lookup: " + name);
  auto it2 = column_shards.find(it->second);
  if (it2 == column_shards.end()) fail("missing shard index for column: " + name);
  return it2->second;
}

const PartReader::FileR& PartReader::values_file(const std::string& name) const {
  if (name.empty()) return message_values_;
  return *shard_values_[shard_index(column_shards_, name_ids_,
                                    ph_.bloom_values_shards_count, ph_.format_version,
                                    name)];
}
const PartReader::FileR& PartReader::bloom_file(const std::string& name) const {
  if (name.empty()) return message_bloom_;
  return *shard_bloom_[shard_index(column_shards_, name_ids_,
                                   ph_.bloom_values_shards_count, ph_.format_version,
                                   name)];
}

void PartReader::read_block_headers(const IndexBlockHeader& ih,
                                    std::vector<BlockHeader>& dst) const {
  // mustReadBlockHeaders (block_search.go:508-533)
  bytes comp(size_t(ih.index_block_size));
  index_.pread_full(comp.data(), comp.size(), ih.index_block_offset);
  bytes data;
  zstd_decompress(data, comp.data(), comp.size());
  size_t pos = 0;
  while (pos < data.size()) {
    BlockHeader bh;
    pos += bh.unmarshal(data.data() + pos, data.size() - pos);
    dst.push_back(bh);
  }
}

std::vector<BlockHeader> PartReader::read_all_block_headers() const {
  std::vector<BlockHeader> out;
  for (const auto& ih : metaindex_) read_block_headers(ih, out);
  return out;
}

void PartReader::read_block_columns(const BlockHeader& bh, BlockColumns& dst) const {
  // readColumnsHeaderIndexBlock / readColumnsHeaderBlock (block_search.go:384-406)
  bytes idx(size_t(bh.columns_header_index_size));
  columns_header_index_.pread_full(idx.data(), idx.size(),
                                   bh.columns_header_index_offset);
  dst.index.unmarshal(idx.data(), idx.size());

  dst.header_data.resize(size_t(bh.columns_header_size));
  columns_header_.pread_full(dst.header_data.data(), dst.header_data.size(),
                             bh.columns_header_offset);
}

bool PartReader::get_column_header(const BlockColumns& bc, const std::string& name,
                                   ColumnHeader* ch) const {
  // getColumnHeader (block_search.go:278-324)
  auto it = name_ids_.find(name);
  if (it == name_ids_.end()) return false;
  for (const auto& ref : bc.index.column_headers_refs) {
    if (ref.column_name_id != it->second) continue;
    if (ref.offset > bc.header_data.size()) fail("column header offset out of range");
    ch->unmarshal(bc.header_data.data() + ref.offset,
                  bc.header_data.size() - ref.offset);
    ch->name = name;
    return true;
  }
  return false;
}

bool PartReader::get_const_column(const BlockColumns& bc, const std::string& name,
                                  std::string* value) const {
  // getConstColumnValue (block_search.go:232-276); Field value-only (v1+)
  auto it = name_ids_.find(name);
  if (it == name_ids_.end()) return false;
  for (const auto& ref : bc.index.const_columns_refs) {
    if (ref.column_name_id != it->second) continue;
    if (ref.offset > bc.header_data.size()) fail("const column offset out of range");
    const uint8_t* p = bc.header_data.data() + ref.offset;
    size_t left = bc.header_data.size() - ref.offset;
    uint64_t len;
    int sz = get_uvarint(p, left, &len);
    if (sz <= 0 || left - sz < len) fail("bad const column value");
    value->assign((const char*)p + sz, size_t(len));
    return true;
  }
  return false;
}

void PartReader::get_all_column_headers(const BlockColumns& bc,
                                        std::vector<ColumnHeader>* chs,
                                        std::vector<ConstColumn>* ccs) const {
  if (chs) {
    for (const auto& ref : bc.index.column_headers_refs) {
      ColumnHeader ch;
      ch.unmarshal(bc.header_data.data() + ref.offset,
                   bc.header_data.size() - ref.offset);
      ch.name = names_.at(size_t(ref.column_name_id));
      chs->push_back(std::move(ch));
    }
  }
  if (ccs) {
    for (const auto& ref : bc.index.const_columns_refs) {
      const uint8_t* p = bc.header_data.data() + ref.offset;
      size_t left = bc.header_data.size() - ref.offset;
      uint64_t len;
      int sz = get_uvarint(p, left, &len);
      if (sz <= 0 || left - sz < len) fail("bad const column value");
      ConstColumn cc;
      cc.name = names_.at(size_t(ref.column_name_id));
      cc.value.assign((const char*)p + sz, size_t(len));
      ccs->push_back(std::move(cc));
    }
  }
}

void PartReader::read_values(const ColumnHeader& ch, uint64_t rows_count,
                             StringsBlockDec& dst) const {
  // getValuesForColumn (block_search.go:444-474)
  bytes raw(size_t(ch.values_size));
  values_file(ch.name).pread_full(raw.data(), raw.size(), ch.values_offset);
  unmarshal_strings_block(dst, raw.data(), raw.size(), rows_count);
}

void PartReader::read_bloom(const ColumnHeader& ch, std::vector<uint64_t>& words) const {
  // getBloomFilterForColumn (block_search.go:411-439)
  bytes raw(size_t(ch.bloom_size));
  bloom_file(ch.name).pread_full(raw.data(), raw.size(), ch.bloom_offset);
  if (!bloom_unmarshal(words, raw.data(), raw.size())) fail("bad bloom block size");
}

void PartReader::read_timestamps(const BlockHeader& bh, std::vector<int64_t>& dst) const {
  // getTimestamps (block_search.go:479-506)
  const TimestampsHeader& th = bh.timestamps_header;
  bytes raw(size_t(th.block_size));
  timestamps_.pread_full(raw.data(), raw.size(), th.block_offset);
  unmarshal_int64_array(dst, raw.data(), raw.size(), th.marshal_type, th.min_timestamp,
                        bh.rows_count);
}

}  // namespace vl
