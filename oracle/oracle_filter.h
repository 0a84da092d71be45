// ============================================================================
// TEST INFRASTRUCTURE — CPU oracle for the block-scan hot path.
//
// This is the CPU restatement of VictoriaLogs' per-block filter evaluation
// (lib/logstorage filter.applyToBlockSearch, driven by
// blockSearch.search, block_search.go:207-226).  It exists ONLY to pin
// parity: tests compare the HIP/GPU path's row bitmaps against this oracle
// word-for-word, and bench.py's cpu_baseline leg times it.  Nothing in the
// product path may call it; the product fails loudly when its HIP extension
// is missing.
//
// Pinned against the reference's own golden vectors (tests/):
//   - bloom hex KATs               bloomfilter_test.go:105-119
//   - TestMatchPhrase truth table  filter_phrase_test.go:9-60
//   - codec round-trips            encoding_test.go:17,106
//   - filter fixtures              filter_test.go fixture pattern
// The reference itself is Go and no Go toolchain exists in this container
// (SURVEY.md §8c), so oracle/_ref is not buildable; parity beyond the ported
// golden vectors is anchored on these KATs (DESIGN.md states this).
// ============================================================================
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "../victorialogs_amd/csrc/core/filter.h"
#include "../victorialogs_amd/csrc/core/part_reader.h"

namespace vl {
namespace oracle {

// bitmap (bitmap.go:28-192): []uint64, LSB-first within each word.
struct Bitmap {
  std::vector<uint64_t> a;
  uint64_t bits_len = 0;

  void init_ones(uint64_t n) {
    // bm.init + bm.setBits (block_search.go:213-214, bitmap.go:47-72)
    bits_len = n;
    a.assign(size_t((n + 63) / 64), ~uint64_t(0));
    uint64_t tail = n % 64;
    if (tail > 0 && !a.empty()) a.back() &= (uint64_t(1) << tail) - 1;
  }
  void reset_bits() { std::fill(a.begin(), a.end(), 0); }
  bool is_zero() const {
    for (uint64_t w : a) {
      if (w) return false;
    }
    return true;
  }
  void and_not(const Bitmap& x) {
    for (size_t i = 0; i < a.size(); i++) a[i] &= ~x.a[i];
  }
  uint64_t ones_count() const {
    uint64_t n = 0;
    for (uint64_t w : a) n += uint64_t(__builtin_popcountll(w));
    return n;
  }
  // forEachSetBit (bitmap.go:128-153): clears bits where f returns false.
  template <typename F>
  void for_each_set_bit(F&& f) {
    for (size_t i = 0; i < a.size(); i++) {
      uint64_t word = a[i];
      if (word == 0) continue;
      uint64_t word_new = word;
      for (int j = 0; j < 64; j++) {
        uint64_t mask = uint64_t(1) << j;
        if ((word & mask) == 0) continue;
        uint64_t idx = uint64_t(i) * 64 + uint64_t(j);
        if (idx >= bits_len) break;
        if (!f(idx)) word_new &= ~mask;
      }
      if (word != word_new) a[i] = word_new;
    }
  }
};

// Lazy per-block context mirroring blockSearch's caches
// (block_search.go:98-148).
struct BlockCtx {
  const PartReader* pr = nullptr;
  const BlockHeader* bh = nullptr;
  PartReader::BlockColumns bc;
  bool bc_loaded = false;

  std::map<std::string, StringsBlockDec> values_cache;
  std::map<std::string, std::vector<uint64_t>> bloom_cache;
  std::vector<int64_t> timestamps;
  bool ts_loaded = false;

  BlockCtx(const PartReader* pr_, const BlockHeader* bh_) : pr(pr_), bh(bh_) {}

  const PartReader::BlockColumns& columns();
  // getConstColumnValue semantics (block_search.go:232-276): empty string for
  // both a missing const column and an empty const value.
  std::string const_value(const std::string& canonical_name);
  bool column_header(const std::string& canonical_name, ColumnHeader* ch);
  const StringsBlockDec& values(const ColumnHeader& ch);
  const std::vector<uint64_t>& bloom(const ColumnHeader& ch);
  const std::vector<int64_t>& get_timestamps();
};

// filter.applyToBlockSearch restatement; bm has bh->rows_count bits.
void apply_filter(const FilterNode& f, BlockCtx& ctx, Bitmap& bm);

// blockSearch.search (block_search.go:207-226) minus blockResult: all-ones
// bitmap -> filter -> final bitmap.
void search_block(const FilterNode& f, const PartReader& pr, const BlockHeader& bh,
                  Bitmap& bm);

}  // namespace oracle
}  // namespace vl
