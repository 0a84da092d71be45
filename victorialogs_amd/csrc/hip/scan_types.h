// Device/host shared types for the block-scan kernels.
//
// The device-side execution model: one workgroup scans one CHUNK of up to
// kChunkRows rows of one block.  The compiled filter program is a postfix op
// sequence over per-(leaf,block) descriptors; leaf predicates are evaluated
// into LDS bitmaps (one 64-bit word per 64 consecutive rows, built with
// 64-lane wavefront ballots) and combined with AND/OR/NOT.  The final chunk
// bitmap is masked to the block's row count and written to HBM; bit i of the
// result equals row i, LSB-first in u64 words (bitmap.go:113-125 layout).
#pragma once

#include <cstdint>

namespace vl {

constexpr uint32_t kChunkRows = 8192;            // rows per workgroup
constexpr uint32_t kChunkWords = kChunkRows / 64;
// Per-wave LDS string tile (see scan_kernels.hip): 1088 16-B slots = 64 rows
// x <=272 B, a multiple of 64*16 B.  Also used by staging to pick the
// small-row super-group size (DevLeafBlock.sg).
constexpr uint32_t kWaveTileBytes = 17408;
constexpr uint32_t kNumWaves = 4;
constexpr int kMaxProgOps = 64;
constexpr int kMaxStackDepth = 8;

enum LeafMode : uint8_t {
  kModeNone = 0,  // predicate false for all rows of the block
  kModeAll = 1,   // predicate true for all rows of the block
  kModeScan = 2,  // evaluate per row
};

enum ScanKind : uint8_t {
  kScanPhraseStr = 0,   // matchPhrase over string rows (filter_phrase.go:211-270)
  kScanEqStr = 1,       // exact string equality (filter_exact.go:286-294)
  kScanEqBin = 2,       // fixed-width binary equality (filter_exact.go:356-364)
  kScanDict = 3,        // 1-byte dict codes vs match mask (filter_phrase.go:272-289)
  kScanTsRange = 4,     // int64 timestamps range (filter_time.go:114-137)
  kScanRangeU = 5,      // BE uint width w range (filter_range.go:267-333)
  kScanRangeI = 6,      // BE zig-zag int64 range (filter_range.go:335-350)
  kScanRangeF = 7,      // BE float64 bits range (filter_range.go:233-246)
  kScanRegexStr = 8,    // regex fast paths over string rows (regex.go:86-212)
  kScanPhraseIp = 9,    // matchPhrase over formatted ipv4 (filter_phrase.go:135-157)
  kScanPhraseIso = 10,  // matchPhrase over formatted iso8601 (filter_phrase.go:113-133)
  kScanRegexU = 11,     // regex over formatted uint (filter_regexp.go:191-241)
  kScanRegexI = 12,     // regex over formatted int64 (filter_regexp.go:243-254)
  kScanRegexIp = 13,    // regex over formatted ipv4 (filter_regexp.go:142-153)
  kScanRegexIso = 14,   // regex over formatted iso8601 (filter_regexp.go:129-140)
  kScanPhraseF64 = 15,  // matchPhrase over Ryu-formatted float64 (filter_phrase.go:159-186)
  kScanRegexF64 = 16,   // regex over Ryu-formatted float64 (filter_regexp.go:155-166)
  kScanRangeStr = 17,   // matchRange via parseMathNumber per row (filter_range.go:261-265,369-372)
  kScanPrefixStr = 18,      // matchPrefix over string rows (filter_prefix.go:318-352)
  kScanExactPrefixStr = 19, // strings.HasPrefix (filter_exact_prefix.go:275-277)
  kScanSeqStr = 20,         // matchSequence (filter_sequence.go:260-269)
  kScanPrefixFmt = 21,      // matchPrefix over formatted value (fmt in flags>>4)
  kScanExactPrefixFmt = 22, // HasPrefix over formatted value
  kScanSeqFmt = 23,         // matchSequence over formatted value
  kScanInStr = 24,          // value in sorted string set (filter_in.go:187-200)
  kScanInBin = 25,          // fixed-width value in sorted binary set
  kScanAnyPhraseStr = 26,   // matchAnyPhrase (filter_contains_any.go:293-300)
  kScanAllPhrasesStr = 27,  // matchAllPhrases (filter_contains_all.go:310-321)
  kScanAnyPhraseFmt = 28,   // matchAnyPhrase over formatted value
  kScanAllPhrasesFmt = 29,  // matchAllPhrases over formatted value
  kScanStrRange = 30,       // s >= min && s < max (filter_string_range.go:225-229)
  kScanStrRangeFmt = 31,    // formatted value string range
  kScanIPv4RangeStr = 32,   // parse ipv4 from string row, range compare
  kScanLenRangeStr = 33,    // rune-count range (filter_len_range.go:333-336)
  kScanLenRangeFmt = 34,    // formatted value length range
  kScanDayRange = 35,       // (ts - offset) % day in [start,end] (filter_day_range.go)
  kScanWeekRange = 36,      // weekday(ts - offset) in [start,end] (filter_week_range.go)
  kScanIPv4RangeBin = 37,   // BE u32 in [vmin,vmax] (filter_ipv4_range.go:166-181)
  kScanAnyCasePhraseStr = 38,  // matchAnyCasePhrase, ASCII rows (filter_any_case_phrase.go:159-181)
  kScanAnyCasePrefixStr = 39,  // matchAnyCasePrefix, ASCII rows (filter_any_case_prefix.go:161-183)
  // two-column filters (filter_eq_field.go, filter_le_field.go); side B's
  // data/offsets ride in the unused bloom-gate fields hashes/bloom
  kScanEqFieldBin = 40,     // fixed-width encoded equality
  kScanEqFieldDict = 41,    // dict codes vs 8x8 equality matrix in vmin
  kScanLeFieldDict = 42,    // dict codes vs 8x8 le matrix in vmin
  kScanLeFieldI64 = 43,     // zig-zag int64 compare (flags bit0 = exclude)
  kScanLeFieldF64 = 44,     // float64 compare (flags bit0 = exclude)
  kScanLeFieldBinStr = 45,  // leValuesString over the ENCODED bytes (quirk)
  kScanEqFieldStr = 46,     // generic string-form equality (sides in operand)
  kScanLeFieldStr = 47,     // generic string-form le (sides in operand)
};

// format source for the *Fmt kinds, stored in flags bits 4..7
enum : uint8_t {
  kFmtU64 = 1,   // BE uint of lb.width bytes -> decimal
  kFmtI64 = 2,   // BE zig-zag int64 -> decimal
  kFmtF64 = 3,   // BE float64 bits -> Ryu shortest 'f'
  kFmtIp = 4,    // BE u32 -> dotted quad
  kFmtIso = 5,   // BE u64 nsecs -> iso8601
};

// phrase flags
enum : uint8_t {
  kPhraseStartsToken = 1,
  kPhraseEndsToken = 2,
};

enum OpKind : uint8_t { kOpLeaf = 0, kOpAnd = 1, kOpOr = 2, kOpNot = 3 };

struct DevOp {
  uint8_t kind;
  uint8_t nargs;  // And/Or child count
  uint16_t leaf;  // leaf index for kOpLeaf
};

// Per-(leaf, block) descriptor, addressed as lbs[block * nleaves + leaf].
struct DevLeafBlock {
  uint8_t mode;
  uint8_t kind;
  uint8_t width;  // fixed-width kinds: 1/2/4/8
  uint8_t flags;
  // bloom gate probe hashes; 0 = no gate.  The any-case string kinds carry
  // no bloom gate and reuse `hashes`/`bloom` as the host-resolved override
  // bitmaps instead (rows with non-ASCII bytes: bit set in hashes=>ovr_mask
  // means the row result is the bloom=>ovr_val bit) — keeps the descriptor
  // at 96 B, which the hot scan loop is sensitive to.
  uint32_t nhashes;
  uint32_t bloom_words;
  uint32_t dict_mask;      // kScanDict: bit i set if dict value i matches
  uint32_t operand_len;
  uint32_t sg;             // string kinds: words per tile fill (2/4/8) for
                           // the small-row loop; 0/1 = per-word tile loop
  const uint64_t* hashes;  // device ptrs
  const uint64_t* bloom;
  const uint8_t* operand;  // phrase bytes / bin value / serialized regex
  const uint8_t* data;     // column payload (bytes / fixed-width / codes)
  const uint32_t* offsets; // strings: u32[rows+1]
  const int64_t* ts;       // kScanTsRange
  uint64_t vmin, vmax;     // range bounds (bit pattern for kScanRangeF)
};

struct DevBlock {
  uint64_t* bitmap_out;  // word 0 = rows [0,64) of the block
  unsigned long long* hits_out;  // per-block matched-row counter
  uint32_t rows;
  uint32_t pad;
};

struct DevChunk {
  uint32_t block;
  uint32_t chunk;  // chunk index within the block
};

// Serialized regex program blob layout (built by stage_regex_blob):
//   u8 flags (bit0 only_prefix, bit1 dot_star, bit2 dot_plus,
//             bit3 substr_star, bit4 substr_plus, bit5 has_or)
//   u16 prefix_len, u16 substr_len, u16 n_or
//   prefix bytes, substr bytes, { u16 len, bytes }[n_or]
enum : uint8_t {
  kReOnlyPrefix = 1,
  kReDotStar = 2,
  kReDotPlus = 4,
  kReSubstrStar = 8,
  kReSubstrPlus = 16,
  kReHasOr = 32,
  kReNfa = 64,      // general class: Glushkov NFA blob follows the or-values
  kReAlways = 128,  // pattern matches the empty string => always true
};

// ---- gather (blockResult materialization) ----
// Compacts the matched rows' values of one column into packed bytes +
// offsets + global row ids, mirroring blockResult.getValues semantics
// (block_result.go:306-478): string columns return raw bytes, fixed-width
// columns return their decoded string forms, dict columns the dict string.

enum GatherSrc : uint8_t {
  kGatherStr = 0,    // data+offsets
  kGatherConst = 1,  // const value (operand ptr, const_len)
  kGatherDict = 2,   // 1-byte codes + dict table (dict_data/dict_offs)
  kGatherFmtU = 3,   // BE uint width w -> decimal
  kGatherFmtI = 4,   // BE zig-zag i64 -> decimal
  kGatherFmtF = 5,   // BE f64 bits -> Ryu 'f'
  kGatherFmtIp = 6,  // BE u32 -> dotted quad
  kGatherFmtIso = 7, // BE u64 -> iso8601
  kGatherMissing = 8,  // column absent in the block -> empty values
};

struct DevGatherCol {
  const uint8_t* data;
  const uint32_t* offsets;
  const uint8_t* dict_data;   // concatenated dict strings
  const uint32_t* dict_offs;  // 9 entries
  const uint8_t* cval;        // const value bytes
  uint32_t cval_len;
  uint8_t src;
  uint8_t width;
  uint8_t pad0, pad1;
};

struct DevChunkCount {
  uint32_t rows;
  uint32_t pad;
  unsigned long long bytes;
};

struct DevChunkBase {
  unsigned long long row_base;   // output row index of the chunk's first match
  unsigned long long byte_base;  // output byte offset
  unsigned long long gid_base;   // global row id of the chunk's row 0
};

}  // namespace vl
