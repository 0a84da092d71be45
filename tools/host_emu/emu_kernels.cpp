// Serial CPU emulation of the scan/gather/bloom kernels over the SAME
// per-row device code (scan_rowops.h) and the same descriptor layouts.
// Mirrors scan_program_kernel's semantics (postfix program over per-block
// bitmap words, bloom gates, any-case overrides, tail masking).
// TEST INFRASTRUCTURE — loaded via VQL_LIB by tests/test_emu_pipeline.py.
#include <cmath>
#include <cstdint>
#include <cstring>
#include <vector>

using std::fma;
using std::trunc;

#define __device__
#define __forceinline__ inline
#define __noinline__ __attribute__((noinline))

#include "../../victorialogs_amd/csrc/core/parse_float.h"
#include "../../victorialogs_amd/csrc/core/ryu.h"
#include "../../victorialogs_amd/csrc/core/xxhash64.h"
#include "../../victorialogs_amd/csrc/hip/scan_types.h"
#include <hip/hip_runtime.h>

namespace vl {
int64_t g_vl_local_tz_nsecs = 0;
#include "../../victorialogs_amd/csrc/hip/scan_rowops.h"
}  // namespace vl

using namespace vl;

extern "C" int vql_set_local_tz_nsecs(long long v) {
  vl::g_vl_local_tz_nsecs = v;
  return 0;
}

namespace {

bool emu_eval_leaf_word(const DevLeafBlock& lb, uint32_t g0, uint32_t g1,
                        uint64_t* word_out) {
  uint64_t word = 0;
  const bool is_str = d_is_string_kind(lb.kind);
  for (uint32_t row = g0; row < g1; row++) {
    bool pred;
    if (is_str) {
      GlobalAcc a{lb.data};
      const long s = lb.offsets[row];
      const long e = lb.offsets[row + 1];
      pred = d_eval_string_row(lb, a, s, e - s);
    } else {
      pred = d_eval_fixed_row(lb, row);
    }
    if (pred) word |= uint64_t(1) << (row - g0);
  }
  // any-case non-ASCII override (word-level merge, as in the kernel)
  if ((lb.kind == kScanAnyCasePhraseStr || lb.kind == kScanAnyCasePrefixStr) &&
      lb.hashes != nullptr) {
    const uint64_t mw = lb.hashes[g0 / 64], vw = lb.bloom[g0 / 64];
    word = (word & ~mw) | (vw & mw);
  }
  *word_out = word;
  return true;
}

}  // namespace

extern "C" hipError_t vql_launch_scan(const DevOp* ops, int nops,
                                      const DevLeafBlock* lbs, int nleaves,
                                      const DevBlock* blocks,
                                      const DevChunk* chunks, uint32_t nchunks,
                                      unsigned long long* hits, hipStream_t) {
  for (uint32_t c = 0; c < nchunks; c++) {
    const DevChunk ck = chunks[c];
    const DevBlock& blk = blocks[ck.block];
    const uint32_t r0 = ck.chunk * kChunkRows;
    const uint32_t r1 = blk.rows < r0 + kChunkRows ? blk.rows : r0 + kChunkRows;
    const uint32_t nwords = (r1 - r0 + 63) / 64;
    std::vector<std::vector<uint64_t>> stack;
    for (int i = 0; i < nops; i++) {
      const DevOp op = ops[i];
      if (op.kind == kOpLeaf) {
        const DevLeafBlock& lb = lbs[size_t(ck.block) * nleaves + op.leaf];
        std::vector<uint64_t> out(nwords, 0);
        if (lb.mode == kModeAll) {
          std::fill(out.begin(), out.end(), ~uint64_t(0));
        } else if (lb.mode == kModeScan) {
          bool bloom_ok = true;
          if (lb.nhashes && lb.bloom_words > 0 &&
              lb.kind != kScanAnyCasePhraseStr &&
              lb.kind != kScanAnyCasePrefixStr) {
            const uint64_t max_bits = uint64_t(lb.bloom_words) * 64;
            for (uint32_t k = 0; k < lb.nhashes; k++) {
              uint64_t idx = lb.hashes[k] % max_bits;
              if (((lb.bloom[idx >> 6] >> (idx & 63)) & 1) == 0) {
                bloom_ok = false;
                break;
              }
            }
          }
          if (bloom_ok) {
            for (uint32_t w = 0; w < nwords; w++) {
              const uint32_t g0 = r0 + w * 64;
              const uint32_t g1 = g0 + 64 < r1 ? g0 + 64 : r1;
              emu_eval_leaf_word(lb, g0, g1, &out[w]);
            }
          }
        }
        stack.push_back(std::move(out));
      } else if (op.kind == kOpNot) {
        auto& top = stack.back();
        for (auto& w : top) w = ~w;
      } else {
        const int n = op.nargs;
        auto& dst = stack[stack.size() - size_t(n)];
        for (uint32_t w = 0; w < nwords; w++) {
          uint64_t acc = dst[w];
          for (int k = 1; k < n; k++) {
            const auto& src = stack[stack.size() - size_t(n) + size_t(k)];
            acc = op.kind == kOpAnd ? (acc & src[w]) : (acc | src[w]);
          }
          dst[w] = acc;
        }
        stack.resize(stack.size() - size_t(n) + 1);
      }
    }
    unsigned long long local = 0;
    for (uint32_t w = 0; w < nwords; w++) {
      uint64_t word = stack[0][w];
      const uint32_t base = r0 + w * 64;
      const uint32_t valid = r1 - base < 64 ? r1 - base : 64;
      if (valid < 64) word &= (uint64_t(1) << valid) - 1;
      blk.bitmap_out[r0 / 64 + w] = word;
      local += (unsigned long long)__builtin_popcountll(word);
    }
    *hits += local;
    if (blk.hits_out) *blk.hits_out += local;
  }
  return hipSuccess;
}

extern "C" hipError_t vql_launch_gather_count(const DevGatherCol* gcols,
                                              const DevBlock* blocks,
                                              const DevChunk* chunks,
                                              uint32_t nchunks,
                                              DevChunkCount* counts,
                                              hipStream_t) {
  for (uint32_t c = 0; c < nchunks; c++) {
    const DevChunk ck = chunks[c];
    const DevBlock& blk = blocks[ck.block];
    const DevGatherCol& gc = gcols[ck.block];
    const uint32_t r0 = ck.chunk * kChunkRows;
    const uint32_t r1 = blk.rows < r0 + kChunkRows ? blk.rows : r0 + kChunkRows;
    uint32_t rows = 0;
    unsigned long long bytes = 0;
    for (uint32_t row = r0; row < r1; row++) {
      if ((blk.bitmap_out[row / 64] >> (row % 64)) & 1) {
        rows++;
        bytes += d_gather_len(gc, row);
      }
    }
    counts[c].rows = rows;
    counts[c].bytes = bytes;
  }
  return hipSuccess;
}

extern "C" hipError_t vql_launch_gather_copy(
    const DevGatherCol* gcols, const DevBlock* blocks, const DevChunk* chunks,
    uint32_t nchunks, const DevChunkBase* bases, uint8_t* out_bytes,
    unsigned long long* out_offs, unsigned long long* out_rowids,
    hipStream_t) {
  for (uint32_t c = 0; c < nchunks; c++) {
    const DevChunk ck = chunks[c];
    const DevBlock& blk = blocks[ck.block];
    const DevGatherCol& gc = gcols[ck.block];
    const uint32_t r0 = ck.chunk * kChunkRows;
    const uint32_t r1 = blk.rows < r0 + kChunkRows ? blk.rows : r0 + kChunkRows;
    unsigned long long ri = bases[c].row_base;
    unsigned long long bi = bases[c].byte_base;
    for (uint32_t row = r0; row < r1; row++) {
      if (((blk.bitmap_out[row / 64] >> (row % 64)) & 1) == 0) continue;
      out_offs[ri] = bi;
      if (out_rowids) out_rowids[ri] = bases[c].gid_base + (row - r0);
      bi += d_gather_write(gc, row, out_bytes + bi);
      ri++;
    }
  }
  return hipSuccess;
}

extern "C" int vql_launch_bloom_tokenize(const void* data, const void* offsets,
                                         unsigned rows, void* slots,
                                         unsigned cap_mask, void* unique_count,
                                         void* overflow, void*) {
  // serial mirror of bloom_tokenize_kernel
  const uint8_t* d = (const uint8_t*)data;
  const uint32_t* offs = (const uint32_t*)offsets;
  unsigned long long* sl = (unsigned long long*)slots;
  unsigned long long* uniq = (unsigned long long*)unique_count;
  int* ovf = (int*)overflow;
  for (unsigned row = 0; row < rows; row++) {
    const uint8_t* p = d + offs[row];
    const long n = long(offs[row + 1]) - long(offs[row]);
    long i = 0;
    bool ascii = true;
    for (long k = 0; k < n; k++) {
      if (p[k] >= 0x80) {
        ascii = false;
        break;
      }
    }
    auto tokchar = [](uint8_t ch) {
      return (ch >= 'a' && ch <= 'z') || (ch >= 'A' && ch <= 'Z') ||
             (ch >= '0' && ch <= '9') || ch == '_';
    };
    while (i < n) {
      long start, end;
      if (ascii) {
        while (i < n && !tokchar(p[i])) i++;
        start = i;
        while (i < n && tokchar(p[i])) i++;
        end = i;
      } else {
        GlobalAcc a{p};
        while (i < n) {
          int sz;
          uint32_t r = d_utf8_decode(a, i, n - i, &sz);
          if (d_is_token_rune(r)) break;
          i += sz;
        }
        start = i;
        while (i < n) {
          int sz;
          uint32_t r = d_utf8_decode(a, i, n - i, &sz);
          if (!d_is_token_rune(r)) break;
          i += sz;
        }
        end = i;
      }
      if (end <= start) break;
      const uint64_t h = vl::xxhash64(p + start, size_t(end - start));
      uint32_t idx = uint32_t(h) & cap_mask;
      for (uint32_t probes = 0;; probes++) {
        if (probes > cap_mask) {
          *ovf = 1;
          return 0;
        }
        if (sl[idx] == 0) {
          sl[idx] = h;
          (*uniq)++;
          break;
        }
        if (sl[idx] == h) break;
        idx = (idx + 1) & cap_mask;
      }
    }
  }
  return 0;
}

extern "C" int vql_launch_bloom_setbits(const void* slots, unsigned cap,
                                        void* bits, unsigned long long max_bits,
                                        void*) {
  const unsigned long long* sl = (const unsigned long long*)slots;
  unsigned long long* b = (unsigned long long*)bits;
  for (unsigned i = 0; i < cap; i++) {
    uint64_t h = sl[i];
    if (h == 0) continue;
    uint64_t buf = h;
    for (int k = 0; k < 6; k++) {
      uint64_t hk = vl::xxhash64(&buf, 8);
      buf++;
      uint64_t idx = hk % max_bits;
      b[idx >> 6] |= 1ULL << (idx & 63);
    }
  }
  return 0;
}
