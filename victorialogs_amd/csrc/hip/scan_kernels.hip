// HIP/CDNA4 (gfx950) kernels for the VictoriaLogs block-scan hot path.
//
// Each workgroup (256 threads = 4 wavefronts) evaluates the compiled filter
// program over one chunk of up to kChunkRows rows of one block.  Leaf
// predicates are computed lane-per-row and assembled into bitmap words with
// 64-lane ballots (one wavefront ballot = one u64 bitmap word covering 64
// consecutive rows, LSB-first — exactly the bitmap.go:113-125 layout the
// drop-in boundary requires).  Bloom gates (bloomfilter.go:173-191) run
// cooperatively per chunk before any row is touched.
//
// String columns are scanned through an LDS tile: 256-row groups are copied
// from HBM with coalesced 16-byte loads (streaming reads == algorithmic
// bytes; the naive per-lane global scan fetched ~3x the data, measured with
// rocprofv3 FETCH_SIZE), then each lane scans its own row from LDS.  The
// tile is bank-swizzled at 16-byte-slot granularity so that the ~256-byte
// row stride does not put all 64 lanes of a wave on the same LDS bank pair
// (cdna_hip_programming.md §2, Guideline 4).
//
// This path is HBM-bandwidth-bound integer/byte work — no MFMA by design
// (BASELINE.json north star).
#include <hip/hip_runtime.h>

#include "../core/parse_float.h"
#include "../core/ryu.h"
#include "scan_types.h"

namespace vl {


// Each wavefront owns a private LDS tile and scans 64-row groups (one bitmap
// word per group) with no workgroup barriers: the wave copies its group's
// bytes with coalesced 16-byte loads (4-deep batches keep ~4 KiB per wave in
// flight), then matches lane-per-row from the tile.
// kWaveTileBytes / kNumWaves live in scan_types.h (staging reads them too)

// host local timezone offset for no-suffix RFC3339 parses (see scan_rowops.h)
__device__ int64_t g_vl_local_tz_nsecs = 0;

#include "scan_rowops.h"

extern "C" int vql_set_local_tz_nsecs(long long v) {
  int64_t x = v;
  return hipMemcpyToSymbol(HIP_SYMBOL(g_vl_local_tz_nsecs), &x, sizeof(x)) ==
                 hipSuccess
             ? 0
             : -1;
}



// The per-wave LDS-tiled string scan loop, templated on the row predicate so
// the hot filters get their own clone with ONLY their matcher inlined (the
// compiler stopped unswitching the kind dispatch out of this loop once the
// kind count grew, costing ~13% on the phrase workload).
// NB: every descriptor field is passed BY VALUE — capturing DevLeafBlock by
// reference made the compiler re-load lb.data/lb.offsets from memory inside
// the group loop under SGPR-spill pressure (global_load_dwordx2 + vmcnt(0)
// per iteration, ~7%% of the phrase workload).
template <bool kOvr, typename EvalFn>
__device__ __forceinline__ void d_string_tile_loop(
    const uint8_t* __restrict__ col_data, const uint32_t* __restrict__ col_offs,
    const uint64_t* __restrict__ ovr_mask, const uint64_t* __restrict__ ovr_val,
    uint8_t* wtile, uint64_t* out, uint32_t r0,
    uint32_t r1, uint32_t nwords, int lane, int wave, int nwaves,
    EvalFn eval) {
  typedef uint32_t v4u __attribute__((ext_vector_type(4)));
  v4u* dst = (v4u*)wtile;
  uint32_t wd = wave;
  // Offsets pipeline: each lane holds offsets[g0+lane] (clamped to g1);
  // a lane's row end is the next lane's start (shfl), lane ng-1's end is
  // the clamped value itself.  The next group's offsets are prefetched
  // while this group's tile copy is in flight.
  uint32_t o_lane = 0, o_end = 0;
  if (wd < nwords) {
    o_lane = col_offs[min(r0 + wd * 64 + uint32_t(lane), r1)];
    if (lane == 0) o_end = col_offs[min(r0 + wd * 64 + 64, r1)];
  }
  while (wd < nwords) {
    const uint32_t g0 = r0 + wd * 64;
    const uint32_t g1 = min(g0 + 64, r1);
    const uint32_t ng = g1 - g0;
    const uint32_t byte0 =
        uint32_t(__builtin_amdgcn_readfirstlane(int(o_lane))) & ~15u;
    const uint32_t byte1 = uint32_t(__shfl(int(o_end), 0, 64));
    const uint32_t nbytes = byte1 - byte0;
    const bool use_tile = nbytes <= kWaveTileBytes;
    if (use_tile) {
      const v4u* src = (const v4u*)(col_data + byte0);
      // round up to a full 64-slot stride: all 64 lanes always load together
      // (the staging arena leaves >=1 KiB of slack after each column)
      const uint32_t n16 = (((nbytes + 15) >> 4) + 63) & ~63u;
      uint32_t k = lane;
      // 16-deep batches: a 64-row group of ~256 B rows is 1024 slots, so one
      // batch puts the whole group's loads in flight per lane (256 B/lane)
      for (; k + 960 < n16; k += 1024) {
        v4u b0 = __builtin_nontemporal_load(src + k);
        v4u b1 = __builtin_nontemporal_load(src + k + 64);
        v4u b2 = __builtin_nontemporal_load(src + k + 128);
        v4u b3 = __builtin_nontemporal_load(src + k + 192);
        v4u b4 = __builtin_nontemporal_load(src + k + 256);
        v4u b5 = __builtin_nontemporal_load(src + k + 320);
        v4u b6 = __builtin_nontemporal_load(src + k + 384);
        v4u b7 = __builtin_nontemporal_load(src + k + 448);
        v4u b8 = __builtin_nontemporal_load(src + k + 512);
        v4u b9 = __builtin_nontemporal_load(src + k + 576);
        v4u b10 = __builtin_nontemporal_load(src + k + 640);
        v4u b11 = __builtin_nontemporal_load(src + k + 704);
        v4u b12 = __builtin_nontemporal_load(src + k + 768);
        v4u b13 = __builtin_nontemporal_load(src + k + 832);
        v4u b14 = __builtin_nontemporal_load(src + k + 896);
        v4u b15 = __builtin_nontemporal_load(src + k + 960);
        dst[k ^ ((k >> 4) & 15)] = b0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = b1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = b2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = b3;
        dst[(k + 256) ^ (((k + 256) >> 4) & 15)] = b4;
        dst[(k + 320) ^ (((k + 320) >> 4) & 15)] = b5;
        dst[(k + 384) ^ (((k + 384) >> 4) & 15)] = b6;
        dst[(k + 448) ^ (((k + 448) >> 4) & 15)] = b7;
        dst[(k + 512) ^ (((k + 512) >> 4) & 15)] = b8;
        dst[(k + 576) ^ (((k + 576) >> 4) & 15)] = b9;
        dst[(k + 640) ^ (((k + 640) >> 4) & 15)] = b10;
        dst[(k + 704) ^ (((k + 704) >> 4) & 15)] = b11;
        dst[(k + 768) ^ (((k + 768) >> 4) & 15)] = b12;
        dst[(k + 832) ^ (((k + 832) >> 4) & 15)] = b13;
        dst[(k + 896) ^ (((k + 896) >> 4) & 15)] = b14;
        dst[(k + 960) ^ (((k + 960) >> 4) & 15)] = b15;
      }
      // 8-deep batches for mid-size remainders
      for (; k + 448 < n16; k += 512) {
        // nt loads: each byte is read once per kernel; keep L2 for
        // the offsets/bitmap traffic (cdna guide: nt-weights row)
        v4u a0 = __builtin_nontemporal_load(src + k);
        v4u a1 = __builtin_nontemporal_load(src + k + 64);
        v4u a2 = __builtin_nontemporal_load(src + k + 128);
        v4u a3 = __builtin_nontemporal_load(src + k + 192);
        v4u a4 = __builtin_nontemporal_load(src + k + 256);
        v4u a5 = __builtin_nontemporal_load(src + k + 320);
        v4u a6 = __builtin_nontemporal_load(src + k + 384);
        v4u a7 = __builtin_nontemporal_load(src + k + 448);
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
        dst[(k + 256) ^ (((k + 256) >> 4) & 15)] = a4;
        dst[(k + 320) ^ (((k + 320) >> 4) & 15)] = a5;
        dst[(k + 384) ^ (((k + 384) >> 4) & 15)] = a6;
        dst[(k + 448) ^ (((k + 448) >> 4) & 15)] = a7;
      }
      for (; k + 192 < n16; k += 256) {
        v4u a0 = src[k], a1 = src[k + 64], a2 = src[k + 128],
            a3 = src[k + 192];
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
      }
      for (; k < n16; k += 64) dst[k ^ ((k >> 4) & 15)] = src[k];
    }
    // prefetch next group's offsets while the copy is in flight
    const uint32_t next_wd = wd + nwaves;
    uint32_t o_next = 0, o_end_next = 0;
    if (next_wd < nwords) {
      o_next = col_offs[min(r0 + next_wd * 64 + uint32_t(lane), r1)];
      if (lane == 0) o_end_next = col_offs[min(r0 + next_wd * 64 + 64, r1)];
    }
    if (use_tile) {
      // every lane's ds_writes must land before cross-lane reads below
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }
    bool pred = false;
    if (uint32_t(lane) < ng) {
      const long s = o_lane;
      const long e = uint32_t(__shfl(int(o_lane), lane + 1, 64));
      const long e_fix = uint32_t(lane) == ng - 1 ? long(byte1) : e;
      if (use_tile) {
        TileAcc a{wtile};
        pred = eval(a, s - byte0, e_fix - s);
      } else {
        GlobalAcc a{col_data};
        pred = eval(a, s, e_fix - s);
      }
    }
    uint64_t word = __ballot(pred);
    if (kOvr && ovr_mask != nullptr) {
      // host-resolved rows (non-ASCII any-case, stored in the unused bloom
      // gate fields): merge at word level
      const uint64_t mw = ovr_mask[wd], vw = ovr_val[wd];
      word = (word & ~mw) | (vw & mw);
    }
    if (lane == 0) out[wd] = word;
    o_lane = o_next;
    o_end = o_end_next;
    wd = next_wd;
  }
}

// Small-row string loop: the per-word tile loop above serializes one HBM
// round trip per 64-row group, which is latency-bound when rows are short
// (a ~30 B-row column moved ~0.8 TB/s vs ~5 TB/s for 256 B rows).  Here a
// wave fills its tile with SG words' bytes at once (SG = lb.sg, chosen at
// staging so SG*64*avg_row fits the tile) and evaluates SG ballot words per
// fill — amortizing the copy latency SG-fold.  Super-groups whose actual
// byte span overflows the tile (row-length variance) fall back to direct
// global evaluation for those words, bit-identically.
// __noinline__: one clone per evaluator, kept out of the hot per-word loop's
// instruction stream (see the I-cache regression note above).
template <bool kOvr, typename EvalFn>
__device__ __noinline__ void d_string_smallrow_loop(
    const uint8_t* __restrict__ col_data, const uint32_t* __restrict__ col_offs,
    const uint64_t* __restrict__ ovr_mask, const uint64_t* __restrict__ ovr_val,
    uint8_t* wtile, uint64_t* out, uint32_t r0, uint32_t r1, uint32_t nwords,
    uint32_t sgw, int lane, int wave, int nwaves, EvalFn eval) {
  typedef uint32_t v4u __attribute__((ext_vector_type(4)));
  v4u* dst = (v4u*)wtile;
  const uint32_t nsg = (nwords + sgw - 1) / sgw;
  for (uint32_t sg = wave; sg < nsg; sg += nwaves) {
    const uint32_t w0 = sg * sgw;
    const uint32_t g0 = r0 + w0 * 64;
    uint32_t o[9];
#pragma unroll
    for (uint32_t j = 0; j < 8; j++) {
      o[j] = j < sgw ? col_offs[min(g0 + j * 64 + uint32_t(lane), r1)] : 0;
    }
    o[8] = 0;
    if (lane == 0) o[8] = col_offs[min(g0 + sgw * 64, r1)];
    const uint32_t byte0 =
        uint32_t(__builtin_amdgcn_readfirstlane(int(o[0]))) & ~15u;
    const uint32_t byte1 = uint32_t(__shfl(int(o[8]), 0, 64));
    const uint32_t nbytes = byte1 - byte0;
    const bool use_tile = nbytes <= kWaveTileBytes;
    if (use_tile) {
      const v4u* src = (const v4u*)(col_data + byte0);
      const uint32_t n16 = (((nbytes + 15) >> 4) + 63) & ~63u;
      uint32_t k = lane;
      for (; k + 448 < n16; k += 512) {
        v4u a0 = __builtin_nontemporal_load(src + k);
        v4u a1 = __builtin_nontemporal_load(src + k + 64);
        v4u a2 = __builtin_nontemporal_load(src + k + 128);
        v4u a3 = __builtin_nontemporal_load(src + k + 192);
        v4u a4 = __builtin_nontemporal_load(src + k + 256);
        v4u a5 = __builtin_nontemporal_load(src + k + 320);
        v4u a6 = __builtin_nontemporal_load(src + k + 384);
        v4u a7 = __builtin_nontemporal_load(src + k + 448);
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
        dst[(k + 256) ^ (((k + 256) >> 4) & 15)] = a4;
        dst[(k + 320) ^ (((k + 320) >> 4) & 15)] = a5;
        dst[(k + 384) ^ (((k + 384) >> 4) & 15)] = a6;
        dst[(k + 448) ^ (((k + 448) >> 4) & 15)] = a7;
      }
      for (; k + 192 < n16; k += 256) {
        v4u a0 = src[k], a1 = src[k + 64], a2 = src[k + 128], a3 = src[k + 192];
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
      }
      for (; k < n16; k += 64) dst[k ^ ((k >> 4) & 15)] = src[k];
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }
#pragma unroll
    for (uint32_t j = 0; j < 8; j++) {
      if (j >= sgw) break;
      const uint32_t w = w0 + j;
      if (w >= nwords) break;
      const uint32_t gj = g0 + j * 64;
      const uint32_t ng = min(64u, r1 - gj);
      const uint32_t wend =
          uint32_t(__shfl(int(j + 1 < sgw ? o[j + 1] : o[8]), 0, 64));
      // all 64 lanes execute the shfl; for the word's LAST valid lane the
      // next lane's o[] may come from an inactive/garbage source, so that
      // lane takes the broadcast word end instead (the per-word tile loop
      // does the same with byte1)
      const uint32_t e0 = uint32_t(__shfl(int(o[j]), lane + 1, 64));
      bool pred = false;
      if (uint32_t(lane) < ng) {
        const long s = o[j];
        const long e = uint32_t(lane) == ng - 1 ? long(wend) : long(e0);
        if (use_tile) {
          TileAcc a{wtile};
          pred = eval(a, s - byte0, e - s);
        } else {
          GlobalAcc a{col_data};
          pred = eval(a, s, e - s);
        }
      }
      uint64_t word = __ballot(pred);
      if (kOvr && ovr_mask != nullptr) {
        const uint64_t mw = ovr_mask[w], vw = ovr_val[w];
        word = (word & ~mw) | (vw & mw);
      }
      if (lane == 0) out[w] = word;
    }
  }
}

// Dual-phrase tile loop: two phrase leaves over the SAME string column
// evaluate from one tile fill, halving the column's HBM reads (an OR/AND of
// two phrases on one column — configs[3]'s or8 — otherwise streams the
// column once per leaf).  Same structure as d_string_tile_loop; the copy is
// shared, the two matchers run back-to-back from LDS.
__device__ __noinline__ void d_string_phrase2_loop(
    const uint8_t* __restrict__ col_data, const uint32_t* __restrict__ col_offs,
    uint8_t* wtile, uint64_t* out1, uint64_t* out2, uint32_t r0, uint32_t r1,
    uint32_t nwords, int lane, int wave, int nwaves, const uint8_t* p1,
    uint32_t n1, uint8_t f1, const uint8_t* p2, uint32_t n2, uint8_t f2) {
  typedef uint32_t v4u __attribute__((ext_vector_type(4)));
  v4u* dst = (v4u*)wtile;
  uint32_t wd = wave;
  uint32_t o_lane = 0, o_end = 0;
  if (wd < nwords) {
    o_lane = col_offs[min(r0 + wd * 64 + uint32_t(lane), r1)];
    if (lane == 0) o_end = col_offs[min(r0 + wd * 64 + 64, r1)];
  }
  while (wd < nwords) {
    const uint32_t g0 = r0 + wd * 64;
    const uint32_t g1 = min(g0 + 64, r1);
    const uint32_t ng = g1 - g0;
    const uint32_t byte0 =
        uint32_t(__builtin_amdgcn_readfirstlane(int(o_lane))) & ~15u;
    const uint32_t byte1 = uint32_t(__shfl(int(o_end), 0, 64));
    const uint32_t nbytes = byte1 - byte0;
    const bool use_tile = nbytes <= kWaveTileBytes;
    if (use_tile) {
      const v4u* src = (const v4u*)(col_data + byte0);
      const uint32_t n16 = (((nbytes + 15) >> 4) + 63) & ~63u;
      uint32_t k = lane;
      for (; k + 448 < n16; k += 512) {
        v4u a0 = __builtin_nontemporal_load(src + k);
        v4u a1 = __builtin_nontemporal_load(src + k + 64);
        v4u a2 = __builtin_nontemporal_load(src + k + 128);
        v4u a3 = __builtin_nontemporal_load(src + k + 192);
        v4u a4 = __builtin_nontemporal_load(src + k + 256);
        v4u a5 = __builtin_nontemporal_load(src + k + 320);
        v4u a6 = __builtin_nontemporal_load(src + k + 384);
        v4u a7 = __builtin_nontemporal_load(src + k + 448);
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
        dst[(k + 256) ^ (((k + 256) >> 4) & 15)] = a4;
        dst[(k + 320) ^ (((k + 320) >> 4) & 15)] = a5;
        dst[(k + 384) ^ (((k + 384) >> 4) & 15)] = a6;
        dst[(k + 448) ^ (((k + 448) >> 4) & 15)] = a7;
      }
      for (; k + 192 < n16; k += 256) {
        v4u a0 = src[k], a1 = src[k + 64], a2 = src[k + 128], a3 = src[k + 192];
        dst[k ^ ((k >> 4) & 15)] = a0;
        dst[(k + 64) ^ (((k + 64) >> 4) & 15)] = a1;
        dst[(k + 128) ^ (((k + 128) >> 4) & 15)] = a2;
        dst[(k + 192) ^ (((k + 192) >> 4) & 15)] = a3;
      }
      for (; k < n16; k += 64) dst[k ^ ((k >> 4) & 15)] = src[k];
    }
    const uint32_t next_wd = wd + nwaves;
    uint32_t o_next = 0, o_end_next = 0;
    if (next_wd < nwords) {
      o_next = col_offs[min(r0 + next_wd * 64 + uint32_t(lane), r1)];
      if (lane == 0) o_end_next = col_offs[min(r0 + next_wd * 64 + 64, r1)];
    }
    if (use_tile) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    }
    bool pred1 = false, pred2 = false;
    if (uint32_t(lane) < ng) {
      const long s = o_lane;
      const long e = uint32_t(__shfl(int(o_lane), lane + 1, 64));
      const long e_fix = uint32_t(lane) == ng - 1 ? long(byte1) : e;
      if (use_tile) {
        TileAcc a{wtile};
        pred1 = d_match_phrase_at(a, s - byte0, e_fix - s, p1, n1, f1);
        pred2 = d_match_phrase_at(a, s - byte0, e_fix - s, p2, n2, f2);
      } else {
        GlobalAcc a{col_data};
        pred1 = d_match_phrase_at(a, s, e_fix - s, p1, n1, f1);
        pred2 = d_match_phrase_at(a, s, e_fix - s, p2, n2, f2);
      }
    }
    const uint64_t word1 = __ballot(pred1);
    const uint64_t word2 = __ballot(pred2);
    if (lane == 0) {
      out1[wd] = word1;
      out2[wd] = word2;
    }
    o_lane = o_next;
    o_end = o_end_next;
    wd = next_wd;
  }
}

// Cooperative bloom gate (bloomfilter.go:173-191); uniform control flow.
__device__ __forceinline__ bool d_bloom_gate_ok(const DevLeafBlock& lb,
                                                int tid, int bdim,
                                                int* shared_flag) {
  if (!lb.nhashes) return true;
  if (tid == 0) *shared_flag = 1;
  __syncthreads();
  if (lb.bloom_words > 0) {
    const uint64_t max_bits = uint64_t(lb.bloom_words) * 64;
    bool miss = false;
    for (uint32_t k = tid; k < lb.nhashes; k += bdim) {
      uint64_t idx = lb.hashes[k] % max_bits;
      if (((lb.bloom[idx >> 6] >> (idx & 63)) & 1) == 0) miss = true;
    }
    if (miss) *shared_flag = 0;
  }
  __syncthreads();
  return *shared_flag != 0;
}

// ---- the program kernel ----

__global__ __launch_bounds__(256) void scan_program_kernel(
    const DevOp* __restrict__ ops, int nops, const DevLeafBlock* __restrict__ lbs,
    int nleaves, const DevBlock* __restrict__ blocks,
    const DevChunk* __restrict__ chunks, unsigned long long* __restrict__ hits) {
  __shared__ uint64_t stack[kMaxStackDepth][kChunkWords];
  __shared__ __attribute__((aligned(16))) uint8_t tile[kNumWaves * kWaveTileBytes];
  __shared__ int bloom_ok;
  __shared__ unsigned long long wave_sums[4];

  const DevChunk ck = chunks[blockIdx.x];
  const DevBlock blk = blocks[ck.block];
  const uint32_t r0 = ck.chunk * kChunkRows;
  const uint32_t r1 = min(blk.rows, r0 + kChunkRows);
  const uint32_t nwords = (r1 - r0 + 63) / 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  int sp = 0;
  for (int i = 0; i < nops; i++) {
    const DevOp op = ops[i];
    __syncthreads();
    if (op.kind == kOpLeaf) {
      const DevLeafBlock& lb = lbs[size_t(ck.block) * nleaves + op.leaf];
      uint64_t* out = stack[sp];
      // phrase-pair fusion: the next op is another phrase leaf over the
      // SAME column -> evaluate both from one tile fill (halves this
      // column's HBM reads).  Falls through to the single path when a
      // bloom gate misses (rare; the single path re-evaluates it).
      if (i + 1 < nops && ops[i + 1].kind == kOpLeaf &&
          lb.kind == kScanPhraseStr && lb.mode == kModeScan && lb.sg <= 1 &&
          sp + 1 < kMaxStackDepth) {
        const DevLeafBlock& lb2 =
            lbs[size_t(ck.block) * nleaves + ops[i + 1].leaf];
        if (lb2.kind == kScanPhraseStr && lb2.mode == kModeScan &&
            lb2.data == lb.data && lb2.offsets == lb.offsets) {
          const bool ok1 = d_bloom_gate_ok(lb, tid, blockDim.x, &bloom_ok);
          const bool ok2 = d_bloom_gate_ok(lb2, tid, blockDim.x, &bloom_ok);
          if (ok1 && ok2) {
            d_string_phrase2_loop(lb.data, lb.offsets,
                                  tile + wave * kWaveTileBytes, stack[sp],
                                  stack[sp + 1], r0, r1, nwords, lane, wave,
                                  nwaves, lb.operand, lb.operand_len, lb.flags,
                                  lb2.operand, lb2.operand_len, lb2.flags);
            sp += 2;
            i++;
            continue;
          }
        }
      }
      if (lb.mode == kModeNone || lb.mode == kModeAll) {
        uint64_t fill = lb.mode == kModeAll ? ~uint64_t(0) : 0;
        for (uint32_t w = tid; w < nwords; w += blockDim.x) out[w] = fill;
        sp++;
        continue;
      }
      // bloom gate (bloomfilter.go:173-191), cooperative over the workgroup
      if (lb.nhashes) {
        if (tid == 0) bloom_ok = 1;
        __syncthreads();
        if (lb.bloom_words > 0) {
          const uint64_t max_bits = uint64_t(lb.bloom_words) * 64;
          bool miss = false;
          for (uint32_t k = tid; k < lb.nhashes; k += blockDim.x) {
            uint64_t idx = lb.hashes[k] % max_bits;
            if (((lb.bloom[idx >> 6] >> (idx & 63)) & 1) == 0) miss = true;
          }
          if (miss) bloom_ok = 0;
        }
        __syncthreads();
        if (!bloom_ok) {
          for (uint32_t w = tid; w < nwords; w += blockDim.x) out[w] = 0;
          sp++;
          continue;
        }
      }

      if (d_is_string_kind(lb.kind)) {
        uint8_t* wtile = tile + wave * kWaveTileBytes;
        const uint8_t* col_data = lb.data;
        const uint32_t* col_offs = lb.offsets;
        const uint8_t* op_ptr = lb.operand;
        const uint32_t op_len = lb.operand_len;
        const uint8_t op_flags = lb.flags;
        const uint32_t sgw = lb.sg;
        if (lb.kind == kScanPhraseStr) {
          // hot clone: only the phrase matcher in the loop body
          auto eval = [=](const auto& a, long s0, long sn) {
            return d_match_phrase_at(a, s0, sn, op_ptr, op_len, op_flags);
          };
          if (sgw > 1) {
            d_string_smallrow_loop<false>(col_data, col_offs, nullptr, nullptr,
                                          wtile, out, r0, r1, nwords, sgw,
                                          lane, wave, nwaves, eval);
          } else {
            d_string_tile_loop<false>(col_data, col_offs, nullptr, nullptr,
                                      wtile, out, r0, r1, nwords, lane, wave,
                                      nwaves, eval);
          }
        } else if (lb.kind == kScanRegexStr) {
          // second hot clone: regex fast paths + NFA (BASELINE config 3)
          auto eval = [=](const auto& a, long s0, long sn) {
            return d_regex_match_at(op_ptr, a, s0, sn);
          };
          if (sgw > 1) {
            d_string_smallrow_loop<false>(col_data, col_offs, nullptr, nullptr,
                                          wtile, out, r0, r1, nwords, sgw,
                                          lane, wave, nwaves, eval);
          } else {
            d_string_tile_loop<false>(col_data, col_offs, nullptr, nullptr,
                                      wtile, out, r0, r1, nwords, lane, wave,
                                      nwaves, eval);
          }
        } else {
          const bool anycase = lb.kind == kScanAnyCasePhraseStr ||
                               lb.kind == kScanAnyCasePrefixStr;
          auto eval = [&](const auto& a, long s0, long sn) {
            return d_eval_string_row(lb, a, s0, sn);
          };
          if (sgw > 1) {
            d_string_smallrow_loop<true>(col_data, col_offs,
                                         anycase ? lb.hashes : nullptr,
                                         anycase ? lb.bloom : nullptr, wtile,
                                         out, r0, r1, nwords, sgw, lane, wave,
                                         nwaves, eval);
          } else {
            d_string_tile_loop<true>(col_data, col_offs,
                                     anycase ? lb.hashes : nullptr,
                                     anycase ? lb.bloom : nullptr, wtile, out,
                                     r0, r1, nwords, lane, wave, nwaves, eval);
          }
        }
      } else if (lb.kind == kScanDict) {
        // word-per-lane: each lane builds one full bitmap word from its 64
        // dict codes with 8 independent u64 loads (64 B/lane stride, fully
        // coalesced across the wave).  The lane-per-row ballot loop was
        // latency-bound here: one byte-load round trip per word serialized
        // the wave (measured 215 GB/s on the dict+time config).  Over-read
        // past r1 within the word is masked by the final tail masking and
        // stays inside the slab tail pad.
        const uint32_t mask = lb.dict_mask;
        const uint64_t* base = (const uint64_t*)(lb.data + r0);
        for (uint32_t w = wave * 64 + lane; w < ((nwords + 63) & ~63u);
             w += nwaves * 64) {
          if (w >= nwords) continue;
          uint64_t word = 0;
          const uint64_t* p = base + w * 8;
          for (int k = 0; k < 8; k++) {
            uint64_t v = p[k];
            uint64_t bits = 0;
            for (int j = 0; j < 8; j++) {
              // dict codes are < 8 (dict <= 8 entries); tail over-read bytes
              // may be arbitrary, but those bits are masked at the end —
              // '& 31' just keeps the shift defined
              bits |= uint64_t((mask >> (v & 31)) & 1) << j;
              v >>= 8;
            }
            word |= bits << (k * 8);
          }
          out[w] = word;
        }
      } else {
        for (uint32_t w = wave; w < nwords; w += nwaves) {
          uint32_t row = r0 + w * 64 + lane;
          bool pred = row < r1 ? d_eval_fixed_row(lb, row) : false;
          uint64_t word = __ballot(pred);
          if (lane == 0) out[w] = word;
        }
      }
      sp++;
    } else if (op.kind == kOpNot) {
      uint64_t* top = stack[sp - 1];
      for (uint32_t w = tid; w < nwords; w += blockDim.x) top[w] = ~top[w];
    } else {
      const int n = op.nargs;
      uint64_t* dst = stack[sp - n];
      for (uint32_t w = tid; w < nwords; w += blockDim.x) {
        uint64_t acc = dst[w];
        if (op.kind == kOpAnd) {
          for (int k = 1; k < n; k++) acc &= stack[sp - n + k][w];
        } else {
          for (int k = 1; k < n; k++) acc |= stack[sp - n + k][w];
        }
        dst[w] = acc;
      }
      sp -= n - 1;
    }
  }
  __syncthreads();

  // mask the tail word, write the result, count hits
  unsigned long long local = 0;
  for (uint32_t w = tid; w < nwords; w += blockDim.x) {
    uint64_t word = stack[0][w];
    uint32_t base = r0 + w * 64;
    uint32_t valid = min(64u, r1 - base);
    if (valid < 64) word &= (uint64_t(1) << valid) - 1;
    blk.bitmap_out[r0 / 64 + w] = word;
    local += __popcll(word);
  }
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off, 64);
  if (lane == 0) wave_sums[wave] = local;
  __syncthreads();
  if (tid == 0) {
    unsigned long long s = 0;
    for (int k = 0; k < nwaves; k++) s += wave_sums[k];
    atomicAdd(hits, s);
    // per-block popcount: the blockResult rowsLen / `| stats count()` fast
    // path (block_result.go:403-413)
    atomicAdd(blk.hits_out, s);
  }
}


// ---- gather kernels (blockResult materialization, §8f row 1) ----

__global__ __launch_bounds__(256) void gather_count_kernel(
    const DevGatherCol* __restrict__ gcols, const DevBlock* __restrict__ blocks,
    const DevChunk* __restrict__ chunks, DevChunkCount* __restrict__ out) {
  __shared__ unsigned long long byte_sums[4];
  __shared__ uint32_t row_sums[4];
  const DevChunk ck = chunks[blockIdx.x];
  const DevBlock blk = blocks[ck.block];
  const DevGatherCol gc = gcols[ck.block];
  const uint32_t r0 = ck.chunk * kChunkRows;
  const uint32_t r1 = min(blk.rows, r0 + kChunkRows);
  const uint32_t nwords = (r1 - r0 + 63) / 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  unsigned long long my_bytes = 0;
  uint32_t my_rows = 0;
  for (uint32_t w = wave; w < nwords; w += nwaves) {
    const uint64_t word = blk.bitmap_out[r0 / 64 + w];
    const uint32_t row = r0 + w * 64 + lane;
    unsigned long long len = 0;
    if ((word >> lane) & 1) {
      len = d_gather_len(gc, row);
      my_rows++;
    }
    my_bytes += len;
  }
  for (int off = 32; off > 0; off >>= 1) {
    my_bytes += __shfl_down(my_bytes, off, 64);
    my_rows += __shfl_down(my_rows, off, 64);
  }
  if (lane == 0) {
    byte_sums[wave] = my_bytes;
    row_sums[wave] = my_rows;
  }
  __syncthreads();
  if (tid == 0) {
    unsigned long long b = 0;
    uint32_t r = 0;
    for (int k = 0; k < nwaves; k++) {
      b += byte_sums[k];
      r += row_sums[k];
    }
    out[blockIdx.x].rows = r;
    out[blockIdx.x].bytes = b;
  }
}

__global__ __launch_bounds__(256) void gather_copy_kernel(
    const DevGatherCol* __restrict__ gcols, const DevBlock* __restrict__ blocks,
    const DevChunk* __restrict__ chunks, const DevChunkBase* __restrict__ bases,
    uint8_t* __restrict__ out_bytes, unsigned long long* __restrict__ out_offs,
    unsigned long long* __restrict__ out_rowids) {
  // per-word (row, byte) bases within the chunk, computed by thread 0
  __shared__ uint32_t word_rows[kChunkWords];
  __shared__ unsigned long long word_bytes[kChunkWords];
  const DevChunk ck = chunks[blockIdx.x];
  const DevBlock blk = blocks[ck.block];
  const DevGatherCol gc = gcols[ck.block];
  const DevChunkBase base = bases[blockIdx.x];
  const uint32_t r0 = ck.chunk * kChunkRows;
  const uint32_t r1 = min(blk.rows, r0 + kChunkRows);
  const uint32_t nwords = (r1 - r0 + 63) / 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int nwaves = blockDim.x >> 6;

  // per-word totals
  for (uint32_t w = wave; w < nwords; w += nwaves) {
    const uint64_t word = blk.bitmap_out[r0 / 64 + w];
    const uint32_t row = r0 + w * 64 + lane;
    unsigned long long len = ((word >> lane) & 1) ? d_gather_len(gc, row) : 0;
    unsigned long long sum = len;
    for (int off = 32; off > 0; off >>= 1) sum += __shfl_down(sum, off, 64);
    if (lane == 0) {
      word_rows[w] = uint32_t(__popcll(word));
      word_bytes[w] = sum;
    }
  }
  __syncthreads();
  if (tid == 0) {
    // exclusive scan over the chunk's words
    uint32_t racc = 0;
    unsigned long long bacc = 0;
    for (uint32_t w = 0; w < nwords; w++) {
      uint32_t r = word_rows[w];
      unsigned long long b = word_bytes[w];
      word_rows[w] = racc;
      word_bytes[w] = bacc;
      racc += r;
      bacc += b;
    }
  }
  __syncthreads();

  for (uint32_t w = wave; w < nwords; w += nwaves) {
    const uint64_t word = blk.bitmap_out[r0 / 64 + w];
    const uint32_t row = r0 + w * 64 + lane;
    const bool mine = (word >> lane) & 1;
    unsigned long long len = mine ? d_gather_len(gc, row) : 0;
    // inclusive scan of lengths over the wave
    unsigned long long incl = len;
    for (int off = 1; off < 64; off <<= 1) {
      unsigned long long up = __shfl_up(incl, off, 64);
      if (lane >= off) incl += up;
    }
    if (mine) {
      const unsigned long long row_out =
          base.row_base + word_rows[w] +
          (unsigned long long)__popcll(word & ((uint64_t(1) << lane) - 1));
      const unsigned long long byte_out = base.byte_base + word_bytes[w] + incl - len;
      d_gather_write(gc, row, out_bytes + byte_out);
      out_offs[row_out] = byte_out;
      if (out_rowids) out_rowids[row_out] = base.gid_base + row;
    }
  }
}

extern "C" hipError_t vql_launch_gather_count(const DevGatherCol* gcols,
                                              const DevBlock* blocks,
                                              const DevChunk* chunks,
                                              uint32_t nchunks, DevChunkCount* out,
                                              hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  hipLaunchKernelGGL(gather_count_kernel, dim3(nchunks), dim3(256), 0, stream,
                     gcols, blocks, chunks, out);
  return hipGetLastError();
}

extern "C" hipError_t vql_launch_gather_copy(
    const DevGatherCol* gcols, const DevBlock* blocks, const DevChunk* chunks,
    uint32_t nchunks, const DevChunkBase* bases, uint8_t* out_bytes,
    unsigned long long* out_offs, unsigned long long* out_rowids,
    hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  hipLaunchKernelGGL(gather_copy_kernel, dim3(nchunks), dim3(256), 0, stream,
                     gcols, blocks, chunks, bases, out_bytes, out_offs, out_rowids);
  return hipGetLastError();
}

// host-side launcher (called from vql_api.cpp)
extern "C" hipError_t vql_launch_scan(const DevOp* ops, int nops,
                                      const DevLeafBlock* lbs, int nleaves,
                                      const DevBlock* blocks, const DevChunk* chunks,
                                      uint32_t nchunks, unsigned long long* hits,
                                      hipStream_t stream) {
  if (nchunks == 0) return hipSuccess;
  hipLaunchKernelGGL(scan_program_kernel, dim3(nchunks), dim3(256), 0, stream, ops,
                     nops, lbs, nleaves, blocks, chunks, hits);
  return hipGetLastError();
}

}  // namespace vl

// ---- GPU ingest-side bloom build ----
// Restates the write path column.mustWriteTo bloom leg (block.go:160-168):
// tokenizeHashes (hash_tokenizer.go:68-166, hash-level dedup) ->
// bloomFilterMarshalHashes (bloomfilter.go:49-55,83-121).  Bloom bits are
// idempotent, so the kernel inserts every token occurrence's XXH64 into a
// global open-addressing set (the dedup gives the unique-hash COUNT that
// sizes the filter), then a second kernel sets the 6 chained probe bits per
// unique hash.

#include "../core/xxhash64.h"

namespace vl {

// one lane per row: tokenize + insert token hashes into the set
__global__ __launch_bounds__(256) void bloom_tokenize_kernel(
    const uint8_t* __restrict__ data, const uint32_t* __restrict__ offsets,
    uint32_t rows, unsigned long long* __restrict__ slots, uint32_t cap_mask,
    unsigned long long* __restrict__ unique_count, int* __restrict__ overflow) {
  const uint32_t row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= rows) return;
  const uint8_t* p = data + offsets[row];
  const long n = long(offsets[row + 1]) - long(offsets[row]);
  bool ascii = true;
  for (long i = 0; i < n; i++) {
    if (p[i] >= 0x80) {
      ascii = false;
      break;
    }
  }
  long i = 0;
  while (i < n) {
    long start, end;
    if (ascii) {
      while (i < n && !d_is_token_char(p[i])) i++;
      start = i;
      while (i < n && d_is_token_char(p[i])) i++;
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
    // open addressing, linear probing; slot 0 == empty (h==0 never occurs in
    // practice; documented)
    uint32_t idx = uint32_t(h) & cap_mask;
    for (uint32_t probes = 0;; probes++) {
      if (probes > cap_mask) {
        *overflow = 1;
        return;
      }
      unsigned long long prev = atomicCAS(&slots[idx], 0ULL, h);
      if (prev == 0) {
        atomicAdd(unique_count, 1ULL);
        break;
      }
      if (prev == h) break;
      idx = (idx + 1) & cap_mask;
    }
  }
}

// one lane per slot: 6 chained probe hashes per unique token hash
// (appendHashesHashes, bloomfilter.go:159-169), bits set with atomicOr
__global__ __launch_bounds__(256) void bloom_setbits_kernel(
    const unsigned long long* __restrict__ slots, uint32_t cap,
    unsigned long long* __restrict__ bits, uint64_t max_bits) {
  const uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= cap) return;
  uint64_t h = slots[i];
  if (h == 0) return;
  uint64_t buf = h;
  for (int k = 0; k < 6; k++) {
    uint64_t hk = vl::xxhash64(&buf, 8);
    buf++;
    uint64_t idx = hk % max_bits;
    atomicOr(&bits[idx >> 6], 1ULL << (idx & 63));
  }
}

extern "C" int vql_launch_bloom_tokenize(const void* data, const void* offsets,
                                         unsigned rows, void* slots,
                                         unsigned cap_mask, void* unique_count,
                                         void* overflow, void* stream) {
  dim3 grid((rows + 255) / 256), block(256);
  hipLaunchKernelGGL(bloom_tokenize_kernel, grid, block, 0,
                     (hipStream_t)stream, (const uint8_t*)data,
                     (const uint32_t*)offsets, rows,
                     (unsigned long long*)slots, cap_mask,
                     (unsigned long long*)unique_count, (int*)overflow);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

extern "C" int vql_launch_bloom_setbits(const void* slots, unsigned cap,
                                        void* bits, unsigned long long max_bits,
                                        void* stream) {
  dim3 grid((cap + 255) / 256), block(256);
  hipLaunchKernelGGL(bloom_setbits_kernel, grid, block, 0, (hipStream_t)stream,
                     (const unsigned long long*)slots, cap,
                     (unsigned long long*)bits, max_bits);
  return hipGetLastError() == hipSuccess ? 0 : -1;
}

}  // namespace vl
