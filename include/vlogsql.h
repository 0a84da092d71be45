/* vlogsql.h — C ABI of the MI355X-native VictoriaLogs block-scan engine.
 *
 * This is the drop-in boundary: the entry points a cgo shim
 * inside VictoriaLogs would bind to replace the per-block filter evaluation
 * invoked at lib/logstorage/block_search.go:215
 * (filter.applyToBlockSearch(bs, bm), interface lib/logstorage/filter.go:8-20)
 * from the search worker loop (lib/logstorage/storage_search.go:1040-1066).
 * See INTEGRATION.md for the Go-side binding a maintainer would add.
 *
 * Bitmap layout contract: bit i of a block's result = row i of the block,
 * LSB-first within little-endian u64 words (lib/logstorage/bitmap.go:113-125).
 * Result bitmaps for a scanned range are concatenated per block in block
 * order, each block padded to a word boundary.
 *
 * Threading: one stage/scan context per caller thread; parts and compiled
 * filters are immutable after creation and may be shared
 * (mirrors blockSearch pooling, block_search.go:79-96).
 *
 * Errors: functions returning pointers return NULL on failure; functions
 * returning counts return -1.  vql_errstr() returns the thread-local message
 * (the reference panics on corruption, block_search.go:264,318 — we surface
 * hard errors instead).
 */
#ifndef VLOGSQL_H
#define VLOGSQL_H

#ifdef __cplusplus
extern "C" {
#endif

const char* vql_errstr(void);
/* 1 if the last error was an unsupported-construct rejection (valid LogsQL
 * outside the engine class, e.g. regex \p{...}) rather than
 * corruption/IO — the cgo shim keeps such filters on the host Go path
 * (INTEGRATION.md); the engine never silently falls back itself. */
int vql_error_unsupported(void);

/* Opens a reference-format part directory (FormatVersion 1..3; the file set
 * of lib/logstorage/filenames.go:3-24, read as part.go:105-173 does). */
void* vql_open_part(const char* dir);
void vql_close_part(void* part);
long vql_part_blocks(void* part);
long long vql_part_rows(void* part);
long vql_block_rows(void* part, long block);

/* Compiles a JSON filter tree (same shapes filter_test.go:34-59 builds
 * programmatically).  Node types: phrase, exact, regexp, prefix,
 * exact_prefix, sequence, any_case_phrase, any_case_prefix, in,
 * contains_any, contains_all, string_range, ipv4_range, len_range,
 * day_range, week_range, value_type, stream_id, eq_field, le_field, and,
 * or, not, time, range, noop — every applyToBlockSearch filter except
 * filter_stream (needs indexdb; stream_id is its resolved form). */
void* vql_compile_filter(const char* json);
void vql_free_filter(void* filter);

/* Steady-state path: decode blocks [block_lo, block_hi) of the part on the
 * host once and stage the needed columns + blooms into HBM of `device`.
 * block_hi < 0 means "all blocks". */
void* vql_stage(void* part, void* filter, int device, long block_lo,
                long block_hi);
/* Stages every block of nparts parts into ONE context/launch (the worker
 * batching of storage_search.go:1035-1067 is part-agnostic).  Bitmaps are
 * concatenated in (part, block) order. */
void* vql_stage_parts(void** parts, int nparts, void* filter, int device);
void vql_stage_free(void* stage);
long long vql_stage_bytes(void* stage);      /* bytes resident in HBM */
long long vql_stage_algo_bytes(void* stage); /* algorithmic bytes per pass */
long long vql_stage_rows(void* stage);
/* Rows of blocks that actually reach the kernel (blocks whose filter
 * program is statically all-zero are pruned from the dispatch, mirroring
 * the reference's header prunes, e.g. filter_time.go:114-137). */
long long vql_stage_live_rows(void* stage);

/* One scan pass over the staged blocks (one kernel launch).  Returns the
 * number of matched rows; bitmaps stay device-resident. */
long long vql_scan_staged(void* stage);
/* Kernel time of the last vql_scan_staged, measured with HIP events on the
 * stream the kernel was launched on. */
double vql_last_kernel_ms(void* stage);
/* Copies result bitmaps to host (concatenated per-block u64 words). */
int vql_fetch_bitmaps(void* stage, unsigned long long* out_words,
                      long long cap_words);
/* Per-block matched-row counts of the last scan: the blockResult rowsLen
 * popcount (block_result.go:403-413) and the `| stats count()` fast path
 * (SURVEY.md §8f row 2). */
int vql_fetch_block_hits(void* stage, unsigned long long* out,
                         long long cap_blocks);

/* blockResult materialization: after a scan, gathers
 * the matched rows' values of `field` (decoded string form, like
 * blockResult.getValues, block_result.go:306-478) into packed bytes +
 * per-row byte offsets (nrows+1) + optional global row ids. */
int vql_gather_sizes(void* stage, const char* field, unsigned long long* nrows,
                     unsigned long long* nbytes);
long long vql_gather(void* stage, const char* field, unsigned char* out_bytes,
                     long long bytes_cap, unsigned long long* out_offs,
                     long long offs_cap, unsigned long long* out_rowids);

/* GPU ingest-side bloom build: builds the marshaled
 * bloom-filter bytes for one column block (the write path's tokenizeHashes +
 * bloomFilterMarshalHashes, lib/logstorage/block.go:160-168), bit-identical
 * to the CPU writer.  offsets = u32[rows+1] over the concatenated value
 * bytes.  Returns the marshaled length; fills `out` when cap suffices. */
long long vql_bloom_build(const unsigned char* data, long long nbytes,
                          const unsigned int* offsets, long long rows,
                          int device, unsigned char* out, long long cap);

/* Whole-query driver (§8b vql_scan_query): shards `nparts` parts across
 * `ngpus` devices (round-robin by part, the reference's independent-worker
 * batching, storage_search.go:1035-1067), stages + scans each shard on its
 * device concurrently, and reduces the per-device matched-row counters (an
 * in-process sum; across processes the same reduction is RCCL all_reduce —
 * bench.py's multi-rank path).  Fills stats{matched_rows, bytes_scanned,
 * rows_scanned, elapsed_ms} and returns matched rows, or -1. */
typedef struct {
  unsigned long long matched_rows;
  unsigned long long rows_scanned;
  unsigned long long bytes_scanned; /* algorithmic bytes over all passes */
  double elapsed_ms;
} vql_stats;
long long vql_scan_query(void** parts, int nparts, void* filter, int ngpus,
                         vql_stats* stats);

/* Cold path (§8b vql_scan_batch): stage + scan + fetch + free in one call. */
long long vql_scan_batch(void* part, void* filter, long block_lo, long block_hi,
                         unsigned long long* out_words, long long cap_words,
                         unsigned long long* out_popcounts);

#ifdef __cplusplus
}
#endif

#endif /* VLOGSQL_H */
