// Device structures + kernel launcher declarations shared between
// kernels.hip (device) and plan.cpp (host).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace pmh {

// Hard capacity limits (validated host-side at plan creation):
//  - PMH_MAX_RUNS sorted runs per section (the reference's disk-spill path,
//    MergeSorter.java:112-125, stays out of scope while real compaction
//    shapes fit — 32 covers C5's 16->1 with 2x headroom);
//  - rows per run < 2^PMH_ROW_BITS (winner packing run:5 | row:27; a run is
//    the concatenation of its files, the cap applies to the run TOTAL).
constexpr int PMH_MAX_RUNS = 32;
constexpr int PMH_ROW_BITS = 27;
constexpr uint32_t PMH_ROW_MASK = ((uint32_t)1 << PMH_ROW_BITS) - 1;
// 3584-row tiles with 512-thread workgroups (~76 KB LDS, 2 workgroups/CU).
// Measured on MI355X: halving to 1792/256 for 4 WGs/CU made every phase
// SLOWER (fused 6.3 -> 7.8 ms; C3 emit +60%) — the doubled tile count costs
// more in per-tile setup/barriers/lookback than the extra resident tiles
// buy in latency hiding. The partition runs two-level regardless (coarse
// every PMH_COARSE_G-th boundary, windowed refine for the rest).
constexpr int PMH_TILE_THREADS = 512;
constexpr int64_t PMH_TILE_ROWS = 3584;
constexpr int PMH_COARSE_G = 16;
// packed-seq sentinel for rows removed by a deletion vector: the merge
// kernels treat such rows as nonexistent (ApplyDeletionVectorReader
// semantics). Real packed values are >= 0 (sequence numbers are
// non-negative counters).
constexpr int64_t PMH_DEAD = INT64_MIN;
constexpr int PMH_TILE_MAX = PMH_TILE_ROWS + PMH_MAX_RUNS;
constexpr int PMH_TILE_ITER =
    (PMH_TILE_MAX + PMH_TILE_THREADS - 1) / PMH_TILE_THREADS;

// A decoded or raw-PLAIN region of one column of one run. Columns staged by
// plan.cpp are CONTIGUOUS (n_pages == 1, direct addressing via addr0 — the
// PLAIN page payloads are packed per column at H2D time); the paged path
// (binary search by start_row) remains for future non-contiguous encodings.
struct DevPage {
    uint64_t addr;      // device pointer to first element
    int64_t start_row;  // first run-row covered by this page
};

struct DevCol {
    uint64_t addr0;        // n_pages == 1 fast path: element 0 address
    uint64_t valid0;       // byte-per-row validity array; 0 = all valid
    const DevPage *pages;  // device array, sorted by start_row
    int32_t n_pages;
    int32_t esize;  // element size in bytes (stored width)
};

// Host-prescanned RLE/bit-packed work unit (split at group boundaries).
struct RleChunk {
    uint64_t src;       // device pointer to first packed group (kind 1)
    int64_t out_start;  // output element index
    int32_t count;
    int32_t kind;       // 0 = RLE run, 1 = bit-packed groups
    uint32_t value;     // RLE literal (kind 0)
    int32_t bit_width;  // per-page bit width (dict-id streams vary per page)
    int64_t aux;        // def-level streams: dense (non-null) values before
                        // this chunk within its (run, column)
    // absolute device addresses (patched at staging finalize) so decode
    // chunks from every (run, column) batch into ONE launch:
    uint64_t out_addr;    // positioned column values
    uint64_t valid_addr;  // byte validity
    uint64_t dense_addr;  // dense (non-null) values
    int32_t esize;
    int32_t _pad2;
};

// Host-prescanned ORC RLEv2 / byte-RLE work unit (one run per chunk; runs
// are <= 512 values and byte-aligned). Restates the ORC v1 spec RLEv2 as
// consumed by the reference through orc-core 1.9.8.
struct Rlev2Chunk {
    uint64_t src;        // packed values (byte-aligned within DATA stream)
    int64_t out_start;   // dense output element index
    int32_t count;
    uint8_t kind;        // 0 SHORT_REPEAT, 1 DIRECT, 2 PATCHED_BASE,
                         // 3 DELTA, 4 BYTE_RUN, 5 BYTE_LITERAL
    uint8_t width;       // packed bit width (0 = fixed-delta / none)
    uint8_t is_signed;   // zigzag-decode DIRECT/SHORT_REPEAT values
    uint8_t out_esize;   // 4 or 8
    int64_t base;        // SR/BYTE_RUN: value; DELTA/PATCHED: base
    int64_t delta;       // DELTA: delta base (sign = direction)
    uint64_t patch_src;  // PATCHED: packed patch entries
    uint16_t patch_pl;   // PATCHED: number of patch entries
    uint8_t patch_pw;    // PATCHED: patch value bits (decoded)
    uint8_t patch_pgw;   // PATCHED: gap bits
    uint8_t patch_cfb;   // PATCHED: packed entry bits
    uint8_t dense_target;  // 0: contig (positioned); 1: dense buffer
                           // (PRESENT scatter follows)
    uint8_t _pad[2];
    uint64_t out_addr;     // absolute output base (patched at finalize)
};

// Host-prescanned DELTA_BINARY_PACKED work unit: one parquet delta BLOCK
// (VectorizedDeltaBinaryPackedReader.java; parquet-format encodings.md):
// per block, <= 8 miniblocks of vpm values each, bit widths packed into
// `widths` (8 bits each), LSB-first bit packing. Blocks chain: the value
// base entering block b = page first_value + sum of all prior blocks'
// delta sums — computed on device (k_delta_sum + per-stream scan).
struct DeltaChunk {
    uint64_t src;       // packed miniblock data (byte-aligned)
    uint64_t out_addr;  // absolute output base (patched at finalize)
    int64_t out_start;  // output index of this block's FIRST delta value
    int64_t min_delta;
    int32_t count;      // real delta values in this block
    int32_t stream;     // delta-stream index (for base resolution)
    uint64_t widths;    // miniblock bit widths, 8 bits each
    int16_t vpm;        // values per miniblock
    int16_t n_mini;
    int32_t out_esize;  // 4 or 8
};

// One DELTA stream (= one column chunk's page chain? no — one PAGE: pages
// are self-contained): chunk range + the page's first value and where it
// lands in the output.
struct DeltaStream {
    int64_t chunk_lo;
    int64_t chunk_hi;
    int64_t first;       // header first value = output element out0
    int64_t out0;        // output index of the first value
    uint64_t out_addr;   // absolute output base (patched at finalize)
    int32_t out_esize;
    int32_t _pad;
};

extern "C" {

// DELTA_BINARY_PACKED decode, three phases batched across all chunks:
// per-block delta sums, per-stream base scan, then unpack + wave scan +
// base add into the output column.
hipError_t pmh_launch_delta_sum(const DeltaChunk *chunks, int64_t n_chunks,
                                int64_t *sums, hipStream_t stream);
hipError_t pmh_launch_delta_scan(const DeltaStream *streams,
                                 int64_t n_streams, const int64_t *sums,
                                 int64_t *bases, hipStream_t stream);
hipError_t pmh_launch_delta_emit(const DeltaChunk *chunks, int64_t n_chunks,
                                 const int64_t *bases, hipStream_t stream);


// Decode ORC RLEv2 / byte-RLE work chunks into a dense typed column
// (int32 or int64 elements per chunk.out_esize). One wave per chunk.

// value-filter term (single-run sections only — MergeFileSplitRead.java:
// 227-239: value filters must not push into overlapping sections).
// op: 0 eq, 1 ne, 2 lt, 3 le, 4 gt, 5 ge, 6 is_null, 7 is_not_null
struct FilterTerm {
    int32_t col;     // index into the plan's column list
    int32_t op;
    int64_t ilit;    // integer/string-id literal
    double dlit;     // float literal
    int32_t is_fp;   // compare as double
    int32_t pad;
};
// mark rows FAILING the conjunction as dead in the tombstone array
hipError_t pmh_launch_filter(const DevCol *cols, int n_cols,
                             const struct FilterTerm *terms, int n_terms,
                             int64_t rows, uint8_t *tomb,
                             hipStream_t stream);

// full-compaction changelog chain (FullChangelogMergeFunctionWrapper):
// finalize compacts the merge pass's provisional per-group entries into
// rows (evaluating row-deduplicate value equality); emit gathers them.
hipError_t pmh_launch_cl_finalize(const DevCol *cols,
                                  const uint8_t *col_dtype, int n_cols,
                                  int first_val, int k,
                                  const uint64_t *cl_entries,
                                  uint64_t *cl_out, int32_t *cl_counts,
                                  int64_t n_tiles, int64_t tile_rows,
                                  hipStream_t stream);
hipError_t pmh_launch_cl_emit(const DevCol *cols, const uint8_t *col_dtype,
                              const uint8_t *col_nullable, int n_cols,
                              int kind_col, const uint64_t *cl_rows,
                              const int32_t *cl_counts,
                              const int64_t *cl_offsets, int64_t n_tiles,
                              int64_t tile_rows, void *const *out_ptrs,
                              uint8_t *const *out_valid, hipStream_t stream);

// on-GPU zstd page decompression (k_zstd_pages): one job per parquet page.
// scratch must hold n_jobs * PZ_SLOT bytes (PZ_SLOT in zstd_core.h);
// status[j] = decompressed bytes or a PZ_ERR_* code.
struct ZstdJob {
    uint64_t src_off, dst_off;  // into the batch src / dst blobs
    uint32_t src_len, dst_len;
};
// on-GPU zstd page COMPRESSION (k_zstd_compress): one job per page;
// scratch holds n * sizeof(PzEnc) bytes; status[j] = compressed size or
// a PZ_ERR_* code.
hipError_t pmh_launch_zstd_compress(const uint8_t *src,
                                    const struct ZstdJob *jobs, int n,
                                    uint8_t *dst, uint8_t *scratch,
                                    int64_t *status, hipStream_t stream);
hipError_t pmh_launch_zstd_pages(const uint8_t *src,
                                 const struct ZstdJob *jobs, int n,
                                 uint8_t *dst, uint8_t *scratch,
                                 int64_t *status, hipStream_t stream);

hipError_t pmh_launch_rlev2(const Rlev2Chunk *chunks, int64_t n_chunks,
                            hipStream_t stream);

hipError_t pmh_launch_partition(const DevCol *keys, const int64_t *lens, int k,
                                int64_t tile_rows, int64_t n_bounds,
                                int64_t total_rows, int32_t *cuts,
                                hipStream_t stream);

hipError_t pmh_launch_merge_tiles(const DevCol *keys, const DevCol *seqs,
                                  const DevCol *kinds, const int64_t *lens,
                                  int k, const int32_t *cuts, int64_t n_tiles,
                                  int64_t tile_rows, int flags,
                                  const uint64_t *tombs,
                                  uint32_t *winners, int32_t *tile_counts,
                                  uint16_t *group_start, uint32_t *err_flag,
                                  const uint8_t *run_levels, int max_level,
                                  uint64_t *cl_entries, int32_t *cl_counts,
                                  hipStream_t stream);

hipError_t pmh_launch_scan_tiles(const int32_t *tile_counts, int64_t n_tiles,
                                 int64_t *tile_offsets, int64_t *total_out,
                                 hipStream_t stream);

// FUSED merge + emit (deduplicate / first-row): ticket-ordered persistent
// workgroups, decoupled-lookback output offsets (packed agent-scope atomic
// per tile in `status`, zeroed before launch along with `ticket`), emission
// from LDS-staged columns. Replaces merge_tiles + scan_tiles + emit for
// non-member-list engines. key_col = -1 for composite keys (key columns
// then emit through the generic column path).
hipError_t pmh_launch_merge_emit(const DevCol *keys, const DevCol *seqs,
                                 const DevCol *kinds, const int64_t *lens,
                                 int k, const int32_t *cuts,
                                 int64_t tile_base, int64_t tile_limit,
                                 int64_t n_tiles,
                                 int64_t tile_rows, int flags,
                                 const uint64_t *tombs,
                                 const DevCol *cols, const uint8_t *col_dtype,
                                 const uint8_t *col_nullable, int n_cols,
                                 int key_col, int seq_col, int kind_col,
                                 const int16_t *useq_cols, int n_useq,
                                 uint64_t *status, uint64_t *ticket,
                                 int64_t *total_out, uint32_t *dense_winners,
                                 void *const *out_ptrs,
                                 uint8_t *const *out_valid,
                                 uint32_t *err_flag, hipStream_t stream);

// SPLIT-mode value emission: gathers value columns by the densely-written
// packed winners (k_merge_emit with dense_winners != null emitted key/seq/
// kind and the winner list; this kernel runs at full occupancy, no LDS).
hipError_t pmh_launch_emit_dense(const DevCol *cols,
                                 const uint8_t *col_dtype,
                                 const uint8_t *col_nullable, int n_cols,
                                 int key_col, int seq_col, int kind_col,
                                 const uint32_t *winners,
                                 int64_t t0, int64_t t1,
                                 const uint64_t *status,
                                 void *const *out_ptrs,
                                 uint8_t *const *out_valid,
                                 hipStream_t stream);

hipError_t pmh_launch_emit(const DevCol *cols, const uint8_t *col_dtype,
                           const uint8_t *col_nullable, int n_cols, int k,
                           const uint32_t *winners,
                           const int32_t *tile_counts,
                           const int64_t *tile_offsets, int64_t n_tiles,
                           int64_t tile_rows, const int64_t *total_out,
                           void *const *out_ptrs, uint8_t *const *out_valid,
                           hipStream_t stream);

hipError_t pmh_launch_rle_decode(const RleChunk *chunks, int64_t n_chunks,
                                 int32_t *out, hipStream_t stream);

// Decode def-level streams (bit width 1) and position the dense PLAIN
// values: valid[row] = level; out[row] = dense[aux + prefix] for valid rows
// (VectorizedColumnReader null handling, VectorizedColumnReader.java:143-241).
hipError_t pmh_launch_level_scatter(const RleChunk *chunks, int64_t n_chunks,
                                    hipStream_t stream);

// PartialUpdate emit: per owned group, overlay non-null fields in ascending
// (seq, isAdd) order (PartialUpdateMergeFunction.java:188-215 + Reducer
// wrapper bypass). members/group_start written by k_merge_tiles in PU mode.
// run_masks: device array of k per-run packed-validity pointers
// (k_pack_valid), or null for the legacy per-column byte walk (>64 cols).
hipError_t pmh_launch_emit_pu(const DevCol *cols, const uint8_t *col_dtype,
                              const uint8_t *col_nullable, int n_cols, int k,
                              int seq_col, int kind_col, int flags,
                              const uint32_t *members,
                              const uint16_t *group_start,
                              const int64_t *tile_offsets, int64_t n_tiles,
                              int64_t tile_rows, const int64_t *total_out,
                              uint64_t *const *run_masks,
                              void *const *out_ptrs,
                              uint8_t *const *out_valid, hipStream_t stream);

// Pack one run's per-column validity bytes into u64 row masks (bit c =
// column c non-null; columns without staged nulls contribute 1).
hipError_t pmh_launch_pack_valid(const DevCol *cols, int n_cols, int64_t rows,
                                 uint64_t *mask, hipStream_t stream);

// Build one run's order-preserving composite key from <= 8 integer key
// columns whose widths sum to <= 64 bits (spec packed 8 bits per sub-key).
hipError_t pmh_launch_composite(const DevCol *keys, int nk, uint64_t shifts,
                                uint64_t bits, int64_t rows, int64_t *ckey,
                                hipStream_t stream);

// Aggregation emit: per owned group, fold members in ascending (seq, isAdd)
// order through per-column FieldAggregators (AggregateMergeFunction.java:
// 82-125; default last_non_null_value, :201). col_agg holds one PMH_AGG_*
// code per output column. INSERT-only streams in v1 (retracts are detected
// by k_merge_tiles in PU mode and fail the read).
enum {
    PMH_AGG_LAST_NON_NULL = 0,  // FieldLastNonNullValueAgg (the default)
    PMH_AGG_LAST_VALUE = 1,     // FieldLastValueAgg
    PMH_AGG_FIRST_VALUE = 2,    // FieldFirstValueAgg
    PMH_AGG_FIRST_NON_NULL = 3, // FieldFirstNonNullValueAgg
    PMH_AGG_SUM = 4,            // FieldSumAgg
    PMH_AGG_MAX = 5,            // FieldMaxAgg
    PMH_AGG_MIN = 6,            // FieldMinAgg
    PMH_AGG_PRIMARY_KEY = 7,    // FieldPrimaryKeyAgg (agg = retract = input)
};
// col_agg bit 0x80: FieldIgnoreRetractAgg wrapper (fields.<f>.ignore-retract
// = true): retract records leave the accumulator untouched
constexpr uint8_t PMH_AGG_IGNORE_RETRACT = 0x80;
// PartialUpdate with sequence groups (PartialUpdateMergeFunction.java:
// 219-377): per-group last-prefix-max-achiever resolution; retracts null
// their group members. col_group[c] = group id or 0xff; sg_fields packs 4
// sequence-field column indices per group; no per-field aggregators in v1.
hipError_t pmh_launch_emit_pu_sg(
    const DevCol *cols, const uint8_t *col_dtype,
    const uint8_t *col_nullable, int n_cols, int k, int seq_col,
    int kind_col, int flags, const uint8_t *col_group,
    const int16_t *sg_fields, const uint8_t *sg_nseq, int n_groups,
    const uint32_t *members, const uint16_t *group_start,
    const int64_t *tile_offsets, int64_t n_tiles, int64_t tile_rows,
    const int64_t *total_out, uint64_t *const *run_masks,
    void *const *out_ptrs, uint8_t *const *out_valid, hipStream_t stream);

hipError_t pmh_launch_emit_agg(const DevCol *cols, const uint8_t *col_dtype,
                               const uint8_t *col_nullable,
                               const uint8_t *col_agg, int n_cols, int k,
                               int seq_col, int kind_col, int flags,
                               const uint32_t *members,
                               const uint16_t *group_start,
                               const int64_t *tile_offsets, int64_t n_tiles,
                               int64_t tile_rows, const int64_t *total_out,
                               uint64_t *const *run_masks,
                               void *const *out_ptrs,
                               uint8_t *const *out_valid, hipStream_t stream);

hipError_t pmh_launch_dict_gather(const int32_t *ids, const void *dict,
                                  int64_t n, void *out, int esize,
                                  hipStream_t stream);

}  // extern "C"

}  // namespace pmh
