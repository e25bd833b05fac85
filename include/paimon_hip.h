/* paimon_hip.h — C-ABI of libpaimon_hip.so: the MI355X-native drop-in for
 * Apache Paimon's merge-on-read hot path.
 *
 * Replaced reference surfaces (apache/paimon @ 2026-08-21), per entry point:
 *  - pmh_plan_create + pmh_read_next replace
 *      SplitRead<KeyValue>.createReader(Split) -> RecordReader<KeyValue>
 *      (paimon-core/.../operation/SplitRead.java:39-63) as implemented by
 *      MergeFileSplitRead.createReader / createMergeReader
 *      (paimon-core/.../operation/MergeFileSplitRead.java:242-270), i.e. the
 *      IntervalPartition sectioning (mergetree/compact/IntervalPartition.java:
 *      67-125), per-section sort-merge (mergetree/compact/SortMergeReader.java:
 *      41-57 with the default LOSER_TREE engine), merge functions
 *      (DeduplicateMergeFunction.java:48-62, PartialUpdateMergeFunction.java:
 *      148-397) wrapped by ReducerMergeFunctionWrapper.java:53-73, and
 *      DropDeleteReader.java:53-61.
 *  - The batch contract mirrors RecordReader<T>
 *      (paimon-common/.../reader/RecordReader.java:40-72): pmh_read_next
 *      returns one batch per section (the loser-tree engine yields one batch
 *      per section, SortMergeReaderWithLoserTree.java:77-83); returned
 *      buffers stay valid until the next pmh_read_next / pmh_plan_close on
 *      the same plan (releaseBatch reuse contract). One plan = one HIP
 *      stream; concurrent plans are allowed (one per bucket).
 *  - pmh_write_parquet is the encode half of the compaction surface
 *      (CompactRewriter.rewrite, mergetree/compact/CompactRewriter.java:
 *      29-56): the merge half runs through a plan, the rolling write-back
 *      through this writer (driven by paimon_amd/compact.py in-repo).
 *
 * JNI precedent in the reference: the Vortex JNI reader
 * (paimon-vortex/.../dev/vortex/jni/NativeRuntime.java:30-34) — a Java-side
 * binding of exactly this shape is sketched in INTEGRATION.md.
 *
 * Error model: functions returning pointers return NULL on error; functions
 * returning int/int64 return negative on error. pmh_last_error() returns a
 * thread-local message (mirrors IOException propagation of RecordReader).
 */
#ifndef PAIMON_HIP_H
#define PAIMON_HIP_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct pmh_session_t pmh_session_t;
typedef struct pmh_plan_t pmh_plan_t;

/* Column dtypes (paimon logical types on the hot path; TINYINT is stored as
 * parquet INT32 and returned here as INT8 — KeyValueSerializer.java:58-66
 * row layout: key cols | _SEQUENCE_NUMBER | _VALUE_KIND | value cols). */
enum pmh_dtype {
    PMH_DT_INT8 = 1,
    PMH_DT_INT16 = 2,
    PMH_DT_INT32 = 3,
    PMH_DT_INT64 = 4,
    PMH_DT_FLOAT32 = 5,
    PMH_DT_FLOAT64 = 6,
    /* Dictionary-backed string column (CHAR/VARCHAR -> parquet BYTE_ARRAY,
     * ParquetSchemaConverter.java:120-128): data = int32 ids into the
     * column's dictionary (dict_data/dict_offsets/dict_len) — the same
     * shape the reference's WritableColumnVector dictionary support and
     * parquet's own dictionary encoding use. DECIMAL(p,s) is not a
     * separate dtype: p <= 9 rides INT32, p <= 18 INT64 (unscaled values,
     * ParquetSchemaConverter.java:153-171) with precision/scale set. */
    PMH_DT_STRING = 7,
};

typedef struct pmh_col {
    const char *name;
    int32_t dtype;         /* enum pmh_dtype */
    const void *data;      /* columnar values (PMH_DT_STRING: int32 ids);
                              device ptr unless host batch */
    const uint8_t *valid;  /* byte-per-row validity (1=non-null); NULL = all
                              valid. (Arrow bitmap export: later round.) */
    /* PMH_DT_STRING: the column dictionary (HOST-resident even for device
     * batches — it is plan-level metadata built at staging). Entry i =
     * dict_data[dict_offsets[i] .. dict_offsets[i+1]). */
    const void *dict_data;
    const int32_t *dict_offsets;  /* dict_len + 1 entries */
    int32_t dict_len;
    /* DECIMAL annotation on INT32/INT64 columns (0 = plain integer). */
    int32_t precision;
    int32_t scale;
} pmh_col;

typedef struct pmh_batch {
    int64_t n_rows;
    int32_t n_cols;        /* key cols, then seq, kind, then value cols */
    int32_t device;        /* HIP device ordinal; -1 = host-resident copy */
    const pmh_col *cols;
} pmh_batch;

typedef struct pmh_stats {
    int64_t rows_in;         /* records fed to the merge */
    int64_t rows_out;        /* records emitted */
    int64_t hbm_bytes_algo;  /* encoded input bytes staged in HBM */
    double decode_ms;        /* device time in decode kernels */
    double partition_ms;     /* device time in merge-path partition */
    double merge_ms;         /* device time in the tile merge kernel */
    double scan_ms;          /* device time in the tile-offset scan */
    double emit_ms;          /* device time in gather/emit kernels */
    double total_device_ms;  /* end-to-end device time of read_next calls */
    double h2d_ms;           /* untimed-region staging cost, informational */
    int64_t path_mode;       /* 0 = 3-kernel chain, 1 = fused in-kernel
                              * emission, 2 = fused + split value emission */
    int64_t gpu_zstd_pages;  /* pages decompressed by k_zstd_pages */
} pmh_stats;

/* Session: owns the device + stream pool. device < 0 opens a host-only
 * session usable for metadata entry points (footer parse, planning). */
pmh_session_t *pmh_open_session(int device);
void pmh_close_session(pmh_session_t *s);

/* plan_json (one bucket's DataSplit, table/source/DataSplit.java:63-75):
 * {
 *   "key_cols":   [{"name": "_KEY_k", "type": "int64"}],
 *   "value_cols": [{"name": "v_k", "type": "int64"}, ...],   // read type
 *   "merge_engine": "deduplicate" | "partial-update",
 *   "drop_delete": true,         // !forceKeepDelete
 *   "ignore_delete": false,      // CoreOptions.IGNORE_DELETE
 *   "output": "device" | "host",
 *   "files": [{"path": "...", "rowCount": N, "minKey": x, "maxKey": y,
 *              "level": L}, ...]                              // DataFileMeta
 * }
 * Staging (file read + H2D of encoded column chunks + page tables) happens
 * here; pmh_read_next launches only device work. */
pmh_plan_t *pmh_plan_create(pmh_session_t *s, const char *plan_json);

/* Returns rows in the batch (one section per call), 0 at end of input,
 * negative on error. Batch buffers valid until next call / plan close. */
int64_t pmh_read_next(pmh_plan_t *p, pmh_batch *out);

int pmh_plan_close(pmh_plan_t *p);

/* Rewind the plan to its first section without restaging: the next
 * pmh_read_next re-runs the full device pipeline on the resident encoded
 * data. Used by benchmarks to repeat the timed region; stats accumulate. */
/* Changelog batch of the LAST pmh_read_next (changelog_producer =
 * "full-compaction" in the plan JSON, with "max_level" and optional
 * "changelog_row_deduplicate"): the FullChangelogMergeFunctionWrapper
 * stream (INSERT / UPDATE_BEFORE / UPDATE_AFTER / DELETE rows in key
 * order, same schema as the main batch). Valid until the next read_next.
 * Returns row count or < 0. */
int64_t pmh_changelog_next(pmh_plan_t *plan, pmh_batch *out);

int pmh_plan_reset(pmh_plan_t *p);

int pmh_stats_get(pmh_plan_t *p, pmh_stats *out);

const char *pmh_last_error(void);

/* ---- host-only debug/metadata entry points (no GPU required) ---- */

/* Parse a parquet footer + page headers; returns a malloc'd JSON string
 * (caller frees with pmh_free_string). Used by CPU tests to pin the native
 * thrift parser against pyarrow metadata. */
char *pmh_debug_footer_json(const char *path);
void pmh_free_string(char *s);

/* Write a Parquet v1 data file (PLAIN pages) from HOST columnar
 * buffers — the compaction write-back half of the CompactRewriter surface
 * (what KeyValueDataFileWriter + the vendored parquet-mr writer do in the
 * reference, io/KeyValueDataFileWriter.java:121-170). cols[i].data/valid
 * are host pointers at the column's output width (TINYINT/SMALLINT widen to
 * the INT32 physical type on write, matching the read path). No GPU
 * required. row_group_rows/page_rows <= 0 pick defaults (1M / 64k).
 * compression: NULL or "NONE" for uncompressed, "zstd" for ZSTD pages.
 * Returns 0, or -1 with pmh_last_error() set. */
int pmh_write_parquet(const pmh_col *cols, int32_t n_cols, int64_t n_rows,
                      const char *path, int64_t row_group_rows,
                      int64_t page_rows, const char *compression);

/* Raw snappy block decode (the library's from-scratch decoder, used for
 * SNAPPY parquet pages / ORC chunks). Returns decompressed size or -1 with
 * pmh_last_error() set. Exposed so CPU tests can pin the decoder against an
 * independent compressor. */
int64_t pmh_debug_snappy(const void *src, int64_t n, void *dst, int64_t cap);

/* Decode one zstd frame with the from-scratch scalar core (zstd_core.h,
 * RFC 8878 restatement) on the host — test/validation entry for the GPU
 * page decoder which shares the same core. Returns bytes or < 0. */
int64_t pmh_debug_zstd_cpu(const void *src, int64_t n, void *dst,
                           int64_t cap);

/* Decode one zstd frame ON THE GPU (k_zstd_pages batch of 1); `expected`
 * is the known decompressed size. Returns expected or < 0. */
int64_t pmh_debug_zstd_gpu(const void *src, int64_t n, void *dst,
                           int64_t expected);

/* Compress one frame with the from-scratch encoder (scalar core; the GPU
 * page compressor shares it). Returns compressed bytes or < 0. */
int64_t pmh_debug_zstd_enc_cpu(const void *src, int64_t n, void *dst,
                               int64_t cap);

/* Same, on the GPU (k_zstd_compress batch of 1). */
int64_t pmh_debug_zstd_enc_gpu(const void *src, int64_t n, void *dst,
                               int64_t cap);

/* Parse one deletion vector from a DV index file slice (DeletionFile
 * {path, offset, length}; BitmapDeletionVector.java:98-112 wrapper around
 * the portable Roaring serialization). Writes up to `cap` deleted
 * positions ascending; returns the total count or -1. CPU-only debug
 * entry pinning the parser against independently-serialized fixtures. */
int64_t pmh_debug_parse_dv(const char *path, int64_t offset, int64_t length,
                           int64_t *out, int64_t cap);

/* Restatement of IntervalPartition.partition() for int64 keys
 * (mergetree/compact/IntervalPartition.java:67-125): given n files'
 * (minKey, maxKey), writes section id and run-within-section id per file
 * (by input order). Returns number of sections, or negative on error. */
int pmh_debug_interval_partition(int n, const int64_t *min_keys,
                                 const int64_t *max_keys,
                                 int32_t *out_section, int32_t *out_run);

#ifdef __cplusplus
}
#endif
#endif /* PAIMON_HIP_H */
