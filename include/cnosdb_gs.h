/* cnosdb_gs — C ABI of the MI355X-native TSM scan/decode engine.
 *
 * This is the drop-in seam described in SURVEY.md §8b: a thin host shim
 * standing where CnosDB's `ColumnGroupReader::read` + `decode_pages` stand
 * (tskv/src/reader/column_group/mod.rs:33-70,195-243 and
 * tskv/src/tsm/reader.rs:494-560) calls these entry points instead of the
 * Rust CPU codec path.  A Rust shim (cxx/bindgen) would reconstitute
 * RecordBatches zero-copy from the output buffers; INTEGRATION.md shows the
 * binding a cnosdb maintainer would add.
 *
 * Conventions:
 *  - all functions return 0 (GS_OK) on success or a negative GsStatus;
 *    gs_last_error() returns a thread-local message for the last failure.
 *  - device buffers are raw HIP device pointers owned by the caller
 *    (e.g. torch tensors); the library only allocates its internal page
 *    store and scratch.
 *  - one GsCtx per device; calls on a ctx are serialized on its HIP stream
 *    and synchronized before returning.
 */
#ifndef CNOSDB_GS_H
#define CNOSDB_GS_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef int32_t GsStatus;
enum {
    GS_OK = 0,
    GS_ERR = -1,          /* generic; see gs_last_error() */
    GS_ERR_NO_GPU = -2,   /* HIP device unavailable — the product path never
                             falls back to CPU; callers must treat this as fatal */
    GS_ERR_FORMAT = -3,   /* malformed page / unknown encoding byte */
    GS_ERR_CRC = -4,      /* page crc32 mismatch (tsm/page.rs:58-76) */
    GS_ERR_CAP = -5,      /* output buffer too small */
    GS_ERR_SENTINEL = -6, /* Gorilla sentinel appeared as input (float.rs:58-60) */
};

/* Encoding byte, first byte of every encoded data buffer
 * (common/models/src/codec.rs:39-54) */
enum {
    GS_ENC_NULL = 1,
    GS_ENC_DELTA = 2,    /* i64: zigzag delta + simple8b/RLE  (codec/integer.rs) */
    GS_ENC_GORILLA = 6,  /* f64 XOR                           (codec/float.rs) */
    GS_ENC_SNAPPY = 7,   /* strings: snappy block             (codec/string.rs) */
    GS_ENC_BITPACK = 10, /* bool                              (codec/boolean.rs) */
    GS_ENC_DELTATS = 11, /* ts: delta + scaled simple8b/RLE   (codec/timestamp.rs) */
};

/* physical column type of a page (PhysicalCType, tsm/reader.rs:658-731) */
enum {
    GS_CT_TIME = 0, /* i64 timestamps */
    GS_CT_I64 = 1,
    GS_CT_F64 = 2,
    GS_CT_BOOL = 3,
    GS_CT_U64 = 4, /* bit-cast to i64, unsigned.rs:20-45 */
    GS_CT_STR = 5, /* snappy string blocks, codec/string.rs */
};

/* Closed time interval (common/models/src/predicate/domain.rs:36-44) */
typedef struct {
    int64_t min_ts;
    int64_t max_ts;
} GsTimeRange;

/* One on-disk page: bytes = [u32 BE bitset_len][u64 BE row_count]
 * [u32 BE crc32(data)][validity bitset, LSB-first][encoded data]
 * (tsm/page.rs:32-94); num_values mirrors PageMeta.num_values. */
typedef struct {
    const uint8_t *bytes;
    uint64_t len;
    uint32_t num_values;
    uint8_t ctype;    /* GS_CT_* */
    uint32_t column_id;
} GsPageSpec;

/* One column group = one series' time slab: a time page plus field pages,
 * all with the same row count (tsm/column_group.rs:10-17). */
typedef struct {
    const GsPageSpec *pages; /* pages[0] MUST be the time page */
    uint32_t npages;
    uint32_t series_id;
} GsColumnGroupDesc;

typedef struct GsCtx GsCtx;
typedef struct GsGroupSet GsGroupSet;

/* ---- context ---- */
GsCtx *gs_ctx_create(int device);
void gs_ctx_destroy(GsCtx *ctx);
const char *gs_last_error(void);
int gs_device_count(void);
const char *gs_version(void);

/* ---- write path (host): the re-encode side of the seam
 * (Page::arrow_array_to_page, tsm/page.rs:100-353 + codec encode fns).
 * Return encoded length, or a negative GsStatus. ---- */
int64_t gs_encode_ts(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_i64(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_f64(const double *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_bool(const uint8_t *src, size_t n, uint8_t *dst, size_t cap);
/* string block: src = concatenated string bytes, lens[i] = byte length of
 * string i (non-null values only).  Produces [7][0x10][snappy raw] per
 * str_snappy_encode (codec/string.rs:32-88); empty input -> 0 bytes.
 * The snappy arithmetic restates the published algorithm of the
 * un-vendored `snap` crate v1.1.1, pinned by the reference's golden
 * vectors (string.rs:529-566). */
int64_t gs_encode_str(const uint8_t *src, const uint64_t *lens, int64_t nstr,
                      uint8_t *dst, size_t cap);
/* assemble a full page (header + crc + bitset + data), tsm/page.rs:488-497 */
int64_t gs_build_page(const uint8_t *bitset, int64_t nrows, const uint8_t *data,
                      size_t data_len, uint8_t *dst, size_t cap);
uint32_t gs_crc32(const uint8_t *data, size_t len); /* CRC-32/ISO-HDLC */

/* ---- device page store ----
 * Uploads raw page bytes of `ngroups` column groups to HBM and builds the
 * device-side page table.  validate_crc runs the crc32 check of
 * Page::crc_validation on upload (host-side).  Row layout: group g's rows
 * occupy [row_offset[g], row_offset[g] + num_values_g) in every output
 * column buffer; gs_set_rows returns the total. */
GsGroupSet *gs_groups_upload(GsCtx *ctx, const GsColumnGroupDesc *groups,
                             size_t ngroups, int validate_crc);
void gs_groups_free(GsGroupSet *set);
int64_t gs_set_rows(const GsGroupSet *set);
int64_t gs_set_groups(const GsGroupSet *set);
/* series-level group count (consecutive same-series groups merged) */
int64_t gs_set_series(const GsGroupSet *set);
/* pushed-down COUNT from page metadata (pushdown_agg_reader.rs:39-106) */
int64_t gs_count_pushdown(const GsGroupSet *set, uint32_t col);
/* copy the per-group row offsets (ngroups entries) into caller buffer */
GsStatus gs_set_row_offsets(const GsGroupSet *set, int64_t *out);

/* ---- decode (hot loop ①, tsm/reader.rs:494-560) ----
 * Decodes column slot `col` (index into each group's pages[]) of every
 * group into d_out (device, 8 B/row for i64/f64/u64, 1 B for bool) at the
 * group row offsets.  Null slots are written as 0 (arrow builder
 * append_null semantics).  d_valid (device, 1 B/row, may be NULL) receives
 * 1 for valid rows, 0 for null rows. */
GsStatus gs_decode(GsCtx *ctx, GsGroupSet *set, uint32_t col, void *d_out,
                   uint8_t *d_valid);

/* ---- memcache rows (MemCacheReader, tskv/src/reader/memcache_reader.rs;
 * rows from mem_cache/series_data.rs:15-27) ----
 * A raw row set gives the unflushed in-memory rows the same series
 * structure as a page set, with NO pages: counts[i] = rows of series i
 * (time-sorted, non-null — pre-filter nulls when building the arrays).
 * gs_scan over a raw set skips the decode phases and filters/aggregates
 * the caller's device-resident spec->d_ts / spec->d_val directly
 * (tombstones and value predicates apply as usual; the fused path never
 * runs).  Raw sets also feed gs_compact_merge, so hot (memcache) and
 * cold (TSM) streams dedup together — newest stream wins at equal ts. */
GsGroupSet *gs_raw_set(GsCtx *ctx, const int64_t *counts, int64_t nseries);

/* ---- string column decode (str_snappy_decode_to_array,
 * codec/string.rs:226-276 via data_buf_to_arrow_array,
 * tsm/reader.rs:658-731) ----
 * Decodes string column slot `col` of every group into Arrow
 * varbinary layout at the set's row offsets: d_offsets (device,
 * int64[rows+1], exclusive prefix of byte lengths; null rows
 * contribute 0) and d_bytes (device, capacity bytes_cap).  d_valid
 * (device, 1 B/row, may be NULL) gets the validity bytes.  On return
 * *total_bytes = d_offsets[rows].  Rows whose page payload ends early
 * decode as null (the reference's builder stops appending).  Handles
 * GS_ENC_SNAPPY and GS_ENC_NULL blocks; empty data region -> all rows
 * null.  Current limit: rows <= 134M per set for the device offset
 * scan. */
GsStatus gs_decode_str(GsCtx *ctx, GsGroupSet *set, uint32_t col,
                       int64_t *d_offsets, uint8_t *d_bytes,
                       int64_t bytes_cap, uint8_t *d_valid,
                       int64_t *total_bytes);

/* ---- tombstone masking (tsm/reader.rs:634-656) ----
 * Clears validity (in d_valid) for rows whose decoded timestamp (d_ts,
 * from gs_decode of the time page) falls in any deleted closed range.
 * Applied per group; `ranges` apply to every group in the set. */
GsStatus gs_apply_tombstone(GsCtx *ctx, GsGroupSet *set, const int64_t *d_ts,
                            uint8_t *d_valid, const GsTimeRange *ranges,
                            size_t nranges);

/* ---- fused time-range scan (hot loops ①+②+④) ----
 * For the TSBS-devops shape: schema = time page + one f64 field page per
 * group (col 1).  Pipeline per group, entirely on device:
 *   decode ts -> decode f64 -> closed-interval time filter
 *   -> EITHER compacted output (d_out_ts/d_out_val at gather offsets)
 *      OR per-bucket aggregates (max/sum/count, fused, no row output).
 * Mirrors TableScanStream semantics for a pure time-range predicate
 * (data_source/batch/tskv.rs:351-371 pushes exactly these) with the
 * downsampling aggregate that stock DataFusion would run above
 * (SURVEY.md §8a row "downsampling agg").
 */
/* Value predicate on the scanned f64 field, evaluated like DataFilter's
 * pushed PhysicalExpr (reader/filter.rs:91-142): rows kept iff the time
 * range AND the predicate hold; a null field value fails the predicate
 * (arrow comparison-with-null semantics), unlike the time-only filter
 * where null field rows survive. */
enum {
    GS_PRED_NONE = 0,
    GS_PRED_GT = 1,
    GS_PRED_GE = 2,
    GS_PRED_LT = 3,
    GS_PRED_LE = 4,
    GS_PRED_EQ = 5,
    GS_PRED_NE = 6,
    GS_PRED_BETWEEN = 7, /* closed [a, b] */
};

typedef struct {
    int32_t op; /* GS_PRED_* */
    double a;
    double b;   /* BETWEEN upper bound */
} GsValuePred;

typedef struct {
    GsTimeRange range;     /* closed; use INT64_MIN/MAX for no filter */
    /* column slot of the f64 field to scan (0 = the first field page,
       i.e. pages[1]).  A multi-metric query (TSBS cpu-max-all-8) scans
       the same set once per field, reusing the uploaded pages; spans are
       recomputed per call (cheap, time-only). */
    int32_t field_col;
    /* deleted time ranges (tombstones) applied to the field column's
       validity before filter/agg (tsm/reader.rs:529-544) */
    const GsTimeRange *tombstones;
    size_t n_tombstones;
    /* aggregate spec; n_buckets == 0 disables aggregation */
    int64_t bucket_ns;     /* e.g. 300_000_000_000 for 5-min buckets */
    int64_t t0;            /* bucket origin: bucket = (ts - t0) / bucket_ns */
    int32_t n_buckets;
    /* device outputs for aggregation (caller-allocated, e.g. torch):
       d_max must be pre-filled with -inf, d_sum/d_count with 0 */
    double *d_agg_max;
    double *d_agg_sum;
    long long *d_agg_count;
    /* device outputs for compaction; NULL to skip row materialization.
       capacity must be >= gs_set_rows(set) rows. */
    int64_t *d_out_ts;
    double *d_out_val;
    /* scratch/outputs the caller provides (device):
       d_ts: decoded time column (gs_set_rows rows)
       d_val: decoded f64 column (gs_set_rows rows) */
    int64_t *d_ts;
    double *d_val;
    /* optional value predicate (general path; disables the fused path) */
    GsValuePred value_pred;
} GsScanSpec;

typedef struct {
    int64_t out_rows;      /* rows selected by the time filter */
    int64_t decoded_rows;  /* total rows decoded (= gs_set_rows) */
    /* per-phase kernel times, ms, measured with HIP events on the engine
       stream (for roofline accounting) */
    double ms_decode_ts;
    double ms_decode_val;
    double ms_filter;
    double ms_compact;
    double ms_agg;
} GsScanResult;

GsStatus gs_scan(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec,
                 GsScanResult *result);

/* Async variant of the fused scan: gs_scan_async enqueues the whole
 * fused pipeline (requires the fused-capable shape: RLE ts pages,
 * all-valid Gorilla field pages, no tombstones, compacted outputs — the
 * TSBS shape; `d_ts`/`d_val` scratch is not used) and returns without
 * synchronizing; gs_scan_wait synchronizes the ctx stream and fills the
 * result.  Lets a caller overlap sub-batches on two ctx streams (the
 * Gorilla decode is ALU-bound, the aggregate HBM-bound). */
/* GROUP BY tag over a completed aggregate scan: keys the
 * per-series bucket partials by the decoded varbinary tag column (tags
 * are SeriesKey members, constant within a series —
 * tskv/src/reader/series.rs:23-100; the group-by above TskvExec is stock
 * DataFusion hash-agg in the reference).  Outputs per-(tag, bucket)
 * max/sum/count in deterministic first-occurrence tag order;
 * tag_rep_row[gid] = a row index whose string value IS the tag.
 * Precondition: gs_scan with the same n_buckets, and gs_decode_str of
 * the tag column, both on this set. */
GsStatus gs_groupby_tag(GsCtx *ctx, GsGroupSet *set, int tag_col,
                        int n_buckets, double *d_out_max, double *d_out_sum,
                        long long *d_out_count, int64_t *tag_rep_row,
                        int cap_gids, int *out_ngids);

/* Fused scan of nf field columns over ONE span/ts pass (TSBS
 * cpu-max-all-8, BASELINE config #3; the reference decodes every
 * projected field column of the column group in one decode_pages pass,
 * tsm/reader.rs:494-560).  Fused-capable shapes only.  Output strides:
 * d_out_val + f*total_rows per field; d_agg_* + f*n_buckets per field. */
GsStatus gs_scan_fields(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec,
                        const int32_t *field_cols, int nf,
                        GsScanResult *result);
/* async variant: pair with gs_scan_wait */
GsStatus gs_scan_fields_async(GsCtx *ctx, GsGroupSet *set,
                              const GsScanSpec *spec,
                              const int32_t *field_cols, int nf);

GsStatus gs_scan_async(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec);
GsStatus gs_scan_wait(GsCtx *ctx, GsGroupSet *set, GsScanResult *result);

/* ---- GPU page re-encode (the write side of compaction/flush:
 * Page::arrow_array_to_page, tsm/page.rs:100-353 + tsm/writer.rs:249-314).
 * kind: 0 = ts (DeltaTs), 1 = i64 (Delta), 2 = f64 (Gorilla).  d_vals is a
 * device column; pages are [h_row_off[p], +h_rows[p]) slices.  Each full
 * page (header+crc+bitset+data, byte-exact with the host encoder) is
 * written at d_out + p*cap_per_page; h_lens receives encoded lengths.
 * Int encoders (kind 0/1) require all-valid slices — the time column is
 * never null; null-carrying i64 re-encode stays on the host encoder. */
GsStatus gs_encode_pages_dev(GsCtx *ctx, int32_t kind, const void *d_vals,
                             const uint8_t *d_valid, const int64_t *h_row_off,
                             const int32_t *h_rows, int32_t npages,
                             uint8_t *d_out, int64_t cap_per_page,
                             int64_t *h_lens);

/* ---- compaction merge (BASELINE config #5) ----
 * k overlapping L0 column-group streams per series -> one merged, deduped
 * (ts, value, validity) stream per series (tskv/src/compaction/compact.rs:
 * 271-404, comapcting_block_meta_group.rs:52-208; dedup rule
 * reader/batch_builder.rs:106-155).  sets[f] (f = 0 oldest .. nsets-1
 * newest, file_id order) must cover the same series list; d_ts/d_val/
 * d_valid[f] are that stream's decoded columns (from gs_decode).  Outputs
 * are caller device buffers sized >= the sum of input rows;
 * h_out_offsets[nseries+1] receives per-series output row offsets. */
GsStatus gs_compact_merge(GsCtx *ctx, GsGroupSet *const *sets, int32_t nsets,
                          const int64_t *const *d_ts,
                          const double *const *d_val,
                          const uint8_t *const *d_valid, int64_t *d_out_ts,
                          double *d_out_val, uint8_t *d_out_valid,
                          int64_t *h_out_offsets, int64_t *out_rows);

/* ---- Arrow C Data Interface export (decode output as
 * ArrowArray so the Rust shim reconstitutes RecordBatches zero-copy from
 * the shim's side).  GsArrowArray is layout-identical to `struct
 * ArrowArray` of the Arrow C data interface; for primitive columns
 * buffers = {validity bitmap or NULL, data}. */
typedef struct GsArrowArray {
    int64_t length;
    int64_t null_count;
    int64_t offset;
    int64_t n_buffers;
    int64_t n_children;
    const void **buffers;
    struct GsArrowArray **children;
    struct GsArrowArray *dictionary;
    void (*release)(struct GsArrowArray *);
    void *private_data;
} GsArrowArray;

/* Copy one group's rows of a decoded device column into a freshly
 * allocated host ArrowArray (validity bytes packed LSB-first into the
 * arrow bitmap; d_valid NULL => no validity buffer, null_count 0).
 * elem_size: 8 for i64/f64/u64, 1 for bool (exported as byte values —
 * the shim repacks to arrow bool bits if needed).  Caller releases via
 * out->release(out). */
GsStatus gs_export_group_column(GsCtx *ctx, GsGroupSet *set, int64_t group,
                                const void *d_col, int32_t elem_size,
                                const uint8_t *d_valid, GsArrowArray *out);

#ifdef __cplusplus
}
#endif
#endif /* CNOSDB_GS_H */
