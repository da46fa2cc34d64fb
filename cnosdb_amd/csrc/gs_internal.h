/* Internal shared declarations of the cnosdb_gs engine. */
#ifndef GS_INTERNAL_H
#define GS_INTERNAL_H

#include <cstdint>
#include <cstddef>

/* Device-side page descriptor. Offsets index the set's blob buffer.
 * data region of each page is 16-byte aligned within the blob. */
struct DevPage {
    uint64_t data_off;
    uint64_t bitset_off;
    int64_t row_off;    /* output row offset of this page's group */
    uint32_t data_len;
    uint32_t nrows;
    uint8_t ctype;
    uint8_t enc;        /* first byte of data buffer, 0 if empty */
    uint8_t sub;        /* sub-tag high nibble for DELTA/DELTATS, else 0 */
    uint8_t all_valid;  /* host-precomputed: every validity bit set */
    uint32_t grp;       /* page-group index (class lists may be re-ordered) */
};

struct DevGroup {
    int64_t row_off;
    int32_t nrows;
    int32_t pad;
};

/* Gorilla chunked decode: reference-shaped pages hold up to
 * max_datablock_size = 102,400 rows (comapcting_block_meta_group.rs:87,
 * storage_config.rs:136-138) and the Gorilla bitstream is strictly
 * sequential (float.rs:445-463), so thread-per-page parallelism collapses
 * on big pages.  At upload, k_gor_sync walks each all-valid Gorilla page
 * once and records the parser state (bit cursor, previous value, XOR
 * window) every GOR_CHUNK values; decode then runs one chunk per lane,
 * and the filtered scan skips chunks wholly outside the selected span
 * (sub-page pruning — the chunk-granularity analog of the reference's
 * page min/max pruning, tskv/src/reader/chunk.rs:12-49).  ~32 B of side
 * table per 4096 values = ~0.1% of typical compressed size. */
#define GOR_CHUNK 2048

/* DevGorChunk.flags bits */
#define GORF_SAFE_STOP 1 /* pre-pass proved the whole chunk decodable (the
                            next chunk's state was recorded): the filtered
                            kernel may stop at the span end without losing
                            the page's error surface (PC_GOR only) */
#define GORF_SENT_SEEN 2 /* the sentinel was already consumed before row0:
                            this chunk holds only trailing null rows
                            (PC_GORN only) */

struct DevGorChunk {
    uint64_t data_off;   /* page data offset in blob */
    uint64_t bitset_off; /* page validity bitset offset (PC_GORN) */
    uint64_t bitpos;     /* bits consumed from stream start; 0 for row0==0
                            (decode reads the page header itself there) */
    uint64_t val;        /* PC_GOR: bits of row (row0-1)'s value.
                            PC_GORN: bits of the value PENDING for the next
                            set-bit row >= row0 (pipelined one ahead). */
    int64_t row_off;     /* output row offset of the page's group */
    uint32_t grp;        /* page-group index */
    uint32_t row0;       /* first row index in page this chunk produces */
    uint32_t cnt;        /* rows this chunk produces */
    uint32_t data_len;   /* page data_len */
    uint8_t trailing;    /* XOR window state at bitpos (pre-pass) */
    uint8_t meaningful;
    uint8_t last;        /* final chunk: page-end semantics (sentinel) */
    uint8_t flags;       /* GORF_* */
};

/* launch-class partition of a column slot's pages */
enum PageClass {
    PC_SEQ = 0,     /* universal sequential thread-per-page decoder */
    PC_RLE_TS = 1,  /* DeltaTs + RLE + all-valid: closed-form parallel */
    PC_RLE_I64 = 2, /* Delta + RLE + all-valid: closed-form parallel */
    PC_GOR = 3,     /* Gorilla + all-valid: LDS-staged cooperative stores */
    PC_S8B = 4,     /* DeltaTs/Delta + simple8b + all-valid: block-parallel */
    PC_RAW = 5,     /* Null encoding (raw BE) + all-valid: coalesced bswap */
    PC_BOOL = 6,    /* BitPack + all-valid: parallel bit extract */
    PC_STR = 7,     /* string blocks (snappy / uncompressed): sequential
                       thread-per-page decompress, see gs_decode_str */
    PC_GORN = 8,    /* Gorilla + nulls: chunk-parallel pending-value
                       pipeline over the validity bitset */
    PC_NCLASS = 9,
};

extern "C" {
int64_t gs_encode_ts(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_i64(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_f64(const double *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_bool(const uint8_t *src, size_t n, uint8_t *dst, size_t cap);
uint32_t gs_crc32(const uint8_t *data, size_t len);
}

#endif
