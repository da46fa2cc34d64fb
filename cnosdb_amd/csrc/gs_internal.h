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
    PC_NCLASS = 8,
};

extern "C" {
int64_t gs_encode_ts(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_i64(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_f64(const double *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_bool(const uint8_t *src, size_t n, uint8_t *dst, size_t cap);
uint32_t gs_crc32(const uint8_t *data, size_t len);
}

#endif
