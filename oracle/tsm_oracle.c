/* ============================================================================
 * ORACLE — TEST INFRASTRUCTURE ONLY.
 *
 * CPU restatement of the CnosDB tskv TSM DataBlock codecs, used exclusively
 * as the parity checker for the GPU product path and as the reported CPU
 * baseline leg of bench.py.  It must never be imported, linked or executed
 * by the product path (cnosdb_amd/): only tests/, __graft_entry__.smoke()
 * and bench.py's cpu_baseline leg may touch it.
 *
 * Parity pinning: the reference (cnosdb/cnosdb, Rust) cannot be compiled in
 * this container (no rustc/cargo).  This restatement is pinned by the
 * reference's own byte-exact test vectors, transcribed into
 * tests/golden/golden_vectors.json:
 *   - InfluxDB RLE block        tskv/src/tsm/codec/integer.rs:438-459
 *   - InfluxDB simple8b block   tskv/src/tsm/codec/integer.rs:461-483
 *   - bool bitpack vectors      tskv/src/tsm/codec/boolean.rs:156-225
 *   - zigzag vectors            tskv/src/tsm/codec/integer.rs:269-280
 *   - float special values      tskv/src/tsm/codec/float.rs:634-665 (round trip)
 *
 * Every function cites the reference file:line whose semantics it restates.
 * Behaviour is matched bit-exactly for all valid inputs, including the
 * documented traps:
 *   - ts RLE count includes the first value, i64 RLE count excludes it
 *     (count is written but never read on decode)
 *   - ts single-value block carries scaler nibble 12 (div loop never runs)
 *   - ts n==2 always takes RLE; i64 needs n>=3
 *   - Gorilla leading-zero count capped at 31 BEFORE the window-reuse
 *     comparison (float.rs:89)
 *   - Gorilla meaningful==64 encodes as 6-bit 0 (float.rs:169-181,546-554)
 *   - empty input encodes to an EMPTY buffer; decode of empty src yields
 *     an all-null array sized to the bitset
 * ==========================================================================*/

#include <stdint.h>
#include <stddef.h>
#include <string.h>
#include <stdlib.h>

#define ORC_OK 0
#define ORC_ERR_BOUNDS (-1)
#define ORC_ERR_FORMAT (-2)
#define ORC_ERR_CAP (-3)
#define ORC_ERR_SENTINEL (-4)
#define ORC_ERR_SHORT (-5)

#define ORC_EXPORT __attribute__((visibility("default")))

/* Encoding enum, common/models/src/codec.rs:39-54 */
enum {
    ENC_DEFAULT = 0,
    ENC_NULL = 1,
    ENC_DELTA = 2,
    ENC_QUANTILE = 3,
    ENC_GORILLA = 6,
    ENC_BITPACK = 10,
    ENC_DELTATS = 11,
    ENC_UNKNOWN = 15,
};

/* sub-tags, tskv/src/tsm/codec/{timestamp.rs:15-19,integer.rs:20-24} */
enum { SUB_UNCOMPRESSED = 0, SUB_SIMPLE8B = 1, SUB_RLE = 2 };

/* ---------------------------------------------------------------- helpers */

static inline uint64_t be64(const uint8_t *p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return __builtin_bswap64(v);
}
static inline void put_be64(uint8_t *p, uint64_t v) {
    for (int i = 7; i >= 0; i--) { p[i] = (uint8_t)(v & 0xff); v >>= 8; }
}
static inline uint32_t be32(const uint8_t *p) {
    return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) | ((uint32_t)p[2] << 8) | p[3];
}
static inline void put_be32(uint8_t *p, uint32_t v) {
    p[0] = (uint8_t)(v >> 24); p[1] = (uint8_t)(v >> 16); p[2] = (uint8_t)(v >> 8); p[3] = (uint8_t)v;
}

/* zig_zag_encode/decode, integer.rs:102-110 */
static inline uint64_t zigzag_enc(int64_t v) { return ((uint64_t)v << 1) ^ (uint64_t)(v >> 63); }
static inline int64_t zigzag_dec(uint64_t v) {
    return (int64_t)((v >> 1) ^ (uint64_t)(-(int64_t)(v & 1)));
}

/* LEB128 unsigned varint (integer_encoding crate encode_var/decode_var) */
static size_t varint_put(uint8_t *dst, uint64_t v) {
    size_t n = 0;
    while (v >= 0x80) { dst[n++] = (uint8_t)(v | 0x80); v >>= 7; }
    dst[n++] = (uint8_t)v;
    return n;
}
static int varint_get(const uint8_t *src, size_t len, uint64_t *out, size_t *nread) {
    uint64_t v = 0;
    int shift = 0;
    for (size_t i = 0; i < len && i < 10; i++) {
        v |= (uint64_t)(src[i] & 0x7f) << shift;
        if (!(src[i] & 0x80)) { *out = v; *nread = i + 1; return ORC_OK; }
        shift += 7;
    }
    return ORC_ERR_FORMAT;
}

/* arrow validity bitset: LSB-first within each byte */
static inline int bit_get(const uint8_t *bs, int64_t i) { return (bs[i >> 3] >> (i & 7)) & 1; }
static inline void bit_clear(uint8_t *bs, int64_t i) { bs[i >> 3] &= (uint8_t)~(1u << (i & 7)); }

/* ------------------------------------------------------------- simple8b
 * tskv/src/tsm/codec/simple8b.rs.  64-bit BE words; selector = top 4 bits:
 * 0 -> 240 ones, 1 -> 120 ones, 2..15 -> NUM_BITS[sel-2]; values packed
 * LSB-first within the word (simple8b.rs:64). */

#define S8B_MAX_VALUE ((((uint64_t)1) << 60) - 1)

static const uint8_t S8B_NUM_BITS[14][2] = {
    {60, 1}, {30, 2}, {20, 3}, {15, 4}, {12, 5}, {10, 6}, {8, 7},
    {7, 8},  {6, 10}, {5, 12}, {4, 15}, {3, 20}, {2, 30}, {1, 60},
};

/* simple8b.rs:26-76 */
static int64_t s8b_encode(const uint64_t *src, size_t n, uint8_t *dst, size_t cap) {
    size_t i = 0, w = 0;
    while (i < n) {
        size_t remain = n - i;
        if (remain >= 120) {
            size_t lim = remain >= 240 ? 240 : 120;
            size_t k = 0;
            while (k < lim && src[i + k] == 1) k++;
            if (k == 240) {
                if (w + 8 > cap) return ORC_ERR_CAP;
                memset(dst + w, 0, 8);
                w += 8; i += 240;
                continue;
            } else if (k >= 120) {
                if (w + 8 > cap) return ORC_ERR_CAP;
                put_be64(dst + w, ((uint64_t)1) << 60);
                w += 8; i += 120;
                continue;
            }
        }
        int packed = 0;
        for (int idx = 0; idx < 14; idx++) {
            size_t int_n = S8B_NUM_BITS[idx][0];
            unsigned bit_n = S8B_NUM_BITS[idx][1];
            if (int_n > remain) continue;
            uint64_t max_val = ((uint64_t)1) << (bit_n & 0x3f);
            uint64_t val = ((uint64_t)idx + 2) << 60;
            int fits = 1;
            for (size_t k = 0; k < int_n; k++) {
                if (src[i + k] >= max_val) { fits = 0; break; }
                val |= src[i + k] << ((k * bit_n) & 0x3f);
            }
            if (!fits) continue;
            if (w + 8 > cap) return ORC_ERR_CAP;
            put_be64(dst + w, val);
            w += 8; i += int_n;
            packed = 1;
            break;
        }
        if (!packed) return ORC_ERR_BOUNDS; /* "value out of bounds" */
    }
    return (int64_t)w;
}

static const uint8_t S8B_COUNT[16] = {240, 120, 60, 30, 20, 15, 12, 10, 8, 7, 6, 5, 4, 3, 2, 1};
static const uint8_t S8B_WIDTH[16] = {0, 0, 1, 2, 3, 4, 5, 6, 7, 8, 10, 12, 15, 20, 30, 60};

/* simple8b.rs:95-208: returns count of values unpacked from one word */
static inline int s8b_decode_word(uint64_t v, uint64_t *dst) {
    unsigned sel = (unsigned)(v >> 60);
    int cnt = S8B_COUNT[sel];
    if (sel <= 1) {
        for (int i = 0; i < cnt; i++) dst[i] = 1;
    } else {
        unsigned bits = S8B_WIDTH[sel];
        uint64_t mask = (bits == 60) ? 0x0fffffffffffffffULL : ((((uint64_t)1) << bits) - 1);
        for (int i = 0; i < cnt; i++) { dst[i] = v & mask; v >>= bits; }
    }
    return cnt;
}

/* simple8b.rs:80-93: dst must have room for (len/8)*240 values worst case */
static size_t s8b_decode(const uint8_t *src, size_t len, uint64_t *dst) {
    size_t j = 0;
    for (size_t i = 0; i + 8 <= len; i += 8) j += (size_t)s8b_decode_word(be64(src + i), dst + j);
    return j;
}

/* --------------------------------------------------------------- ts codec
 * ts_zigzag_simple8b_encode, timestamp.rs:51-122 (no zigzag despite name). */
ORC_EXPORT int64_t orc_ts_encode(const int64_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    if (cap < 2 + 8 * n + 16) return ORC_ERR_CAP;
    size_t w = 0;
    dst[w++] = ENC_DELTATS;

    uint64_t *deltas = (uint64_t *)malloc(n * sizeof(uint64_t));
    if (!deltas) return ORC_ERR_CAP;
    for (size_t i = 0; i < n; i++) deltas[i] = (uint64_t)src[i];
    uint64_t max = 0;
    if (n > 1) {
        for (size_t i = n - 1; i >= 1; i--) {
            deltas[i] = deltas[i] - deltas[i - 1];
            if (deltas[i] > max) max = deltas[i];
        }
        int use_rle = 1;
        for (size_t i = 2; i < n; i++)
            if (deltas[1] != deltas[i]) { use_rle = 0; break; }
        if (use_rle) {
            /* encode_rle, timestamp.rs:136-175; count = n INCLUDING first value */
            dst[w++] = 0; /* scaler byte placeholder */
            put_be64(dst + w, deltas[0]);
            w += 8;
            uint64_t div = 1000000000000ULL;
            while (div > 1 && deltas[1] % div != 0) div /= 10;
            if (div > 1) {
                unsigned scaler = 0;
                for (uint64_t d = div; d > 1; d /= 10) scaler++;
                dst[1] |= (uint8_t)scaler;
                w += varint_put(dst + w, deltas[1] / div);
            } else {
                w += varint_put(dst + w, deltas[1]);
            }
            w += varint_put(dst + w, (uint64_t)n);
            dst[1] |= (uint8_t)(SUB_RLE << 4);
            free(deltas);
            return (int64_t)w;
        }
    }
    if (max > S8B_MAX_VALUE) {
        /* uncompressed: raw u64 deltas incl. first (timestamp.rs:84-95) */
        dst[w++] = (uint8_t)(SUB_UNCOMPRESSED << 4);
        for (size_t i = 0; i < n; i++) { put_be64(dst + w, deltas[i]); w += 8; }
        free(deltas);
        return (int64_t)w;
    }
    /* simple8b with divisor scaling (timestamp.rs:97-121).
     * NOTE: for n==1 the div loop never runs so div stays 1e12, scaler 12. */
    uint64_t div = 1000000000000ULL;
    for (size_t i = 1; i < n; i++) {
        if (div <= 1) break;
        while (div > 1 && deltas[i] % div != 0) div /= 10;
    }
    if (div > 1)
        for (size_t i = 1; i < n; i++) deltas[i] /= div;
    unsigned scaler = 0;
    for (uint64_t d = div; d > 1; d /= 10) scaler++;
    dst[w] = (uint8_t)(SUB_SIMPLE8B << 4);
    dst[w] |= (uint8_t)scaler;
    w++;
    put_be64(dst + w, deltas[0]);
    w += 8;
    int64_t s8 = s8b_encode(deltas + 1, n - 1, dst + w, cap - w);
    free(deltas);
    if (s8 < 0) return s8;
    return (int64_t)(w + (size_t)s8);
}

/* i64_zigzag_simple8b_encode, integer.rs:40-96 */
ORC_EXPORT int64_t orc_i64_encode(const int64_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    if (cap < 2 + 8 * n + 16) return ORC_ERR_CAP;
    size_t w = 0;
    dst[w++] = ENC_DELTA;

    uint64_t *deltas = (uint64_t *)malloc(n * sizeof(uint64_t));
    if (!deltas) return ORC_ERR_CAP;
    for (size_t i = 0; i < n; i++) deltas[i] = (uint64_t)src[i];
    uint64_t max = 0;
    for (size_t i = n - 1; i >= 1; i--) {
        deltas[i] = zigzag_enc((int64_t)(deltas[i] - deltas[i - 1]));
        if (deltas[i] > max) max = deltas[i];
        if (i == 1) break;
    }
    deltas[0] = zigzag_enc(src[0]);

    if (n > 2) {
        int use_rle = 1;
        for (size_t i = 2; i < n; i++)
            if (deltas[1] != deltas[i]) { use_rle = 0; break; }
        if (use_rle) {
            /* encode_rle, integer.rs:124-140; count EXCLUDES first value; no divisor */
            dst[w++] = 0;
            put_be64(dst + w, deltas[0]);
            w += 8;
            w += varint_put(dst + w, deltas[1]);
            w += varint_put(dst + w, (uint64_t)n - 1);
            dst[1] |= (uint8_t)(SUB_RLE << 4);
            free(deltas);
            return (int64_t)w;
        }
    }
    if (max > S8B_MAX_VALUE) {
        /* uncompressed: ALL values stored as zigzag deltas incl. first (integer.rs:77-88) */
        dst[w++] = (uint8_t)(SUB_UNCOMPRESSED << 4);
        for (size_t i = 0; i < n; i++) { put_be64(dst + w, deltas[i]); w += 8; }
        free(deltas);
        return (int64_t)w;
    }
    dst[w++] = (uint8_t)(SUB_SIMPLE8B << 4);
    put_be64(dst + w, deltas[0]);
    w += 8;
    int64_t s8 = s8b_encode(deltas + 1, n - 1, dst + w, cap - w);
    free(deltas);
    if (s8 < 0) return s8;
    return (int64_t)(w + (size_t)s8);
}

/* ts_without_compress_encode, timestamp.rs:21-31 (also used for i64/f64 Null) */
ORC_EXPORT int64_t orc_null_encode_i64(const int64_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    if (cap < 1 + 8 * n) return ORC_ERR_CAP;
    dst[0] = ENC_NULL;
    for (size_t i = 0; i < n; i++) put_be64(dst + 1 + 8 * i, (uint64_t)src[i]);
    return (int64_t)(1 + 8 * n);
}

/* ----------------------------------------------------- i64/ts page decode
 * Decode one integer/timestamp data buffer (first byte = Encoding) through
 * the validity bitset into out[nrows]; null slots get 0 (arrow builder
 * append_null semantics).  Mirrors:
 *   ts_zigzag_simple8b_decode_to_array   timestamp.rs:177-299
 *   i64_zigzag_simple8b_decode_to_array  integer.rs:142-248
 *   ts/i64_without_compress_decode       timestamp.rs:301-323
 */
ORC_EXPORT int orc_decode_i64(const uint8_t *src, size_t len, const uint8_t *bitset,
                              int64_t nrows, int64_t *out) {
    if (len == 0) { /* all-null (timestamp.rs:181-185) */
        for (int64_t i = 0; i < nrows; i++) out[i] = 0;
        return ORC_OK;
    }
    uint8_t enc = src[0];
    if (enc == ENC_NULL) {
        /* raw BE i64 per valid slot (timestamp.rs:301-323) */
        const uint8_t *p = src + 1;
        size_t avail = (len - 1) / 8, used = 0;
        for (int64_t r = 0; r < nrows; r++) {
            if (bit_get(bitset, r) && used < avail) { out[r] = (int64_t)be64(p + 8 * used); used++; }
            else out[r] = 0;
        }
        return ORC_OK;
    }
    int is_ts = (enc == ENC_DELTATS);
    if (!is_ts && enc != ENC_DELTA) return ORC_ERR_FORMAT;
    if (len < 2) return ORC_ERR_FORMAT;
    const uint8_t *s = src + 1; /* sub-tag byte at s[0] */
    size_t slen = len - 1;
    unsigned sub = s[0] >> 4;

    if (sub == SUB_UNCOMPRESSED) {
        /* ts: raw delta prefix-sum (timestamp.rs:201-224);
         * i64: zigzag deltas incl. first (integer.rs:165-184) */
        const uint8_t *p = s + 1;
        size_t plen = slen - 1;
        if (plen == 0 || (plen & 7)) return ORC_ERR_FORMAT;
        size_t avail = plen / 8, used = 0;
        int64_t prev = 0;
        for (int64_t r = 0; r < nrows; r++) {
            if (!bit_get(bitset, r)) { out[r] = 0; continue; }
            if (used >= avail) {
                if (is_ts) { out[r] = 0; continue; } /* ts silently stops (timestamp.rs:216) */
                return ORC_ERR_SHORT;
            }
            uint64_t raw = be64(p + 8 * used);
            used++;
            if (is_ts) prev = (int64_t)((uint64_t)prev + raw);
            else prev = (int64_t)((uint64_t)prev + (uint64_t)zigzag_dec(raw));
            out[r] = prev;
        }
        return ORC_OK;
    }
    if (sub == SUB_RLE) {
        /* ts: timestamp.rs:226-259 (scaler nibble); i64: integer.rs:186-214 */
        uint64_t scaler = 1;
        const uint8_t *p;
        size_t plen;
        if (is_ts) {
            unsigned s10 = s[0] & 0x0f;
            for (unsigned i = 0; i < s10; i++) scaler *= 10;
            p = s + 1; plen = slen - 1;
            if (slen < 9) return ORC_ERR_FORMAT;
        } else {
            p = s + 1; plen = slen - 1;
            if (plen < 8) return ORC_ERR_FORMAT;
        }
        uint64_t first_raw = be64(p);
        uint64_t dv; size_t nr;
        if (varint_get(p + 8, plen - 8, &dv, &nr) != ORC_OK) return ORC_ERR_FORMAT;
        int64_t cur, delta;
        if (is_ts) { cur = (int64_t)first_raw; delta = (int64_t)(dv * scaler); }
        else { cur = zigzag_dec(first_raw); delta = zigzag_dec(dv); }
        int first = 1;
        for (int64_t r = 0; r < nrows; r++) {
            if (!bit_get(bitset, r)) { out[r] = 0; continue; }
            if (first) { out[r] = cur; first = 0; continue; }
            cur = (int64_t)((uint64_t)cur + (uint64_t)delta);
            out[r] = cur;
        }
        return ORC_OK;
    }
    if (sub == SUB_SIMPLE8B) {
        /* ts: timestamp.rs:261-299; i64: integer.rs:216-248 */
        uint64_t scaler = 1;
        const uint8_t *p;
        size_t plen;
        if (is_ts) {
            unsigned s10 = s[0] & 0x0f;
            for (unsigned i = 0; i < s10; i++) scaler *= 10;
            if (slen < 9) return ORC_ERR_FORMAT;
            p = s + 1; plen = slen - 1;
        } else {
            if (slen < 9) return ORC_ERR_SHORT;
            p = s + 1; plen = slen - 1;
        }
        size_t nwords = (plen - 8) / 8;
        uint64_t *vals = (uint64_t *)malloc((nwords * 240 + 1) * sizeof(uint64_t));
        if (!vals) return ORC_ERR_CAP;
        size_t nvals = s8b_decode(p + 8, plen - 8, vals);
        uint64_t first_raw = be64(p);
        int64_t cur = is_ts ? (int64_t)first_raw : zigzag_dec(first_raw);
        int first = 1;
        size_t vi = 0;
        for (int64_t r = 0; r < nrows; r++) {
            if (!bit_get(bitset, r)) { out[r] = 0; continue; }
            if (first) { out[r] = cur; first = 0; continue; }
            if (vi >= nvals) { out[r] = 0; continue; } /* iterator exhaustion: builder skips */
            uint64_t v = vals[vi++];
            if (is_ts) cur = (int64_t)((uint64_t)cur + v * scaler);
            else cur = (int64_t)((uint64_t)cur + (uint64_t)zigzag_dec(v));
            out[r] = cur;
        }
        free(vals);
        return ORC_OK;
    }
    return ORC_ERR_FORMAT;
}

/* ------------------------------------------------------------ gorilla f64
 * f64_gorilla_encode, float.rs:32-243.  Block: [0x06][0x10][8B BE first]
 * [bitstream MSB-first], terminated by encoded SENTINEL. */

#define GORILLA_SENTINEL 0x7ff8000000000ffULL /* float.rs:16 (0x7ff8_0000_0000_00ff) */

typedef struct {
    uint8_t *dst;
    size_t cap;
    size_t n; /* bit cursor, bits relative to dst[1] (header-byte convention, float.rs:43) */
    int err;
} BitWriter;

static inline void bw_put_bit(BitWriter *bw, int bit) {
    size_t byte = (bw->n >> 3) + 1;
    if (byte >= bw->cap) { bw->err = 1; return; }
    if (bit) bw->dst[byte] |= (uint8_t)(128u >> (bw->n & 7));
    bw->n++;
}
/* write the top `l` bits of v (MSB-aligned), float.rs:105-130 pattern */
static inline void bw_put_top_bits(BitWriter *bw, uint64_t v, unsigned l) {
    while (l > 0) {
        size_t byte = (bw->n >> 3) + 1;
        if (byte >= bw->cap) { bw->err = 1; return; }
        unsigned m = bw->n & 7;
        unsigned take = 8 - m;
        if (take > l) take = l;
        bw->dst[byte] |= (uint8_t)((v >> 56) >> m);
        v <<= take;
        bw->n += take;
        l -= take;
    }
}

ORC_EXPORT int64_t orc_f64_encode(const double *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    size_t need = 2 + 8 + (n + 1) * 10 + 16; /* worst case ~77 bits/value */
    if (cap < need) return ORC_ERR_CAP;
    memset(dst, 0, need);
    dst[0] = ENC_GORILLA;
    dst[1] = 1 << 4;
    uint64_t prev;
    memcpy(&prev, &src[0], 8);
    put_be64(dst + 2, prev);
    BitWriter bw = {dst, cap, 8 + 64, 0};
    uint64_t prev_leading = ~0ULL, prev_trailing = 0;
    for (size_t i = 1; i <= n; i++) {
        uint64_t cur;
        if (i < n) {
            memcpy(&cur, &src[i], 8);
            if (cur == GORILLA_SENTINEL) return ORC_ERR_SENTINEL; /* float.rs:58-60 */
        } else {
            cur = GORILLA_SENTINEL;
        }
        uint64_t v_delta = cur ^ prev;
        if (v_delta == 0) { bw_put_bit(&bw, 0); prev = cur; continue; }
        bw_put_bit(&bw, 1);
        uint64_t leading = (uint64_t)__builtin_clzll(v_delta);
        uint64_t trailing = (uint64_t)__builtin_ctzll(v_delta);
        leading &= 0x1f; /* cap at 31, float.rs:89 */
        if (prev_leading != ~0ULL && leading >= prev_leading && trailing >= prev_trailing) {
            bw_put_bit(&bw, 0);
            unsigned l = (unsigned)(64 - prev_leading - prev_trailing);
            uint64_t v = (v_delta >> prev_trailing) << (64 - l);
            bw_put_top_bits(&bw, v, l);
        } else {
            prev_leading = leading;
            prev_trailing = trailing;
            bw_put_bit(&bw, 1);
            bw_put_top_bits(&bw, leading << 59, 5);
            uint64_t sig_bits = 64 - leading - trailing;
            bw_put_top_bits(&bw, sig_bits << 58, 6); /* 64 wraps to 0 (float.rs:169-181) */
            unsigned l = (unsigned)sig_bits;
            uint64_t v = (l == 64) ? (v_delta >> trailing)
                                   : ((v_delta >> trailing) << (64 - l));
            bw_put_top_bits(&bw, v, l);
        }
        prev = cur;
    }
    if (bw.err) return ORC_ERR_CAP;
    size_t length = (bw.n >> 3) + 1;
    if (bw.n & 7) length += 1;
    return (int64_t)length;
}

/* decode_with_sentinel, float.rs:418-606.  MSB-first bit reader with 64-bit
 * cache; stops when XOR-accumulate yields the sentinel. */
typedef struct {
    const uint8_t *src;
    size_t len;
    size_t i;       /* next byte */
    uint64_t cache; /* rotate-left cursor (float.rs:445-463) */
    uint8_t valid;
} BitReader;

static int br_refill(BitReader *br) {
    size_t rem = br->len - br->i;
    if (rem >= 8) {
        br->cache = be64(br->src + br->i);
        br->valid = 64;
        br->i += 8;
        return ORC_OK;
    } else if (rem > 0) {
        uint64_t v = 0;
        for (size_t k = br->i; k < br->len; k++) v = (v << 8) | br->src[k];
        unsigned bits = (unsigned)(rem * 8);
        /* rotate_right(valid): valid bits end up at the TOP */
        br->cache = (v >> bits) | (v << (64 - bits));
        br->valid = (uint8_t)bits;
        br->i = br->len;
        return ORC_OK;
    }
    return ORC_ERR_SHORT; /* "unexpected end of block" */
}

static inline uint64_t rotl64(uint64_t x, unsigned c) {
    c &= 63; /* rotate by 64 == rotate by 0 == identity, matching u64::rotate_left */
    return c ? (x << c) | (x >> (64 - c)) : x;
}

/* read `cnt` bits (1..=64), MSB-first, mirroring the rotate-left scheme incl.
 * the split-read masking at float.rs:518-543/557-582 */
static int br_read(BitReader *br, unsigned cnt, uint64_t *out) {
    if (br->valid == 0) { int e = br_refill(br); if (e) return e; }
    if (br->valid >= cnt) {
        br->valid -= (uint8_t)cnt;
        br->cache = rotl64(br->cache, cnt);
        *out = br->cache; /* caller masks */
        return ORC_OK;
    }
    unsigned m_bits = cnt;
    uint64_t bits = 0;
    if (br->valid > 0) {
        m_bits -= br->valid;
        bits = rotl64(br->cache, cnt);
    }
    int e = br_refill(br);
    if (e) return e;
    br->cache = rotl64(br->cache, m_bits);
    br->valid = (uint8_t)(br->valid - m_bits);
    /* BIT_MASK[(m_bits & 0x3f)]: index 0 (m_bits==64) is all-ones (float.rs:269-283) */
    uint64_t mask = (m_bits & 0x3f) ? ((((uint64_t)1) << (m_bits & 0x3f)) - 1) : ~0ULL;
    bits &= ~mask;
    bits |= br->cache & mask;
    *out = bits;
    return ORC_OK;
}

ORC_EXPORT int orc_decode_f64(const uint8_t *src, size_t len, const uint8_t *bitset,
                              int64_t nrows, double *out) {
    if (len == 0) {
        for (int64_t i = 0; i < nrows; i++) out[i] = 0.0;
        return ORC_OK;
    }
    uint8_t enc = src[0];
    if (enc == ENC_NULL) { /* f64_without_compress_decode, float.rs:387-413 */
        const uint8_t *p = src + 1;
        size_t avail = (len - 1) / 8, used = 0;
        for (int64_t r = 0; r < nrows; r++) {
            if (bit_get(bitset, r)) {
                if (used >= avail) return ORC_ERR_SHORT;
                uint64_t u = be64(p + 8 * used); used++;
                memcpy(&out[r], &u, 8);
            } else out[r] = 0.0;
        }
        return ORC_OK;
    }
    if (enc != ENC_GORILLA) return ORC_ERR_FORMAT;
    const uint8_t *s = src + 1; /* skip encoding byte (float.rs:357) */
    size_t slen = len - 1;
    if (slen < 9) return ORC_ERR_SHORT;
    /* s[0] is the 0x10 compression-type byte; first value at s[1..9] */
    uint64_t val = be64(s + 1);
    BitReader br = {s, slen, 9, 0, 0};
    int e = br_refill(&br);
    if (e) return e;
    uint8_t trailing_n = 0, meaningful_n = 64;

    /* stream values through the bitset as they are produced */
    int64_t r = 0;
    int64_t produced = 0;
#define EMIT(bits_u64)                                              \
    do {                                                            \
        while (r < nrows && !bit_get(bitset, r)) { out[r] = 0.0; r++; } \
        if (r < nrows) { uint64_t _u = (bits_u64); memcpy(&out[r], &_u, 8); r++; } \
        produced++;                                                 \
    } while (0)

    EMIT(val);
    for (;;) {
        uint64_t b;
        e = br_read(&br, 1, &b);
        if (e) return e;
        if ((b & 1) == 0) { EMIT(val); continue; }
        e = br_read(&br, 1, &b);
        if (e) return e;
        if (b & 1) {
            uint64_t lm;
            e = br_read(&br, 11, &lm);
            if (e) return e;
            lm &= 0x7ff;
            uint8_t leading_n = (uint8_t)((lm >> 6) & 0x1f);
            meaningful_n = (uint8_t)(lm & 0x3f);
            if (meaningful_n > 0) trailing_n = (uint8_t)(64 - leading_n - meaningful_n);
            else { trailing_n = 0; meaningful_n = 64; }
        }
        uint64_t s_bits;
        e = br_read(&br, meaningful_n, &s_bits);
        if (e) return e;
        if ((meaningful_n & 0x3f) != 0) s_bits &= ((((uint64_t)1) << (meaningful_n & 0x3f)) - 1);
        val ^= s_bits << (trailing_n & 0x3f);
        if (val == GORILLA_SENTINEL) break;
        EMIT(val);
    }
#undef EMIT
    /* remaining rows must be null (else: mismatch, float.rs:594-599) */
    for (; r < nrows; r++) {
        if (bit_get(bitset, r)) return ORC_ERR_SHORT;
        out[r] = 0.0;
    }
    return ORC_OK;
}

/* ---------------------------------------------------------------- boolean
 * bool_bitpack_encode/decode, boolean.rs:24-110 */
ORC_EXPORT int64_t orc_bool_encode(const uint8_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    size_t size = 1 + 8 + (n + 7) / 8;
    if (cap < size + 2) return ORC_ERR_CAP;
    memset(dst, 0, size + 2);
    dst[0] = ENC_BITPACK;
    dst[1] = 1 << 4;
    size_t vi = varint_put(dst + 2, (uint64_t)n);
    size_t nbit = 8 + vi * 8;
    for (size_t k = 0; k < n; k++) {
        size_t idx = nbit >> 3;
        if (src[k]) dst[idx + 1] |= (uint8_t)(128u >> (nbit & 7));
        nbit++;
    }
    size_t length = nbit >> 3;
    if (nbit & 7) length += 1;
    return (int64_t)(length + 1);
}

ORC_EXPORT int orc_decode_bool(const uint8_t *src, size_t len, const uint8_t *bitset,
                               int64_t nrows, uint8_t *out) {
    if (len == 0) {
        for (int64_t i = 0; i < nrows; i++) out[i] = 0;
        return ORC_OK;
    }
    uint8_t enc = src[0];
    if (enc == ENC_NULL) { /* boolean.rs:111-136 */
        const uint8_t *p = src + 1;
        size_t avail = len - 1, used = 0;
        for (int64_t r = 0; r < nrows; r++) {
            if (bit_get(bitset, r) && used < avail) { out[r] = (p[used] == 1); used++; }
            else out[r] = 0;
        }
        return ORC_OK;
    }
    if (enc != ENC_BITPACK) return ORC_ERR_FORMAT;
    const uint8_t *s = src + 1;
    size_t slen = len - 1;
    if (slen < 1 || s[0] != (1 << 4)) return ORC_ERR_FORMAT;
    uint64_t count; size_t nr;
    if (varint_get(s + 1, slen - 1, &count, &nr) != ORC_OK) return ORC_ERR_FORMAT;
    const uint8_t *bits = s + 1 + nr;
    uint64_t bi = 0;
    for (int64_t r = 0; r < nrows; r++) {
        if (!bit_get(bitset, r)) { out[r] = 0; continue; }
        if (bi >= count) return ORC_ERR_SHORT; /* boolean.rs:97-99 */
        out[r] = (bits[bi / 8] >> (7 - (bi % 8))) & 1;
        bi++;
    }
    return ORC_OK;
}

/* ------------------------------------------------------------------ crc32
 * crc32fast = CRC-32/ISO-HDLC (page.rs:58-76) */
static uint32_t crc_table[256];
static int crc_init_done = 0;
static void crc_init(void) {
    for (uint32_t i = 0; i < 256; i++) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        crc_table[i] = c;
    }
    crc_init_done = 1;
}
ORC_EXPORT uint32_t orc_crc32(const uint8_t *data, size_t len) {
    if (!crc_init_done) crc_init();
    uint32_t c = 0xFFFFFFFFu;
    for (size_t i = 0; i < len; i++) c = crc_table[(c ^ data[i]) & 0xFF] ^ (c >> 8);
    return c ^ 0xFFFFFFFFu;
}

/* ------------------------------------------------------------------ page
 * Page layout (page.rs:32-38,488-497):
 * [u32 BE bitset_len][u64 BE data_len(=row count)][u32 BE crc32(data)]
 * [bitset LSB-first][encoded data] */
ORC_EXPORT int64_t orc_build_page(const uint8_t *bitset, int64_t nrows,
                                  const uint8_t *data, size_t data_len,
                                  uint8_t *dst, size_t cap) {
    size_t bitset_len = (size_t)((nrows + 7) / 8);
    size_t total = 16 + bitset_len + data_len;
    if (cap < total) return ORC_ERR_CAP;
    put_be32(dst, (uint32_t)bitset_len);
    put_be64(dst + 4, (uint64_t)nrows);
    put_be32(dst + 12, orc_crc32(data, data_len));
    memcpy(dst + 16, bitset, bitset_len);
    memcpy(dst + 16 + bitset_len, data, data_len);
    return (int64_t)total;
}

/* validate crc + return pointers; mirrors Page::crc_validation (page.rs:58-76) */
ORC_EXPORT int orc_page_check(const uint8_t *page, size_t page_len,
                              int64_t *nrows, uint32_t *bitset_off, uint64_t *data_off) {
    if (page_len < 16) return ORC_ERR_FORMAT;
    uint32_t bl = be32(page);
    uint64_t dl = be64(page + 4);
    uint32_t crc = be32(page + 12);
    if (16 + (size_t)bl > page_len) return ORC_ERR_FORMAT;
    uint32_t calc = orc_crc32(page + 16 + bl, page_len - 16 - bl);
    if (calc != crc) return ORC_ERR_FORMAT;
    *nrows = (int64_t)dl;
    *bitset_off = 16;
    *data_off = 16 + bl;
    return ORC_OK;
}

/* -------------------------------------------------------------- tombstone
 * update_nullbits_by_time_range, tsm/reader.rs:634-656: for each deleted
 * closed range, binary-search the decoded (sorted) ts values and clear
 * validity bits in [start, end).  start = partition point of ts < min;
 * end = index_of(max)+1 if found else partition point. */
ORC_EXPORT void orc_update_nullbits(const int64_t *ts, int64_t nrows,
                                    const int64_t *ranges, size_t nranges,
                                    uint8_t *bitset) {
    for (size_t g = 0; g < nranges; g++) {
        int64_t mn = ranges[2 * g], mx = ranges[2 * g + 1];
        /* lower_bound(mn) */
        int64_t lo = 0, hi = nrows;
        while (lo < hi) { int64_t mid = (lo + hi) / 2; if (ts[mid] < mn) lo = mid + 1; else hi = mid; }
        int64_t start = lo;
        /* binary_search(mx): found -> idx+1 else insertion point */
        lo = 0; hi = nrows;
        while (lo < hi) { int64_t mid = (lo + hi) / 2; if (ts[mid] < mx) lo = mid + 1; else hi = mid; }
        int64_t end = (lo < nrows && ts[lo] == mx) ? lo + 1 : lo;
        for (int64_t i = start; i < end; i++) bit_clear(bitset, i);
    }
}

/* ------------------------------------------------- batch decode (baseline)
 * OpenMP-threaded decode of many pages: the cpu_baseline leg of bench.py.
 * Each page: data buffer (already sliced), its bitset, nrows, column type.
 * ctype: 0 = i64/ts, 1 = f64, 2 = bool. Outputs to per-page offsets in a
 * contiguous out array (8 B/row for i64/f64, 1 B for bool).
 */
typedef struct {
    const uint8_t *data;
    uint64_t data_len;
    const uint8_t *bitset;
    int64_t nrows;
    uint64_t out_off; /* row offset into out */
    uint8_t ctype;
} OrcPageDesc;

ORC_EXPORT int orc_decode_pages_omp(const OrcPageDesc *pages, size_t npages,
                                    void *out, int nthreads) {
    int err = 0;
#pragma omp parallel for schedule(dynamic, 4) num_threads(nthreads)
    for (size_t p = 0; p < npages; p++) {
        const OrcPageDesc *pg = &pages[p];
        int e = 0;
        if (pg->ctype == 0)
            e = orc_decode_i64(pg->data, pg->data_len, pg->bitset, pg->nrows,
                               (int64_t *)out + pg->out_off);
        else if (pg->ctype == 1)
            e = orc_decode_f64(pg->data, pg->data_len, pg->bitset, pg->nrows,
                               (double *)out + pg->out_off);
        else
            e = orc_decode_bool(pg->data, pg->data_len, pg->bitset, pg->nrows,
                                (uint8_t *)out + pg->out_off);
        if (e) {
#pragma omp atomic write
            err = e;
        }
    }
    return err;
}

/* ------------------------------------------------- k-way merge (baseline)
 * CPU restatement of the compaction merge for the cpu_baseline leg of
 * bench.py --mode compact: k time-sorted streams (oldest..newest, file_id
 * order), equal-ts rows collapse, the newest non-null value wins
 * (reader/sort_merge.rs:152-343 + reader/batch_builder.rs:106-155).
 * Simple k-pointer merge per series, OpenMP over series. */
typedef struct {
    const int64_t *ts;
    const double *val;
    const uint8_t *valid; /* may be NULL */
    int64_t n;
} OrcStream;

/* merge one series' k streams; returns merged row count */
ORC_EXPORT int64_t orc_merge_dedup(const OrcStream *streams, int k,
                                   int64_t *out_ts, double *out_val,
                                   uint8_t *out_valid) {
    int64_t pos[16];
    if (k > 16) return -1;
    for (int f = 0; f < k; f++) pos[f] = 0;
    int64_t w = 0;
    for (;;) {
        int64_t mn = INT64_MAX;
        int any = 0;
        for (int f = 0; f < k; f++)
            if (pos[f] < streams[f].n) {
                any = 1;
                if (streams[f].ts[pos[f]] < mn) mn = streams[f].ts[pos[f]];
            }
        if (!any) break;
        double v = 0.0;
        uint8_t ok = 0;
        for (int f = k - 1; f >= 0; f--) { /* newest stream first */
            if (pos[f] < streams[f].n && streams[f].ts[pos[f]] == mn) {
                if (!ok && (!streams[f].valid || streams[f].valid[pos[f]])) {
                    v = streams[f].val[pos[f]];
                    ok = 1;
                }
                pos[f]++;
            }
        }
        out_ts[w] = mn;
        out_val[w] = ok ? v : 0.0;
        if (out_valid) out_valid[w] = ok;
        w++;
    }
    return w;
}

/* OpenMP over series: streams laid out [series][k]; outputs at
 * out_offsets[series] (caller-provided, e.g. upper bounds) */
ORC_EXPORT int orc_merge_dedup_many(const OrcStream *streams, int k,
                                    int64_t nseries,
                                    const int64_t *out_offsets,
                                    int64_t *out_ts, double *out_val,
                                    int64_t *out_counts, int nthreads) {
    int err = 0;
#pragma omp parallel for schedule(dynamic, 1) num_threads(nthreads)
    for (int64_t s = 0; s < nseries; s++) {
        int64_t w = orc_merge_dedup(streams + s * k, k,
                                    out_ts + out_offsets[s],
                                    out_val + out_offsets[s], NULL);
        if (w < 0) {
#pragma omp atomic write
            err = -1;
        } else {
            out_counts[s] = w;
        }
    }
    return err;
}

/* time-range filter count: closed interval [mn,mx] over sorted ts
 * (reader/filter.rs semantics for a pure time-range PhysicalExpr;
 * TimeRange is a closed interval, common/models/src/predicate/domain.rs:36-44) */
ORC_EXPORT void orc_ts_span(const int64_t *ts, int64_t nrows, int64_t mn, int64_t mx,
                            int64_t *start, int64_t *count) {
    int64_t lo = 0, hi = nrows;
    while (lo < hi) { int64_t mid = (lo + hi) / 2; if (ts[mid] < mn) lo = mid + 1; else hi = mid; }
    int64_t s = lo;
    lo = 0; hi = nrows;
    while (lo < hi) { int64_t mid = (lo + hi) / 2; if (ts[mid] <= mx) lo = mid + 1; else hi = mid; }
    *start = s;
    *count = lo - s;
}

/* ======================= string codec (Snappy) =========================
 * Restatement of the reference string block codec
 * (tskv/src/tsm/codec/string.rs:32-88 encode, :185-276 decode).  Block
 * layout: [Encoding::Snappy=7][STRING_COMPRESSED_SNAPPY<<4 = 0x10]
 * [snappy RAW stream of payload], payload = per string
 * [LEB128 varint len][bytes].  The Snappy arithmetic itself lives in the
 * un-vendored `snap` crate v1.1.1 (Cargo.lock) — a port of Google's
 * published snappy reference algorithm — so the compressor below
 * restates that published algorithm (64 KiB fragments, 256..16384-entry
 * hash table of 4-byte loads, hash mul 0x1e35a7bd, skip-32 search
 * acceleration, 15-byte input margin) and is pinned byte-exactly by the
 * reference's own golden vectors (string.rs:529-566 encode_single/
 * multi_compressed/unicode/invalid_utf8, transcribed in tests/golden/).
 * The uncompressed variant (Encoding::Null=1, [u64 BE len][bytes] per
 * string) follows string.rs:169-183. */

#define SNAP_BLOCK 65536
#define SNAP_MARGIN 15
#define SNAP_MAX_TABLE 16384

static uint32_t snap_load32(const uint8_t *p) {
    uint32_t v; memcpy(&v, p, 4); return v; /* little-endian hosts only */
}

static uint8_t *snap_emit_literal(uint8_t *op, const uint8_t *lit, size_t n) {
    size_t n1 = n - 1;
    if (n1 < 60) {
        *op++ = (uint8_t)(n1 << 2);
    } else {
        uint8_t *tag = op++;
        int c = 0;
        size_t v = n1;
        while (v > 0) { *op++ = v & 0xff; v >>= 8; c++; }
        *tag = (uint8_t)((59 + c) << 2);
    }
    memcpy(op, lit, n);
    return op + n;
}

static uint8_t *snap_emit_copy_upto64(uint8_t *op, size_t offset, size_t len) {
    if (len < 12 && offset < 2048) {
        *op++ = (uint8_t)(1 | ((len - 4) << 2) | ((offset >> 8) << 5));
        *op++ = (uint8_t)(offset & 0xff);
    } else {
        *op++ = (uint8_t)(2 | ((len - 1) << 2));
        *op++ = (uint8_t)(offset & 0xff);
        *op++ = (uint8_t)((offset >> 8) & 0xff);
    }
    return op;
}

static uint8_t *snap_emit_copy(uint8_t *op, size_t offset, size_t len) {
    while (len >= 68) { op = snap_emit_copy_upto64(op, offset, 64); len -= 64; }
    if (len > 64) { op = snap_emit_copy_upto64(op, offset, 60); len -= 60; }
    return snap_emit_copy_upto64(op, offset, len);
}

static uint8_t *snap_compress_fragment(const uint8_t *input, size_t n,
                                       uint8_t *op, uint16_t *table) {
    size_t table_size = 256;
    while (table_size < SNAP_MAX_TABLE && table_size < n) table_size <<= 1;
    int shift = 32;
    for (size_t t = table_size; t > 1; t >>= 1) shift--;
    memset(table, 0, table_size * sizeof(uint16_t));
    const uint8_t *ip = input, *ip_end = input + n;
    const uint8_t *next_emit = ip;
    if (n >= SNAP_MARGIN) {
        const uint8_t *ip_limit = ip_end - SNAP_MARGIN;
        ip++;
        uint32_t next_hash = (snap_load32(ip) * 0x1e35a7bdu) >> shift;
        for (;;) {
            uint32_t skip = 32;
            const uint8_t *next_ip = ip;
            const uint8_t *candidate;
            do {
                ip = next_ip;
                uint32_t h = next_hash;
                uint32_t adv = skip >> 5;
                skip += adv;
                next_ip = ip + adv;
                if (next_ip > ip_limit) goto emit_remainder;
                next_hash = (snap_load32(next_ip) * 0x1e35a7bdu) >> shift;
                candidate = input + table[h];
                table[h] = (uint16_t)(ip - input);
            } while (snap_load32(ip) != snap_load32(candidate));
            op = snap_emit_literal(op, next_emit, (size_t)(ip - next_emit));
            uint32_t cand32;
            do {
                const uint8_t *base = ip;
                size_t matched = 4;
                {
                    const uint8_t *s1 = candidate + 4, *s2 = ip + 4;
                    while (s2 < ip_end && *s1 == *s2) { s1++; s2++; matched++; }
                }
                ip += matched;
                op = snap_emit_copy(op, (size_t)(base - candidate), matched);
                next_emit = ip;
                if (ip >= ip_limit) goto emit_remainder;
                uint32_t ph = (snap_load32(ip - 1) * 0x1e35a7bdu) >> shift;
                table[ph] = (uint16_t)(ip - 1 - input);
                uint32_t ch = (snap_load32(ip) * 0x1e35a7bdu) >> shift;
                candidate = input + table[ch];
                cand32 = snap_load32(candidate);
                table[ch] = (uint16_t)(ip - input);
            } while (snap_load32(ip) == cand32);
            ip++;
            next_hash = (snap_load32(ip) * 0x1e35a7bdu) >> shift;
        }
    }
emit_remainder:
    if (next_emit < ip_end)
        op = snap_emit_literal(op, next_emit, (size_t)(ip_end - next_emit));
    return op;
}

ORC_EXPORT int64_t orc_snappy_max_compress_len(int64_t n) {
    return 32 + n + n / 6; /* snap::raw::max_compress_len formula */
}

ORC_EXPORT int64_t orc_snappy_compress(const uint8_t *src, size_t n,
                                       uint8_t *dst, size_t cap) {
    if (cap < (size_t)orc_snappy_max_compress_len((int64_t)n)) return -1;
    uint8_t *op = dst;
    size_t v = n; /* preamble: varint uncompressed length */
    while (v >= 0x80) { *op++ = (uint8_t)(v | 0x80); v >>= 7; }
    *op++ = (uint8_t)v;
    uint16_t table[SNAP_MAX_TABLE];
    for (size_t off = 0; off < n; off += SNAP_BLOCK) {
        size_t frag = n - off < SNAP_BLOCK ? n - off : SNAP_BLOCK;
        op = snap_compress_fragment(src + off, frag, op, table);
    }
    if (n == 0) { /* empty input: preamble only (snappy of "" is [0]) */ }
    return op - dst;
}

ORC_EXPORT int64_t orc_snappy_uncompressed_len(const uint8_t *src, size_t len,
                                               int *consumed) {
    uint64_t v = 0;
    int shift_ = 0, i = 0;
    while ((size_t)i < len && i < 10) {
        uint8_t b = src[i++];
        v |= (uint64_t)(b & 0x7f) << shift_;
        if (!(b & 0x80)) { if (consumed) *consumed = i; return (int64_t)v; }
        shift_ += 7;
    }
    return -1;
}

ORC_EXPORT int64_t orc_snappy_decompress(const uint8_t *src, size_t len,
                                         uint8_t *dst, size_t cap) {
    int hdr = 0;
    int64_t ulen = orc_snappy_uncompressed_len(src, len, &hdr);
    if (ulen < 0 || (size_t)ulen > cap) return -1;
    const uint8_t *ip = src + hdr, *end = src + len;
    uint8_t *op = dst, *op_end = dst + ulen;
    while (ip < end) {
        uint8_t tag = *ip++;
        size_t l, off;
        switch (tag & 3) {
        case 0:
            l = tag >> 2;
            if (l >= 60) {
                int c = (int)l - 59;
                if (ip + c > end) return -1;
                l = 0;
                for (int i = 0; i < c; i++) l |= (size_t)ip[i] << (8 * i);
                ip += c;
            }
            l += 1;
            if (ip + l > end || op + l > op_end) return -1;
            memcpy(op, ip, l);
            ip += l;
            op += l;
            continue;
        case 1:
            if (ip >= end) return -1;
            l = ((tag >> 2) & 7) + 4;
            off = ((size_t)(tag >> 5) << 8) | *ip++;
            break;
        case 2:
            if (ip + 2 > end) return -1;
            l = (tag >> 2) + 1;
            off = (size_t)ip[0] | ((size_t)ip[1] << 8);
            ip += 2;
            break;
        default:
            if (ip + 4 > end) return -1;
            l = (tag >> 2) + 1;
            off = (size_t)ip[0] | ((size_t)ip[1] << 8) |
                  ((size_t)ip[2] << 16) | ((size_t)ip[3] << 24);
            ip += 4;
            break;
        }
        if (off == 0 || (size_t)(op - dst) < off || op + l > op_end) return -1;
        const uint8_t *cp = op - off; /* may overlap: byte-by-byte */
        for (size_t i = 0; i < l; i++) op[i] = cp[i];
        op += l;
    }
    return (op == op_end && ip == end) ? ulen : -1;
}

/* strings -> block bytes (string.rs:32-88).  src = concatenated string
 * bytes; lens[i] = byte length of string i.  Empty input -> 0 bytes
 * (string.rs:33-35). */
ORC_EXPORT int64_t orc_str_encode(const uint8_t *src, const uint64_t *lens,
                                  int64_t nstr, uint8_t *dst, size_t cap) {
    if (nstr == 0) return 0;
    size_t payload = 0;
    for (int64_t i = 0; i < nstr; i++) {
        uint64_t v = lens[i];
        payload += lens[i];
        do { payload++; v >>= 7; } while (v > 0);
    }
    uint8_t *buf = (uint8_t *)malloc(payload);
    if (!buf) return -1;
    size_t p = 0, s = 0;
    for (int64_t i = 0; i < nstr; i++) {
        uint64_t v = lens[i];
        while (v >= 0x80) { buf[p++] = (uint8_t)(v | 0x80); v >>= 7; }
        buf[p++] = (uint8_t)v;
        memcpy(buf + p, src + s, lens[i]);
        p += lens[i];
        s += lens[i];
    }
    if (cap < 2 + (size_t)orc_snappy_max_compress_len((int64_t)payload)) {
        free(buf);
        return -1;
    }
    dst[0] = 7;    /* Encoding::Snappy, models/src/codec.rs:48 */
    dst[1] = 0x10; /* STRING_COMPRESSED_SNAPPY << 4, string.rs:21,79 */
    int64_t c = orc_snappy_compress(buf, payload, dst + 2, cap - 2);
    free(buf);
    return c < 0 ? -1 : c + 2;
}

/* block bytes + bitset -> concatenated strings + per-row lengths
 * (str_snappy_decode_to_array, string.rs:226-276; nulls consume nothing
 * from the payload; empty src -> all rows null; Encoding::Null blocks
 * use [u64 BE len][bytes] per string, string.rs:169-183,289-300).
 * lens_out[row] = -1 for null rows.  Returns total payload bytes. */
ORC_EXPORT int64_t orc_str_decode(const uint8_t *src, size_t len,
                                  const uint8_t *bitset, int64_t nrows,
                                  uint8_t *bytes_out, size_t bytes_cap,
                                  int64_t *lens_out) {
    for (int64_t r = 0; r < nrows; r++) lens_out[r] = -1;
    if (len == 0) return 0;
    uint8_t *payload = NULL;
    const uint8_t *pl;
    size_t pn;
    int be_lens = 0;
    if (src[0] == 7) { /* Snappy */
        if (len < 2) return -1;
        int64_t ulen = orc_snappy_uncompressed_len(src + 2, len - 2, NULL);
        if (ulen < 0) return -1;
        payload = (uint8_t *)malloc(ulen ? (size_t)ulen : 1);
        if (!payload) return -1;
        if (orc_snappy_decompress(src + 2, len - 2, payload, (size_t)ulen) != ulen) {
            free(payload);
            return -1;
        }
        pl = payload;
        pn = (size_t)ulen;
    } else if (src[0] == 1) { /* Encoding::Null: uncompressed */
        pl = src + 1;
        pn = len - 1;
        be_lens = 1;
    } else {
        return -1;
    }
    size_t i = 0, w = 0;
    int64_t rc = 0;
    for (int64_t r = 0; r < nrows; r++) {
        if (bitset && !((bitset[r >> 3] >> (r & 7)) & 1)) continue;
        if (i >= pn) break; /* reference stops silently at payload end */
        uint64_t slen;
        if (be_lens) {
            if (i + 8 > pn) { rc = -1; break; }
            slen = 0;
            for (int k = 0; k < 8; k++) slen = (slen << 8) | pl[i + k];
            i += 8;
        } else {
            slen = 0;
            int sh = 0, ok = 0;
            while (i < pn) {
                uint8_t b = pl[i++];
                slen |= (uint64_t)(b & 0x7f) << sh;
                sh += 7;
                if (!(b & 0x80)) { ok = 1; break; }
            }
            if (!ok) { rc = -1; break; }
        }
        if (i + slen > pn || w + slen > bytes_cap) { rc = -1; break; }
        memcpy(bytes_out + w, pl + i, slen);
        lens_out[r] = (int64_t)slen;
        i += slen;
        w += slen;
    }
    if (payload) free(payload);
    return rc < 0 ? -1 : (int64_t)w;
}

/* cpu_baseline leg for the strings bench: decode many string blocks in
 * parallel (mirrors decode_pages fan-out over ColumnGroups,
 * tskv/src/reader/column_group/mod.rs:202-243).  blocks[i]/lens[i] =
 * page data region; all pages all-valid with nrows rows.  Returns total
 * payload bytes or -1. */
ORC_EXPORT int64_t orc_str_decode_pages_omp(const uint8_t **blocks,
                                            const size_t *lens, int npages,
                                            int64_t nrows, int64_t cap_per_page,
                                            int nthreads) {
    int64_t total = 0;
    int err = 0;
#pragma omp parallel num_threads(nthreads)
    {
        uint8_t *bytes_buf = (uint8_t *)malloc((size_t)cap_per_page);
        int64_t *lens_buf = (int64_t *)malloc((size_t)nrows * 8);
        int64_t mine = 0;
#pragma omp for schedule(dynamic, 4)
        for (int p = 0; p < npages; p++) {
            if (!bytes_buf || !lens_buf) { err = 1; continue; }
            int64_t w = orc_str_decode(blocks[p], lens[p], NULL, nrows,
                                       bytes_buf, (size_t)cap_per_page,
                                       lens_buf);
            if (w < 0) err = 1;
            else mine += w;
        }
#pragma omp atomic
        total += mine;
        free(bytes_buf);
        free(lens_buf);
    }
    return err ? -1 : total;
}
