/* Host-side write path of the cnosdb_gs engine: the TSM DataBlock encoders
 * and page assembly.  This is the product's mirror of the reference's
 * encode side (tskv/src/tsm/codec/{timestamp,integer,simple8b,float,
 * boolean}.rs encode fns and Page::arrow_array_to_page, tsm/page.rs:
 * 100-353,488-497); byte-exact with the reference encoder, pinned by the
 * InfluxDB golden blocks (integer.rs:438-483) and the boolean vectors
 * (boolean.rs:156-181).  Used by the GPU read path's fixture/flush side;
 * runs on host CPU exactly as the reference writer does.
 */
#include "gs_internal.h"
#include <cstring>
#include <vector>

namespace {

enum {
    ENC_NULL = 1,
    ENC_DELTA = 2,
    ENC_GORILLA = 6,
    ENC_BITPACK = 10,
    ENC_DELTATS = 11,
};
enum { SUB_UNCOMPRESSED = 0, SUB_SIMPLE8B = 1, SUB_RLE = 2 };

constexpr uint64_t S8B_MAX = (1ULL << 60) - 1;
constexpr uint64_t SENTINEL = 0x7ff8000000000ffULL; /* float.rs:16 */

inline void put_be64(uint8_t *p, uint64_t v) {
    for (int i = 7; i >= 0; i--) { p[i] = uint8_t(v); v >>= 8; }
}
inline uint64_t zz(int64_t v) { return (uint64_t(v) << 1) ^ uint64_t(v >> 63); }

size_t varint_put(uint8_t *dst, uint64_t v) {
    size_t n = 0;
    while (v >= 0x80) { dst[n++] = uint8_t(v | 0x80); v >>= 7; }
    dst[n++] = uint8_t(v);
    return n;
}

/* simple8b greedy packer, simple8b.rs:26-76 */
const uint8_t NUM_BITS[14][2] = {
    {60, 1}, {30, 2}, {20, 3}, {15, 4}, {12, 5}, {10, 6}, {8, 7},
    {7, 8},  {6, 10}, {5, 12}, {4, 15}, {3, 20}, {2, 30}, {1, 60},
};

int64_t s8b_encode(const uint64_t *src, size_t n, uint8_t *dst, size_t cap) {
    size_t i = 0, w = 0;
    while (i < n) {
        size_t remain = n - i;
        if (remain >= 120) {
            size_t lim = remain >= 240 ? 240 : 120;
            size_t k = 0;
            while (k < lim && src[i + k] == 1) k++;
            if (k == 240) {
                if (w + 8 > cap) return -5;
                memset(dst + w, 0, 8); w += 8; i += 240; continue;
            }
            if (k >= 120) {
                if (w + 8 > cap) return -5;
                put_be64(dst + w, 1ULL << 60); w += 8; i += 120; continue;
            }
        }
        bool packed = false;
        for (int idx = 0; idx < 14; idx++) {
            size_t int_n = NUM_BITS[idx][0];
            unsigned bit_n = NUM_BITS[idx][1];
            if (int_n > remain) continue;
            uint64_t max_val = 1ULL << (bit_n & 0x3f);
            uint64_t word = (uint64_t(idx) + 2) << 60;
            bool fits = true;
            for (size_t k = 0; k < int_n; k++) {
                if (src[i + k] >= max_val) { fits = false; break; }
                word |= src[i + k] << ((k * bit_n) & 0x3f);
            }
            if (!fits) continue;
            if (w + 8 > cap) return -5;
            put_be64(dst + w, word); w += 8; i += int_n;
            packed = true;
            break;
        }
        if (!packed) return -1; /* value out of bounds */
    }
    return int64_t(w);
}

} // namespace

extern "C" {

/* ts_zigzag_simple8b_encode, timestamp.rs:51-122 (raw wrapping deltas, no
 * zigzag; RLE count includes the first value; divisor scaling in the
 * simple8b/RLE sub-paths; single-value blocks carry scaler nibble 12) */
int64_t gs_encode_ts(const int64_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    if (cap < 2 + 8 * n + 16) return -5;
    size_t w = 0;
    dst[w++] = ENC_DELTATS;
    std::vector<uint64_t> d(n);
    for (size_t i = 0; i < n; i++) d[i] = uint64_t(src[i]);
    uint64_t max = 0;
    if (n > 1) {
        for (size_t i = n - 1; i >= 1; i--) {
            d[i] -= d[i - 1];
            if (d[i] > max) max = d[i];
            if (i == 1) break;
        }
        bool use_rle = true;
        for (size_t i = 2; i < n; i++)
            if (d[1] != d[i]) { use_rle = false; break; }
        if (use_rle) {
            dst[w++] = 0;
            put_be64(dst + w, d[0]); w += 8;
            uint64_t div = 1000000000000ULL;
            while (div > 1 && d[1] % div != 0) div /= 10;
            if (div > 1) {
                unsigned sc = 0;
                for (uint64_t x = div; x > 1; x /= 10) sc++;
                dst[1] |= uint8_t(sc);
                w += varint_put(dst + w, d[1] / div);
            } else {
                w += varint_put(dst + w, d[1]);
            }
            w += varint_put(dst + w, uint64_t(n));
            dst[1] |= uint8_t(SUB_RLE << 4);
            return int64_t(w);
        }
    }
    if (max > S8B_MAX) {
        dst[w++] = uint8_t(SUB_UNCOMPRESSED << 4);
        for (size_t i = 0; i < n; i++) { put_be64(dst + w, d[i]); w += 8; }
        return int64_t(w);
    }
    uint64_t div = 1000000000000ULL;
    for (size_t i = 1; i < n && div > 1; i++)
        while (div > 1 && d[i] % div != 0) div /= 10;
    if (div > 1)
        for (size_t i = 1; i < n; i++) d[i] /= div;
    unsigned sc = 0;
    for (uint64_t x = div; x > 1; x /= 10) sc++;
    dst[w++] = uint8_t((SUB_SIMPLE8B << 4) | sc);
    put_be64(dst + w, d[0]); w += 8;
    int64_t s = s8b_encode(d.data() + 1, n - 1, dst + w, cap - w);
    if (s < 0) return s;
    return int64_t(w + size_t(s));
}

/* i64_zigzag_simple8b_encode, integer.rs:40-96 (zigzag deltas; RLE needs
 * n>=3 and its count excludes the first value; no divisor) */
int64_t gs_encode_i64(const int64_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    if (cap < 2 + 8 * n + 16) return -5;
    size_t w = 0;
    dst[w++] = ENC_DELTA;
    std::vector<uint64_t> d(n);
    for (size_t i = 0; i < n; i++) d[i] = uint64_t(src[i]);
    uint64_t max = 0;
    for (size_t i = n - 1; i >= 1; i--) {
        d[i] = zz(int64_t(d[i] - d[i - 1]));
        if (d[i] > max) max = d[i];
        if (i == 1) break;
    }
    d[0] = zz(src[0]);
    if (n > 2) {
        bool use_rle = true;
        for (size_t i = 2; i < n; i++)
            if (d[1] != d[i]) { use_rle = false; break; }
        if (use_rle) {
            dst[w++] = 0;
            put_be64(dst + w, d[0]); w += 8;
            w += varint_put(dst + w, d[1]);
            w += varint_put(dst + w, uint64_t(n) - 1);
            dst[1] |= uint8_t(SUB_RLE << 4);
            return int64_t(w);
        }
    }
    if (max > S8B_MAX) {
        dst[w++] = uint8_t(SUB_UNCOMPRESSED << 4);
        for (size_t i = 0; i < n; i++) { put_be64(dst + w, d[i]); w += 8; }
        return int64_t(w);
    }
    dst[w++] = uint8_t(SUB_SIMPLE8B << 4);
    put_be64(dst + w, d[0]); w += 8;
    int64_t s = s8b_encode(d.data() + 1, n - 1, dst + w, cap - w);
    if (s < 0) return s;
    return int64_t(w + size_t(s));
}

/* f64_gorilla_encode, float.rs:32-243: XOR bitstream, MSB-first; leading
 * capped at 31 before the window comparison; meaningful 64 encoded as 0;
 * encoded NaN sentinel appended as terminator */
int64_t gs_encode_f64(const double *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    size_t need = 2 + 8 + (n + 1) * 10 + 16;
    if (cap < need) return -5;
    memset(dst, 0, need);
    dst[0] = ENC_GORILLA;
    dst[1] = 1 << 4;
    uint64_t prev;
    memcpy(&prev, &src[0], 8);
    put_be64(dst + 2, prev);
    size_t nb = 8 + 64; /* bit cursor relative to dst[1] */
    auto put_bit = [&](int bit) {
        if (bit) dst[(nb >> 3) + 1] |= uint8_t(128u >> (nb & 7));
        nb++;
    };
    auto put_top = [&](uint64_t v, unsigned l) {
        while (l > 0) {
            unsigned m = nb & 7;
            unsigned take = 8 - m;
            if (take > l) take = l;
            dst[(nb >> 3) + 1] |= uint8_t((v >> 56) >> m);
            v <<= take;
            nb += take;
            l -= take;
        }
    };
    uint64_t prev_lead = ~0ULL, prev_trail = 0;
    for (size_t i = 1; i <= n; i++) {
        uint64_t cur;
        if (i < n) {
            memcpy(&cur, &src[i], 8);
            if (cur == SENTINEL) return -6;
        } else {
            cur = SENTINEL;
        }
        uint64_t x = cur ^ prev;
        if (x == 0) { put_bit(0); prev = cur; continue; }
        put_bit(1);
        uint64_t lead = uint64_t(__builtin_clzll(x)) & 0x1f;
        uint64_t trail = uint64_t(__builtin_ctzll(x));
        if (prev_lead != ~0ULL && lead >= prev_lead && trail >= prev_trail) {
            put_bit(0);
            unsigned l = unsigned(64 - prev_lead - prev_trail);
            uint64_t v = (l == 64) ? (x >> prev_trail) : ((x >> prev_trail) << (64 - l));
            put_top(v, l);
        } else {
            prev_lead = lead;
            prev_trail = trail;
            put_bit(1);
            put_top(lead << 59, 5);
            uint64_t sig = 64 - lead - trail;
            put_top(sig << 58, 6);
            unsigned l = unsigned(sig);
            uint64_t v = (l == 64) ? (x >> trail) : ((x >> trail) << (64 - l));
            put_top(v, l);
        }
        prev = cur;
    }
    size_t length = (nb >> 3) + 1;
    if (nb & 7) length += 1;
    return int64_t(length);
}

/* bool_bitpack_encode, boolean.rs:24-64 */
int64_t gs_encode_bool(const uint8_t *src, size_t n, uint8_t *dst, size_t cap) {
    if (n == 0) return 0;
    size_t size = 1 + 8 + (n + 7) / 8;
    if (cap < size + 2) return -5;
    memset(dst, 0, size + 2);
    dst[0] = ENC_BITPACK;
    dst[1] = 1 << 4;
    size_t vi = varint_put(dst + 2, uint64_t(n));
    size_t nb = 8 + vi * 8;
    for (size_t k = 0; k < n; k++) {
        if (src[k]) dst[(nb >> 3) + 1] |= uint8_t(128u >> (nb & 7));
        nb++;
    }
    size_t length = nb >> 3;
    if (nb & 7) length += 1;
    return int64_t(length + 1);
}

/* CRC-32/ISO-HDLC (crc32fast), page.rs:58-76 */
uint32_t gs_crc32(const uint8_t *data, size_t len) {
    static uint32_t table[256];
    static bool init = false;
    if (!init) {
        for (uint32_t i = 0; i < 256; i++) {
            uint32_t c = i;
            for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
            table[i] = c;
        }
        init = true;
    }
    uint32_t c = 0xFFFFFFFFu;
    for (size_t i = 0; i < len; i++) c = table[(c ^ data[i]) & 0xFF] ^ (c >> 8);
    return c ^ 0xFFFFFFFFu;
}

/* ---------------- string block encoder (codec/string.rs:32-88) ----------
 * [Encoding::Snappy=7][0x10][snappy raw stream] over a payload of
 * [LEB128 varint len][bytes] per string.  The snappy compressor restates
 * the published reference algorithm used by the un-vendored `snap` crate
 * v1.1.1 (64 KiB fragments, power-of-two hash table in [256,16384] of
 * 4-byte little-endian loads hashed by *0x1e35a7bd>>shift, skip-32
 * acceleration, 15-byte tail margin); its output is pinned byte-exactly
 * by the reference's own golden vectors (string.rs:529-566), checked in
 * tests/test_encoders.py.  Independent of oracle/tsm_oracle.c. */

namespace {

inline uint32_t sload32(const uint8_t *p) {
    uint32_t v;
    memcpy(&v, p, 4);
    return v;
}

inline uint8_t *slit(uint8_t *op, const uint8_t *p, size_t n) {
    if (n - 1 < 60) {
        *op++ = uint8_t((n - 1) << 2);
    } else {
        uint8_t *tag = op++;
        size_t v = n - 1;
        int nb = 0;
        for (; v; v >>= 8) *op++ = uint8_t(v), nb++;
        *tag = uint8_t((59 + nb) << 2);
    }
    memcpy(op, p, n);
    return op + n;
}

inline uint8_t *scopy1(uint8_t *op, size_t off, size_t len) {
    if (len < 12 && off < 2048) {
        *op++ = uint8_t(1 | ((len - 4) << 2) | ((off >> 8) << 5));
        *op++ = uint8_t(off);
    } else {
        *op++ = uint8_t(2 | ((len - 1) << 2));
        *op++ = uint8_t(off);
        *op++ = uint8_t(off >> 8);
    }
    return op;
}

inline uint8_t *scopy(uint8_t *op, size_t off, size_t len) {
    for (; len >= 68; len -= 64) op = scopy1(op, off, 64);
    if (len > 64) op = scopy1(op, off, 60), len -= 60;
    return scopy1(op, off, len);
}

uint8_t *snappy_fragment(const uint8_t *in, size_t n, uint8_t *op,
                         uint16_t *tab) {
    size_t tsz = 256;
    while (tsz < 16384 && tsz < n) tsz <<= 1;
    int shift = 32 - __builtin_ctzll(tsz);
    memset(tab, 0, tsz * sizeof(uint16_t));
    const uint8_t *ip = in, *iend = in + n, *nmit = in;
    if (n >= 15) {
        const uint8_t *ilim = iend - 15;
        uint32_t nh = (sload32(++ip) * 0x1e35a7bdu) >> shift;
        for (;;) {
            uint32_t skip = 32;
            const uint8_t *nip = ip, *cand;
            do {
                ip = nip;
                uint32_t h = nh, adv = skip >> 5;
                skip += adv;
                nip = ip + adv;
                if (nip > ilim) goto tail;
                nh = (sload32(nip) * 0x1e35a7bdu) >> shift;
                cand = in + tab[h];
                tab[h] = uint16_t(ip - in);
            } while (sload32(ip) != sload32(cand));
            op = slit(op, nmit, size_t(ip - nmit));
            uint32_t c32;
            do {
                const uint8_t *b = ip;
                const uint8_t *m1 = cand + 4, *m2 = ip + 4;
                while (m2 < iend && *m1 == *m2) m1++, m2++;
                size_t mlen = size_t(m2 - ip);
                ip += mlen;
                op = scopy(op, size_t(b - cand), mlen);
                nmit = ip;
                if (ip >= ilim) goto tail;
                uint32_t hp = (sload32(ip - 1) * 0x1e35a7bdu) >> shift;
                tab[hp] = uint16_t(ip - 1 - in);
                uint32_t hc = (sload32(ip) * 0x1e35a7bdu) >> shift;
                cand = in + tab[hc];
                c32 = sload32(cand);
                tab[hc] = uint16_t(ip - in);
            } while (sload32(ip) == c32);
            nh = (sload32(++ip) * 0x1e35a7bdu) >> shift;
        }
    }
tail:
    if (nmit < iend) op = slit(op, nmit, size_t(iend - nmit));
    return op;
}

} // namespace

int64_t gs_encode_str(const uint8_t *src, const uint64_t *lens, int64_t nstr,
                      uint8_t *dst, size_t cap) {
    if (nstr == 0) return 0;
    size_t payload = 0;
    for (int64_t i = 0; i < nstr; i++) {
        uint64_t v = lens[i];
        payload += size_t(lens[i]);
        do payload++; while ((v >>= 7));
    }
    std::vector<uint8_t> buf(payload);
    size_t p = 0, s = 0;
    for (int64_t i = 0; i < nstr; i++) {
        uint64_t v = lens[i];
        for (; v >= 0x80; v >>= 7) buf[p++] = uint8_t(v | 0x80);
        buf[p++] = uint8_t(v);
        memcpy(buf.data() + p, src + s, size_t(lens[i]));
        p += size_t(lens[i]);
        s += size_t(lens[i]);
    }
    if (cap < 2 + 32 + payload + payload / 6) return -5;
    dst[0] = 7;    /* Encoding::Snappy */
    dst[1] = 0x10; /* STRING_COMPRESSED_SNAPPY << 4 */
    uint8_t *op = dst + 2;
    size_t v = payload;
    for (; v >= 0x80; v >>= 7) *op++ = uint8_t(v | 0x80);
    *op++ = uint8_t(v);
    std::vector<uint16_t> tab(16384);
    for (size_t off = 0; off < payload; off += 65536) {
        size_t frag = payload - off < 65536 ? payload - off : 65536;
        op = snappy_fragment(buf.data() + off, frag, op, tab.data());
    }
    return op - dst;
}

/* page assembly, tsm/page.rs:488-497 */
int64_t gs_build_page(const uint8_t *bitset, int64_t nrows, const uint8_t *data,
                      size_t data_len, uint8_t *dst, size_t cap) {
    size_t bl = size_t((nrows + 7) / 8);
    size_t total = 16 + bl + data_len;
    if (cap < total) return -5;
    dst[0] = uint8_t(bl >> 24); dst[1] = uint8_t(bl >> 16);
    dst[2] = uint8_t(bl >> 8); dst[3] = uint8_t(bl);
    put_be64(dst + 4, uint64_t(nrows));
    uint32_t crc = gs_crc32(data, data_len);
    dst[12] = uint8_t(crc >> 24); dst[13] = uint8_t(crc >> 16);
    dst[14] = uint8_t(crc >> 8); dst[15] = uint8_t(crc);
    memcpy(dst + 16, bitset, bl);
    memcpy(dst + 16 + bl, data, data_len);
    return int64_t(total);
}

/* OpenMP batch Gorilla encode: one page per chunk of `rows_per_page`
 * consecutive values (fixture/flush helper; mirrors the writer loop of
 * TsmWriter::write_record_batch, tsm/writer.rs:249-314). Each page is
 * written at dst + p*cap_per_page; returns 0 and fills out_lens. */
int32_t gs_encode_f64_pages_omp(const double *vals, int64_t rows_per_page,
                                int64_t npages, uint8_t *dst,
                                int64_t cap_per_page, int64_t *out_lens,
                                int32_t nthreads) {
    (void)nthreads; /* used only via the OMP clause below */
    int32_t err = 0;
#pragma omp parallel for schedule(dynamic, 1) num_threads(nthreads)
    for (int64_t p = 0; p < npages; p++) {
        int64_t n = gs_encode_f64(vals + p * rows_per_page, size_t(rows_per_page),
                                  dst + p * cap_per_page, size_t(cap_per_page));
        out_lens[p] = n;
        if (n < 0) {
#pragma omp atomic write
            err = int32_t(n);
        }
    }
    return err;
}

const char *gs_version(void) { return "cnosdb_gs 0.1 (gfx950)"; }

} // extern "C"
