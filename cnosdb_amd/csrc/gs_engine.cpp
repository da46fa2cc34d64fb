/* cnosdb_gs engine: HIP/CDNA4 (gfx950) kernels for the TSM DataBlock
 * decode + predicate-filtered time-range scan + downsampling aggregates,
 * and the C-ABI host orchestration around them.
 *
 * Decode semantics mirror, bit-exactly:
 *   tskv/src/tsm/codec/timestamp.rs:177-299   (DeltaTs: sub-tag + scaler)
 *   tskv/src/tsm/codec/integer.rs:142-248     (Delta: zigzag)
 *   tskv/src/tsm/codec/simple8b.rs:80-208
 *   tskv/src/tsm/codec/float.rs:351-606       (Gorilla, sentinel-terminated)
 *   tskv/src/tsm/codec/boolean.rs:79-110
 *   tskv/src/tsm/reader.rs:494-560,634-656    (decode_pages + tombstone)
 *
 * MI355X mapping (this path is HBM-bandwidth-bound integer/bit work — no
 * MFMA): page-level parallelism across the 256 CUs; inherently sequential
 * bitstreams (Gorilla) decode one page per thread with many pages in
 * flight; closed-form encodings (RLE) decode one workgroup per page fully
 * parallel.  All launches are grid-strided and sized per Guideline 11.
 */
#include <hip/hip_runtime.h>
#include "../../include/cnosdb_gs.h"
#include "gs_internal.h"

/* hipError_t returns are checked with HIP_TRY on every path that can fail
 * user-visibly; cleanup paths (hipFree/hipEventDestroy in destructors) and
 * async enqueues deliberately drop the return — every enqueue is followed
 * by a checked hipStreamSynchronize on the same stream, which surfaces any
 * deferred error.  Silence the nodiscard noise for those. */
#pragma clang diagnostic ignored "-Wunused-value"

#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <vector>
#include <algorithm>
#include <string>

/* ------------------------------------------------------------- error glue */
static thread_local char g_err[512] = "";
static GsStatus fail(GsStatus s, const char *msg) {
    snprintf(g_err, sizeof(g_err), "%s", msg);
    return s;
}
extern "C" const char *gs_last_error(void) { return g_err; }

#define HIP_TRY(expr)                                                          \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess) {                                                \
            snprintf(g_err, sizeof(g_err), "%s: %s", #expr,                    \
                     hipGetErrorString(_e));                                   \
            return GS_ERR;                                                     \
        }                                                                      \
    } while (0)

#define HIP_TRY_NULL(expr)                                                     \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess) {                                                \
            snprintf(g_err, sizeof(g_err), "%s: %s", #expr,                    \
                     hipGetErrorString(_e));                                   \
            return nullptr;                                                    \
        }                                                                      \
    } while (0)

/* ------------------------------------------------------------ device utils */

__device__ __forceinline__ uint64_t dev_be64(const uint8_t *p) {
    uint64_t v;
    __builtin_memcpy(&v, p, 8);
    return __builtin_bswap64(v);
}
__device__ __forceinline__ int64_t dev_zzdec(uint64_t v) {
    return int64_t((v >> 1) ^ uint64_t(-int64_t(v & 1)));
}
__device__ __forceinline__ int dev_bit(const uint8_t *bs, uint32_t i) {
    return (bs[i >> 3] >> (i & 7)) & 1;
}
__device__ __forceinline__ uint64_t dev_rotl64(uint64_t x, unsigned c) {
    c &= 63;
    return c ? (x << c) | (x >> (64 - c)) : x;
}

/* device varint (LEB128), integer_encoding crate semantics */
__device__ __forceinline__ bool dev_varint(const uint8_t *p, uint32_t len,
                                           uint64_t *out, uint32_t *nread) {
    uint64_t v = 0;
    int shift = 0;
    for (uint32_t i = 0; i < len && i < 10; i++) {
        v |= uint64_t(p[i] & 0x7f) << shift;
        if (!(p[i] & 0x80)) { *out = v; *nread = i + 1; return true; }
        shift += 7;
    }
    return false;
}

/* error codes accumulated into d_err */
#define DERR_FORMAT 1u
#define DERR_SHORT 2u
#define DERR_NONMONO 4u /* a CRC-valid but non-monotonic time
                           page would silently mis-select rows via the
                           binary-search span/tombstone kernels; fail
                           loudly instead (the reference's row-wise
                           filtering cannot mis-select) */

/* --------------------------------------------- pull-style value iterators */

/* simple8b word stream (simple8b.rs:95-208), value extraction by index */
struct DevS8b {
    const uint8_t *p;
    uint32_t len, pos;
    uint64_t word;
    int idx, cnt;
    unsigned bits;
    __device__ void init(const uint8_t *p_, uint32_t len_) {
        p = p_; len = len_; pos = 0; idx = 0; cnt = 0; bits = 0; word = 0;
    }
    __device__ bool next(uint64_t *out) {
        while (idx >= cnt) {
            if (pos + 8 > len) return false;
            word = dev_be64(p + pos);
            pos += 8;
            unsigned sel = unsigned(word >> 60);
            const uint8_t COUNT[16] = {240, 120, 60, 30, 20, 15, 12, 10,
                                       8, 7, 6, 5, 4, 3, 2, 1};
            const uint8_t WIDTH[16] = {0, 0, 1, 2, 3, 4, 5, 6,
                                       7, 8, 10, 12, 15, 20, 30, 60};
            cnt = COUNT[sel];
            bits = WIDTH[sel];
            idx = 0;
        }
        if (bits == 0) { *out = 1; }
        else {
            uint64_t mask = (bits == 60) ? 0x0fffffffffffffffULL
                                         : ((1ULL << bits) - 1);
            *out = (word >> (unsigned(idx) * bits)) & mask;
        }
        idx++;
        return true;
    }
};


#define GORILLA_SENTINEL 0x7ff8000000000ffULL

/* ---------------------------------------------------------------- kernels */

/* Universal sequential decoder for int64-family pages (TIME/I64/U64):
 * one thread per page, grid-strided.  Handles DeltaTs/Delta with all three
 * sub-tags plus Null (raw BE), with validity-bitset scatter (null -> 0). */
__global__ void k_seq_i64(const uint8_t *__restrict__ blob,
                          const DevPage *__restrict__ pages, int npages,
                          int64_t *__restrict__ out,
                          uint8_t *__restrict__ valid,
                          unsigned *__restrict__ err) {
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        DevPage pg = pages[p];
        const uint8_t *data = blob + pg.data_off;
        const uint8_t *bs = blob + pg.bitset_off;
        int64_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;

        if (pg.data_len == 0) { /* all-null (timestamp.rs:181-185) */
            for (uint32_t r = 0; r < n; r++) { o[r] = 0; if (vd) vd[r] = 0; }
            continue;
        }
        uint8_t enc = pg.enc;
        if (enc == GS_ENC_NULL) { /* raw BE per valid slot */
            const uint8_t *q = data + 1;
            uint32_t avail = (pg.data_len - 1) / 8, used = 0;
            int64_t prev_ts = INT64_MIN;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (v && used < avail) {
                    int64_t x = int64_t(dev_be64(q + 8ull * used)); used++;
                    if (pg.ctype == GS_CT_TIME) {
                        if (x < prev_ts) atomicOr(err, DERR_NONMONO);
                        prev_ts = x;
                    }
                    o[r] = x;
                }
                else { o[r] = 0; v = 0; }
                if (vd) vd[r] = uint8_t(v);
            }
            continue;
        }
        bool is_ts = (enc == GS_ENC_DELTATS);
        if (!is_ts && enc != GS_ENC_DELTA) { atomicOr(err, DERR_FORMAT); continue; }
        if (pg.data_len < 2) { atomicOr(err, DERR_FORMAT); continue; }
        const uint8_t *s = data + 1;
        uint32_t slen = pg.data_len - 1;
        unsigned sub = s[0] >> 4;

        if (sub == 0) { /* uncompressed deltas */
            const uint8_t *q = s + 1;
            uint32_t plen = slen - 1;
            if (plen == 0 || (plen & 7)) { atomicOr(err, DERR_FORMAT); continue; }
            uint32_t avail = plen / 8, used = 0;
            int64_t prev = 0;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (!v) { o[r] = 0; if (vd) vd[r] = 0; continue; }
                if (used >= avail) {
                    if (!is_ts) { atomicOr(err, DERR_SHORT); }
                    o[r] = 0; if (vd) vd[r] = uint8_t(is_ts ? 1 : 0);
                    continue;
                }
                uint64_t raw = dev_be64(q + 8ull * used);
                used++;
                int64_t nxt = is_ts ? int64_t(uint64_t(prev) + raw)
                                    : int64_t(uint64_t(prev) +
                                              uint64_t(dev_zzdec(raw)));
                if (is_ts && nxt < prev && used > 1)
                    atomicOr(err, DERR_NONMONO);
                prev = nxt;
                o[r] = prev;
                if (vd) vd[r] = 1;
            }
            continue;
        }
        if (sub == 2) { /* RLE */
            uint64_t scaler = 1;
            if (is_ts) {
                unsigned s10 = s[0] & 0x0f;
                for (unsigned k = 0; k < s10; k++) scaler *= 10;
            }
            const uint8_t *q = s + 1;
            uint32_t plen = slen - 1;
            if (plen < 9) { atomicOr(err, DERR_FORMAT); continue; }
            uint64_t first_raw = dev_be64(q);
            uint64_t dv; uint32_t nr;
            if (!dev_varint(q + 8, plen - 8, &dv, &nr)) { atomicOr(err, DERR_FORMAT); continue; }
            int64_t cur, delta;
            if (is_ts) { cur = int64_t(first_raw); delta = int64_t(dv * scaler); }
            else { cur = dev_zzdec(first_raw); delta = dev_zzdec(dv); }
            if (is_ts && delta < 0 && n > 1) atomicOr(err, DERR_NONMONO);
            bool first = true;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (!v) { o[r] = 0; if (vd) vd[r] = 0; continue; }
                if (first) { o[r] = cur; first = false; }
                else { cur = int64_t(uint64_t(cur) + uint64_t(delta)); o[r] = cur; }
                if (vd) vd[r] = 1;
            }
            continue;
        }
        if (sub == 1) { /* simple8b */
            uint64_t scaler = 1;
            if (is_ts) {
                unsigned s10 = s[0] & 0x0f;
                for (unsigned k = 0; k < s10; k++) scaler *= 10;
            }
            const uint8_t *q = s + 1;
            uint32_t plen = slen - 1;
            if (plen < 8) { atomicOr(err, DERR_SHORT); continue; }
            uint64_t first_raw = dev_be64(q);
            int64_t cur = is_ts ? int64_t(first_raw) : dev_zzdec(first_raw);
            DevS8b it;
            it.init(q + 8, plen - 8);
            bool first = true;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (!v) { o[r] = 0; if (vd) vd[r] = 0; continue; }
                if (first) { o[r] = cur; first = false; if (vd) vd[r] = 1; continue; }
                uint64_t u;
                if (!it.next(&u)) { o[r] = 0; if (vd) vd[r] = 1; continue; } /* iterator exhaustion: builder skips */
                int64_t nxt2 = is_ts ? int64_t(uint64_t(cur) + u * scaler)
                                     : int64_t(uint64_t(cur) +
                                               uint64_t(dev_zzdec(u)));
                if (is_ts && nxt2 < cur) atomicOr(err, DERR_NONMONO);
                cur = nxt2;
                o[r] = cur;
                if (vd) vd[r] = 1;
            }
            continue;
        }
        atomicOr(err, DERR_FORMAT);
    }
}

/* Gorilla f64 pages: one thread per page (the bitstream carries a strict
 * sequential dependency — float.rs:445-463; parallelism comes from the
 * page count).  The hot loop is a flat 128-bit-window parser: one 13-bit
 * peek covers ctrl bits + leading/meaningful, whole-word refills run over
 * the blob's tail pad with a bit-budget guard standing in for the
 * reference's end-of-block refill error — identical bit consumption and
 * identical error surface on truncated streams. */
template <bool ALLVALID>
__device__ __forceinline__ void gorilla_decode_page(
    const uint8_t *data, uint32_t data_len, const uint8_t *bs, uint32_t n,
    double *o, uint8_t *vd, unsigned *err) {
    const uint8_t *s = data + 1;
    uint32_t slen = data_len - 1;
    if (slen < 9) { atomicOr(err, DERR_SHORT); return; }
    uint64_t val = dev_be64(s + 1);
    const uint8_t *p = s + 9;
    int64_t budget = int64_t(slen - 9) * 8; /* bits the stream really holds */
    uint64_t hi = 0, lo = 0;
    int nb = 0; /* valid window bits (may exceed budget over the pad) */
    uint32_t trailing_n = 0, meaningful_n = 64;
    uint32_t r = 0;
    bool bad = false;

    auto emit = [&](uint64_t bits_) {
        if (!ALLVALID) {
            while (r < n && !dev_bit(bs, r)) { o[r] = 0.0; if (vd) vd[r] = 0; r++; }
        }
        if (r < n) {
            o[r] = __longlong_as_double(int64_t(bits_));
            if (vd) vd[r] = 1;
            r++;
        }
    };
    /* one-word prefetch: the load for the NEXT refill is issued as soon as
       the previous word is consumed, so its latency hides under ~7 values
       of ALU work instead of stalling every refill (the PMC profile showed
       67% of wave cycles parked on exactly that wait) */
    uint64_t nextw = dev_be64(p);
    p += 8;
    auto topup = [&]() { /* call only with nb < 64 (x >> 64 is UB/mod-64) */
        uint64_t x = nextw;
        nextw = dev_be64(p);
        p += 8;
        if (nb == 0) { hi = x; lo = 0; }
        else { hi |= x >> nb; lo = x << (64 - nb); }
        nb += 64;
    };
    auto consume = [&](unsigned k) { /* k in 1..=64 */
        hi = (k == 64) ? lo : ((hi << k) | (lo >> (64 - k)));
        lo = (k == 64) ? 0 : (lo << k);
        nb -= int(k);
        budget -= int64_t(k);
    };

    emit(val);
    for (;;) {
        if (nb < 64) topup(); /* >= 64 bits: enough for any single field */
        if (budget <= 0) { bad = true; break; }
        uint32_t top13 = uint32_t(hi >> 51);
        if (!(top13 & 0x1000)) { /* ctrl 0: repeat */
            consume(1);
        } else {
            if (top13 & 0x0800) { /* ctrl 11: new window */
                uint32_t lead = (top13 >> 6) & 0x1f;
                meaningful_n = top13 & 0x3f;
                if (meaningful_n > 0) trailing_n = 64 - lead - meaningful_n;
                else { trailing_n = 0; meaningful_n = 64; }
                consume(13);
            } else { /* ctrl 10: reuse window */
                consume(2);
            }
            while (nb < int(meaningful_n)) topup();
            uint64_t sbits = (meaningful_n == 64) ? hi : (hi >> (64 - meaningful_n));
            consume(meaningful_n);
            if (budget < 0) { bad = true; break; }
            val ^= sbits << trailing_n;
            if (val == GORILLA_SENTINEL) break;
        }
        emit(val); /* single store site per iteration */
    }
    if (bad) { atomicOr(err, DERR_SHORT); return; }
    for (; r < n; r++) {
        if (ALLVALID || dev_bit(bs, r)) { atomicOr(err, DERR_SHORT); break; }
        o[r] = 0.0;
        if (vd) vd[r] = 0;
    }
}

#define GS_RING 16
#define GS_GOR_BLOCK 256 /* 4 waves; RING 8/16/32 x block 128/256 swept: 16/256 optimal */

/* --- Gorilla chunked decode (see DevGorChunk in gs_internal.h) ---
 * k_gor_sync: one-time upload pre-pass.  One thread walks each multi-chunk
 * all-valid Gorilla page with the same window parser as decode (no value
 * stores) and records the parser state every chunk_rows values, so decode
 * can start mid-stream.  Errors are NOT raised here: a truncated or
 * early-sentinel page poisons its unrecorded chunks (bitpos past the
 * stream end), and the decode of that chunk raises DERR_SHORT exactly
 * where the page-sequential kernel would have — same error surface,
 * surfaced at decode/scan time like the reference's decode errors.
 *
 * Budget accounting (all Gorilla kernels here): instead of decrementing a
 * bit budget per value (2 ops/value + 2 checks/value on the round-1
 * parser), the cursor advances freely and correctness is enforced by
 * (a) a load clamp: words at/after p_clamp read as zero, so no access
 *     ever leaves the blob's 48-B tail pad;
 * (b) an overrun flag: once the window would hold only pad (entering
 *     word index >= W+2), the lane stops with DERR_SHORT — bounds every
 *     loop on corrupt input within ~128 pad bits;
 * (c) exact end checks, evaluated once per chunk (at the sentinel, or on
 *     completing a non-final chunk): bits consumed > total_bits means the
 *     parse ran past the real stream -> DERR_SHORT, reproducing the
 *     reference's "unexpected end of block" (float.rs:462) surface. */
__global__ void k_gor_sync(const uint8_t *__restrict__ blob,
                           const DevPage *__restrict__ pages, int npages,
                           const int32_t *__restrict__ chunk_base,
                           DevGorChunk *__restrict__ chunks,
                           uint32_t chunk_rows, int il_stride) {
    for (int pi = blockIdx.x * blockDim.x + threadIdx.x; pi < npages;
         pi += gridDim.x * blockDim.x) {
        DevPage pg = pages[pi];
        int cb = chunk_base[pi];
        int nch = chunk_base[pi + 1] - cb;
        if (nch <= 1) continue; /* single chunk: decode reads the header */
        /* chunk k of page pi lives at cb+k (page-major) or
           k*il_stride+pi (interleaved) */
        auto cidx = [&](int k) {
            return il_stride ? k * il_stride + pi : cb + k;
        };
        const uint8_t *data = blob + pg.data_off;
        int next_k = 1;
        bool ok = false;
        if (pg.data_len >= 10) {
            int64_t total_bits = int64_t(pg.data_len - 10) * 8;
            const uint8_t *stream = data + 10;
            uint64_t W = (uint64_t(total_bits) + 63) >> 6;
            const uint8_t *p_clamp = stream + 8 * W + 24;
            const uint8_t *p_over = stream + 8 * W + 40;
            uint64_t val = dev_be64(data + 2);
            const uint8_t *p = stream;
            uint64_t hi = 0, lo = 0;
            int nb = 0;
            uint32_t trailing = 0, meaningful = 64;
            uint32_t r = 1; /* header value = row 0 */
            bool over = false;
            uint64_t nextw = dev_be64(p);
            uint64_t nextw2 = dev_be64(p + 8);
            p += 16;
            auto topup = [&]() { /* only with nb < 64 */
                uint64_t x = nextw;
                nextw = nextw2;
                over |= (p >= p_over);
                nextw2 = (p < p_clamp) ? dev_be64(p) : 0;
                p += 8;
                if (nb == 0) { hi = x; lo = 0; }
                else { hi |= x >> nb; lo = x << (64 - nb); }
                nb += 64;
            };
            auto consume = [&](unsigned k) { /* k in 1..=64 */
                hi = (k == 64) ? lo : ((hi << k) | (lo >> (64 - k)));
                lo = (k == 64) ? 0 : (lo << k);
                nb -= int(k);
            };
            ok = true;
            for (;;) {
                if (r == uint32_t(next_k) * chunk_rows) {
                    int64_t used = int64_t(p - stream) * 8 - 128 - nb;
                    if (used > total_bits) { ok = false; break; }
                    DevGorChunk &c = chunks[cidx(next_k)];
                    c.bitpos = uint64_t(used);
                    c.val = val;
                    c.trailing = uint8_t(trailing);
                    c.meaningful = uint8_t(meaningful);
                    /* the previous chunk's whole range is now known
                       decodable: the filtered kernel may stop it early */
                    chunks[cidx(next_k - 1)].flags = GORF_SAFE_STOP;
                    next_k++;
                    if (next_k == nch) break; /* tail chunk parses itself */
                }
                if (nb < 64) topup();
                if (over) { ok = false; break; }
                uint32_t top13 = uint32_t(hi >> 51);
                if (!(top13 & 0x1000)) { /* ctrl 0: repeat */
                    consume(1);
                } else {
                    unsigned shift = 2;
                    if (top13 & 0x0800) { /* ctrl 11: new window */
                        uint32_t lead = (top13 >> 6) & 0x1f;
                        meaningful = top13 & 0x3f;
                        if (meaningful > 0) trailing = 64 - lead - meaningful;
                        else { trailing = 0; meaningful = 64; }
                        shift = 13;
                    }
                    unsigned need = shift + meaningful; /* <= 77 */
                    uint64_t sb;
                    if (int(need) <= nb) { /* window holds the whole value */
                        uint64_t w = (hi << shift) | (lo >> (64 - shift));
                        sb = (meaningful == 64) ? w : (w >> (64 - meaningful));
                        if (need <= 64) consume(need);
                        else { consume(shift); consume(meaningful); }
                    } else { /* short window: two-step (topup needs nb<64) */
                        consume(shift);
                        while (nb < int(meaningful)) topup();
                        sb = (meaningful == 64) ? hi : (hi >> (64 - meaningful));
                        consume(meaningful);
                    }
                    val ^= sb << trailing;
                    /* sentinel before the last sync point: rows missing */
                    if (val == GORILLA_SENTINEL) { ok = false; break; }
                }
                r++;
            }
        }
        if (!ok) { /* poison unrecorded chunks -> DERR_SHORT at decode */
            for (int k = next_k; k < nch; k++) {
                DevGorChunk &c = chunks[cidx(k)];
                c.bitpos = uint64_t(pg.data_len) * 8;
                c.val = 0;
                c.trailing = 0;
                c.meaningful = 64;
                c.flags = 0;
            }
        }
    }
}

/* Parser window state of one chunk, positioned at DevGorChunk.bitpos. */
struct GorChunkState {
    const uint8_t *p;       /* next 8-byte word to prefetch */
    const uint8_t *stream;  /* bitstream start (page data + 10) */
    const uint8_t *p_clamp; /* loads at/after here read as zero */
    const uint8_t *p_over;  /* reaching here = stream overrun */
    uint64_t hi, lo, nextw, nextw2;
    uint64_t val;
    int64_t total_bits;
    int nb;
    uint32_t trailing, meaningful;
    bool bad;
};

__device__ __forceinline__ GorChunkState
gor_chunk_init(const uint8_t *__restrict__ blob, const DevGorChunk &c) {
    GorChunkState st;
    st.bad = false;
    const uint8_t *data = blob + c.data_off;
    /* page data: [enc=6][0x10][first f64 BE][bitstream] (float.rs:418-444) */
    if (c.data_len < 10) { st.bad = true; return st; }
    const uint8_t *stream = data + 10;
    st.stream = stream;
    st.total_bits = int64_t(c.data_len - 10) * 8;
    uint64_t W = (uint64_t(st.total_bits) + 63) >> 6;
    st.p_clamp = stream + 8 * W + 24;
    st.p_over = stream + 8 * W + 40;
    if (c.row0 == 0) {
        st.val = dev_be64(data + 2);
        st.trailing = 0;
        st.meaningful = 64;
        st.hi = 0;
        st.lo = 0;
        st.nb = 0;
        st.nextw = dev_be64(stream);
        st.nextw2 = dev_be64(stream + 8);
        st.p = stream + 16;
    } else {
        if (int64_t(c.bitpos) >= st.total_bits) { /* poisoned / truncated */
            st.bad = true;
            return st;
        }
        st.val = c.val;
        st.trailing = c.trailing;
        st.meaningful = c.meaningful;
        const uint8_t *p = stream + (c.bitpos >> 6) * 8;
        unsigned rem = unsigned(c.bitpos & 63);
        uint64_t w0 = dev_be64(p);
        p += 8;
        st.hi = (rem == 0) ? w0 : (w0 << rem);
        st.lo = 0;
        st.nb = int(64 - rem);
        st.nextw = dev_be64(p);
        st.nextw2 = dev_be64(p + 8);
        st.p = p + 16;
    }
    return st;
}

/* Chunk-parallel Gorilla decode with LDS-staged output: each lane decodes
 * one chunk, staging GS_RING values in LDS; the wave flushes cooperatively
 * with chunk-contiguous coalesced stores.  The direct per-lane 8-B stores
 * of the naive version touch 64 distinct cache lines per wave-store and
 * were measured address-bound (profiles/: storeonly 11.8 ms vs lds-staged
 * ~3 ms for the same bytes).  Lanes run in lockstep (one value per
 * iteration), so rings fill together; lanes whose chunk ended keep
 * cooperating in flushes until all are done.  Chunks are page-major in
 * the table, so a wave's lanes hold NEIGHBORING regions of one page. */
__global__ void k_gor_chunks(const uint8_t *__restrict__ blob,
                             const DevGorChunk *__restrict__ chunks,
                             int nchunks, double *__restrict__ out,
                             unsigned *__restrict__ err) {
    __shared__ double ring[GS_GOR_BLOCK / 64][GS_RING][64 + 1];
    __shared__ uint64_t fdesc[GS_GOR_BLOCK / 64][64][2];
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    auto rslot = ring[wv];
    int stride = gridDim.x * blockDim.x;
    int base_id = blockIdx.x * blockDim.x + threadIdx.x;
    int rounds = (nchunks + stride - 1) / stride;
    for (int rd = 0; rd < rounds; rd++) {
        int ci = base_id + rd * stride;
        bool have = ci < nchunks;
        /* branch-free prologue: clamp to chunk 0 so every load issues
           unconditionally (see k_gor_chunks_filtered) */
        DevGorChunk ch = chunks[have ? ci : 0];
        double *o = out + ch.row_off;
        uint32_t r = ch.row0;
        uint32_t end = ch.row0 + ch.cnt;
        if (ch.cnt == 0) have = false; /* interleave-padding entry */
        GorChunkState st = gor_chunk_init(blob, ch);
        int rfill = 0;
        bool done = !have, over = false;
        if (!have) { r = 0; end = 0; }
        if (have && st.bad) { atomicOr(err, DERR_SHORT); done = true; }
        auto topup = [&]() { /* only with nb < 64 */
            uint64_t x = st.nextw;
            st.nextw = st.nextw2;
            over |= (st.p >= st.p_over);
            st.nextw2 = (st.p < st.p_clamp) ? dev_be64(st.p) : 0;
            st.p += 8;
            if (st.nb == 0) { st.hi = x; st.lo = 0; }
            else { st.hi |= x >> st.nb; st.lo = x << (64 - st.nb); }
            st.nb += 64;
        };
        auto consume = [&](unsigned k) { /* k in 1..=64 */
            st.hi = (k == 64) ? st.lo : ((st.hi << k) | (st.lo >> (64 - k)));
            st.lo = (k == 64) ? 0 : (st.lo << k);
            st.nb -= int(k);
        };
        auto used_bits = [&]() {
            return int64_t(st.p - st.stream) * 8 - 128 - st.nb;
        };
        auto fd = fdesc[wv];
        const int f_idx = lane & (GS_RING - 1);
        const int f_sq = lane / GS_RING;
        double *rcur = &rslot[0][lane]; /* strength-reduced ring
                                           cursor (row stride 65*8 B) */
        auto flush = [&]() {
            fd[lane][0] = (uint64_t)(uintptr_t)(o + (int64_t(r) - rfill));
            fd[lane][1] = uint64_t(rfill);
            __builtin_amdgcn_wave_barrier();
            constexpr int SRCP = 64 / GS_RING; /* sources per store */
            for (int src0 = 0; src0 < 64; src0 += SRCP * 4) {
                double vbuf[4];
                uint64_t ob[4];
                int cnt[4];
                for (int t = 0; t < 4; t++) {
                    int src = src0 + f_sq + t * SRCP;
                    vbuf[t] = rslot[f_idx][src];
                    ob[t] = fd[src][0];
                    cnt[t] = int(fd[src][1]);
                }
                for (int t = 0; t < 4; t++)
                    if (f_idx < cnt[t])
                        ((double *)(uintptr_t)ob[t])[f_idx] = vbuf[t];
            }
            rfill = 0;
            rcur = &rslot[0][lane];
        };
        if (!done && ch.row0 == 0 && r < end) { /* header value = row 0 */
            *rcur = __longlong_as_double((long long)st.val);
            rcur += 65;
            rfill++;
            r++;
        }
        unsigned it = 1; /* header stage counts toward the first cadence */
        while (!__all(done)) {
            if (!done) {
                if (st.nb < 64) topup();
                if (over) { atomicOr(err, DERR_SHORT); done = true; }
            }
            if (!done) {
                uint32_t top13 = uint32_t(st.hi >> 51);
                bool stg = true;
                {
                    /* uniform path (no repeat/xor branch): a repeat is an
                       XOR value with 0 meaningful bits (sb=0), so every
                       lane runs the same straight-line code and the wave
                       diverges only on the rare need>64 case — measured
                       the round-1 two-branch parse at ~20% VALU
                       utilization, divergence being the dominant loss */
                    unsigned bit0 = (top13 >> 12) & 1; /* 1 = XOR value */
                    unsigned bit1 = (top13 >> 11) & 1; /* 1 = new window */
                    unsigned nw = bit0 & bit1;
                    uint32_t lead = (top13 >> 6) & 0x1f;
                    uint32_t mg_raw = top13 & 0x3f;
                    uint32_t mg_new = mg_raw ? mg_raw : 64;
                    uint32_t tr_new = mg_raw ? (64 - lead - mg_raw) : 0;
                    st.meaningful = nw ? mg_new : st.meaningful;
                    st.trailing = nw ? tr_new : st.trailing;
                    unsigned shift = bit0 ? (bit1 ? 13u : 2u) : 1u;
                    unsigned m_eff = bit0 ? st.meaningful : 0u;
                    unsigned need = shift + m_eff;
                    uint64_t sb;
                    if (__builtin_expect(need < 64, 1)) {
                        /* nb >= 64 after the loop-top topup; need < 64
                           keeps the consume branch-free (no k==64 case) */
                        uint64_t w =
                            (st.hi << shift) | (st.lo >> (64 - shift));
                        sb = m_eff ? (w >> ((64 - m_eff) & 63)) : 0;
                        st.hi = (st.hi << need) | (st.lo >> (64 - need));
                        st.lo <<= need;
                        st.nb -= int(need);
                    } else { /* m_eff >= 51: rare */
                        consume(shift);
                        while (st.nb < int(m_eff)) topup();
                        sb = (m_eff == 64) ? st.hi
                                           : (st.hi >> (64 - m_eff));
                        consume(m_eff);
                    }
                    st.val ^= sb << st.trailing;
                    if (bit0 && st.val == GORILLA_SENTINEL) {
                        done = true;
                        stg = false;
                        if (used_bits() > st.total_bits)
                            atomicOr(err, DERR_SHORT);
                    }
                }
                if (stg && r < end) {
                    *rcur = __longlong_as_double((long long)st.val);
                    rcur += 65;
                    rfill++;
                    r++;
                    /* non-last chunk: all rows produced, stop; the last
                       chunk keeps parsing to the sentinel like the page-
                       sequential kernel (a missing sentinel is an error) */
                    if (r == end && !ch.last) {
                        done = true;
                        if (used_bits() > st.total_bits)
                            atomicOr(err, DERR_SHORT);
                    }
                }
            }
            /* uniform cadence: <=1 staged/iteration, so a flush every
               GS_RING iterations can never overflow the ring — no ballot,
               no divergent branch */
            if ((++it & (GS_RING - 1)) == 0) flush();
        }
        flush();
        if (have && r < end) atomicOr(err, DERR_SHORT);
    }
}

/* --- Gorilla chunked decode for NULL-carrying pages (PC_GORN) ---
 * The encoded stream holds only non-null values, scattered to set bits of
 * the validity bitset (tsm/reader.rs:763-825 / float.rs decode via
 * bitset-driven builder).  The parser state is pipelined ONE VALUE AHEAD:
 * `val` always holds the value pending for the next set-bit row, so a
 * chunk boundary at an arbitrary row needs only (bitpos, pending val,
 * window).  The page header's first f64 is the initial pending value.
 * Chunks are row-ranges; null rows stage 0.0 through the same LDS ring
 * (arrow append_null semantics); validity BYTES are produced separately
 * by k_valid_expand (coalesced), not by this kernel. */
__global__ void k_gor_sync_null(const uint8_t *__restrict__ blob,
                                const DevPage *__restrict__ pages, int npages,
                                const int32_t *__restrict__ chunk_base,
                                DevGorChunk *__restrict__ chunks,
                                uint32_t chunk_rows) {
    for (int pi = blockIdx.x * blockDim.x + threadIdx.x; pi < npages;
         pi += gridDim.x * blockDim.x) {
        DevPage pg = pages[pi];
        int cb = chunk_base[pi];
        int nch = chunk_base[pi + 1] - cb;
        if (nch <= 1) continue;
        const uint8_t *data = blob + pg.data_off;
        const uint8_t *bs = blob + pg.bitset_off;
        int next_k = 1;
        bool ok = false;
        bool sent_seen = false;
        if (pg.data_len >= 10) {
            int64_t total_bits = int64_t(pg.data_len - 10) * 8;
            const uint8_t *stream = data + 10;
            uint64_t W = (uint64_t(total_bits) + 63) >> 6;
            const uint8_t *p_clamp = stream + 8 * W + 24;
            const uint8_t *p_over = stream + 8 * W + 40;
            uint64_t val = dev_be64(data + 2); /* pending for 1st set row */
            const uint8_t *p = stream;
            uint64_t hi = 0, lo = 0;
            int nb = 0;
            uint32_t trailing = 0, meaningful = 64;
            uint32_t r = 0; /* ROW cursor */
            bool over = false;
            uint64_t nextw = dev_be64(p);
            uint64_t nextw2 = dev_be64(p + 8);
            p += 16;
            auto topup = [&]() {
                uint64_t x = nextw;
                nextw = nextw2;
                over |= (p >= p_over);
                nextw2 = (p < p_clamp) ? dev_be64(p) : 0;
                p += 8;
                if (nb == 0) { hi = x; lo = 0; }
                else { hi |= x >> nb; lo = x << (64 - nb); }
                nb += 64;
            };
            auto consume = [&](unsigned k) {
                hi = (k == 64) ? lo : ((hi << k) | (lo >> (64 - k)));
                lo = (k == 64) ? 0 : (lo << k);
                nb -= int(k);
            };
            ok = true;
            for (;;) {
                if (r == uint32_t(next_k) * chunk_rows) {
                    int64_t used = int64_t(p - stream) * 8 - 128 - nb;
                    if (!sent_seen && used > total_bits) { ok = false; break; }
                    DevGorChunk &c = chunks[cb + next_k];
                    c.bitpos = uint64_t(used);
                    c.val = val;
                    c.trailing = uint8_t(trailing);
                    c.meaningful = uint8_t(meaningful);
                    c.flags = sent_seen ? GORF_SENT_SEEN : 0;
                    next_k++;
                    if (next_k == nch) break;
                    if (sent_seen) continue; /* tail: just mark chunks */
                }
                if (sent_seen) { r++; continue; } /* walk rows to boundary */
                if (nb < 64) topup();
                if (over) { ok = false; break; }
                if (dev_bit(bs, r)) {
                    /* this set row consumes the pending value: parse the
                       next one (uniform path, see k_gor_chunks) */
                    uint32_t top13 = uint32_t(hi >> 51);
                    unsigned bit0 = (top13 >> 12) & 1;
                    unsigned bit1 = (top13 >> 11) & 1;
                    unsigned nw = bit0 & bit1;
                    uint32_t lead = (top13 >> 6) & 0x1f;
                    uint32_t mg_raw = top13 & 0x3f;
                    uint32_t mg_new = mg_raw ? mg_raw : 64;
                    uint32_t tr_new = mg_raw ? (64 - lead - mg_raw) : 0;
                    meaningful = nw ? mg_new : meaningful;
                    trailing = nw ? tr_new : trailing;
                    unsigned shift = bit0 ? (bit1 ? 13u : 2u) : 1u;
                    unsigned m_eff = bit0 ? meaningful : 0u;
                    unsigned need = shift + m_eff;
                    uint64_t sb;
                    if (need <= 64) {
                        uint64_t w = (hi << shift) | (lo >> (64 - shift));
                        sb = m_eff ? (w >> ((64 - m_eff) & 63)) : 0;
                        consume(need);
                    } else {
                        consume(shift);
                        while (nb < int(m_eff)) topup();
                        sb = (m_eff == 64) ? hi : (hi >> (64 - m_eff));
                        consume(m_eff);
                    }
                    val ^= sb << trailing;
                    if (bit0 && val == GORILLA_SENTINEL) {
                        int64_t used = int64_t(p - stream) * 8 - 128 - nb;
                        if (used > total_bits) { ok = false; break; }
                        sent_seen = true;
                    }
                }
                r++;
            }
        }
        if (!ok) { /* poison unrecorded chunks -> DERR_SHORT at decode */
            for (int k = next_k; k < nch; k++) {
                DevGorChunk &c = chunks[cb + k];
                c.bitpos = uint64_t(pg.data_len) * 8;
                c.val = 0;
                c.trailing = 0;
                c.meaningful = 64;
                c.flags = 0;
            }
        }
    }
}

/* Chunk-parallel decode of null-carrying Gorilla pages: per iteration one
 * ROW is staged (parsed pending value for set bits, 0.0 for nulls) and,
 * when the row consumed a value, the next value is parsed ahead.  Same
 * LDS ring/flush as k_gor_chunks. */
__global__ void k_gor_chunks_null(const uint8_t *__restrict__ blob,
                                  const DevGorChunk *__restrict__ chunks,
                                  int nchunks, double *__restrict__ out,
                                  unsigned *__restrict__ err) {
    __shared__ double ring[GS_GOR_BLOCK / 64][GS_RING][64 + 1];
    __shared__ uint64_t fdesc[GS_GOR_BLOCK / 64][64][2];
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    auto rslot = ring[wv];
    int stride = gridDim.x * blockDim.x;
    int base_id = blockIdx.x * blockDim.x + threadIdx.x;
    int rounds = (nchunks + stride - 1) / stride;
    for (int rd = 0; rd < rounds; rd++) {
        int ci = base_id + rd * stride;
        bool have = ci < nchunks;
        DevGorChunk ch = chunks[have ? ci : 0];
        const uint8_t *bs = blob + ch.bitset_off;
        double *o = out + ch.row_off;
        uint32_t r = ch.row0;
        /* per-8-row bitset byte, prefetched one ahead like the stream
           refills (measured neutral: the kernel is ~72% memory-parked
           with or without it — the stall is elsewhere, see DESIGN.md
           round-3 notes).  Byte loads are always aligned (the 8-B word
           variant misread flakily at high blob offsets); the +1 read
           tops out one byte past the bitset into the page's own data. */
        uint32_t bb_idx = ch.row0 >> 3;
        uint32_t bb = bs[bb_idx];
        uint32_t bb_next = bs[bb_idx + 1];
        uint32_t end = ch.row0 + ch.cnt;
        bool sent_seen = (ch.flags & GORF_SENT_SEEN) != 0;
        GorChunkState st;
        st.bad = false;
        if (sent_seen) { /* tail chunk after the sentinel: nulls only */
            st.p = st.stream = st.p_clamp = st.p_over = nullptr;
            st.hi = st.lo = st.nextw = st.nextw2 = st.val = 0;
            st.total_bits = 0;
            st.nb = 0;
            st.trailing = 0;
            st.meaningful = 64;
        } else {
            st = gor_chunk_init(blob, ch);
        }
        int rfill = 0;
        bool done = !have, over = false;
        if (!have) { r = 0; end = 0; }
        if (have && st.bad) { atomicOr(err, DERR_SHORT); done = true; }
        auto topup = [&]() {
            uint64_t x = st.nextw;
            st.nextw = st.nextw2;
            over |= (st.p >= st.p_over);
            st.nextw2 = (st.p < st.p_clamp) ? dev_be64(st.p) : 0;
            st.p += 8;
            if (st.nb == 0) { st.hi = x; st.lo = 0; }
            else { st.hi |= x >> st.nb; st.lo = x << (64 - st.nb); }
            st.nb += 64;
        };
        auto consume = [&](unsigned k) {
            st.hi = (k == 64) ? st.lo : ((st.hi << k) | (st.lo >> (64 - k)));
            st.lo = (k == 64) ? 0 : (st.lo << k);
            st.nb -= int(k);
        };
        auto used_bits = [&]() {
            return int64_t(st.p - st.stream) * 8 - 128 - st.nb;
        };
        auto parse_one = [&]() { /* uniform path; updates st / sent_seen */
            uint32_t top13 = uint32_t(st.hi >> 51);
            unsigned bit0 = (top13 >> 12) & 1;
            unsigned bit1 = (top13 >> 11) & 1;
            unsigned nw = bit0 & bit1;
            uint32_t lead = (top13 >> 6) & 0x1f;
            uint32_t mg_raw = top13 & 0x3f;
            uint32_t mg_new = mg_raw ? mg_raw : 64;
            uint32_t tr_new = mg_raw ? (64 - lead - mg_raw) : 0;
            st.meaningful = nw ? mg_new : st.meaningful;
            st.trailing = nw ? tr_new : st.trailing;
            unsigned shift = bit0 ? (bit1 ? 13u : 2u) : 1u;
            unsigned m_eff = bit0 ? st.meaningful : 0u;
            unsigned need = shift + m_eff;
            uint64_t sb;
            if (__builtin_expect(need < 64, 1)) {
                uint64_t w = (st.hi << shift) | (st.lo >> (64 - shift));
                sb = m_eff ? (w >> ((64 - m_eff) & 63)) : 0;
                st.hi = (st.hi << need) | (st.lo >> (64 - need));
                st.lo <<= need;
                st.nb -= int(need);
            } else {
                consume(shift);
                while (st.nb < int(m_eff)) topup();
                sb = (m_eff == 64) ? st.hi : (st.hi >> (64 - m_eff));
                consume(m_eff);
            }
            st.val ^= sb << st.trailing;
            if (bit0 && st.val == GORILLA_SENTINEL) {
                if (used_bits() > st.total_bits) atomicOr(err, DERR_SHORT);
                sent_seen = true;
            }
        };
        auto fd = fdesc[wv];
        const int f_idx = lane & (GS_RING - 1);
        const int f_sq = lane / GS_RING;
        auto flush = [&]() {
            fd[lane][0] = (uint64_t)(uintptr_t)(o + (int64_t(r) - rfill));
            fd[lane][1] = uint64_t(rfill);
            __builtin_amdgcn_wave_barrier();
            constexpr int SRCP = 64 / GS_RING;
            for (int src0 = 0; src0 < 64; src0 += SRCP * 4) {
                double vbuf[4];
                uint64_t ob[4];
                int cnt[4];
                for (int t = 0; t < 4; t++) {
                    int src = src0 + f_sq + t * SRCP;
                    vbuf[t] = rslot[f_idx][src];
                    ob[t] = fd[src][0];
                    cnt[t] = int(fd[src][1]);
                }
                for (int t = 0; t < 4; t++)
                    if (f_idx < cnt[t])
                        ((double *)(uintptr_t)ob[t])[f_idx] = vbuf[t];
            }
            rfill = 0;
        };
        unsigned it = 0;
        while (!__all(done)) {
            if (!done && !sent_seen) {
                if (st.nb < 64) topup();
                if (over) { atomicOr(err, DERR_SHORT); done = true; }
            }
            if (!done) {
                if (r < end) {
                    const uint32_t bi = r >> 3;
                    if (bi != bb_idx) {
                        bb = bb_next;
                        bb_idx = bi;
                        bb_next = bs[bi + 1];
                    }
                    const int bit = int((bb >> (r & 7)) & 1);
                    uint64_t sbits = 0;
                    bool hole = false; /* set row after the sentinel */
                    if (bit) {
                        if (sent_seen) {
                            atomicOr(err, DERR_SHORT);
                            done = true;
                            hole = true;
                        } else {
                            sbits = st.val;
                        }
                    }
                    if (!hole) {
                        rslot[rfill][lane] =
                            __longlong_as_double((long long)sbits);
                        rfill++;
                        r++;
                        if (bit && !sent_seen) parse_one();
                        if (r == end) {
                            if (!ch.last) {
                                done = true;
                                if (!sent_seen &&
                                    used_bits() > st.total_bits)
                                    atomicOr(err, DERR_SHORT);
                            }
                            /* last chunk: keep draining to the sentinel */
                        }
                    }
                } else { /* last chunk drain: parse to the sentinel */
                    if (sent_seen) done = true;
                    else parse_one();
                }
            }
            if ((++it & (GS_RING - 1)) == 0) flush();
        }
        flush();
        if (have && r < end) atomicOr(err, DERR_SHORT);
    }
}

/* validity bytes, decoupled from the decode kernels (coalesced):
 * k_valid_fill: vd=1 over all-valid pages; k_valid_expand: vd=bitset. */
__global__ void k_valid_fill(const DevPage *__restrict__ pages, int npages,
                             uint8_t *__restrict__ valid) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        uint8_t *vd = valid + pg.row_off;
        for (uint32_t r = threadIdx.x; r < pg.nrows; r += blockDim.x)
            vd[r] = 1;
    }
}

__global__ void k_valid_expand(const uint8_t *__restrict__ blob,
                               const DevPage *__restrict__ pages, int npages,
                               uint8_t *__restrict__ valid) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *bs = blob + pg.bitset_off;
        uint8_t *vd = valid + pg.row_off;
        const uint32_t nfull = pg.nrows >> 3; /* whole bytes -> u64 writes */
        for (uint32_t b = threadIdx.x; b < nfull; b += blockDim.x) {
            uint64_t ex = 0;
            uint8_t byte = bs[b];
            for (int k = 0; k < 8; k++)
                ex |= uint64_t((byte >> k) & 1) << (8 * k);
            __builtin_memcpy(vd + (size_t(b) << 3), &ex, 8);
        }
        for (uint32_t r = nfull * 8 + threadIdx.x; r < pg.nrows;
             r += blockDim.x)
            vd[r] = uint8_t(dev_bit(bs, r));
    }
}

__global__ void k_seq_f64(const uint8_t *__restrict__ blob,
                          const DevPage *__restrict__ pages, int npages,
                          double *__restrict__ out,
                          uint8_t *__restrict__ valid,
                          unsigned *__restrict__ err) {
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        DevPage pg = pages[p];
        const uint8_t *data = blob + pg.data_off;
        const uint8_t *bs = blob + pg.bitset_off;
        double *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;

        if (pg.data_len == 0) {
            for (uint32_t r = 0; r < n; r++) { o[r] = 0.0; if (vd) vd[r] = 0; }
            continue;
        }
        uint8_t enc = pg.enc;
        if (enc == GS_ENC_NULL) {
            const uint8_t *q = data + 1;
            uint32_t avail = (pg.data_len - 1) / 8, used = 0;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (v && used < avail) {
                    uint64_t u = dev_be64(q + 8ull * used); used++;
                    o[r] = __longlong_as_double(int64_t(u));
                } else { o[r] = 0.0; v = 0; }
                if (vd) vd[r] = uint8_t(v);
            }
            continue;
        }
        if (enc != GS_ENC_GORILLA) { atomicOr(err, DERR_FORMAT); continue; }
        if (pg.all_valid)
            gorilla_decode_page<true>(data, pg.data_len, bs, n, o, vd, err);
        else
            gorilla_decode_page<false>(data, pg.data_len, bs, n, o, vd, err);
    }
}

/* bool pages: one thread per page */
__global__ void k_seq_bool(const uint8_t *__restrict__ blob,
                           const DevPage *__restrict__ pages, int npages,
                           uint8_t *__restrict__ out,
                           uint8_t *__restrict__ valid,
                           unsigned *__restrict__ err) {
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        DevPage pg = pages[p];
        const uint8_t *data = blob + pg.data_off;
        const uint8_t *bs = blob + pg.bitset_off;
        uint8_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;
        if (pg.data_len == 0) {
            for (uint32_t r = 0; r < n; r++) { o[r] = 0; if (vd) vd[r] = 0; }
            continue;
        }
        if (pg.enc == GS_ENC_NULL) { /* boolean.rs:111-136 */
            const uint8_t *q = data + 1;
            uint32_t avail = pg.data_len - 1, used = 0;
            for (uint32_t r = 0; r < n; r++) {
                int v = pg.all_valid ? 1 : dev_bit(bs, r);
                if (v && used < avail) { o[r] = (q[used] == 1); used++; }
                else { o[r] = 0; v = 0; }
                if (vd) vd[r] = uint8_t(v);
            }
            continue;
        }
        if (pg.enc != GS_ENC_BITPACK || pg.data_len < 2 ||
            data[1] != (1 << 4)) { atomicOr(err, DERR_FORMAT); continue; }
        uint64_t count; uint32_t nr;
        if (!dev_varint(data + 2, pg.data_len - 2, &count, &nr)) {
            atomicOr(err, DERR_FORMAT); continue;
        }
        const uint8_t *bits = data + 2 + nr;
        uint64_t bi = 0;
        for (uint32_t r = 0; r < n; r++) {
            int v = pg.all_valid ? 1 : dev_bit(bs, r);
            if (!v) { o[r] = 0; if (vd) vd[r] = 0; continue; }
            if (bi >= count) { atomicOr(err, DERR_SHORT); o[r] = 0; if (vd) vd[r] = 0; continue; }
            o[r] = (bits[bi >> 3] >> (7 - (bi & 7))) & 1;
            bi++;
            if (vd) vd[r] = 1;
        }
    }
}

/* Closed-form parallel RLE decode (all-valid pages): value[r] =
 * first + r*delta — one workgroup per page, coalesced 8-B stores.
 * (timestamp.rs:226-259 / integer.rs:186-214 semantics.) */
__global__ void k_rle_par(const uint8_t *__restrict__ blob,
                          const DevPage *__restrict__ pages, int npages,
                          int64_t *__restrict__ out,
                          uint8_t *__restrict__ valid, int is_ts,
                          unsigned *__restrict__ err) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *s = blob + pg.data_off + 1;
        uint32_t plen = pg.data_len - 2;
        /* every thread parses the tiny header redundantly (no sync needed) */
        uint64_t scaler = 1;
        if (is_ts) {
            unsigned s10 = s[0] & 0x0f;
            for (unsigned k = 0; k < s10; k++) scaler *= 10;
        }
        const uint8_t *q = s + 1;
        if (plen < 9) { if (threadIdx.x == 0) atomicOr(err, DERR_FORMAT); continue; }
        uint64_t first_raw = dev_be64(q);
        uint64_t dv; uint32_t nr;
        if (!dev_varint(q + 8, plen - 8, &dv, &nr)) {
            if (threadIdx.x == 0) atomicOr(err, DERR_FORMAT);
            continue;
        }
        int64_t first, delta;
        if (is_ts) { first = int64_t(first_raw); delta = int64_t(dv * scaler); }
        else { first = dev_zzdec(first_raw); delta = dev_zzdec(dv); }
        if (is_ts && delta < 0 && pg.nrows > 1 && threadIdx.x == 0)
            atomicOr(err, DERR_NONMONO);
        int64_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        for (uint32_t r = threadIdx.x; r < pg.nrows; r += blockDim.x) {
            o[r] = int64_t(uint64_t(first) + uint64_t(r) * uint64_t(delta));
            if (vd) vd[r] = 1;
        }
    }
}

/* Block-parallel simple8b delta decode (all-valid DeltaTs/Delta pages,
 * sub-tag 1; timestamp.rs:261-299 / integer.rs:216-248): one workgroup
 * per page.  Words are fixed 8-byte units, so the block scans per-word
 * (value count, delta sum) pairs with a carry, then each thread unpacks
 * its word directly to the right output positions — the delta prefix-sum
 * that is sequential in the reference becomes a block scan. */
__global__ void k_s8b_par(const uint8_t *__restrict__ blob,
                          const DevPage *__restrict__ pages, int npages,
                          int64_t *__restrict__ out,
                          uint8_t *__restrict__ valid,
                          unsigned *__restrict__ err) {
    __shared__ uint32_t scnt[256];
    __shared__ uint64_t ssum[256];
    __shared__ uint64_t carry[2]; /* [0]=value offset, [1]=delta prefix */
    const uint8_t S8B_COUNT[16] = {240, 120, 60, 30, 20, 15, 12, 10,
                                   8, 7, 6, 5, 4, 3, 2, 1};
    const uint8_t S8B_WIDTH[16] = {0, 0, 1, 2, 3, 4, 5, 6,
                                   7, 8, 10, 12, 15, 20, 30, 60};
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *s = blob + pg.data_off + 1; /* sub-tag byte */
        bool is_ts = pg.enc == GS_ENC_DELTATS;
        uint64_t scaler = 1;
        if (is_ts) {
            unsigned s10 = s[0] & 0x0f;
            for (unsigned k = 0; k < s10; k++) scaler *= 10;
        }
        const uint8_t *q = s + 1;
        uint32_t plen = pg.data_len - 2;
        if (plen < 8) { if (threadIdx.x == 0) atomicOr(err, DERR_SHORT); continue; }
        uint64_t first_raw = dev_be64(q);
        int64_t first = is_ts ? int64_t(first_raw) : dev_zzdec(first_raw);
        const uint8_t *words = q + 8;
        uint32_t nwords = (plen - 8) / 8;
        int64_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;
        if (threadIdx.x == 0) {
            if (n > 0) o[0] = first;
            carry[0] = 1; /* first value occupies row 0 */
            carry[1] = 0;
        }
        if (vd)
            for (uint32_t r = threadIdx.x; r < n; r += blockDim.x) vd[r] = 1;
        __syncthreads();
        for (uint32_t tile = 0; tile < nwords; tile += blockDim.x) {
            uint32_t wi = tile + threadIdx.x;
            uint64_t w = 0;
            uint32_t cnt = 0;
            uint64_t dsum = 0;
            unsigned sel = 0, bits = 0;
            uint64_t mask = 0;
            if (wi < nwords) {
                w = dev_be64(words + 8ull * wi);
                sel = unsigned(w >> 60);
                cnt = S8B_COUNT[sel];
                bits = S8B_WIDTH[sel];
                if (sel <= 1) {
                    /* runs of "1": delta sum = cnt (x scaler for ts,
                       zigzag-decoded 1 -> -1 for i64) */
                    dsum = is_ts ? uint64_t(cnt) * scaler
                                 : uint64_t(int64_t(-1)) * cnt;
                } else {
                    mask = (bits == 60) ? 0x0fffffffffffffffULL
                                        : ((1ULL << bits) - 1);
                    uint64_t v = w;
                    for (uint32_t k = 0; k < cnt; k++) {
                        uint64_t d = v & mask;
                        dsum += is_ts ? d * scaler : uint64_t(dev_zzdec(d));
                        v >>= bits;
                    }
                }
            }
            /* block exclusive scan of (cnt, dsum) */
            scnt[threadIdx.x] = cnt;
            ssum[threadIdx.x] = dsum;
            __syncthreads();
            for (int off = 1; off < int(blockDim.x); off <<= 1) {
                uint32_t c2 = threadIdx.x >= unsigned(off)
                                  ? scnt[threadIdx.x - off] : 0;
                uint64_t s2 = threadIdx.x >= unsigned(off)
                                  ? ssum[threadIdx.x - off] : 0;
                __syncthreads();
                scnt[threadIdx.x] += c2;
                ssum[threadIdx.x] += s2;
                __syncthreads();
            }
            uint64_t voff = carry[0] +
                            (threadIdx.x ? scnt[threadIdx.x - 1] : 0);
            uint64_t dpre = carry[1] +
                            (threadIdx.x ? ssum[threadIdx.x - 1] : 0);
            if (wi < nwords && cnt) {
                uint64_t acc = uint64_t(first) + dpre;
                if (sel <= 1) {
                    uint64_t step = is_ts ? scaler : uint64_t(int64_t(-1));
                    for (uint32_t k = 0; k < cnt && voff + k < n; k++) {
                        acc += step;
                        o[voff + k] = int64_t(acc);
                    }
                } else {
                    uint64_t v = w;
                    int64_t pr = int64_t(acc);
                    for (uint32_t k = 0; k < cnt; k++) {
                        uint64_t d = v & mask;
                        acc += is_ts ? d * scaler : uint64_t(dev_zzdec(d));
                        /* ts: i64 wrap of the delta prefix = corrupt page
                           (times must not decrease) */
                        if (is_ts && int64_t(acc) < pr)
                            atomicOr(err, DERR_NONMONO);
                        pr = int64_t(acc);
                        if (voff + k < n) o[voff + k] = int64_t(acc);
                        v >>= bits;
                    }
                }
            }
            __syncthreads();
            if (threadIdx.x == 0) {
                carry[0] += scnt[blockDim.x - 1];
                carry[1] += ssum[blockDim.x - 1];
            }
            __syncthreads();
            if (carry[0] >= n) break; /* enough values for every row */
        }
        __syncthreads();
        /* iterator exhaustion: rows past the decoded count are 0
           (k_seq_i64 / Int64Builder skip semantics) */
        uint64_t total = carry[0];
        for (uint64_t r = total + threadIdx.x; r < n; r += blockDim.x)
            o[r] = 0;
        __syncthreads();
    }
}

/* Null-encoded (raw BE) all-valid pages: one workgroup per page, fully
 * coalesced byteswap copy (ts_without_compress_decode_to_array,
 * timestamp.rs:301-323; uncompressed ts carries NO deltas under Null).
 * Works for i64/u64/f64 alike (bit pattern passthrough). */
__global__ void k_raw_par(const uint8_t *__restrict__ blob,
                          const DevPage *__restrict__ pages, int npages,
                          int64_t *__restrict__ out,
                          uint8_t *__restrict__ valid,
                          unsigned *__restrict__ err) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *q = blob + pg.data_off + 1;
        uint32_t avail = (pg.data_len - 1) / 8;
        int64_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;
        const bool ck_ts = pg.ctype == GS_CT_TIME;
        for (uint32_t r = threadIdx.x; r < n; r += blockDim.x) {
            int64_t x = (r < avail) ? int64_t(dev_be64(q + 8ull * r)) : 0;
            /* absolute (Null-encoded) time pages may be unsorted: the
               previous input word is an L1 hit */
            if (ck_ts && r > 0 && r < avail &&
                x < int64_t(dev_be64(q + 8ull * (r - 1))))
                atomicOr(err, DERR_NONMONO);
            o[r] = x;
            if (vd) vd[r] = 1;
        }
    }
}

/* BitPack bool pages, all-valid: parallel bit extract (boolean.rs:79-110;
 * MSB-first bits after the varint count) */
__global__ void k_bool_par(const uint8_t *__restrict__ blob,
                           const DevPage *__restrict__ pages, int npages,
                           uint8_t *__restrict__ out,
                           uint8_t *__restrict__ valid,
                           unsigned *__restrict__ err) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *data = blob + pg.data_off;
        uint64_t count;
        uint32_t nr;
        if (pg.data_len < 3 || data[1] != (1 << 4) ||
            !dev_varint(data + 2, pg.data_len - 2, &count, &nr)) {
            if (threadIdx.x == 0) atomicOr(err, DERR_FORMAT);
            continue;
        }
        const uint8_t *bits = data + 2 + nr;
        uint8_t *o = out + pg.row_off;
        uint8_t *vd = valid ? valid + pg.row_off : nullptr;
        uint32_t n = pg.nrows;
        if (count < n && threadIdx.x == 0) atomicOr(err, DERR_SHORT);
        for (uint32_t r = threadIdx.x; r < n; r += blockDim.x) {
            o[r] = (r < count) ? ((bits[r >> 3] >> (7 - (r & 7))) & 1) : 0;
            if (vd) vd[r] = 1;
        }
    }
}

/* per-group closed-interval span on the sorted decoded ts
 * (TimeRange semantics, domain.rs:36-44) */
__global__ void k_spans(const DevGroup *__restrict__ groups, int n,
                        const int64_t *__restrict__ ts, int64_t mn, int64_t mx,
                        int64_t *__restrict__ sp_start,
                        int64_t *__restrict__ sp_cnt) {
    for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < n;
         g += gridDim.x * blockDim.x) {
        const int64_t *t = ts + groups[g].row_off;
        int64_t nr = groups[g].nrows;
        int64_t lo = 0, hi = nr;
        while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < mn) lo = m + 1; else hi = m; }
        int64_t s = lo;
        lo = 0; hi = nr;
        while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] <= mx) lo = m + 1; else hi = m; }
        sp_start[g] = s;
        sp_cnt[g] = lo - s;
    }
}

/* tombstone masking (tsm/reader.rs:634-656): clear validity for rows whose
 * ts lies in a deleted closed range.  Block per group. */
__global__ void k_tombstone(const DevGroup *__restrict__ groups, int n,
                            const int64_t *__restrict__ ts,
                            uint8_t *__restrict__ valid,
                            const GsTimeRange *__restrict__ ranges,
                            int nranges) {
    for (int g = blockIdx.x; g < n; g += gridDim.x) {
        const int64_t *t = ts + groups[g].row_off;
        uint8_t *vd = valid + groups[g].row_off;
        int64_t nr = groups[g].nrows;
        for (int q = 0; q < nranges; q++) {
            int64_t mn = ranges[q].min_ts, mx = ranges[q].max_ts;
            int64_t lo = 0, hi = nr;
            while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < mn) lo = m + 1; else hi = m; }
            int64_t s = lo;
            lo = 0; hi = nr;
            while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < mx) lo = m + 1; else hi = m; }
            int64_t e = (lo < nr && t[lo] == mx) ? lo + 1 : lo;
            for (int64_t r = s + threadIdx.x; r < e; r += blockDim.x) vd[r] = 0;
        }
    }
}

/* value-predicate mask over the selected span (DataFilter semantics,
 * reader/filter.rs:91-142): mask = validity AND pred(value); per-group
 * selected count by block reduction */
template <typename T>
__device__ __forceinline__ bool dev_pred_t(int op, T a, T b, T x) {
    switch (op) {
    case GS_PRED_GT: return x > a;
    case GS_PRED_GE: return x >= a;
    case GS_PRED_LT: return x < a;
    case GS_PRED_LE: return x <= a;
    case GS_PRED_EQ: return x == a;
    case GS_PRED_NE: return x != a;
    case GS_PRED_BETWEEN: return x >= a && x <= b;
    default: return true;
    }
}

/* vt: 0 = f64, 1 = i64, 2 = u64 (bit-cast slot, unsigned.rs:20-45) — the
 * reference's DataFilter evaluates the pushed expr in the column's own
 * type (reader/filter.rs:91-142) */
__device__ __forceinline__ bool dev_pred(int op, double a, double b,
                                         double x, int vt) {
    if (vt == 1)
        return dev_pred_t<int64_t>(op, int64_t(a), int64_t(b),
                                   __double_as_longlong(x));
    if (vt == 2)
        return dev_pred_t<uint64_t>(op, uint64_t(a), uint64_t(b),
                                    uint64_t(__double_as_longlong(x)));
    return dev_pred_t<double>(op, a, b, x);
}

__global__ void k_vmask(const DevGroup *__restrict__ groups, int n,
                        const double *__restrict__ val,
                        const uint8_t *__restrict__ valid,
                        const int64_t *__restrict__ sp_start,
                        const int64_t *__restrict__ sp_cnt, int op, double a,
                        double b, int vt, uint8_t *__restrict__ mask,
                        int64_t *__restrict__ sel_cnt) {
    __shared__ long long sred[256];
    for (int g = blockIdx.x; g < n; g += gridDim.x) {
        int64_t base = groups[g].row_off + sp_start[g];
        int64_t cnt = sp_cnt[g];
        long long c = 0;
        for (int64_t r = threadIdx.x; r < cnt; r += blockDim.x) {
            bool ok = (!valid || valid[base + r]) &&
                      dev_pred(op, a, b, val[base + r], vt);
            mask[base + r] = ok;
            c += ok;
        }
        sred[threadIdx.x] = c;
        __syncthreads();
        for (int w = blockDim.x >> 1; w > 0; w >>= 1) {
            if (threadIdx.x < unsigned(w)) sred[threadIdx.x] += sred[threadIdx.x + w];
            __syncthreads();
        }
        if (threadIdx.x == 0) sel_cnt[g] = sred[0];
        __syncthreads();
    }
}

/* masked compaction: block-scan of the mask within the span, gather the
 * selected rows (filter_record_batch's mask-then-gather,
 * reader/filter.rs:130-142) */
__global__ void k_compact_masked(const DevGroup *__restrict__ groups, int n,
                                 const int64_t *__restrict__ ts,
                                 const double *__restrict__ val,
                                 const uint8_t *__restrict__ mask,
                                 const int64_t *__restrict__ sp_start,
                                 const int64_t *__restrict__ sp_cnt,
                                 const int64_t *__restrict__ out_off,
                                 int64_t *__restrict__ out_ts,
                                 double *__restrict__ out_val) {
    __shared__ int32_t sh[256];
    __shared__ int64_t carry;
    for (int g = blockIdx.x; g < n; g += gridDim.x) {
        int64_t base = groups[g].row_off + sp_start[g];
        int64_t cnt = sp_cnt[g];
        int64_t dst = out_off[g];
        if (threadIdx.x == 0) carry = 0;
        __syncthreads();
        for (int64_t tile = 0; tile < cnt; tile += blockDim.x) {
            int64_t r = tile + threadIdx.x;
            int32_t m = (r < cnt) ? mask[base + r] : 0;
            sh[threadIdx.x] = m;
            __syncthreads();
            for (int off = 1; off < int(blockDim.x); off <<= 1) {
                int32_t u = threadIdx.x >= unsigned(off)
                                ? sh[threadIdx.x - off] : 0;
                __syncthreads();
                sh[threadIdx.x] += u;
                __syncthreads();
            }
            if (m) {
                int64_t pos = dst + carry + sh[threadIdx.x] - 1;
                out_ts[pos] = ts[base + r];
                out_val[pos] = val[base + r];
            }
            __syncthreads();
            if (threadIdx.x == 0) carry += sh[blockDim.x - 1];
            __syncthreads();
        }
    }
}

/* compacting copy of the selected span (filter_record_batch semantics,
 * reader/filter.rs:130-142: row filter on time; field nulls travel) */
__global__ void k_compact(const DevGroup *__restrict__ groups, int n,
                          const int64_t *__restrict__ ts,
                          const double *__restrict__ val,
                          const int64_t *__restrict__ sp_start,
                          const int64_t *__restrict__ sp_cnt,
                          const int64_t *__restrict__ out_off,
                          int64_t *__restrict__ out_ts,
                          double *__restrict__ out_val) {
    for (int g = blockIdx.x; g < n; g += gridDim.x) {
        int64_t base = groups[g].row_off + sp_start[g];
        int64_t cnt = sp_cnt[g];
        int64_t dst = out_off[g];
        for (int64_t r = threadIdx.x; r < cnt; r += blockDim.x) {
            out_ts[dst + r] = ts[base + r];
            out_val[dst + r] = val[base + r];
        }
    }
}

/* Fused per-bucket max/sum/count over the selected span, two phases with
 * NO atomics and a deterministic (fixed-tree) float reduction order —
 * BASELINE config #3 requires max exact and sum within stated tolerance,
 * and the reference's DataFusion hash-agg is sequential per partition.
 *
 * Phase 1 (k_agg_partial): block per AGGREGATION GROUP (consecutive
 * page-groups of one series merged at upload — their rows are contiguous
 * and time-ordered, tsm/column_group.rs:55-63).  The decoded ts of a
 * group is sorted, so each bucket's rows form a contiguous range found by
 * binary search (clipped to the scan's closed time range); the block's
 * waves split the bucket list, each wave reduces its bucket's rows with a
 * shuffle tree and writes the (group,bucket) partial.  Every cell is
 * written exactly once — no init needed, no atomics anywhere.
 *
 * Phase 2 (k_agg_merge): one wave per bucket strides the groups, reducing
 * partials in a fixed lane order, and accumulates into the caller's
 * global bucket arrays (sequential across sub-batches => deterministic).
 */
__global__ void k_agg_partial(const DevGroup *__restrict__ groups, int n,
                              const int64_t *__restrict__ ts,
                              const double *__restrict__ val,
                              const uint8_t *__restrict__ valid,
                              int64_t range_lo, int64_t range_hi, int64_t t0,
                              int64_t bucket_ns, int nbuckets,
                              double *__restrict__ pmax,
                              double *__restrict__ psum,
                              long long *__restrict__ pcnt) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int nwaves = blockDim.x >> 6;
    /* bucket-boundary row offsets staged in LDS: one cooperative binary
       search per boundary (256 in flight per block) instead of 2 dependent
       searches per (wave,bucket) — the searches were the latency hot spot */
    /* dynamic LDS: sized to nbuckets+1 at launch so occupancy is not
       capped by the worst-case table (32.8 KB static -> 4 blocks/CU even
       for small bucket counts) */
    extern __shared__ __attribute__((aligned(16))) uint32_t soff[];
    constexpr int MAXB = 8192;
    const bool use_lds = nbuckets <= MAXB;
    const int64_t hi_open = (range_hi == INT64_MAX) ? INT64_MAX : range_hi + 1;
    for (int g = blockIdx.x; g < n; g += gridDim.x) {
        const int64_t base = groups[g].row_off;
        const int64_t cnt = groups[g].nrows;
        const int64_t *t = ts + base;
        const double *v = val + base;
        const uint8_t *vd = valid ? valid + base : nullptr;
        if (use_lds) {
            for (int bi = threadIdx.x; bi <= nbuckets; bi += blockDim.x) {
                int64_t bound = t0 + int64_t(bi) * bucket_ns;
                if (bound < range_lo) bound = range_lo;
                if (bound > hi_open) bound = hi_open;
                int64_t lo = 0, hi = cnt;
                while (lo < hi) {
                    int64_t m = (lo + hi) >> 1;
                    if (t[m] < bound) lo = m + 1; else hi = m;
                }
                soff[bi] = uint32_t(lo);
            }
            __syncthreads();
        }
        for (int b = wave; b < nbuckets; b += nwaves) {
            int64_t s, e;
            if (use_lds) {
                s = soff[b];
                e = soff[b + 1];
            } else {
                int64_t blo = t0 + int64_t(b) * bucket_ns;
                int64_t bhi = blo + bucket_ns;
                if (blo < range_lo) blo = range_lo;
                int64_t bhi_cl = (bhi < hi_open) ? bhi : hi_open;
                int64_t lo = 0, hi = cnt;
                while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < blo) lo = m + 1; else hi = m; }
                s = lo;
                lo = s; hi = cnt;
                while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < bhi_cl) lo = m + 1; else hi = m; }
                e = lo;
            }
            double mx = -__builtin_inf(), sm = 0.0;
            long long c = 0;
            for (int64_t r = s + lane; r < e; r += 64) {
                if (vd && !vd[r]) continue;
                double x = v[r];
                if (x > mx) mx = x;
                sm += x;
                c++;
            }
            /* wave shuffle tree (fixed order) */
            for (int off = 32; off > 0; off >>= 1) {
                double omx = __shfl_down(mx, off, 64);
                double osm = __shfl_down(sm, off, 64);
                long long oc = __shfl_down(c, off, 64);
                if (omx > mx) mx = omx;
                sm += osm;
                c += oc;
            }
            if (lane == 0) {
                size_t idx = size_t(g) * nbuckets + b;
                pmax[idx] = mx;
                psum[idx] = sm;
                pcnt[idx] = c;
            }
        }
        if (use_lds) __syncthreads();
    }
}

__global__ void k_agg_merge(int ngroups, int nbuckets,
                            const double *__restrict__ pmax,
                            const double *__restrict__ psum,
                            const long long *__restrict__ pcnt,
                            double *__restrict__ gmax,
                            double *__restrict__ gsum,
                            long long *__restrict__ gcnt) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int nwaves = blockDim.x >> 6;
    for (int b = blockIdx.x * nwaves + wave; b < nbuckets;
         b += gridDim.x * nwaves) {
        double mx = -__builtin_inf(), sm = 0.0;
        long long c = 0;
        for (int g = lane; g < ngroups; g += 64) {
            size_t idx = size_t(g) * nbuckets + b;
            double omx = pmax[idx];
            if (omx > mx) mx = omx;
            sm += psum[idx];
            c += pcnt[idx];
        }
        for (int off = 32; off > 0; off >>= 1) {
            double omx = __shfl_down(mx, off, 64);
            double osm = __shfl_down(sm, off, 64);
            long long oc = __shfl_down(c, off, 64);
            if (omx > mx) mx = omx;
            sm += osm;
            c += oc;
        }
        if (lane == 0) {
            if (mx > gmax[b]) gmax[b] = mx;
            gsum[b] += sm;
            gcnt[b] += c;
        }
    }
}

/* Fused-path aggregate partial with CLOSED-FORM bucket boundaries.  The
 * fused scan requires every ts page to be RLE (fused_capable), so the
 * compacted output of each page-group is an affine ts sequence
 * (t0sel + i*delta) recorded by k_spans_rle.  Instead of 1 global binary
 * search over out_ts per bucket boundary (the latency + traffic hot spot
 * of k_agg_partial: ~log2(rows) dependent DRAM loads each), boundaries
 * are found by a search over the sgroup's page-group table staged in LDS
 * followed by one division — no out_ts reads at all.  The reduction and
 * partial-cell layout are identical to k_agg_partial (deterministic,
 * every cell written once). */
__global__ void k_agg_partial_rle(const DevGroup *__restrict__ sg, int nsg,
                                  const int32_t *__restrict__ sgfirst,
                                  const int64_t *__restrict__ g_t0sel,
                                  const int64_t *__restrict__ g_delta,
                                  const int64_t *__restrict__ out_off,
                                  const double *__restrict__ val,
                                  int64_t t0, int64_t bucket_ns, int nbuckets,
                                  int max_span,
                                  double *__restrict__ pmax,
                                  double *__restrict__ psum,
                                  long long *__restrict__ pcnt) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int nwaves = blockDim.x >> 6;
    extern __shared__ __attribute__((aligned(16))) uint8_t smem[];
    int64_t *t0s = (int64_t *)smem;              /* [max_span] */
    int64_t *dlt = t0s + max_span;               /* [max_span] */
    uint32_t *rel = (uint32_t *)(dlt + max_span); /* [max_span+1] */
    uint32_t *soff = rel + max_span + 1;         /* [nbuckets+1] */
    for (int g = blockIdx.x; g < nsg; g += gridDim.x) {
        const int64_t base = sg[g].row_off;
        const int32_t nrows = sg[g].nrows;
        const int gf = sgfirst[g];
        const int span = sgfirst[g + 1] - gf;
        for (int j = threadIdx.x; j < span; j += blockDim.x) {
            rel[j] = uint32_t(out_off[gf + j] - base);
            t0s[j] = g_t0sel[gf + j];
            dlt[j] = g_delta[gf + j];
        }
        if (threadIdx.x == 0) rel[span] = uint32_t(nrows);
        __syncthreads();
        for (int bi = threadIdx.x; bi <= nbuckets; bi += blockDim.x) {
            const int64_t bound = t0 + int64_t(bi) * bucket_ns;
            /* count of selected rows with ts < bound; page-groups of one
               series are time-ordered and non-overlapping */
            int lo = 0, hi = span;
            while (lo < hi) {
                int m = (lo + hi) >> 1;
                if (t0s[m] < bound) lo = m + 1; else hi = m;
            }
            uint32_t off = 0;
            if (lo > 0) {
                const int k = lo - 1;
                const uint32_t cnt = rel[k + 1] - rel[k];
                const int64_t d = dlt[k], b = bound - t0s[k]; /* b > 0 */
                uint32_t w;
                if (d <= 0) {
                    w = cnt; /* constant ts < bound: all rows qualify */
                } else {
                    __int128 q = ((__int128)b + d - 1) / d; /* #i: i*d < b */
                    w = q >= cnt ? cnt : uint32_t(q);
                }
                off = rel[k] + w;
            }
            soff[bi] = off;
        }
        __syncthreads();
        const double *v = val + base;
        /* CONTIGUOUS bucket range per wave, cut at ROW quantiles via the
           soff table: each wave sweeps one monotone row stream of ~equal
           length (HW prefetch friendly + load-balanced even when the
           time range empties the outer buckets).  Empty buckets skip the
           shuffle tree. */
        int b0, b1;
        {
            const uint32_t q0 = uint32_t(int64_t(nrows) * wave / nwaves);
            const uint32_t q1 = uint32_t(int64_t(nrows) * (wave + 1) / nwaves);
            int lo2 = 0, hi2 = nbuckets + 1;
            while (lo2 < hi2) {
                int m = (lo2 + hi2) >> 1;
                if (soff[m] < q0) lo2 = m + 1; else hi2 = m;
            }
            b0 = lo2 < nbuckets ? lo2 : nbuckets;
            hi2 = nbuckets + 1;
            while (lo2 < hi2) {
                int m = (lo2 + hi2) >> 1;
                if (soff[m] < q1) lo2 = m + 1; else hi2 = m;
            }
            b1 = lo2 < nbuckets ? lo2 : nbuckets;
            /* trailing empty buckets (soff hits nrows early) still need
               their partial cells written: stretch the last wave */
            if (wave == nwaves - 1) b1 = nbuckets;
        }
        for (int b = b0; b < b1; b++) {
            const int64_t s = soff[b], e = soff[b + 1];
            size_t idx = size_t(g) * nbuckets + b;
            if (e <= s) {
                if (lane == 0) {
                    pmax[idx] = -__builtin_inf();
                    psum[idx] = 0.0;
                    pcnt[idx] = 0;
                }
                continue;
            }
            /* fused-path rows are all-valid (fused_capable), so the count
               is e-s outright — no count accumulator, no count tree */
            double mx = -__builtin_inf(), sm = 0.0;
            /* 16-B double2 loads: halve the load-instruction traffic of
               the strided 8-B version (the kernel was instruction-bound
               at 3.4 TB/s vs the ~6.3 achievable).  Output buffers are
               16-B aligned; an odd start row is peeled. */
            const int head = int((base + s) & 1);
            if (head && lane == 0) {
                double x = v[s];
                mx = x;
                sm = x;
            }
            const int64_t n2 = (e - s - head) >> 1;
            const double2 *vp = (const double2 *)(v + s + head);
            int64_t q = lane;
            for (; q + 64 < n2; q += 128) { /* 2 loads in flight per iter */
                double2 x0 = vp[q], x1 = vp[q + 64];
                if (x0.x > mx) mx = x0.x;
                if (x0.y > mx) mx = x0.y;
                if (x1.x > mx) mx = x1.x;
                if (x1.y > mx) mx = x1.y;
                sm += (x0.x + x0.y) + (x1.x + x1.y);
            }
            if (q < n2) {
                double2 x = vp[q];
                if (x.x > mx) mx = x.x;
                if (x.y > mx) mx = x.y;
                sm += x.x + x.y;
            }
            if (((e - s - head) & 1) && lane == 63) {
                double x = v[e - 1];
                if (x > mx) mx = x;
                sm += x;
            }
            for (int off2 = 32; off2 > 0; off2 >>= 1) {
                double omx = __shfl_down(mx, off2, 64);
                double osm = __shfl_down(sm, off2, 64);
                if (omx > mx) mx = omx;
                sm += osm;
            }
            if (lane == 0) {
                pmax[idx] = mx;
                psum[idx] = sm;
                pcnt[idx] = e - s;
            }
        }
        __syncthreads();
    }
}

/* --------------------- string column decode (PC_STR) ---------------------
 * codec/string.rs snappy blocks ([7][0x10][snappy raw of varint-prefixed
 * strings]) and uncompressed blocks ([1][u64 BE len][bytes]...).  The
 * snappy bitstream is strictly sequential, so (like Gorilla) parallelism
 * is pages in flight: one thread decompresses one page into its scratch
 * slab, then walks the payload against the validity bitset recording
 * (pos,len) per row; a device scan of the lengths builds the Arrow
 * offsets and a gather packs the bytes. */

__device__ int64_t dev_snappy_decompress(const uint8_t *__restrict__ src,
                                         uint32_t len,
                                         uint8_t *__restrict__ dst) {
    uint64_t ulen = 0;
    uint32_t i = 0;
    int sh = 0, ok = 0;
    while (i < len && i < 10) {
        uint8_t b = src[i++];
        ulen |= uint64_t(b & 0x7f) << sh;
        sh += 7;
        if (!(b & 0x80)) { ok = 1; break; }
    }
    if (!ok) return -1;
    const uint8_t *ip = src + i, *end = src + len;
    uint8_t *op = dst, *op_end = dst + ulen;
    while (ip < end) {
        uint8_t tag = *ip++;
        uint32_t l, off;
        switch (tag & 3) {
        case 0: {
            l = tag >> 2;
            if (l >= 60) {
                int c = int(l) - 59;
                if (ip + c > end) return -1;
                l = 0;
                for (int k = 0; k < c; k++) l |= uint32_t(ip[k]) << (8 * k);
                ip += c;
            }
            l += 1;
            if (ip + l > end || op + l > op_end) return -1;
            { /* word-wide literal copy (disjoint buffers; bounds above
                 guarantee the 8-B reads/writes stay inside [ip,ip+l) /
                 [op,op+l)) */
                uint32_t k = 0;
                for (; k + 8 <= l; k += 8) {
                    uint64_t w;
                    __builtin_memcpy(&w, ip + k, 8);
                    __builtin_memcpy(op + k, &w, 8);
                }
                for (; k < l; k++) op[k] = ip[k];
            }
            ip += l;
            op += l;
            continue;
        }
        case 1:
            if (ip >= end) return -1;
            l = ((tag >> 2) & 7) + 4;
            off = (uint32_t(tag >> 5) << 8) | *ip++;
            break;
        case 2:
            if (ip + 2 > end) return -1;
            l = (tag >> 2) + 1;
            off = uint32_t(ip[0]) | (uint32_t(ip[1]) << 8);
            ip += 2;
            break;
        default:
            if (ip + 4 > end) return -1;
            l = (tag >> 2) + 1;
            off = uint32_t(ip[0]) | (uint32_t(ip[1]) << 8) |
                  (uint32_t(ip[2]) << 16) | (uint32_t(ip[3]) << 24);
            ip += 4;
            break;
        }
        if (off == 0 || uint64_t(op - dst) < off || op + l > op_end) return -1;
        const uint8_t *cp = op - off; /* may overlap: strictly in order */
        if (off >= 8) { /* no overlap within a word: copy 8 B at a time */
            uint32_t k = 0;
            for (; k + 8 <= l; k += 8) {
                uint64_t w;
                __builtin_memcpy(&w, cp + k, 8);
                __builtin_memcpy(op + k, &w, 8);
            }
            for (; k < l; k++) op[k] = cp[k];
        } else { /* short offset: byte-serial repeat semantics */
            for (uint32_t k = 0; k < l; k++) op[k] = cp[k];
        }
        op += l;
    }
    return (op == op_end && ip == end) ? int64_t(ulen) : -1;
}

__global__ void k_str_decode(const uint8_t *__restrict__ blob,
                             const DevPage *__restrict__ pages, int n,
                             const int64_t *__restrict__ scr_off,
                             uint8_t *__restrict__ scratch,
                             int64_t *__restrict__ sz,
                             int64_t *__restrict__ pos,
                             uint8_t *__restrict__ valid,
                             unsigned *__restrict__ err) {
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < n;
         p += gridDim.x * blockDim.x) {
        DevPage pg = pages[p];
        const uint8_t *s = blob + pg.data_off;
        const uint8_t *bs = blob + pg.bitset_off;
        const int64_t base = pg.row_off;
        uint8_t *dst = scratch + scr_off[p];
        for (uint32_t r = 0; r < pg.nrows; r++) {
            sz[base + r] = 0;
            pos[base + r] = 0;
            if (valid) valid[base + r] = 0;
        }
        if (pg.data_len == 0) continue; /* empty src -> all null,
                                           string.rs:231-236 */
        /* normalize: the payload lands in this page's scratch slab for
           both encodings, so the walk and the gather read one source */
        int64_t pn;
        int be = 0;
        if (pg.enc == GS_ENC_SNAPPY) {
            if (pg.data_len < 2) { atomicOr(err, DERR_FORMAT); continue; }
            pn = dev_snappy_decompress(s + 2, pg.data_len - 2, dst);
            if (pn < 0) { atomicOr(err, DERR_FORMAT); continue; }
        } else if (pg.enc == GS_ENC_NULL) { /* string.rs:169-183 */
            pn = int64_t(pg.data_len) - 1;
            for (int64_t k = 0; k < pn; k++) dst[k] = s[1 + k];
            be = 1;
        } else {
            atomicOr(err, DERR_FORMAT);
            continue;
        }
        uint64_t i = 0;
        for (uint32_t r = 0; r < pg.nrows; r++) {
            int v = pg.all_valid ? 1 : dev_bit(bs, r);
            if (!v) continue;
            if (i >= uint64_t(pn)) { /* payload exhausted with valid rows
                                        remaining: the reference emits a
                                        SHORTER array here (rows dropped,
                                        str_snappy_decode_to_array); fail
                                        loudly instead of silently
                                        nulling */
                atomicOr(err, DERR_SHORT);
                break;
            }
            uint64_t slen = 0;
            if (be) {
                if (i + 8 > uint64_t(pn)) { atomicOr(err, DERR_FORMAT); break; }
                for (int k = 0; k < 8; k++) slen = (slen << 8) | dst[i + k];
                i += 8;
            } else {
                int sh = 0, ok = 0;
                while (i < uint64_t(pn)) {
                    uint8_t b = dst[i++];
                    slen |= uint64_t(b & 0x7f) << sh;
                    sh += 7;
                    if (!(b & 0x80)) { ok = 1; break; }
                }
                if (!ok) { atomicOr(err, DERR_FORMAT); break; }
            }
            if (i + slen > uint64_t(pn)) { atomicOr(err, DERR_FORMAT); break; }
            pos[base + r] = scr_off[p] + int64_t(i);
            sz[base + r] = int64_t(slen);
            if (valid) valid[base + r] = 1;
            i += slen;
        }
    }
}

__global__ void k_str_gather(const uint8_t *__restrict__ scratch,
                             const int64_t *__restrict__ pos,
                             const int64_t *__restrict__ sz,
                             const int64_t *__restrict__ off, int64_t rows,
                             uint8_t *__restrict__ out) {
    for (int64_t r = blockIdx.x * int64_t(blockDim.x) + threadIdx.x; r < rows;
         r += int64_t(gridDim.x) * blockDim.x) {
        const int64_t n2 = sz[r];
        const uint8_t *s = scratch + pos[r];
        uint8_t *d = out + off[r];
        int64_t k = 0;
        for (; k + 8 <= n2; k += 8) { /* word-wide gather */
            uint64_t w;
            __builtin_memcpy(&w, s + k, 8);
            __builtin_memcpy(d + k, &w, 8);
        }
        for (; k < n2; k++) d[k] = s[k];
    }
}

/* ------------------------------------------------- compaction merge (k-way)
 * Config #5: k overlapping L0 column groups per series -> one merged,
 * deduped stream (compact.rs:271-404, comapcting_block_meta_group.rs:
 * 52-208).  Streams are ordered oldest -> newest (file_id order,
 * iterator.rs:488); equal timestamps collapse, per column the newest
 * non-null value wins (batch_builder.rs:106-155).
 *
 * GPU mapping: no loser tree — ranks by binary search.  An element is the
 * row OWNER iff no newer stream contains its ts; the merged position of
 * an owner is the sum over streams of "owners with smaller ts", obtained
 * from per-stream exclusive scans of the owner flags.  Dedup value
 * selection walks streams newest -> oldest at scatter time. */

#define GS_MAX_STREAMS 16

struct CompactArgs {
    const int64_t *ts[GS_MAX_STREAMS];
    const double *val[GS_MAX_STREAMS];
    const uint8_t *valid[GS_MAX_STREAMS];
    const DevGroup *groups[GS_MAX_STREAMS];
    uint8_t *flags[GS_MAX_STREAMS];
    int32_t *prefix[GS_MAX_STREAMS];
    int64_t *counts; /* [nsets][nseries] owner counts */
    int64_t *out_off; /* [nseries] output row offsets */
};

__device__ __forceinline__ bool dev_ts_contains(const int64_t *t, int64_t n,
                                                int64_t x, int64_t *pos) {
    int64_t lo = 0, hi = n;
    while (lo < hi) { int64_t m = (lo + hi) >> 1; if (t[m] < x) lo = m + 1; else hi = m; }
    *pos = lo;
    return lo < n && t[lo] == x;
}

/* block per (set, series): owner flags + per-series owner count.
 *
 * LDS-staged tiled merge (round-2 rework): per 1024-element tile of
 * stream f, the matching window of each NEWER stream is block-loaded
 * into LDS ONCE (coalesced), and every lane searches LDS instead of
 * global memory.  Round-1 searched global per element (log2(n) loads,
 * tolerable only because bisection's top tree levels are L1-shared);
 * three per-lane-probe variants (tiles+gallop, strided+gallop,
 * interpolation) all measured SLOWER than that — staging is the only
 * scheme whose global rank traffic is "each t2 element loaded once".
 * All block-wide loops are uniform (lanes idle within chunks, never
 * diverge across a __syncthreads). */
#define CM_TILE 2048 /* stream-f elements per tile (8 per lane; 1024 and 4096 measured slower) */
#define CM_W 2048    /* LDS window entries (16 KiB) per chunk */

__global__ void k_cm_flags(CompactArgs a, int nsets, int nseries,
                           unsigned *__restrict__ err) {
    __shared__ int64_t wts[CM_W];
    __shared__ int64_t wb_sh[2];
    __shared__ long long sred[256];
    int f = blockIdx.y;
    for (int s = blockIdx.x; s < nseries; s += gridDim.x) {
        DevGroup g = a.groups[f][s];
        const int64_t *t = a.ts[f] + g.row_off;
        uint8_t *fl = a.flags[f] + g.row_off;
        long long cnt = 0;
        for (int64_t j0 = 0; j0 < g.nrows; j0 += CM_TILE) {
            const int64_t je = j0 + CM_TILE < g.nrows ? j0 + CM_TILE
                                                      : g.nrows;
            /* this lane's elements: j0+tid, j0+tid+256, ... (sorted) */
            int64_t x[CM_TILE / 256];
            bool own[CM_TILE / 256];
            int ne = 0;
            for (int64_t j = j0 + threadIdx.x; j < je; j += blockDim.x) {
                x[ne] = t[j];
                own[ne] = true;
                if (j > 0 && t[j - 1] >= x[ne])
                    atomicOr(err, DERR_FORMAT); /* "not sorted" */
                ne++;
            }
            const int64_t x_lo = t[j0], x_hi = t[je - 1];
            for (int f2 = f + 1; f2 < nsets; f2++) {
                DevGroup g2 = a.groups[f2][s];
                const int64_t *t2 = a.ts[f2] + g2.row_off;
                const int64_t n2 = g2.nrows;
                if (n2 == 0) continue;
                /* window [wl, wh) = rows of t2 that can equal a tile x */
                if (threadIdx.x == 0) {
                    int64_t lo = 0, hi = n2;
                    while (lo < hi) {
                        int64_t m = (lo + hi) >> 1;
                        if (t2[m] < x_lo) lo = m + 1; else hi = m;
                    }
                    wb_sh[0] = lo;
                    int64_t lo2 = lo, hi2 = n2;
                    while (lo2 < hi2) {
                        int64_t m = (lo2 + hi2) >> 1;
                        if (t2[m] <= x_hi) lo2 = m + 1; else hi2 = m;
                    }
                    wb_sh[1] = lo2;
                }
                __syncthreads();
                const int64_t wl = wb_sh[0], wend = wb_sh[1];
                __syncthreads();
                int k = 0; /* this lane's next unresolved element */
                for (int64_t wbase = wl; wbase < wend; wbase += CM_W) {
                    const int wcnt = int(wend - wbase < CM_W ? wend - wbase
                                                             : CM_W);
                    for (int i = threadIdx.x; i < wcnt; i += blockDim.x)
                        wts[i] = t2[wbase + i];
                    __syncthreads();
                    const int64_t w_last = wts[wcnt - 1];
                    const bool last_chunk = wbase + wcnt == wend;
                    while (k < ne && (x[k] <= w_last || last_chunk)) {
                        int lo = 0, hi = wcnt; /* LDS lower bound */
                        while (lo < hi) {
                            int m = (lo + hi) >> 1;
                            if (wts[m] < x[k]) lo = m + 1; else hi = m;
                        }
                        if (lo < wcnt && wts[lo] == x[k]) own[k] = false;
                        k++;
                    }
                    __syncthreads();
                }
            }
            for (int i = 0; i < ne; i++) {
                fl[j0 + threadIdx.x + int64_t(i) * blockDim.x] = own[i];
                cnt += own[i];
            }
            __syncthreads();
        }
        /* block reduce cnt -> counts[f][s] */
        sred[threadIdx.x] = cnt;
        __syncthreads();
        for (int w = blockDim.x >> 1; w > 0; w >>= 1) {
            if (threadIdx.x < unsigned(w)) sred[threadIdx.x] += sred[threadIdx.x + w];
            __syncthreads();
        }
        if (threadIdx.x == 0)
            a.counts[size_t(f) * nseries + s] = sred[0];
        __syncthreads();
    }
}

/* block per (set, series): exclusive block-scan of owner flags */
__global__ void k_cm_prefix(CompactArgs a, int nsets, int nseries) {
    __shared__ int32_t sh[256];
    __shared__ int32_t carry;
    int total = nsets * nseries;
    for (int idx = blockIdx.x; idx < total; idx += gridDim.x) {
        int f = idx / nseries, s = idx % nseries;
        DevGroup g = a.groups[f][s];
        const uint8_t *fl = a.flags[f] + g.row_off;
        int32_t *pf = a.prefix[f] + g.row_off;
        if (threadIdx.x == 0) carry = 0;
        __syncthreads();
        for (int64_t tile = 0; tile < g.nrows; tile += blockDim.x) {
            int64_t j = tile + threadIdx.x;
            int32_t v = (j < g.nrows) ? fl[j] : 0;
            sh[threadIdx.x] = v;
            __syncthreads();
            for (int off = 1; off < int(blockDim.x); off <<= 1) {
                int32_t u = threadIdx.x >= unsigned(off)
                                ? sh[threadIdx.x - off] : 0;
                __syncthreads();
                sh[threadIdx.x] += u;
                __syncthreads();
            }
            if (j < g.nrows)
                pf[j] = carry + sh[threadIdx.x] - v; /* exclusive */
            __syncthreads();
            if (threadIdx.x == 0) carry += sh[blockDim.x - 1];
            __syncthreads();
        }
    }
}

/* block per (set, series): scatter owners to merged positions with
 * newest-non-null value selection */
/* LDS-staged tiled scatter (see k_cm_flags): the merged position (rank
 * over all streams) and the dedup value come from the same LDS windows.
 * The dedup walk is folded into a DESCENDING stream pass: the newest
 * non-null among {f, older hits} wins (batch_builder.rs:106-155), so the
 * first hit with a valid value while walking f-1, f-2, ... resolves it
 * without a per-element hit array. */
__global__ void k_cm_scatter(CompactArgs a, int nsets, int nseries,
                             int64_t *__restrict__ out_ts,
                             double *__restrict__ out_val,
                             uint8_t *__restrict__ out_valid) {
    __shared__ int64_t wts[CM_W];
    __shared__ int32_t wpf[CM_W]; /* prefix window: the per-owner rank adds
                                     were ~645M scattered 4/1-B global
                                     loads — staged once, coalesced */
    __shared__ uint8_t wfl[CM_W];
    __shared__ int64_t wb_sh[2];
    int f = blockIdx.y;
    for (int s = blockIdx.x; s < nseries; s += gridDim.x) {
        DevGroup g = a.groups[f][s];
        const int64_t *t = a.ts[f] + g.row_off;
        const uint8_t *fl = a.flags[f] + g.row_off;
        const int32_t *pf = a.prefix[f] + g.row_off;
        const uint8_t *vdf = a.valid[f];
        const int64_t base = a.out_off[s];
        for (int64_t j0 = 0; j0 < g.nrows; j0 += CM_TILE) {
            const int64_t je = j0 + CM_TILE < g.nrows ? j0 + CM_TILE
                                                      : g.nrows;
            int64_t x[CM_TILE / 256];
            int64_t pos[CM_TILE / 256];
            double v[CM_TILE / 256];
            bool own[CM_TILE / 256], okv[CM_TILE / 256];
            int ne = 0;
            for (int64_t j = j0 + threadIdx.x; j < je; j += blockDim.x) {
                x[ne] = t[j];
                own[ne] = fl[j] != 0;
                pos[ne] = base + pf[j];
                const int64_t row = g.row_off + j;
                const bool o = !vdf || vdf[row];
                v[ne] = o ? a.val[f][row] : 0.0;
                okv[ne] = o;
                ne++;
            }
            const int64_t x_lo = t[j0], x_hi = t[je - 1];
            /* descending pass: newer streams contribute rank only; older
               streams contribute rank + the first valid dedup value */
            for (int f2 = nsets - 1; f2 >= 0; f2--) {
                if (f2 == f) continue;
                DevGroup g2 = a.groups[f2][s];
                const int64_t *t2 = a.ts[f2] + g2.row_off;
                const int32_t *pf2 = a.prefix[f2] + g2.row_off;
                const uint8_t *fl2 = a.flags[f2] + g2.row_off;
                const int64_t n2 = g2.nrows;
                if (n2 == 0) continue;
                if (threadIdx.x == 0) {
                    int64_t lo = 0, hi = n2;
                    while (lo < hi) {
                        int64_t m = (lo + hi) >> 1;
                        if (t2[m] < x_lo) lo = m + 1; else hi = m;
                    }
                    wb_sh[0] = lo;
                    int64_t lo2 = lo, hi2 = n2;
                    while (lo2 < hi2) {
                        int64_t m = (lo2 + hi2) >> 1;
                        if (t2[m] <= x_hi) lo2 = m + 1; else hi2 = m;
                    }
                    wb_sh[1] = lo2;
                }
                __syncthreads();
                const int64_t wl = wb_sh[0], wend = wb_sh[1];
                __syncthreads();
                if (wl == wend) { /* tile range absent from t2: rank only */
                    if (wl > 0)
                        for (int i = 0; i < ne; i++)
                            if (own[i])
                                pos[i] += pf2[wl - 1] + fl2[wl - 1];
                    continue;
                }
                int k = 0;
                for (int64_t wbase = wl; wbase < wend; wbase += CM_W) {
                    const int wcnt = int(wend - wbase < CM_W ? wend - wbase
                                                             : CM_W);
                    for (int i = threadIdx.x; i < wcnt; i += blockDim.x) {
                        wts[i] = t2[wbase + i];
                        wpf[i] = pf2[wbase + i];
                        wfl[i] = fl2[wbase + i];
                    }
                    __syncthreads();
                    const int64_t w_last = wts[wcnt - 1];
                    const bool last_chunk = wbase + wcnt == wend;
                    while (k < ne && (x[k] <= w_last || last_chunk)) {
                        if (own[k]) {
                            int lo = 0, hi = wcnt; /* LDS lower bound */
                            while (lo < hi) {
                                int m = (lo + hi) >> 1;
                                if (wts[m] < x[k]) lo = m + 1; else hi = m;
                            }
                            const int64_t p2 = wbase + lo;
                            if (lo > 0)
                                pos[k] += wpf[lo - 1] + wfl[lo - 1];
                            else if (p2 > 0)
                                pos[k] += pf2[p2 - 1] + fl2[p2 - 1];
                            if (f2 < f && !okv[k] && lo < wcnt &&
                                wts[lo] == x[k]) {
                                const int64_t row2 = g2.row_off + p2;
                                const uint8_t *vd2 = a.valid[f2];
                                if (!vd2 || vd2[row2]) {
                                    v[k] = a.val[f2][row2];
                                    okv[k] = true;
                                }
                            }
                        }
                        k++;
                    }
                    __syncthreads();
                }
            }
            for (int i = 0; i < ne; i++) {
                if (!own[i]) continue;
                out_ts[pos[i]] = x[i];
                out_val[pos[i]] = okv[i] ? v[i] : 0.0;
                if (out_valid) out_valid[pos[i]] = okv[i];
            }
            __syncthreads();
        }
    }
}

/* --------------------------------------------------- fused filtered scan
 * Fast path of gs_scan when every ts page is RLE (the TSBS shape), every
 * field page is all-valid Gorilla and no tombstones apply: the span of
 * the closed time range is computed from the RLE header in closed form
 * (no ts materialization), the Gorilla decoder writes ONLY the selected
 * rows, already compacted, and the aggregate runs over the compacted
 * output — the decode work is identical (every record is still parsed;
 * only the stores of filtered-out rows are skipped), which is exactly the
 * operator fusion the north star asks for. */

/* two-level device exclusive scan of int64 counts (n <= 2048*2048);
 * out has n+1 entries, out[n] = total */
#define SCAN_BLOCK 256
#define SCAN_ITEMS 8
__global__ void k_scan_partials(const int64_t *__restrict__ in, int n,
                                int64_t *__restrict__ out,
                                int64_t *__restrict__ blocksums) {
    __shared__ int64_t sh[SCAN_BLOCK];
    int chunk = SCAN_BLOCK * SCAN_ITEMS;
    int base = blockIdx.x * chunk;
    int64_t vals[SCAN_ITEMS];
    int64_t acc = 0;
    for (int k = 0; k < SCAN_ITEMS; k++) {
        int i = base + threadIdx.x * SCAN_ITEMS + k;
        vals[k] = acc;
        acc += (i < n) ? in[i] : 0;
    }
    sh[threadIdx.x] = acc;
    __syncthreads();
    /* block scan (exclusive) over per-thread sums */
    for (int off = 1; off < SCAN_BLOCK; off <<= 1) {
        int64_t v = (threadIdx.x >= unsigned(off)) ? sh[threadIdx.x - off] : 0;
        __syncthreads();
        sh[threadIdx.x] += v;
        __syncthreads();
    }
    int64_t tbase = (threadIdx.x > 0) ? sh[threadIdx.x - 1] : 0;
    for (int k = 0; k < SCAN_ITEMS; k++) {
        int i = base + threadIdx.x * SCAN_ITEMS + k;
        if (i < n) out[i] = tbase + vals[k];
    }
    if (threadIdx.x == SCAN_BLOCK - 1) blocksums[blockIdx.x] = sh[threadIdx.x];
}

__global__ void k_scan_fixup(int n, int64_t *__restrict__ out,
                             int64_t *__restrict__ blocksums, int nblocks) {
    /* one wave: shuffle-scan the block sums in 64-wide tiles with a
       running carry (exact integer prefix — order fixed).  Handles the
       large-row string decode scans (tens of thousands of blocks)
       without a serial single-thread walk. */
    if (blockIdx.x != 0 || threadIdx.x >= 64) return;
    const int lane = threadIdx.x;
    int64_t carry = 0;
    for (int t = 0; t < nblocks; t += 64) {
        int i = t + lane;
        int64_t v = (i < nblocks) ? blocksums[i] : 0;
        int64_t x = v;
        for (int off = 1; off < 64; off <<= 1) {
            int64_t u = __shfl_up(x, off, 64);
            if (lane >= off) x += u;
        }
        if (i < nblocks) blocksums[i] = carry + x - v; /* exclusive */
        carry += __shfl(x, 63, 64);
    }
    if (lane == 0) out[n] = carry; /* total */
}

__global__ void k_scan_add(int n, int64_t *__restrict__ out,
                           const int64_t *__restrict__ blocksums) {
    int chunk = SCAN_BLOCK * SCAN_ITEMS;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += gridDim.x * blockDim.x)
        out[i] += blocksums[i / chunk];
}

/* series-group layout of the fused compacted output, built on device:
 * sgroup s covers page-groups [first[s], first[s+1]) */
__global__ void k_build_sgroups_out(const int32_t *__restrict__ first,
                                    int nsg,
                                    const int64_t *__restrict__ out_off,
                                    DevGroup *__restrict__ sg) {
    for (int s = blockIdx.x * blockDim.x + threadIdx.x; s < nsg;
         s += gridDim.x * blockDim.x) {
        int64_t a = out_off[first[s]];
        int64_t b = out_off[first[s + 1]];
        sg[s].row_off = a;
        sg[s].nrows = int32_t(b - a);
        sg[s].pad = 0;
    }
}

/* closed-form span of closed [lo,hi] on an RLE ts page */
__global__ void k_spans_rle(const uint8_t *__restrict__ blob,
                            const DevPage *__restrict__ pages, int npages,
                            int64_t lo, int64_t hi,
                            int64_t *__restrict__ sp_start,
                            int64_t *__restrict__ sp_cnt,
                            int64_t *__restrict__ g_t0sel,
                            int64_t *__restrict__ g_delta,
                            unsigned *__restrict__ err) {
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        DevPage pg = pages[p];
        const uint8_t *s = blob + pg.data_off + 1;
        uint64_t scaler = 1;
        unsigned s10 = s[0] & 0x0f;
        for (unsigned k = 0; k < s10; k++) scaler *= 10;
        const uint8_t *q = s + 1;
        int64_t first = int64_t(dev_be64(q));
        uint64_t dv; uint32_t nr;
        dev_varint(q + 8, pg.data_len - 10, &dv, &nr);
        int64_t delta = int64_t(dv * scaler);
        int64_t n = pg.nrows;
        int64_t st, en; /* selected rows = [st, en) */
        if (delta < 0) { atomicOr(err, DERR_NONMONO); st = en = 0; }
        else if (delta == 0) {
            bool in = first >= lo && first <= hi;
            st = 0; en = in ? n : 0;
        } else {
            __int128 d = delta;
            __int128 a = (__int128)lo - first;
            __int128 b = (__int128)hi - first;
            __int128 s0 = a <= 0 ? 0 : (a + d - 1) / d; /* ceil */
            __int128 e0 = b < 0 ? 0 : b / d + 1;        /* floor + 1 */
            if (s0 > n) s0 = n;
            if (e0 > n) e0 = n;
            st = int64_t(s0);
            en = int64_t(e0);
        }
        if (en < st) en = st;
        sp_start[pg.grp] = st;
        sp_cnt[pg.grp] = en - st;
        /* RLE grid of the SELECTED rows, for the closed-form aggregate
           boundary computation (k_agg_partial_rle) */
        g_t0sel[pg.grp] = first + st * delta;
        g_delta[pg.grp] = delta;
    }
}

/* RLE ts generation of only the selected span, compacted */
__global__ void k_rle_ts_filtered(const uint8_t *__restrict__ blob,
                                  const DevPage *__restrict__ pages,
                                  int npages,
                                  const int64_t *__restrict__ sp_start,
                                  const int64_t *__restrict__ sp_cnt,
                                  const int64_t *__restrict__ out_off,
                                  int64_t *__restrict__ out_ts) {
    for (int p = blockIdx.x; p < npages; p += gridDim.x) {
        DevPage pg = pages[p];
        const uint8_t *s = blob + pg.data_off + 1;
        uint64_t scaler = 1;
        unsigned s10 = s[0] & 0x0f;
        for (unsigned k = 0; k < s10; k++) scaler *= 10;
        const uint8_t *q = s + 1;
        int64_t first = int64_t(dev_be64(q));
        uint64_t dv; uint32_t nr;
        dev_varint(q + 8, pg.data_len - 10, &dv, &nr);
        int64_t delta = int64_t(dv * scaler);
        int64_t st = sp_start[pg.grp], cnt = sp_cnt[pg.grp];
        int64_t *o = out_ts + out_off[pg.grp];
        /* paired 16-B stores: closed-form generation is pure store
           bandwidth; halving the store-instruction count lifted the
           measured rate (output buffers are 16-B aligned; out_off parity
           decides the 1-row peel) */
        const uint64_t base = uint64_t(first) + uint64_t(st) * uint64_t(delta);
        const int64_t head = (out_off[pg.grp] & 1) ? 1 : 0;
        if (head && threadIdx.x == 0 && cnt > 0) o[0] = int64_t(base);
        const int64_t n2 = (cnt - head) >> 1;
        longlong2 *op = (longlong2 *)(o + head);
        for (int64_t q2 = threadIdx.x; q2 < n2; q2 += blockDim.x) {
            const uint64_t r0 = uint64_t(head) + 2 * uint64_t(q2);
            longlong2 v;
            v.x = int64_t(base + r0 * uint64_t(delta));
            v.y = int64_t(base + (r0 + 1) * uint64_t(delta));
            op[q2] = v;
        }
        if (((cnt - head) & 1) && threadIdx.x == 0 && cnt > 0)
            o[cnt - 1] = int64_t(base + uint64_t(cnt - 1) * uint64_t(delta));
    }
}

/* Sub-page pruning for the fused scan: compact the indices of chunks whose
 * row range overlaps their group's selected span (the chunk-granularity
 * analog of the reference's page min/max pruning, reader/chunk.rs:12-49 —
 * a chunk wholly outside the time range is never parsed at all).  Output
 * order is wave-compacted (near table order); chunks write disjoint
 * output rows so order never affects results. */
__global__ void k_gor_active(const DevGorChunk *__restrict__ chunks,
                             int nchunks,
                             const int64_t *__restrict__ sp_start,
                             const int64_t *__restrict__ sp_cnt,
                             int *__restrict__ active,
                             int *__restrict__ n_active) {
    const int lane = threadIdx.x & 63;
    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < nchunks;
         i += gridDim.x * blockDim.x) {
        uint32_t grp = chunks[i].grp;
        int64_t row0 = int64_t(chunks[i].row0);
        int64_t rend = row0 + int64_t(chunks[i].cnt);
        int64_t lo = sp_start[grp];
        int64_t hi = lo + sp_cnt[grp];
        bool act = row0 < hi && rend > lo;
        uint64_t m = __ballot(act);
        if (m) {
            int leader = __ffsll((unsigned long long)m) - 1;
            int base = 0;
            if (lane == leader) base = atomicAdd(n_active, __popcll(m));
            base = __shfl(base, leader);
            if (act)
                active[base + __popcll(m & ((1ull << lane) - 1))] = i;
        }
    }
}

/* Chunk-parallel Gorilla decode writing only the selected span, compacted
 * (same LDS flush as k_gor_chunks; each lane's staged run is contiguous
 * in the output by construction).  Runs over the k_gor_active index list;
 * rows before the span are parsed but not staged (the bitstream is
 * sequential), and a chunk whose decodability was proven by the upload
 * pre-pass (safe_stop) stops as soon as the span's end is passed. */
__global__ void k_gor_chunks_filtered(const uint8_t *__restrict__ blob,
                                      const DevGorChunk *__restrict__ chunks,
                                      const int *__restrict__ active,
                                      const int *__restrict__ n_active,
                                      const int64_t *__restrict__ sp_start,
                                      const int64_t *__restrict__ sp_cnt,
                                      const int64_t *__restrict__ out_off,
                                      double *__restrict__ out,
                                      unsigned *__restrict__ err) {
    __shared__ double ring[GS_GOR_BLOCK / 64][GS_RING][64 + 1];
    /* per-flush lane descriptors: {dst pointer, staged count} packed so the
       flush loop reads ONE broadcast ds_read per source lane instead of
       three cross-lane shuffles */
    __shared__ uint64_t fdesc[GS_GOR_BLOCK / 64][64][2];
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    auto rslot = ring[wv];
    const int nact = *n_active;
    int stride = gridDim.x * blockDim.x;
    int base_id = blockIdx.x * blockDim.x + threadIdx.x;
    int rounds = (nact + stride - 1) / stride;
    for (int rd = 0; rd < rounds; rd++) {
        int ai = base_id + rd * stride;
        bool have = ai < nact;
        /* branch-free prologue: clamp to entry 0 so every load issues
           unconditionally and the compiler batches them under one wait
           (the predicated version serialized ~5 dependent vmcnt(0)s) */
        DevGorChunk ch = chunks[active[have ? ai : 0]];
        /* spans/rows are page-local (< 2^31): 32-bit compares in the
           per-value hot loop */
        int32_t sel_lo = int32_t(sp_start[ch.grp]);
        int32_t sel_hi = int32_t(sel_lo + sp_cnt[ch.grp]);
        if (!have) { sel_lo = 0; sel_hi = 0; }
        double *o = out + out_off[ch.grp] - sel_lo; /* o[r] valid in span */
        int32_t r = int32_t(ch.row0);
        int32_t end = r + int32_t(ch.cnt);
        GorChunkState st = gor_chunk_init(blob, ch);
        int rfill = 0;
        int32_t run0 = 0; /* output row of first staged entry */
        bool done = !have, over = false, clean_stop = false;
        if (have && st.bad) { atomicOr(err, DERR_SHORT); done = true; }
        auto topup = [&]() { /* only with nb < 64 */
            uint64_t x = st.nextw;
            st.nextw = st.nextw2;
            over |= (st.p >= st.p_over);
            st.nextw2 = (st.p < st.p_clamp) ? dev_be64(st.p) : 0;
            st.p += 8;
            if (st.nb == 0) { st.hi = x; st.lo = 0; }
            else { st.hi |= x >> st.nb; st.lo = x << (64 - st.nb); }
            st.nb += 64;
        };
        auto consume = [&](unsigned k) { /* k in 1..=64 */
            st.hi = (k == 64) ? st.lo : ((st.hi << k) | (st.lo >> (64 - k)));
            st.lo = (k == 64) ? 0 : (st.lo << k);
            st.nb -= int(k);
        };
        auto used_bits = [&]() {
            return int64_t(st.p - st.stream) * 8 - 128 - st.nb;
        };
        auto fd = fdesc[wv];
        /* 4 source lanes per store instruction: lane = (sq:2, idx:4), sq
           picks the source sub-lane, idx the ring slot — the ring is 16
           deep, so a one-source-per-instruction flush leaves 48/64 lanes
           idle; this shape keeps every lane useful and issues 16 stores
           per flush instead of 64 (reads batched 4 deep before their
           stores, the earlier register-batching lesson) */
        const int f_idx = lane & (GS_RING - 1);
        const int f_sq = lane / GS_RING;
        double *rcur = &rslot[0][lane]; /* strength-reduced ring cursor */
        auto flush = [&]() {
            fd[lane][0] = (uint64_t)(uintptr_t)(o + int64_t(run0));
            fd[lane][1] = uint64_t(rfill);
            __builtin_amdgcn_wave_barrier();
            constexpr int SRCP = 64 / GS_RING; /* sources per store */
            for (int src0 = 0; src0 < 64; src0 += SRCP * 4) {
                double vbuf[4];
                uint64_t ob[4];
                int cnt[4];
                for (int t = 0; t < 4; t++) {
                    int src = src0 + f_sq + t * SRCP;
                    vbuf[t] = rslot[f_idx][src];
                    ob[t] = fd[src][0];
                    cnt[t] = int(fd[src][1]);
                }
                for (int t = 0; t < 4; t++)
                    if (f_idx < cnt[t])
                        ((double *)(uintptr_t)ob[t])[f_idx] = vbuf[t];
            }
            rfill = 0;
            rcur = &rslot[0][lane];
        };
        auto stage_row = [&](uint64_t bits_) {
            if (r >= sel_lo && r < sel_hi) {
                if (rfill == 0) run0 = r;
                *rcur = __longlong_as_double((long long)bits_);
                rcur += 65;
                rfill++;
            }
            r++;
            /* all rows produced (non-last chunk), or span passed on a
               chunk the pre-pass proved decodable: stop parsing */
            if (r == end && !ch.last) {
                done = true;
                if (used_bits() > st.total_bits) atomicOr(err, DERR_SHORT);
            } else if (r >= sel_hi && (ch.flags & GORF_SAFE_STOP)) {
                done = true;
                clean_stop = true;
            }
        };
        if (!done && ch.row0 == 0 && r < end) stage_row(st.val);
        unsigned it = 1; /* header stage counts toward the first cadence */
        while (!__all(done)) {
            if (!done) {
                if (st.nb < 64) topup();
                if (over) { atomicOr(err, DERR_SHORT); done = true; }
            }
            if (!done) {
                uint32_t top13 = uint32_t(st.hi >> 51);
                bool stg = true;
                {
                    /* uniform path (no repeat/xor branch): a repeat is an
                       XOR value with 0 meaningful bits (sb=0), so every
                       lane runs the same straight-line code and the wave
                       diverges only on the rare need>64 case — measured
                       the round-1 two-branch parse at ~20% VALU
                       utilization, divergence being the dominant loss */
                    unsigned bit0 = (top13 >> 12) & 1; /* 1 = XOR value */
                    unsigned bit1 = (top13 >> 11) & 1; /* 1 = new window */
                    unsigned nw = bit0 & bit1;
                    uint32_t lead = (top13 >> 6) & 0x1f;
                    uint32_t mg_raw = top13 & 0x3f;
                    uint32_t mg_new = mg_raw ? mg_raw : 64;
                    uint32_t tr_new = mg_raw ? (64 - lead - mg_raw) : 0;
                    st.meaningful = nw ? mg_new : st.meaningful;
                    st.trailing = nw ? tr_new : st.trailing;
                    unsigned shift = bit0 ? (bit1 ? 13u : 2u) : 1u;
                    unsigned m_eff = bit0 ? st.meaningful : 0u;
                    unsigned need = shift + m_eff;
                    uint64_t sb;
                    if (__builtin_expect(need < 64, 1)) {
                        /* nb >= 64 after the loop-top topup; need < 64
                           keeps the consume branch-free (no k==64 case) */
                        uint64_t w =
                            (st.hi << shift) | (st.lo >> (64 - shift));
                        sb = m_eff ? (w >> ((64 - m_eff) & 63)) : 0;
                        st.hi = (st.hi << need) | (st.lo >> (64 - need));
                        st.lo <<= need;
                        st.nb -= int(need);
                    } else { /* m_eff >= 51: rare */
                        consume(shift);
                        while (st.nb < int(m_eff)) topup();
                        sb = (m_eff == 64) ? st.hi
                                           : (st.hi >> (64 - m_eff));
                        consume(m_eff);
                    }
                    st.val ^= sb << st.trailing;
                    if (bit0 && st.val == GORILLA_SENTINEL) {
                        done = true;
                        stg = false;
                        if (used_bits() > st.total_bits)
                            atomicOr(err, DERR_SHORT);
                    }
                }
                if (stg && r < end) stage_row(st.val);
            }
            if ((++it & (GS_RING - 1)) == 0) flush();
        }
        flush();
        if (have && r < end && !clean_stop) atomicOr(err, DERR_SHORT);
    }
}

/* --------------------------------------------------- GPU page re-encode
 * The write side of compaction (tsm/writer.rs:249-314 via
 * Page::arrow_array_to_page, tsm/page.rs:100-353): one thread per output
 * page, byte-exact with the host encoders in gs_encode.cpp (which are
 * pinned by the reference's golden vectors).  ts/i64 multi-pass encoders
 * require all-valid input (time columns are never null; i64 nulls are a
 * later row); Gorilla streams past nulls inline. */

struct EncPageSpec {
    int64_t row_off;
    int32_t nrows;
    int32_t pad;
};

/* MSB-first bit appender emitting bytes to global memory */
struct DevBitWriter {
    uint8_t *dst;
    int64_t nbytes;
    uint64_t acc; /* filled from the top */
    int nfill;
    __device__ void init(uint8_t *d) { dst = d; nbytes = 0; acc = 0; nfill = 0; }
    __device__ __forceinline__ void put(uint64_t x, int l) { /* l <= 32 */
        acc |= (x & ((l == 64) ? ~0ULL : ((1ULL << l) - 1))) << (64 - nfill - l);
        nfill += l;
        while (nfill >= 8) {
            dst[nbytes++] = uint8_t(acc >> 56);
            acc <<= 8;
            nfill -= 8;
        }
    }
    __device__ __forceinline__ void put64(uint64_t x, int l) {
        if (l > 32) { put(x >> 32, l - 32); put(x & 0xffffffffULL, 32); }
        else if (l > 0) put(x, l);
    }
    __device__ int64_t finish() { /* pad final partial byte with zeros */
        if (nfill > 0) { dst[nbytes++] = uint8_t(acc >> 56); acc = 0; nfill = 0; }
        return nbytes;
    }
};

__device__ __forceinline__ void dev_put_be64(uint8_t *p, uint64_t v) {
    for (int i = 7; i >= 0; i--) { p[i] = uint8_t(v); v >>= 8; }
}
__device__ __forceinline__ uint64_t dev_zzenc(int64_t v) {
    return (uint64_t(v) << 1) ^ uint64_t(v >> 63);
}
__device__ int dev_varint_put(uint8_t *dst, uint64_t v) {
    int n = 0;
    while (v >= 0x80) { dst[n++] = uint8_t(v | 0x80); v >>= 7; }
    dst[n++] = uint8_t(v);
    return n;
}

__constant__ uint8_t DEV_S8B_NUM_BITS[14][2] = {
    {60, 1}, {30, 2}, {20, 3}, {15, 4}, {12, 5}, {10, 6}, {8, 7},
    {7, 8},  {6, 10}, {5, 12}, {4, 15}, {3, 20}, {2, 30}, {1, 60},
};

/* Gorilla f64 encode of one page's non-null values (float.rs:32-243);
 * returns data length or -1 (sentinel input) */
__device__ int64_t dev_enc_gorilla(const double *v, const uint8_t *valid,
                                   int64_t row_off, int32_t nrows,
                                   uint8_t *dst) {
    /* find first non-null */
    int32_t r0 = 0;
    if (valid) while (r0 < nrows && !valid[row_off + r0]) r0++;
    if (r0 >= nrows) return 0; /* empty input -> empty buffer */
    dst[0] = GS_ENC_GORILLA;
    dst[1] = 1 << 4;
    uint64_t prev = (uint64_t)__double_as_longlong(v[row_off + r0]);
    dev_put_be64(dst + 2, prev);
    DevBitWriter bw;
    bw.init(dst + 10);
    uint64_t prev_lead = ~0ULL, prev_trail = 0;
    int32_t r = r0 + 1;
    for (;;) {
        uint64_t cur;
        if (r < nrows) {
            if (valid && !valid[row_off + r]) { r++; continue; }
            cur = (uint64_t)__double_as_longlong(v[row_off + r]);
            r++;
            if (cur == GORILLA_SENTINEL) return -1;
        } else {
            cur = GORILLA_SENTINEL;
        }
        uint64_t x = cur ^ prev;
        if (x == 0) {
            bw.put(0, 1);
            prev = cur;
            if (cur == GORILLA_SENTINEL) break;
            continue;
        }
        bw.put(1, 1);
        uint64_t lead = uint64_t(__builtin_clzll(x)) & 0x1f;
        uint64_t trail = uint64_t(__builtin_ctzll(x));
        if (prev_lead != ~0ULL && lead >= prev_lead && trail >= prev_trail) {
            bw.put(0, 1);
            int l = int(64 - prev_lead - prev_trail);
            bw.put64(x >> prev_trail, l);
        } else {
            prev_lead = lead;
            prev_trail = trail;
            bw.put(1, 1);
            bw.put(lead, 5);
            uint64_t sig = 64 - lead - trail;
            bw.put(sig & 0x3f, 6); /* 64 encodes as 0 */
            bw.put64(x >> trail, int(sig));
        }
        prev = cur;
        if (cur == GORILLA_SENTINEL) break;
    }
    return 10 + bw.finish();
}

/* delta/zigzag int encode of one all-valid page (timestamp.rs:51-122 /
 * integer.rs:40-96), multi-pass over the input; returns data length */
__device__ int64_t dev_enc_int(const int64_t *v, int64_t row_off,
                               int32_t n, uint8_t *dst, bool is_ts) {
    if (n == 0) return 0;
    const int64_t *s = v + row_off;
    auto rawd = [&](int32_t i) { /* wrapping diff, u64 */
        return uint64_t(s[i]) - uint64_t(s[i - 1]);
    };
    size_t w = 0;
    dst[w++] = is_ts ? GS_ENC_DELTATS : GS_ENC_DELTA;
    /* pass 1: max + RLE check */
    uint64_t max = 0, d1 = 0;
    bool use_rle = true;
    for (int32_t i = 1; i < n; i++) {
        uint64_t d = rawd(i);
        if (!is_ts) d = dev_zzenc(int64_t(d));
        if (i == 1) d1 = d;
        else if (d != d1) use_rle = false;
        if (d > max) max = d;
    }
    uint64_t d0 = is_ts ? uint64_t(s[0]) : dev_zzenc(s[0]);
    if ((is_ts && n > 1 && use_rle) || (!is_ts && n > 2 && use_rle)) {
        dst[w++] = 0;
        dev_put_be64(dst + w, d0);
        w += 8;
        if (is_ts) {
            uint64_t div = 1000000000000ULL;
            while (div > 1 && d1 % div != 0) div /= 10;
            if (div > 1) {
                unsigned sc = 0;
                for (uint64_t x = div; x > 1; x /= 10) sc++;
                dst[1] |= uint8_t(sc);
                w += dev_varint_put(dst + w, d1 / div);
            } else {
                w += dev_varint_put(dst + w, d1);
            }
            w += dev_varint_put(dst + w, uint64_t(n));
        } else {
            w += dev_varint_put(dst + w, d1);
            w += dev_varint_put(dst + w, uint64_t(n) - 1);
        }
        dst[1] |= uint8_t(2 << 4);
        return int64_t(w);
    }
    if (max > ((1ULL << 60) - 1)) { /* uncompressed */
        dst[w++] = 0;
        dev_put_be64(dst + w, d0);
        w += 8;
        for (int32_t i = 1; i < n; i++) {
            uint64_t d = rawd(i);
            if (!is_ts) d = dev_zzenc(int64_t(d));
            dev_put_be64(dst + w, d);
            w += 8;
        }
        return int64_t(w);
    }
    /* simple8b; ts applies the power-of-10 divisor (timestamp.rs:97-121) */
    uint64_t div = 1;
    unsigned sc = 0;
    if (is_ts) {
        div = 1000000000000ULL;
        for (int32_t i = 1; i < n && div > 1; i++)
            while (div > 1 && rawd(i) % div != 0) div /= 10;
        for (uint64_t x = div; x > 1; x /= 10) sc++;
    }
    auto packed = [&](int32_t j) { /* j in 0..n-2 */
        uint64_t d = rawd(j + 1);
        return is_ts ? d / div : dev_zzenc(int64_t(d));
    };
    dst[w++] = uint8_t((1 << 4) | sc);
    dev_put_be64(dst + w, d0);
    w += 8;
    /* simple8b greedy packer (simple8b.rs:26-76) */
    int32_t m = n - 1, i = 0;
    while (i < m) {
        int32_t remain = m - i;
        if (remain >= 120) {
            int32_t lim = remain >= 240 ? 240 : 120;
            int32_t k = 0;
            while (k < lim && packed(i + k) == 1) k++;
            if (k == 240) {
                for (int q = 0; q < 8; q++) dst[w + q] = 0;
                w += 8; i += 240;
                continue;
            }
            if (k >= 120) {
                dev_put_be64(dst + w, 1ULL << 60);
                w += 8; i += 120;
                continue;
            }
        }
        bool ok = false;
        for (int idx = 0; idx < 14; idx++) {
            int32_t int_n = DEV_S8B_NUM_BITS[idx][0];
            unsigned bit_n = DEV_S8B_NUM_BITS[idx][1];
            if (int_n > remain) continue;
            uint64_t max_val = 1ULL << (bit_n & 0x3f);
            uint64_t word = (uint64_t(idx) + 2) << 60;
            bool fits = true;
            for (int32_t q = 0; q < int_n; q++) {
                uint64_t pv = packed(i + q);
                if (pv >= max_val) { fits = false; break; }
                word |= pv << ((unsigned(q) * bit_n) & 0x3f);
            }
            if (!fits) continue;
            dev_put_be64(dst + w, word);
            w += 8; i += int_n;
            ok = true;
            break;
        }
        if (!ok) return -2; /* value out of bounds */
    }
    return int64_t(w);
}

/* one thread per output page: bitset + encode + crc + header
 * (page layout tsm/page.rs:488-497) */
__global__ void k_encode_pages(int kind /*0=ts 1=i64 2=f64*/,
                               const void *__restrict__ vals,
                               const uint8_t *__restrict__ valid,
                               const EncPageSpec *__restrict__ pages,
                               int npages, uint8_t *__restrict__ out,
                               int64_t cap, int64_t *__restrict__ lens,
                               unsigned *__restrict__ err) {
    /* crc32 table built cooperatively in LDS (CRC-32/ISO-HDLC) */
    __shared__ uint32_t crct[256];
    for (int i = threadIdx.x; i < 256; i += blockDim.x) {
        uint32_t c = i;
        for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        crct[i] = c;
    }
    __syncthreads();
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < npages;
         p += gridDim.x * blockDim.x) {
        EncPageSpec ps = pages[p];
        uint8_t *pg = out + int64_t(p) * cap;
        uint32_t bl = uint32_t(ps.nrows + 7) / 8;
        /* validity bitset, LSB-first */
        uint8_t *bs = pg + 16;
        for (uint32_t b = 0; b < bl; b++) {
            uint8_t byte = 0;
            for (int k = 0; k < 8; k++) {
                int32_t r = int32_t(b) * 8 + k;
                if (r < ps.nrows && (!valid || valid[ps.row_off + r]))
                    byte |= uint8_t(1u << k);
            }
            bs[b] = byte;
        }
        uint8_t *data = pg + 16 + bl;
        int64_t dl;
        if (kind == 2) {
            dl = dev_enc_gorilla((const double *)vals, valid, ps.row_off,
                                 ps.nrows, data);
        } else {
            if (valid) { /* int encoders need all-valid input (see header) */
                bool av = true;
                for (int32_t r = 0; r < ps.nrows; r++)
                    if (!valid[ps.row_off + r]) { av = false; break; }
                if (!av) { atomicOr(err, DERR_FORMAT); lens[p] = -3; continue; }
            }
            dl = dev_enc_int((const int64_t *)vals, ps.row_off, ps.nrows,
                             data, kind == 0);
        }
        if (dl < 0) {
            atomicOr(err, DERR_FORMAT);
            lens[p] = dl;
            continue;
        }
        uint32_t crc = 0xFFFFFFFFu;
        for (int64_t i = 0; i < dl; i++)
            crc = crct[(crc ^ data[i]) & 0xFF] ^ (crc >> 8);
        crc ^= 0xFFFFFFFFu;
        pg[0] = uint8_t(bl >> 24); pg[1] = uint8_t(bl >> 16);
        pg[2] = uint8_t(bl >> 8); pg[3] = uint8_t(bl);
        dev_put_be64(pg + 4, uint64_t(ps.nrows));
        pg[12] = uint8_t(crc >> 24); pg[13] = uint8_t(crc >> 16);
        pg[14] = uint8_t(crc >> 8); pg[15] = uint8_t(crc);
        lens[p] = 16 + int64_t(bl) + dl;
    }
}

/* ------------------------------------------------------------- host state */

struct GsCtx {
    int device;
    hipStream_t stream;
    unsigned *d_err;
};

static int grid_for(int work, int per_block) {
    int blocks = (work + per_block - 1) / per_block;
    if (blocks > 2048) blocks = 2048; /* grid-stride beyond (Guideline 11) */
    if (blocks < 1) blocks = 1;
    return blocks;
}

struct SlotPages {
    uint8_t ctype;
    std::vector<DevPage> host[PC_NCLASS];
    DevPage *dev[PC_NCLASS] = {nullptr, nullptr, nullptr};
    int n[PC_NCLASS] = {0, 0, 0};
    /* string pages only: per-PC_STR-page scratch offsets (host order of
       host[PC_STR]) + total; caps parsed from the block headers at
       upload (snappy preamble varint / uncompressed payload length) */
    std::vector<int64_t> str_scr;
    int64_t str_total = 0;
    int64_t *d_str_scr = nullptr;
    /* Gorilla chunk table (page-major over host[PC_GOR]; states filled by
       the k_gor_sync upload pre-pass) */
    DevGorChunk *d_gor_chunks = nullptr;
    int n_gor_chunks = 0;
    int32_t *d_gor_chunk_base = nullptr; /* [n[PC_GOR]+1] */
    DevGorChunk *d_gorn_chunks = nullptr;
    int n_gorn_chunks = 0;
    int32_t *d_gorn_chunk_base = nullptr; /* [n[PC_GORN]+1] */
};

struct GsGroupSet {
    GsCtx *ctx = nullptr;
    uint8_t *d_blob = nullptr;
    size_t blob_len = 0;
    int64_t total_rows = 0;
    size_t ngroups = 0;
    uint32_t ncols = 0;
    std::vector<int64_t> row_offsets;
    std::vector<SlotPages> slots;
    DevGroup *d_groups = nullptr;
    DevGroup *d_sgroups = nullptr; /* series-level (consecutive same-series
                                      groups merged) for aggregation */
    int nsgroups = 0;
    std::vector<int32_t> sgroup_span; /* page-groups per series-group */
    std::vector<int64_t> sgroup_row_off; /* first row of each series-group */
    DevGroup *d_sgroups_out = nullptr; /* series-group layout of the fused
                                          compacted output (rebuilt per scan) */
    int32_t *d_sgroup_first = nullptr; /* [nsgroups+1] first page-group idx */
    int64_t *d_blocksums = nullptr;    /* device-scan scratch */
    hipEvent_t sev[6];                 /* fused-scan phase events */
    bool sev_init = false;
    bool pending = false;              /* an async fused scan is in flight */
    int64_t *d_sp_start = nullptr;
    int64_t *d_sp_cnt = nullptr;
    int64_t *d_out_off = nullptr;
    int64_t *d_g_t0sel = nullptr; /* per page-group: ts of first selected
                                     row / RLE delta (k_spans_rle out) */
    int64_t *d_g_delta = nullptr;
    int max_span = 0; /* max page-groups per series-group */
    /* fused-scan active-chunk list (lazy; sized to the largest slot's
       chunk table) */
    int *d_gor_active = nullptr;
    int *d_gor_nactive = nullptr;
    size_t gor_active_cap = 0;
    /* string decode scratch (lazy, cached across gs_decode_str calls) */
    uint8_t *d_str_scratch = nullptr;
    size_t str_scratch_cap = 0;
    int64_t *d_str_sz = nullptr;  /* per-row string byte length (0=null) */
    int64_t *d_str_pos = nullptr; /* per-row start index into scratch */
    int64_t *d_str_bsums = nullptr; /* offset-scan block sums over rows */
    size_t str_bsums_cap = 0;
    GsTimeRange *d_ranges = nullptr;
    size_t ranges_cap = 0;
    uint8_t *d_valid = nullptr; /* lazily allocated internal validity bytes */
    uint8_t *d_mask = nullptr;  /* value-predicate row mask (lazy) */
    int64_t *d_sel_cnt = nullptr; /* per-group masked selected counts (lazy) */
    bool any_nulls_field = false;
    /* agg partials: ngroups x nbuckets cells, cached across scans */
    double *d_pmax = nullptr;
    double *d_psum = nullptr;
    long long *d_pcnt = nullptr;
    size_t partials_cap = 0;
};

extern "C" {

int gs_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

GsCtx *gs_ctx_create(int device) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess || n == 0) {
        fail(GS_ERR_NO_GPU, "no HIP device available — cnosdb_gs has no CPU fallback");
        return nullptr;
    }
    if (device < 0 || device >= n) {
        fail(GS_ERR, "device index out of range");
        return nullptr;
    }
    HIP_TRY_NULL(hipSetDevice(device));
    GsCtx *ctx = new GsCtx();
    ctx->device = device;
    if (hipStreamCreate(&ctx->stream) != hipSuccess ||
        hipMalloc(&ctx->d_err, sizeof(unsigned)) != hipSuccess) {
        fail(GS_ERR, "ctx init failed");
        delete ctx;
        return nullptr;
    }
    hipMemset(ctx->d_err, 0, sizeof(unsigned));
    return ctx;
}

void gs_ctx_destroy(GsCtx *ctx) {
    if (!ctx) return;
    hipSetDevice(ctx->device);
    hipStreamSynchronize(ctx->stream);
    hipFree(ctx->d_err);
    hipStreamDestroy(ctx->stream);
    delete ctx;
}

GsGroupSet *gs_groups_upload(GsCtx *ctx, const GsColumnGroupDesc *groups,
                             size_t ngroups, int validate_crc) {
    if (!ctx || !groups || ngroups == 0) {
        fail(GS_ERR, "bad args to gs_groups_upload");
        return nullptr;
    }
    HIP_TRY_NULL(hipSetDevice(ctx->device));
    uint32_t ncols = groups[0].npages;
    for (size_t g = 0; g < ngroups; g++) {
        if (groups[g].npages != ncols || ncols == 0) {
            fail(GS_ERR, "all groups must have the same page schema");
            return nullptr;
        }
        if (groups[g].pages[0].ctype != GS_CT_TIME) {
            fail(GS_ERR, "pages[0] of every group must be the time page");
            return nullptr;
        }
    }

    GsGroupSet *set = new GsGroupSet();
    set->ctx = ctx;
    set->ngroups = ngroups;
    set->ncols = ncols;
    set->slots.resize(ncols);
    set->row_offsets.resize(ngroups);

    /* pass 1: layout + validation + classification */
    struct HostPage {
        const uint8_t *bitset_src;
        const uint8_t *data_src;
        uint32_t bitset_len;
        uint32_t col;
        DevPage dp;
    };
    std::vector<HostPage> staged;
    staged.reserve(ngroups * ncols);
    size_t blob = 0;
    int64_t rows = 0;
    auto align16 = [](size_t x) { return (x + 15) & ~size_t(15); };
    for (size_t g = 0; g < ngroups; g++) {
        set->row_offsets[g] = rows;
        uint32_t nrows_g = groups[g].pages[0].num_values;
        for (uint32_t c = 0; c < ncols; c++) {
            const GsPageSpec &ps = groups[g].pages[c];
            if (ps.len < 16) { fail(GS_ERR_FORMAT, "page too short"); gs_groups_free(set); return nullptr; }
            uint32_t bl = (uint32_t(ps.bytes[0]) << 24) | (uint32_t(ps.bytes[1]) << 16) |
                          (uint32_t(ps.bytes[2]) << 8) | ps.bytes[3];
            uint64_t dl64 = 0;
            for (int k = 0; k < 8; k++) dl64 = (dl64 << 8) | ps.bytes[4 + k];
            if (16ull + bl > ps.len) { fail(GS_ERR_FORMAT, "bitset overruns page"); gs_groups_free(set); return nullptr; }
            if (dl64 != ps.num_values || ps.num_values != nrows_g) {
                fail(GS_ERR_FORMAT, "row count mismatch in group");
                gs_groups_free(set); return nullptr;
            }
            uint64_t data_len = ps.len - 16 - bl;
            const uint8_t *bitset = ps.bytes + 16;
            const uint8_t *data = ps.bytes + 16 + bl;
            if (validate_crc) {
                uint32_t crc = (uint32_t(ps.bytes[12]) << 24) | (uint32_t(ps.bytes[13]) << 16) |
                               (uint32_t(ps.bytes[14]) << 8) | ps.bytes[15];
                if (gs_crc32(data, data_len) != crc) {
                    fail(GS_ERR_CRC, "page crc32 mismatch");
                    gs_groups_free(set); return nullptr;
                }
            }
            /* all_valid: full bytes 0xff, partial last byte has low bits set */
            bool av = true;
            uint32_t full = ps.num_values / 8, restbits = ps.num_values % 8;
            if (bl * 8 < ps.num_values) av = false;
            else {
                for (uint32_t k = 0; k < full; k++)
                    if (bitset[k] != 0xff) { av = false; break; }
                if (av && restbits) {
                    uint8_t mask = uint8_t((1u << restbits) - 1);
                    if ((bitset[full] & mask) != mask) av = false;
                }
            }
            HostPage hp;
            hp.bitset_src = bitset;
            hp.data_src = data;
            hp.bitset_len = bl;
            hp.col = c;
            hp.dp.bitset_off = blob;
            blob += bl;
            blob = align16(blob);
            hp.dp.data_off = blob;
            blob += data_len;
            blob = align16(blob);
            blob += 48; /* tail pad: the Gorilla fast path prefetches whole
                           words up to ~32 B past the data end (guarded by
                           its bit budget, garbage never interpreted) */
            hp.dp.row_off = rows;
            hp.dp.data_len = uint32_t(data_len);
            hp.dp.nrows = ps.num_values;
            hp.dp.ctype = ps.ctype;
            hp.dp.enc = data_len ? data[0] : 0;
            hp.dp.sub = (data_len >= 2 && (hp.dp.enc == GS_ENC_DELTA ||
                                           hp.dp.enc == GS_ENC_DELTATS))
                            ? (data[1] >> 4) : 0;
            hp.dp.all_valid = av ? 1 : 0;
            hp.dp.grp = uint32_t(g);
            if (c == 0 && !av) {
                fail(GS_ERR_FORMAT, "time page must be fully valid");
                gs_groups_free(set); return nullptr;
            }
            if (c > 0 && !av) set->any_nulls_field = true;
            staged.push_back(hp);
        }
        rows += nrows_g;
    }
    set->total_rows = rows;
    set->blob_len = blob;

    /* allocate + upload blob through chunked pinned staging (double-
     * buffered; the pages sit in arbitrary user memory, so per-page
     * hipMemcpyAsync would cost ~4 us apiece — 246k copies measured) */
    if (hipMalloc(&set->d_blob, blob ? blob : 16) != hipSuccess) {
        fail(GS_ERR, "hipMalloc blob failed (out of HBM?)");
        gs_groups_free(set); return nullptr;
    }
    {
        const size_t CHUNK = size_t(256) << 20;
        uint8_t *stage[2] = {nullptr, nullptr};
        hipEvent_t evs[2];
        for (int k = 0; k < 2; k++) {
            if (hipHostMalloc(&stage[k], CHUNK) != hipSuccess) {
                fail(GS_ERR, "hipHostMalloc staging failed");
                gs_groups_free(set); return nullptr;
            }
            hipEventCreate(&evs[k]);
            hipEventRecord(evs[k], ctx->stream);
        }
        int cur = 0;
        size_t base = 0, fill_end = 0;
        size_t pi = 0;
        while (pi < staged.size()) {
            hipEventSynchronize(evs[cur]);
            base = staged[pi].dp.bitset_off;
            fill_end = base;
            size_t start_pi = pi;
            while (pi < staged.size()) {
                const HostPage &hp = staged[pi];
                size_t page_end = hp.dp.data_off + hp.dp.data_len;
                if (page_end - base > CHUNK) break;
                memcpy(stage[cur] + (hp.dp.bitset_off - base), hp.bitset_src,
                       hp.bitset_len);
                if (hp.dp.data_len)
                    memcpy(stage[cur] + (hp.dp.data_off - base), hp.data_src,
                           hp.dp.data_len);
                fill_end = page_end;
                pi++;
            }
            if (pi == start_pi) { /* single page larger than CHUNK */
                fail(GS_ERR, "page larger than staging chunk");
                for (int k = 0; k < 2; k++) { hipHostFree(stage[k]); hipEventDestroy(evs[k]); }
                gs_groups_free(set); return nullptr;
            }
            hipMemcpyAsync(set->d_blob + base, stage[cur], fill_end - base,
                           hipMemcpyHostToDevice, ctx->stream);
            hipEventRecord(evs[cur], ctx->stream);
            cur ^= 1;
        }
        hipStreamSynchronize(ctx->stream);
        for (int k = 0; k < 2; k++) { hipHostFree(stage[k]); hipEventDestroy(evs[k]); }
    }

    /* classify + upload page tables */
    for (uint32_t c = 0; c < ncols; c++) set->slots[c].ctype = 255;
    for (auto &hp : staged) {
        SlotPages &sp = set->slots[hp.col];
        if (sp.ctype == 255) sp.ctype = hp.dp.ctype;
        if (hp.dp.ctype != sp.ctype) {
            fail(GS_ERR, "mixed ctypes in one column slot");
            gs_groups_free(set); return nullptr;
        }
        int cls = PC_SEQ;
        bool int_ct = hp.dp.ctype == GS_CT_TIME || hp.dp.ctype == GS_CT_I64 ||
                      hp.dp.ctype == GS_CT_U64;
        if (hp.dp.ctype == GS_CT_STR) {
            cls = PC_STR;
            /* scratch cap: decoded payload length (snappy preamble
               varint, string.rs:197) or the raw payload (Null enc) */
            int64_t cap = 0;
            if (hp.dp.data_len >= 3 && hp.dp.enc == GS_ENC_SNAPPY) {
                uint64_t v = 0;
                int sh = 0;
                for (uint32_t i = 2; i < hp.dp.data_len && i < 12; i++) {
                    uint8_t b = hp.data_src[i];
                    v |= uint64_t(b & 0x7f) << sh;
                    sh += 7;
                    if (!(b & 0x80)) break;
                }
                cap = int64_t(v);
            } else if (hp.dp.data_len >= 1 && hp.dp.enc == GS_ENC_NULL) {
                cap = int64_t(hp.dp.data_len) - 1;
            }
            sp.str_scr.push_back(sp.str_total);
            sp.str_total += (cap + 15) & ~int64_t(15);
        } else if (hp.dp.all_valid && hp.dp.sub == 2 && int_ct && hp.dp.data_len > 2) {
            if (hp.dp.enc == GS_ENC_DELTATS) cls = PC_RLE_TS;
            else if (hp.dp.enc == GS_ENC_DELTA) cls = PC_RLE_I64;
        } else if (hp.dp.all_valid && hp.dp.sub == 1 && int_ct &&
                   (hp.dp.enc == GS_ENC_DELTATS || hp.dp.enc == GS_ENC_DELTA) &&
                   hp.dp.data_len > 2) {
            cls = PC_S8B;
        } else if (hp.dp.enc == GS_ENC_GORILLA &&
                   hp.dp.ctype == GS_CT_F64 && hp.dp.data_len >= 10) {
            cls = hp.dp.all_valid ? PC_GOR : PC_GORN;
        } else if (hp.dp.all_valid && hp.dp.enc == GS_ENC_NULL &&
                   hp.dp.ctype != GS_CT_BOOL && hp.dp.data_len >= 1) {
            cls = PC_RAW;
        } else if (hp.dp.all_valid && hp.dp.enc == GS_ENC_BITPACK &&
                   hp.dp.ctype == GS_CT_BOOL && hp.dp.data_len >= 3) {
            cls = PC_BOOL;
        }
        sp.host[cls].push_back(hp.dp);
    }
    for (uint32_t c = 0; c < ncols; c++) {
        SlotPages &sp = set->slots[c];
        /* longest Gorilla pages first: decode time scales with compressed
           bits, so launching the slow pages early removes the ragged tail */
        std::stable_sort(sp.host[PC_GOR].begin(), sp.host[PC_GOR].end(),
                         [](const DevPage &a, const DevPage &b) {
                             return a.data_len > b.data_len;
                         });
        for (int k = 0; k < PC_NCLASS; k++) {
            sp.n[k] = int(sp.host[k].size());
            if (sp.n[k]) {
                if (hipMalloc(&sp.dev[k], sp.n[k] * sizeof(DevPage)) != hipSuccess) {
                    fail(GS_ERR, "hipMalloc page table failed");
                    gs_groups_free(set); return nullptr;
                }
                hipMemcpyAsync(sp.dev[k], sp.host[k].data(),
                               sp.n[k] * sizeof(DevPage),
                               hipMemcpyHostToDevice, ctx->stream);
            }
        }
        if (sp.n[PC_STR]) {
            if (hipMalloc(&sp.d_str_scr, sp.n[PC_STR] * sizeof(int64_t)) != hipSuccess) {
                fail(GS_ERR, "hipMalloc str scratch table failed");
                gs_groups_free(set); return nullptr;
            }
            hipMemcpyAsync(sp.d_str_scr, sp.str_scr.data(),
                           sp.n[PC_STR] * sizeof(int64_t),
                           hipMemcpyHostToDevice, ctx->stream);
        }
        for (int gcls = 0; gcls < 2; gcls++) {
            /* chunk skeletons (page-major) for the two Gorilla classes,
               states filled by k_gor_sync / k_gor_sync_null.  Chunk size
               is tunable for sweeps (GS_GOR_CHUNK env). */
            const int pc = gcls == 0 ? PC_GOR : PC_GORN;
            if (!sp.n[pc]) continue;
            static uint32_t chunk_rows = [] {
                const char *e = getenv("GS_GOR_CHUNK");
                long v = e ? atol(e) : 0;
                return uint32_t(v >= 64 ? v : GOR_CHUNK);
            }();
            std::vector<int32_t> cbase(sp.n[pc] + 1);
            std::vector<DevGorChunk> hch;
            int32_t acc = 0;
            for (int i = 0; i < sp.n[pc]; i++) {
                const DevPage &pg = sp.host[pc][i];
                cbase[i] = acc;
                uint32_t nch =
                    pg.nrows ? (pg.nrows + chunk_rows - 1) / chunk_rows : 1;
                for (uint32_t k = 0; k < nch; k++) {
                    DevGorChunk c{};
                    c.data_off = pg.data_off;
                    c.bitset_off = pg.bitset_off;
                    c.row_off = pg.row_off;
                    c.grp = pg.grp;
                    c.row0 = k * chunk_rows;
                    c.cnt = pg.nrows > c.row0
                                ? (pg.nrows - c.row0 < chunk_rows
                                       ? pg.nrows - c.row0 : chunk_rows)
                                : 0;
                    c.data_len = pg.data_len;
                    c.meaningful = 64;
                    c.last = uint8_t(k == nch - 1);
                    hch.push_back(c);
                }
                acc += int32_t(nch);
            }
            cbase[sp.n[pc]] = acc;
            /* layout experiment (GS_GOR_ORDER=interleave, PC_GOR only):
               chunk-position-major table — a wave's 64 lanes then write
               64 DIFFERENT series' output regions instead of one series'
               contiguous region (DRAM channel spread).  Ragged pages are
               padded with cnt=0 entries the decode kernels skip. */
            static int il = [] {
                const char *e = getenv("GS_GOR_ORDER");
                return e && strcmp(e, "interleave") == 0;
            }();
            int il_stride = 0;
            if (il && gcls == 0 && sp.n[pc] > 0) {
                uint32_t maxc = 0;
                for (int i = 0; i < sp.n[pc]; i++)
                    maxc = uint32_t(cbase[i + 1] - cbase[i]) > maxc
                               ? uint32_t(cbase[i + 1] - cbase[i]) : maxc;
                if (maxc > 1) {
                    il_stride = sp.n[pc];
                    std::vector<DevGorChunk> h2(size_t(maxc) * sp.n[pc]);
                    for (auto &c : h2) { c = DevGorChunk{}; c.cnt = 0; }
                    for (int i = 0; i < sp.n[pc]; i++) {
                        int nchi = cbase[i + 1] - cbase[i];
                        for (int k = 0; k < nchi; k++)
                            h2[size_t(k) * sp.n[pc] + i] = hch[cbase[i] + k];
                    }
                    hch.swap(h2);
                    acc = int32_t(hch.size());
                }
            }
            DevGorChunk **dchp = gcls == 0 ? &sp.d_gor_chunks
                                           : &sp.d_gorn_chunks;
            int32_t **dcbp = gcls == 0 ? &sp.d_gor_chunk_base
                                       : &sp.d_gorn_chunk_base;
            (gcls == 0 ? sp.n_gor_chunks : sp.n_gorn_chunks) = acc;
            if (hipMalloc(dchp, hch.size() * sizeof(DevGorChunk)) !=
                    hipSuccess ||
                hipMalloc(dcbp, cbase.size() * sizeof(int32_t)) !=
                    hipSuccess) {
                fail(GS_ERR, "hipMalloc gor chunk table failed");
                gs_groups_free(set); return nullptr;
            }
            hipMemcpyAsync(*dchp, hch.data(),
                           hch.size() * sizeof(DevGorChunk),
                           hipMemcpyHostToDevice, ctx->stream);
            hipMemcpyAsync(*dcbp, cbase.data(),
                           cbase.size() * sizeof(int32_t),
                           hipMemcpyHostToDevice, ctx->stream);
            /* hch/cbase are stack-local: wait for the staged copies before
               they go out of scope (upload path, not scan time) */
            hipStreamSynchronize(ctx->stream);
            /* one-time sync-point pre-pass (upload/setup, not scan time) */
            if (gcls == 0)
                hipLaunchKernelGGL(k_gor_sync,
                                   dim3(grid_for(sp.n[pc], 256)), dim3(256),
                                   0, ctx->stream, set->d_blob, sp.dev[pc],
                                   sp.n[pc], *dcbp, *dchp, chunk_rows,
                                   il_stride);
            else
                hipLaunchKernelGGL(k_gor_sync_null,
                                   dim3(grid_for(sp.n[pc], 256)), dim3(256),
                                   0, ctx->stream, set->d_blob, sp.dev[pc],
                                   sp.n[pc], *dcbp, *dchp, chunk_rows);
        }
    }

    /* group table + span scratch */
    std::vector<DevGroup> hg(ngroups);
    for (size_t g = 0; g < ngroups; g++) {
        hg[g].row_off = set->row_offsets[g];
        hg[g].nrows = int32_t(groups[g].pages[0].num_values);
        hg[g].pad = 0;
    }
    /* series-level groups: merge consecutive page-groups of one series
       (their rows are contiguous and time-ordered) */
    std::vector<DevGroup> hsg;
    for (size_t g = 0; g < ngroups; g++) {
        if (!hsg.empty() && g > 0 &&
            groups[g].series_id == groups[g - 1].series_id) {
            hsg.back().nrows += hg[g].nrows;
            set->sgroup_span.back()++;
        } else {
            hsg.push_back(hg[g]);
            set->sgroup_span.push_back(1);
            set->sgroup_row_off.push_back(hg[g].row_off);
        }
    }
    set->nsgroups = int(hsg.size());
    if (hipMalloc(&set->d_sgroups, hsg.size() * sizeof(DevGroup)) != hipSuccess) {
        fail(GS_ERR, "hipMalloc sgroup table failed");
        gs_groups_free(set); return nullptr;
    }
    hipMemcpyAsync(set->d_sgroups, hsg.data(), hsg.size() * sizeof(DevGroup),
                   hipMemcpyHostToDevice, ctx->stream);
    size_t nblocks = (ngroups + SCAN_BLOCK * SCAN_ITEMS - 1) /
                     (SCAN_BLOCK * SCAN_ITEMS);
    std::vector<int32_t> sgfirst(hsg.size() + 1);
    {
        int32_t acc32 = 0;
        for (size_t s = 0; s < hsg.size(); s++) {
            sgfirst[s] = acc32;
            acc32 += set->sgroup_span[s];
        }
        sgfirst[hsg.size()] = acc32;
    }
    if (hipMalloc(&set->d_groups, ngroups * sizeof(DevGroup)) != hipSuccess ||
        hipMalloc(&set->d_sp_start, ngroups * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_sp_cnt, ngroups * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_out_off, (ngroups + 1) * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_blocksums, (nblocks + 1) * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_sgroup_first, sgfirst.size() * sizeof(int32_t)) != hipSuccess ||
        hipMalloc(&set->d_sgroups_out, hsg.size() * sizeof(DevGroup)) != hipSuccess ||
        hipMalloc(&set->d_g_t0sel, ngroups * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_g_delta, ngroups * sizeof(int64_t)) != hipSuccess) {
        fail(GS_ERR, "hipMalloc group tables failed");
        gs_groups_free(set); return nullptr;
    }
    for (int32_t sp : set->sgroup_span)
        if (sp > set->max_span) set->max_span = sp;
    hipMemcpyAsync(set->d_sgroup_first, sgfirst.data(),
                   sgfirst.size() * sizeof(int32_t), hipMemcpyHostToDevice,
                   ctx->stream);
    hipMemcpyAsync(set->d_groups, hg.data(), ngroups * sizeof(DevGroup),
                   hipMemcpyHostToDevice, ctx->stream);
    HIP_TRY_NULL(hipStreamSynchronize(ctx->stream));
    return set;
}

void gs_groups_free(GsGroupSet *set) {
    if (!set) return;
    hipSetDevice(set->ctx->device);
    hipStreamSynchronize(set->ctx->stream);
    for (auto &sp : set->slots) {
        for (int k = 0; k < PC_NCLASS; k++)
            if (sp.dev[k]) hipFree(sp.dev[k]);
        if (sp.d_str_scr) hipFree(sp.d_str_scr);
        if (sp.d_gor_chunks) hipFree(sp.d_gor_chunks);
        if (sp.d_gor_chunk_base) hipFree(sp.d_gor_chunk_base);
        if (sp.d_gorn_chunks) hipFree(sp.d_gorn_chunks);
        if (sp.d_gorn_chunk_base) hipFree(sp.d_gorn_chunk_base);
    }
    if (set->d_gor_active) hipFree(set->d_gor_active);
    if (set->d_gor_nactive) hipFree(set->d_gor_nactive);
    if (set->d_str_scratch) hipFree(set->d_str_scratch);
    if (set->d_str_sz) hipFree(set->d_str_sz);
    if (set->d_str_pos) hipFree(set->d_str_pos);
    if (set->d_str_bsums) hipFree(set->d_str_bsums);
    hipFree(set->d_blob);
    hipFree(set->d_groups);
    if (set->d_sgroups) hipFree(set->d_sgroups);
    if (set->d_sgroups_out) hipFree(set->d_sgroups_out);
    if (set->d_sgroup_first) hipFree(set->d_sgroup_first);
    if (set->d_blocksums) hipFree(set->d_blocksums);
    if (set->sev_init)
        for (int k = 0; k < 6; k++) hipEventDestroy(set->sev[k]);
    hipFree(set->d_sp_start);
    hipFree(set->d_sp_cnt);
    hipFree(set->d_out_off);
    if (set->d_g_t0sel) hipFree(set->d_g_t0sel);
    if (set->d_g_delta) hipFree(set->d_g_delta);
    if (set->d_ranges) hipFree(set->d_ranges);
    if (set->d_valid) hipFree(set->d_valid);
    if (set->d_mask) hipFree(set->d_mask);
    if (set->d_sel_cnt) hipFree(set->d_sel_cnt);
    if (set->d_pmax) hipFree(set->d_pmax);
    if (set->d_psum) hipFree(set->d_psum);
    if (set->d_pcnt) hipFree(set->d_pcnt);
    delete set;
}

/* debug: copy a slot's GORN chunk table to host (repro tooling only) */
int64_t gs_debug_gorn_table(GsCtx *ctx, GsGroupSet *set, uint32_t col,
                            DevGorChunk *out, int64_t cap) {
    if (!ctx || !set || col >= set->ncols) return -1;
    SlotPages &sp = set->slots[col];
    int64_t n = sp.n_gorn_chunks;
    if (!out || cap < n) return n;
    hipSetDevice(ctx->device);
    hipStreamSynchronize(ctx->stream);
    if (hipMemcpy(out, sp.d_gorn_chunks, size_t(n) * sizeof(DevGorChunk),
                  hipMemcpyDeviceToHost) != hipSuccess)
        return -1;
    return n;
}

int64_t gs_debug_read_blob(GsCtx *ctx, GsGroupSet *set, uint64_t off,
                           uint64_t len, uint8_t *out) {
    if (!ctx || !set || off + len > set->blob_len) return -1;
    hipSetDevice(ctx->device);
    hipStreamSynchronize(ctx->stream);
    if (hipMemcpy(out, set->d_blob + off, len, hipMemcpyDeviceToHost) !=
        hipSuccess)
        return -1;
    return int64_t(len);
}

int64_t gs_debug_rerun_gorn_sync(GsCtx *ctx, GsGroupSet *set,
                                 uint32_t col) {
    if (!ctx || !set || col >= set->ncols) return -1;
    SlotPages &sp = set->slots[col];
    if (!sp.n[PC_GORN]) return 0;
    hipSetDevice(ctx->device);
    const char *e = getenv("GS_GOR_CHUNK");
    long v = e ? atol(e) : 0;
    uint32_t chunk_rows = uint32_t(v >= 64 ? v : GOR_CHUNK);
    hipLaunchKernelGGL(k_gor_sync_null, dim3(grid_for(sp.n[PC_GORN], 256)),
                       dim3(256), 0, ctx->stream, set->d_blob,
                       sp.dev[PC_GORN], sp.n[PC_GORN], sp.d_gorn_chunk_base,
                       sp.d_gorn_chunks, chunk_rows);
    hipStreamSynchronize(ctx->stream);
    return 1;
}

__global__ void k_debug_crc(const uint8_t *__restrict__ p, uint64_t len,
                            uint64_t *__restrict__ out) {
    /* simple order-independent mix so any byte difference shows */
    uint64_t acc = 0;
    for (uint64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < len;
         i += uint64_t(gridDim.x) * blockDim.x)
        acc += (uint64_t(p[i]) + 1) * (i + 0x9e3779b97f4a7c15ULL);
    atomicAdd((unsigned long long *)out, (unsigned long long)acc);
}

int64_t gs_debug_kernel_crc(GsCtx *ctx, GsGroupSet *set, uint64_t off,
                            uint64_t len, uint64_t *out_sum) {
    if (!ctx || !set || off + len > set->blob_len) return -1;
    hipSetDevice(ctx->device);
    uint64_t *d;
    if (hipMalloc(&d, 8) != hipSuccess) return -1;
    hipMemset(d, 0, 8);
    hipLaunchKernelGGL(k_debug_crc, dim3(64), dim3(256), 0, ctx->stream,
                       set->d_blob + off, len, d);
    hipStreamSynchronize(ctx->stream);
    hipMemcpy(out_sum, d, 8, hipMemcpyDeviceToHost);
    hipFree(d);
    return 0;
}

int64_t gs_set_rows(const GsGroupSet *set) { return set ? set->total_rows : -1; }
int64_t gs_set_series(const GsGroupSet *set) { return set ? set->nsgroups : -1; }

/* Pushed-down COUNT answered from page metadata, never from decoded data
 * (PushDownAggregateReader, tskv/src/reader/pushdown_agg_reader.rs:39-106:
 * sums num_values — rows INCLUDING nulls — of the column's pages; SURVEY.md
 * A.10.12). */
int64_t gs_count_pushdown(const GsGroupSet *set, uint32_t col) {
    if (!set || col >= set->ncols) return -1;
    int64_t n = 0;
    const SlotPages &sp = set->slots[col];
    for (int k = 0; k < PC_NCLASS; k++)
        for (const DevPage &p : sp.host[k]) n += p.nrows;
    return n;
}
int64_t gs_set_groups(const GsGroupSet *set) { return set ? int64_t(set->ngroups) : -1; }
GsStatus gs_set_row_offsets(const GsGroupSet *set, int64_t *out) {
    if (!set || !out) return fail(GS_ERR, "bad args");
    memcpy(out, set->row_offsets.data(), set->ngroups * sizeof(int64_t));
    return GS_OK;
}

/* --------------------------------------------- GROUP BY tag
 * TSBS GROUP-BY-hostname over the decoded varbinary tag column.  In the
 * reference a tag is part of the SeriesKey and therefore CONSTANT within
 * a series (tag columns are materialized per series from the key,
 * reader/series.rs:23-100); the GPU group-by exploits exactly that:
 * hash the tag of each series' first row into a device dictionary
 * (first-occurrence = min series index, deterministic), then merge the
 * per-series bucket partials of the preceding aggregate scan per
 * (tag, bucket) in series order — a deterministic fixed-order reduction,
 * like the aggregate itself. */
__device__ __forceinline__ uint64_t dev_fnv1a(const uint8_t *p, int64_t n) {
    uint64_t h = 1469598103934665603ULL;
    for (int64_t i = 0; i < n; i++) {
        h ^= p[i];
        h *= 1099511628211ULL;
    }
    return h;
}

/* dictionary slot, one atomic u64: [fp32 (nonzero) | min_series32].
 * Claim by CAS from 0; equal-tag series then atomicMin the whole word —
 * the high fp bits are equal so the min acts on the series index,
 * keeping the deterministic first-occurrence owner.  fp collisions are
 * resolved by byte comparison against the slot owner's tag (the owner
 * field is valid from the moment the slot exists — single-word claim). */
__global__ void k_tag_series(const DevGroup *__restrict__ sg, int nsg,
                             const int64_t *__restrict__ str_pos,
                             const int64_t *__restrict__ str_sz,
                             const uint8_t *__restrict__ scratch,
                             unsigned long long *__restrict__ table,
                             int cap_mask, int32_t *__restrict__ slot_of,
                             unsigned *__restrict__ err) {
    for (int g = blockIdx.x * blockDim.x + threadIdx.x; g < nsg;
         g += gridDim.x * blockDim.x) {
        const int64_t r0 = sg[g].row_off; /* tag of the series' first row */
        const uint8_t *tb = scratch + str_pos[r0];
        const int64_t tn = str_sz[r0];
        const uint64_t fphi = (dev_fnv1a(tb, tn) >> 32) | 1ULL;
        const unsigned long long claim =
            (fphi << 32) | (unsigned long long)(uint32_t)g;
        int slot = int(fphi * 0x9e3779b1u) & cap_mask;
        bool placed = false;
        for (int probes = 0; probes <= cap_mask && !placed; probes++) {
            unsigned long long cur = table[slot];
            if (cur == 0) cur = atomicCAS(&table[slot], 0ULL, claim);
            if (cur == 0 || (cur >> 32) == fphi) {
                bool same = true;
                if (cur != 0) { /* verify bytes vs the slot owner */
                    const int own = int(cur & 0xffffffffULL);
                    if (own != g) {
                        const int64_t ro = sg[own].row_off;
                        if (str_sz[ro] != tn) same = false;
                        else {
                            const uint8_t *ob = scratch + str_pos[ro];
                            for (int64_t i = 0; i < tn && same; i++)
                                same = ob[i] == tb[i];
                        }
                    }
                }
                if (same) {
                    atomicMin(&table[slot], claim);
                    slot_of[g] = slot;
                    placed = true;
                    break;
                }
            }
            slot = (slot + 1) & cap_mask;
        }
        if (!placed) atomicOr(err, DERR_FORMAT); /* table full */
    }
}

/* thread per (gid, bucket): ordered walk of the gid's series list */
__global__ void k_agg_tag(const int32_t *__restrict__ csr_off,
                          const int32_t *__restrict__ csr_sg, int ngids,
                          int nbuckets, const double *__restrict__ pmax,
                          const double *__restrict__ psum,
                          const long long *__restrict__ pcnt,
                          double *__restrict__ out_max,
                          double *__restrict__ out_sum,
                          long long *__restrict__ out_cnt) {
    int64_t total = int64_t(ngids) * nbuckets;
    for (int64_t i = blockIdx.x * int64_t(blockDim.x) + threadIdx.x;
         i < total; i += int64_t(gridDim.x) * blockDim.x) {
        const int gid = int(i / nbuckets);
        const int b = int(i % nbuckets);
        double mx = -__builtin_inf(), sm = 0.0;
        long long c = 0;
        for (int32_t k = csr_off[gid]; k < csr_off[gid + 1]; k++) {
            const size_t cell = size_t(csr_sg[k]) * nbuckets + b;
            const long long pc = pcnt[cell];
            if (pc <= 0) continue;
            const double m2 = pmax[cell];
            if (m2 > mx) mx = m2;
            sm += psum[cell];
            c += pc;
        }
        out_max[i] = mx;
        out_sum[i] = sm;
        out_cnt[i] = c;
    }
}

static GsStatus check_dev_err(GsCtx *ctx) {
    unsigned e = 0;
    HIP_TRY(hipMemcpy(&e, ctx->d_err, sizeof(unsigned), hipMemcpyDeviceToHost));
    if (e) {
        hipMemset(ctx->d_err, 0, sizeof(unsigned));
        if (e & DERR_NONMONO)
            return fail(GS_ERR_FORMAT,
                        "decode: non-monotonic time page (corrupt input)");
        if (e & DERR_SHORT) return fail(GS_ERR_FORMAT, "decode: stream shorter than validity demands");
        return fail(GS_ERR_FORMAT, "decode: malformed page data");
    }
    return GS_OK;
}


GsStatus gs_decode(GsCtx *ctx, GsGroupSet *set, uint32_t col, void *d_out,
                   uint8_t *d_valid) {
    if (!ctx || !set || col >= set->ncols || !d_out)
        return fail(GS_ERR, "bad args to gs_decode");
    HIP_TRY(hipSetDevice(ctx->device));
    SlotPages &sp = set->slots[col];
    uint8_t ct = sp.ctype;
    if (sp.n[PC_STR])
        return fail(GS_ERR, "string column: use gs_decode_str");

    if (sp.n[PC_SEQ]) {
        int n = sp.n[PC_SEQ];
        int blocks = grid_for(n, 256);
        if (ct == GS_CT_F64)
            hipLaunchKernelGGL(k_seq_f64, dim3(blocks), dim3(256), 0, ctx->stream,
                               set->d_blob, sp.dev[PC_SEQ], n, (double *)d_out,
                               d_valid, ctx->d_err);
        else if (ct == GS_CT_BOOL)
            hipLaunchKernelGGL(k_seq_bool, dim3(blocks), dim3(256), 0, ctx->stream,
                               set->d_blob, sp.dev[PC_SEQ], n, (uint8_t *)d_out,
                               d_valid, ctx->d_err);
        else
            hipLaunchKernelGGL(k_seq_i64, dim3(blocks), dim3(256), 0, ctx->stream,
                               set->d_blob, sp.dev[PC_SEQ], n, (int64_t *)d_out,
                               d_valid, ctx->d_err);
    }
    if (sp.n[PC_GOR]) {
        int n = sp.n_gor_chunks;
        hipLaunchKernelGGL(k_gor_chunks, dim3(grid_for(n, GS_GOR_BLOCK)),
                           dim3(GS_GOR_BLOCK), 0, ctx->stream, set->d_blob,
                           sp.d_gor_chunks, n, (double *)d_out, ctx->d_err);
        if (d_valid)
            hipLaunchKernelGGL(k_valid_fill,
                               dim3(sp.n[PC_GOR] > 2048 ? 2048 : sp.n[PC_GOR]),
                               dim3(256), 0, ctx->stream, sp.dev[PC_GOR],
                               sp.n[PC_GOR], d_valid);
    }
    if (sp.n[PC_GORN]) {
        int n = sp.n_gorn_chunks;
        hipLaunchKernelGGL(k_gor_chunks_null, dim3(grid_for(n, GS_GOR_BLOCK)),
                           dim3(GS_GOR_BLOCK), 0, ctx->stream, set->d_blob,
                           sp.d_gorn_chunks, n, (double *)d_out, ctx->d_err);
        if (d_valid)
            hipLaunchKernelGGL(
                k_valid_expand,
                dim3(sp.n[PC_GORN] > 2048 ? 2048 : sp.n[PC_GORN]), dim3(256),
                0, ctx->stream, set->d_blob, sp.dev[PC_GORN], sp.n[PC_GORN],
                d_valid);
    }
    if (sp.n[PC_S8B]) {
        int n = sp.n[PC_S8B];
        hipLaunchKernelGGL(k_s8b_par, dim3(n > 65535 ? 65535 : n), dim3(256),
                           0, ctx->stream, set->d_blob, sp.dev[PC_S8B], n,
                           (int64_t *)d_out, d_valid, ctx->d_err);
    }
    if (sp.n[PC_BOOL]) {
        int n = sp.n[PC_BOOL];
        hipLaunchKernelGGL(k_bool_par, dim3(n > 65535 ? 65535 : n), dim3(256),
                           0, ctx->stream, set->d_blob, sp.dev[PC_BOOL], n,
                           (uint8_t *)d_out, d_valid, ctx->d_err);
    }
    if (sp.n[PC_RAW]) {
        int n = sp.n[PC_RAW];
        hipLaunchKernelGGL(k_raw_par, dim3(n > 65535 ? 65535 : n), dim3(256),
                           0, ctx->stream, set->d_blob, sp.dev[PC_RAW], n,
                           (int64_t *)d_out, d_valid, ctx->d_err);
    }
    if (sp.n[PC_RLE_TS])
        hipLaunchKernelGGL(k_rle_par,
                           dim3(sp.n[PC_RLE_TS] > 2048 ? 2048 : sp.n[PC_RLE_TS]),
                           dim3(256), 0, ctx->stream, set->d_blob,
                           sp.dev[PC_RLE_TS], sp.n[PC_RLE_TS], (int64_t *)d_out,
                           d_valid, 1, ctx->d_err);
    if (sp.n[PC_RLE_I64])
        hipLaunchKernelGGL(k_rle_par,
                           dim3(sp.n[PC_RLE_I64] > 2048 ? 2048 : sp.n[PC_RLE_I64]),
                           dim3(256), 0, ctx->stream, set->d_blob,
                           sp.dev[PC_RLE_I64], sp.n[PC_RLE_I64], (int64_t *)d_out,
                           d_valid, 0, ctx->d_err);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return check_dev_err(ctx);
}

/* GROUP BY tag over a completed aggregate scan: the tag column (decoded
 * by gs_decode_str: per-row pos/sz into its scratch) keys a dictionary of
 * the per-series tag (constant within a series, SeriesKey semantics);
 * outputs per-(tag, bucket) max/sum/count in deterministic
 * first-occurrence tag order.  out arrays are caller device buffers of
 * cap_gids * n_buckets; tag_rep_row[gid] (host) = a row whose string is
 * the tag value.  Requires the previous gs_scan on this set to have run
 * with the same n_buckets (the per-series partials are reused). */
GsStatus gs_groupby_tag(GsCtx *ctx, GsGroupSet *set, int tag_col,
                        int n_buckets, double *d_out_max, double *d_out_sum,
                        long long *d_out_count, int64_t *tag_rep_row,
                        int cap_gids, int *out_ngids) {
    if (!ctx || !set || !d_out_max || !d_out_sum || !d_out_count ||
        !out_ngids || n_buckets <= 0)
        return fail(GS_ERR, "bad args to gs_groupby_tag");
    if (!set->d_pmax ||
        set->partials_cap < size_t(set->nsgroups) * size_t(n_buckets))
        return fail(GS_ERR, "gs_groupby_tag needs a prior aggregate scan");
    if (!set->d_str_sz || !set->d_str_pos || !set->d_str_scratch)
        return fail(GS_ERR, "gs_groupby_tag needs a decoded string column");
    (void)tag_col;
    HIP_TRY(hipSetDevice(ctx->device));
    const int nsg = set->nsgroups;
    int cap = 64;
    while (cap < 2 * nsg) cap <<= 1;
    unsigned long long *d_table = nullptr;
    int32_t *d_slot_of = nullptr;
    if (hipMalloc(&d_table, size_t(cap) * 8) != hipSuccess ||
        hipMalloc(&d_slot_of, size_t(nsg) * 4) != hipSuccess) {
        if (d_table) hipFree(d_table);
        return fail(GS_ERR, "hipMalloc tag table failed");
    }
    HIP_TRY(hipMemsetAsync(d_table, 0, size_t(cap) * 8, ctx->stream));
    hipLaunchKernelGGL(k_tag_series, dim3(grid_for(nsg, 256)), dim3(256), 0,
                       ctx->stream, set->d_sgroups, nsg, set->d_str_pos,
                       set->d_str_sz, set->d_str_scratch, d_table, cap - 1,
                       d_slot_of, ctx->d_err);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    GsStatus st = check_dev_err(ctx);
    if (st != GS_OK) { hipFree(d_table); hipFree(d_slot_of); return st; }
    /* host finalize (tiny, deterministic): dense ids in first-occurrence
       order + CSR of the series of each tag */
    std::vector<unsigned long long> table(cap);
    std::vector<int32_t> slot_of(nsg);
    HIP_TRY(hipMemcpy(table.data(), d_table, size_t(cap) * 8,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(slot_of.data(), d_slot_of, size_t(nsg) * 4,
                      hipMemcpyDeviceToHost));
    hipFree(d_table);
    hipFree(d_slot_of);
    std::vector<std::pair<int, int>> owners; /* (min series, slot) */
    for (int sl = 0; sl < cap; sl++)
        if (table[sl])
            owners.push_back({int(table[sl] & 0xffffffffULL), sl});
    std::sort(owners.begin(), owners.end());
    const int ngids = int(owners.size());
    *out_ngids = ngids;
    if (ngids > cap_gids)
        return fail(GS_ERR, "gs_groupby_tag: cap_gids too small");
    std::vector<int32_t> gid_of_slot(cap, -1);
    for (int gidx = 0; gidx < ngids; gidx++)
        gid_of_slot[owners[gidx].second] = gidx;
    std::vector<int32_t> csr_off(ngids + 1, 0), csr_sg(nsg);
    for (int g = 0; g < nsg; g++) csr_off[gid_of_slot[slot_of[g]] + 1]++;
    for (int gidx = 0; gidx < ngids; gidx++) csr_off[gidx + 1] += csr_off[gidx];
    {
        std::vector<int32_t> cur(csr_off.begin(), csr_off.end() - 1);
        for (int g = 0; g < nsg; g++) /* series order within a tag */
            csr_sg[cur[gid_of_slot[slot_of[g]]]++] = g;
    }
    if (tag_rep_row)
        for (int gidx = 0; gidx < ngids; gidx++) {
            int sgi = owners[gidx].first;
            tag_rep_row[gidx] = size_t(sgi) < set->sgroup_row_off.size()
                                    ? set->sgroup_row_off[sgi] : -1;
        }
    int32_t *d_csr_off = nullptr, *d_csr_sg = nullptr;
    if (hipMalloc(&d_csr_off, (ngids + 1) * 4) != hipSuccess ||
        hipMalloc(&d_csr_sg, size_t(nsg) * 4) != hipSuccess) {
        if (d_csr_off) hipFree(d_csr_off);
        return fail(GS_ERR, "hipMalloc tag csr failed");
    }
    hipMemcpyAsync(d_csr_off, csr_off.data(), (ngids + 1) * 4,
                   hipMemcpyHostToDevice, ctx->stream);
    hipMemcpyAsync(d_csr_sg, csr_sg.data(), size_t(nsg) * 4,
                   hipMemcpyHostToDevice, ctx->stream);
    hipStreamSynchronize(ctx->stream);
    const int64_t total = int64_t(ngids) * n_buckets;
    const int blocks = int(total > int64_t(2048) * 256
                               ? 2048 : (total + 255) / 256);
    hipLaunchKernelGGL(k_agg_tag, dim3(blocks < 1 ? 1 : blocks), dim3(256),
                       0, ctx->stream, d_csr_off, d_csr_sg, ngids, n_buckets,
                       set->d_pmax, set->d_psum, set->d_pcnt, d_out_max,
                       d_out_sum, d_out_count);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    hipFree(d_csr_off);
    hipFree(d_csr_sg);
    return check_dev_err(ctx);
}

GsStatus gs_decode_str(GsCtx *ctx, GsGroupSet *set, uint32_t col,
                       int64_t *d_offsets, uint8_t *d_bytes,
                       int64_t bytes_cap, uint8_t *d_valid,
                       int64_t *total_bytes) {
    if (!ctx || !set || col >= set->ncols || !d_offsets || !d_bytes ||
        !total_bytes)
        return fail(GS_ERR, "bad args to gs_decode_str");
    SlotPages &sp = set->slots[col];
    if (sp.ctype != GS_CT_STR || sp.n[PC_STR] != int(sp.host[PC_STR].size()))
        return fail(GS_ERR, "column is not a string column");
    if (set->total_rows > int64_t(SCAN_BLOCK) * SCAN_ITEMS * 131072)
        return fail(GS_ERR, "string decode row limit exceeded (268M)");
    HIP_TRY(hipSetDevice(ctx->device));
    int n = sp.n[PC_STR];
    if (!set->d_str_scratch || set->str_scratch_cap < size_t(sp.str_total)) {
        if (set->d_str_scratch) hipFree(set->d_str_scratch);
        size_t cap = sp.str_total > 0 ? size_t(sp.str_total) : 16;
        if (hipMalloc(&set->d_str_scratch, cap) != hipSuccess)
            return fail(GS_ERR, "hipMalloc string scratch failed");
        set->str_scratch_cap = cap;
    }
    if (!set->d_str_sz) {
        if (hipMalloc(&set->d_str_sz, set->total_rows * 8) != hipSuccess ||
            hipMalloc(&set->d_str_pos, set->total_rows * 8) != hipSuccess)
            return fail(GS_ERR, "hipMalloc string row tables failed");
    }
    int nblocks = int((set->total_rows + SCAN_BLOCK * SCAN_ITEMS - 1) /
                      (SCAN_BLOCK * SCAN_ITEMS));
    if (set->str_bsums_cap < size_t(nblocks) + 1) {
        if (set->d_str_bsums) hipFree(set->d_str_bsums);
        if (hipMalloc(&set->d_str_bsums, (size_t(nblocks) + 1) * 8) != hipSuccess)
            return fail(GS_ERR, "hipMalloc string scan sums failed");
        set->str_bsums_cap = size_t(nblocks) + 1;
    }
    hipLaunchKernelGGL(k_str_decode, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, set->d_blob, sp.dev[PC_STR], n,
                       sp.d_str_scr, set->d_str_scratch, set->d_str_sz,
                       set->d_str_pos, d_valid, ctx->d_err);
    hipLaunchKernelGGL(k_scan_partials, dim3(nblocks), dim3(SCAN_BLOCK), 0,
                       ctx->stream, set->d_str_sz, int(set->total_rows),
                       d_offsets, set->d_str_bsums);
    hipLaunchKernelGGL(k_scan_fixup, dim3(1), dim3(64), 0, ctx->stream,
                       int(set->total_rows), d_offsets, set->d_str_bsums,
                       nblocks);
    hipLaunchKernelGGL(k_scan_add, dim3(grid_for(int(set->total_rows), 256)),
                       dim3(256), 0, ctx->stream, int(set->total_rows),
                       d_offsets, set->d_str_bsums);
    int64_t total = 0;
    HIP_TRY(hipMemcpyAsync(&total, d_offsets + set->total_rows, 8,
                           hipMemcpyDeviceToHost, ctx->stream));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    GsStatus st = check_dev_err(ctx);
    if (st != GS_OK) return st;
    if (total > bytes_cap)
        return fail(GS_ERR, "string bytes buffer too small");
    hipLaunchKernelGGL(k_str_gather, dim3(grid_for(int(set->total_rows), 256)),
                       dim3(256), 0, ctx->stream, set->d_str_scratch,
                       set->d_str_pos, set->d_str_sz, d_offsets,
                       set->total_rows, d_bytes);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    *total_bytes = total;
    return check_dev_err(ctx);
}

GsStatus gs_apply_tombstone(GsCtx *ctx, GsGroupSet *set, const int64_t *d_ts,
                            uint8_t *d_valid, const GsTimeRange *ranges,
                            size_t nranges) {
    if (!ctx || !set || !d_ts || !d_valid || !ranges || nranges == 0)
        return fail(GS_ERR, "bad args to gs_apply_tombstone");
    HIP_TRY(hipSetDevice(ctx->device));
    if (set->ranges_cap < nranges) {
        if (set->d_ranges) hipFree(set->d_ranges);
        HIP_TRY(hipMalloc(&set->d_ranges, nranges * sizeof(GsTimeRange)));
        set->ranges_cap = nranges;
    }
    HIP_TRY(hipMemcpyAsync(set->d_ranges, ranges, nranges * sizeof(GsTimeRange),
                           hipMemcpyHostToDevice, ctx->stream));
    int blocks = int(set->ngroups > 2048 ? 2048 : set->ngroups);
    hipLaunchKernelGGL(k_tombstone, dim3(blocks), dim3(256), 0, ctx->stream,
                       set->d_groups, int(set->ngroups), d_ts, d_valid,
                       set->d_ranges, int(nranges));
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    return GS_OK;
}

static bool fused_capable_fields(GsGroupSet *set, const GsScanSpec *spec,
                                 const int32_t *fields, int nf) {
    if (!(spec->d_out_ts && spec->d_out_val && spec->n_tombstones == 0 &&
          spec->value_pred.op == GS_PRED_NONE && !set->any_nulls_field &&
          set->slots[0].n[PC_RLE_TS] == int(set->ngroups)))
        return false;
    for (int f = 0; f < nf; f++) {
        uint32_t fc = 1 + uint32_t(fields[f]);
        if (fc >= set->ncols ||
            set->slots[fc].n[PC_GOR] != int(set->ngroups))
            return false;
    }
    return true;
}

static bool fused_capable(GsGroupSet *set, const GsScanSpec *spec) {
    int32_t f0 = spec->field_col;
    return fused_capable_fields(set, spec, &f0, 1);
}

/* Fused scan over ONE span/ts pass and nf field columns (TSBS
 * cpu-max-all-8, BASELINE config #3: the per-field loop of round 1 paid
 * the span search and the 8 B/row ts generation once PER FIELD).
 * Per-field outputs are strided: field i's compacted values go to
 * d_out_val + i*total_rows; its aggregates to d_agg_* + i*n_buckets. */
static GsStatus scan_fused_launch(GsCtx *ctx, GsGroupSet *set,
                                  const GsScanSpec *spec,
                                  const int32_t *fields, int nf) {
    HIP_TRY(hipSetDevice(ctx->device));
    if (!set->sev_init) {
        for (int k = 0; k < 6; k++) HIP_TRY(hipEventCreate(&set->sev[k]));
        set->sev_init = true;
    }
    hipEvent_t *ev = set->sev;
    int ng = int(set->ngroups);
    const DevPage *ts_pages = set->slots[0].dev[PC_RLE_TS];
    int nch_max = 0;
    for (int f = 0; f < nf; f++) {
        int nc = set->slots[1 + fields[f]].n_gor_chunks;
        nch_max = nc > nch_max ? nc : nch_max;
    }
    if (!set->d_gor_nactive &&
        hipMalloc(&set->d_gor_nactive, sizeof(int)) != hipSuccess)
        return fail(GS_ERR, "hipMalloc active counter failed");
    if (set->gor_active_cap < size_t(nch_max)) {
        if (set->d_gor_active) hipFree(set->d_gor_active);
        if (hipMalloc(&set->d_gor_active, size_t(nch_max) * sizeof(int)) !=
            hipSuccess)
            return fail(GS_ERR, "hipMalloc active list failed");
        set->gor_active_cap = size_t(nch_max);
    }
    int nblocks = (ng + SCAN_BLOCK * SCAN_ITEMS - 1) /
                  (SCAN_BLOCK * SCAN_ITEMS);
    if (nblocks > 2048)
        return fail(GS_ERR, "too many groups for the fused span scan");
    HIP_TRY(hipEventRecord(ev[0], ctx->stream));
    hipLaunchKernelGGL(k_spans_rle, dim3(grid_for(ng, 256)), dim3(256), 0,
                       ctx->stream, set->d_blob, ts_pages, ng,
                       spec->range.min_ts, spec->range.max_ts,
                       set->d_sp_start, set->d_sp_cnt, set->d_g_t0sel,
                       set->d_g_delta, ctx->d_err);
    /* device exclusive scan of span counts -> output offsets (+total) */
    hipLaunchKernelGGL(k_scan_partials, dim3(nblocks), dim3(SCAN_BLOCK), 0,
                       ctx->stream, set->d_sp_cnt, ng, set->d_out_off,
                       set->d_blocksums);
    hipLaunchKernelGGL(k_scan_fixup, dim3(1), dim3(64), 0, ctx->stream, ng,
                       set->d_out_off, set->d_blocksums, nblocks);
    hipLaunchKernelGGL(k_scan_add, dim3(grid_for(ng, 256)), dim3(256), 0,
                       ctx->stream, ng, set->d_out_off, set->d_blocksums);
    HIP_TRY(hipEventRecord(ev[1], ctx->stream));
    hipLaunchKernelGGL(k_rle_ts_filtered, dim3(ng > 2048 ? 2048 : ng),
                       dim3(256), 0, ctx->stream, set->d_blob, ts_pages, ng,
                       set->d_sp_start, set->d_sp_cnt, set->d_out_off,
                       spec->d_out_ts);
    HIP_TRY(hipEventRecord(ev[2], ctx->stream));
    int nsg = set->nsgroups;
    if (spec->n_buckets > 0) {
        if (!spec->d_agg_max || !spec->d_agg_sum || !spec->d_agg_count)
            return fail(GS_ERR, "agg outputs missing");
        hipLaunchKernelGGL(k_build_sgroups_out, dim3(grid_for(nsg, 256)),
                           dim3(256), 0, ctx->stream, set->d_sgroup_first,
                           nsg, set->d_out_off, set->d_sgroups_out);
        size_t cells = size_t(nsg) * size_t(spec->n_buckets);
        if (set->partials_cap < cells) {
            if (set->d_pmax) hipFree(set->d_pmax);
            if (set->d_psum) hipFree(set->d_psum);
            if (set->d_pcnt) hipFree(set->d_pcnt);
            if (hipMalloc(&set->d_pmax, cells * 8) != hipSuccess ||
                hipMalloc(&set->d_psum, cells * 8) != hipSuccess ||
                hipMalloc(&set->d_pcnt, cells * 8) != hipSuccess)
                return fail(GS_ERR, "hipMalloc agg partials failed");
            set->partials_cap = cells;
        }
    }
    for (int f = 0; f < nf; f++) {
        SlotPages &fsp = set->slots[1 + fields[f]];
        int nch = fsp.n_gor_chunks;
        double *d_val_f = spec->d_out_val + size_t(f) * set->total_rows;
        HIP_TRY(hipMemsetAsync(set->d_gor_nactive, 0, sizeof(int),
                               ctx->stream));
        hipLaunchKernelGGL(k_gor_active, dim3(grid_for(nch, 256)), dim3(256),
                           0, ctx->stream, fsp.d_gor_chunks, nch,
                           set->d_sp_start, set->d_sp_cnt, set->d_gor_active,
                           set->d_gor_nactive);
        hipLaunchKernelGGL(k_gor_chunks_filtered,
                           dim3(grid_for(nch, GS_GOR_BLOCK)),
                           dim3(GS_GOR_BLOCK), 0, ctx->stream, set->d_blob,
                           fsp.d_gor_chunks, set->d_gor_active,
                           set->d_gor_nactive, set->d_sp_start,
                           set->d_sp_cnt, set->d_out_off, d_val_f,
                           ctx->d_err);
        if (f == 0) HIP_TRY(hipEventRecord(ev[3], ctx->stream));
        if (spec->n_buckets > 0) {
            /* closed-form boundaries via the RLE group table when it
               fits in LDS (it always does for realistic pages-per-
               series); otherwise the out_ts binary-search kernel */
            size_t shm_rle =
                size_t(set->max_span) * 16 +
                (size_t(set->max_span) + size_t(spec->n_buckets) + 2) * 4;
            if (spec->n_buckets <= 8192 && shm_rle <= 64 * 1024) {
                static int agg_block = [] { /* sweepable (GS_AGG_BLOCK) */
                    const char *e = getenv("GS_AGG_BLOCK");
                    long v = e ? atol(e) : 0;
                    return int(v == 128 || v == 512 ? v : 256);
                }();
                hipLaunchKernelGGL(k_agg_partial_rle,
                                   dim3(nsg > 65535 ? 65535 : nsg),
                                   dim3(agg_block), shm_rle, ctx->stream,
                                   set->d_sgroups_out, nsg,
                                   set->d_sgroup_first, set->d_g_t0sel,
                                   set->d_g_delta, set->d_out_off, d_val_f,
                                   spec->t0, spec->bucket_ns,
                                   spec->n_buckets, set->max_span,
                                   set->d_pmax, set->d_psum, set->d_pcnt);
            } else {
                size_t agg_shm = spec->n_buckets <= 8192
                                     ? (size_t(spec->n_buckets) + 1) * 4
                                     : 0;
                hipLaunchKernelGGL(
                    k_agg_partial, dim3(nsg > 65535 ? 65535 : nsg),
                    dim3(256), agg_shm, ctx->stream, set->d_sgroups_out,
                    nsg, spec->d_out_ts, d_val_f, nullptr, INT64_MIN,
                    INT64_MAX, spec->t0, spec->bucket_ns, spec->n_buckets,
                    set->d_pmax, set->d_psum, set->d_pcnt);
            }
            int mb = (spec->n_buckets + 3) / 4;
            hipLaunchKernelGGL(k_agg_merge, dim3(mb > 2048 ? 2048 : mb),
                               dim3(256), 0, ctx->stream, nsg,
                               spec->n_buckets, set->d_pmax, set->d_psum,
                               set->d_pcnt,
                               spec->d_agg_max + size_t(f) * spec->n_buckets,
                               spec->d_agg_sum + size_t(f) * spec->n_buckets,
                               spec->d_agg_count +
                                   size_t(f) * spec->n_buckets);
        }
    }
    HIP_TRY(hipEventRecord(ev[4], ctx->stream));
    HIP_TRY(hipEventRecord(ev[5], ctx->stream));
    set->pending = true;
    return GS_OK;
}

static GsStatus scan_fused_wait(GsCtx *ctx, GsGroupSet *set,
                                GsScanResult *result) {
    if (!set->pending)
        return fail(GS_ERR, "no fused scan in flight on this set");
    set->pending = false;
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    GsStatus st = check_dev_err(ctx);
    if (st != GS_OK) return st;
    int64_t acc = 0;
    HIP_TRY(hipMemcpy(&acc, set->d_out_off + set->ngroups, sizeof(int64_t),
                      hipMemcpyDeviceToHost));
    hipEvent_t *ev = set->sev;
    float ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[0], ev[1]));
    result->ms_filter = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[1], ev[2]));
    result->ms_decode_ts = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[2], ev[3]));
    result->ms_decode_val = ms; /* nf>1: field 0 only */
    result->ms_compact = 0.0;
    HIP_TRY(hipEventElapsedTime(&ms, ev[3], ev[4]));
    result->ms_agg = ms; /* nf>1: field 0's agg + remaining fields */
    result->out_rows = acc;
    result->decoded_rows = set->total_rows;
    return GS_OK;
}

GsStatus gs_scan_async(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec) {
    if (!ctx || !set || !spec)
        return fail(GS_ERR, "bad args to gs_scan_async");
    if (!fused_capable(set, spec))
        return fail(GS_ERR, "gs_scan_async requires the fused-capable shape "
                            "(RLE ts + all-valid Gorilla, no tombstones, "
                            "compacted outputs)");
    int32_t f0 = spec->field_col;
    return scan_fused_launch(ctx, set, spec, &f0, 1);
}

GsStatus gs_scan_wait(GsCtx *ctx, GsGroupSet *set, GsScanResult *result) {
    if (!ctx || !set || !result)
        return fail(GS_ERR, "bad args to gs_scan_wait");
    return scan_fused_wait(ctx, set, result);
}

/* Raw row set: series structure over caller-resident DEVICE arrays with
 * no pages — the memcache leg (MemCacheReader rows,
 * mem_cache/series_data.rs:15-27).  gs_scan over such a set skips the
 * decode phases and filters/aggregates spec->d_ts / spec->d_val
 * directly; rows are taken as non-null (pre-filter nulls when building
 * the arrays).  counts[i] = rows of series i (time-sorted). */
GsGroupSet *gs_raw_set(GsCtx *ctx, const int64_t *counts, int64_t nseries) {
    if (!ctx || !counts || nseries <= 0) {
        fail(GS_ERR, "bad args to gs_raw_set");
        return nullptr;
    }
    if (hipSetDevice(ctx->device) != hipSuccess) {
        fail(GS_ERR, "hipSetDevice failed");
        return nullptr;
    }
    GsGroupSet *set = new GsGroupSet();
    set->ctx = ctx;
    set->ngroups = size_t(nseries);
    set->ncols = 0;
    set->row_offsets.resize(nseries);
    std::vector<DevGroup> hg(nseries);
    int64_t rows = 0;
    for (int64_t i = 0; i < nseries; i++) {
        if (counts[i] < 0 || counts[i] > INT32_MAX) {
            fail(GS_ERR, "raw set series row count out of range");
            gs_groups_free(set);
            return nullptr;
        }
        set->row_offsets[i] = rows;
        hg[i].row_off = rows;
        hg[i].nrows = int32_t(counts[i]);
        hg[i].pad = 0;
        rows += counts[i];
    }
    set->total_rows = rows;
    set->nsgroups = int(nseries);
    set->sgroup_span.assign(nseries, 1);
    set->max_span = 1;
    std::vector<int32_t> sgfirst(nseries + 1);
    for (int64_t i = 0; i <= nseries; i++) sgfirst[i] = int32_t(i);
    size_t nblocks = (size_t(nseries) + SCAN_BLOCK * SCAN_ITEMS - 1) /
                     (SCAN_BLOCK * SCAN_ITEMS);
    if (hipMalloc(&set->d_groups, nseries * sizeof(DevGroup)) != hipSuccess ||
        hipMalloc(&set->d_sgroups, nseries * sizeof(DevGroup)) != hipSuccess ||
        hipMalloc(&set->d_sgroups_out, nseries * sizeof(DevGroup)) != hipSuccess ||
        hipMalloc(&set->d_sgroup_first, sgfirst.size() * sizeof(int32_t)) != hipSuccess ||
        hipMalloc(&set->d_sp_start, nseries * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_sp_cnt, nseries * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_out_off, (nseries + 1) * sizeof(int64_t)) != hipSuccess ||
        hipMalloc(&set->d_blocksums, (nblocks + 1) * sizeof(int64_t)) != hipSuccess) {
        fail(GS_ERR, "hipMalloc raw set tables failed");
        gs_groups_free(set);
        return nullptr;
    }
    hipMemcpyAsync(set->d_groups, hg.data(), nseries * sizeof(DevGroup),
                   hipMemcpyHostToDevice, ctx->stream);
    hipMemcpyAsync(set->d_sgroups, hg.data(), nseries * sizeof(DevGroup),
                   hipMemcpyHostToDevice, ctx->stream);
    hipMemcpyAsync(set->d_sgroup_first, sgfirst.data(),
                   sgfirst.size() * sizeof(int32_t), hipMemcpyHostToDevice,
                   ctx->stream);
    if (hipStreamSynchronize(ctx->stream) != hipSuccess) {
        fail(GS_ERR, "raw set upload failed");
        gs_groups_free(set);
        return nullptr;
    }
    return set;
}

/* Fused scan of nf field columns over ONE span/ts pass (TSBS
 * cpu-max-all-8, BASELINE config #3; the reference decodes each field
 * column of the column group in the same decode_pages pass,
 * tsm/reader.rs:494-560).  Fused-capable shapes only.  Strides:
 * d_out_val + f*total_rows per field; d_agg_* + f*n_buckets per field. */
GsStatus gs_scan_fields(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec,
                        const int32_t *field_cols, int nf,
                        GsScanResult *result) {
    if (!ctx || !set || !spec || !result || !field_cols || nf <= 0)
        return fail(GS_ERR, "bad args to gs_scan_fields");
    if (!fused_capable_fields(set, spec, field_cols, nf))
        return fail(GS_ERR, "gs_scan_fields needs a fused-capable shape "
                            "(all-RLE ts, all-valid Gorilla fields, no "
                            "tombstones/predicates, compacted outputs)");
    GsStatus st = scan_fused_launch(ctx, set, spec, field_cols, nf);
    if (st != GS_OK) return st;
    return scan_fused_wait(ctx, set, result);
}

/* async variant: pair with gs_scan_wait */
GsStatus gs_scan_fields_async(GsCtx *ctx, GsGroupSet *set,
                              const GsScanSpec *spec,
                              const int32_t *field_cols, int nf) {
    if (!ctx || !set || !spec || !field_cols || nf <= 0)
        return fail(GS_ERR, "bad args to gs_scan_fields_async");
    if (!fused_capable_fields(set, spec, field_cols, nf))
        return fail(GS_ERR, "gs_scan_fields needs a fused-capable shape");
    return scan_fused_launch(ctx, set, spec, field_cols, nf);
}

GsStatus gs_scan(GsCtx *ctx, GsGroupSet *set, const GsScanSpec *spec,
                 GsScanResult *result) {
    if (!ctx || !set || !spec || !result || !spec->d_ts || !spec->d_val)
        return fail(GS_ERR, "bad args to gs_scan");
    const bool raw = set->ncols == 0; /* gs_raw_set: memcache rows already
                                         resident in spec->d_ts/d_val */
    if (!raw) {
        if (set->ncols < 2)
            return fail(GS_ERR, "gs_scan needs a time page + one f64 field page");
        const uint32_t fcol = 1 + uint32_t(spec->field_col);
        if (spec->field_col < 0 || fcol >= set->ncols)
            return fail(GS_ERR, "gs_scan field column out of range");
        const uint8_t fct2 = set->slots[fcol].ctype;
        if (fct2 != GS_CT_F64 && fct2 != GS_CT_I64 && fct2 != GS_CT_U64)
            return fail(GS_ERR,
                        "gs_scan field column must be f64/i64/u64");
        if (fct2 != GS_CT_F64 && spec->n_buckets > 0)
            return fail(GS_ERR, "aggregates are f64-only (TSBS path); "
                                "decode+filter+compact support i64/u64");
    }
    HIP_TRY(hipSetDevice(ctx->device));

    /* per-phase timing on the engine stream (HIP events) */
    hipEvent_t ev[6];
    for (int k = 0; k < 6; k++) HIP_TRY(hipEventCreate(&ev[k]));
    struct EvGuard {
        hipEvent_t *e;
        ~EvGuard() { for (int k = 0; k < 6; k++) hipEventDestroy(e[k]); }
    } guard{ev};

    /* fused filtered path: spans closed-form from RLE ts headers, Gorilla
     * decode writes only the selected rows already compacted, aggregate
     * runs over the compacted output.  Preconditions: every ts page RLE,
     * every field page all-valid Gorilla, no tombstones, compacted
     * outputs requested.  Falls back to the general path otherwise. */
    if (fused_capable(set, spec)) {
        int32_t f0 = spec->field_col;
        GsStatus fst = scan_fused_launch(ctx, set, spec, &f0, 1);
        if (fst != GS_OK) return fst;
        return scan_fused_wait(ctx, set, result);
    }

    /* validity bytes needed if the field has nulls or tombstones apply */
    uint8_t *d_valid = nullptr;
    bool need_valid = set->any_nulls_field || spec->n_tombstones > 0;
    if (need_valid) {
        if (!set->d_valid)
            HIP_TRY(hipMalloc(&set->d_valid, size_t(set->total_rows)));
        d_valid = set->d_valid;
    }

    HIP_TRY(hipEventRecord(ev[0], ctx->stream));
    GsStatus st;
    if (!raw) {
        st = gs_decode(ctx, set, 0, spec->d_ts, nullptr);
        if (st != GS_OK) return st;
    }
    HIP_TRY(hipEventRecord(ev[1], ctx->stream));
    if (!raw) {
        st = gs_decode(ctx, set, 1 + uint32_t(spec->field_col), spec->d_val,
                       d_valid);
        if (st != GS_OK) return st;
    } else if (d_valid) {
        /* raw rows are non-null; tombstones clear from an all-valid base */
        HIP_TRY(hipMemsetAsync(d_valid, 1, size_t(set->total_rows),
                               ctx->stream));
    }
    HIP_TRY(hipEventRecord(ev[2], ctx->stream));

    if (spec->n_tombstones > 0) {
        st = gs_apply_tombstone(ctx, set, spec->d_ts, d_valid,
                                spec->tombstones, spec->n_tombstones);
        if (st != GS_OK) return st;
    }

    int ng = int(set->ngroups);
    hipLaunchKernelGGL(k_spans, dim3(grid_for(ng, 256)), dim3(256), 0,
                       ctx->stream, set->d_groups, ng, spec->d_ts,
                       spec->range.min_ts, spec->range.max_ts, set->d_sp_start,
                       set->d_sp_cnt);

    /* optional value predicate (DataFilter, reader/filter.rs:91-142) */
    uint8_t *d_mask = nullptr;
    const int64_t *d_counts = set->d_sp_cnt;
    if (spec->value_pred.op != GS_PRED_NONE) {
        if (!set->d_mask)
            HIP_TRY(hipMalloc(&set->d_mask, size_t(set->total_rows)));
        if (!set->d_sel_cnt)
            HIP_TRY(hipMalloc(&set->d_sel_cnt,
                              set->ngroups * sizeof(int64_t)));
        d_mask = set->d_mask;
        int vt = 0; /* raw (memcache) sets have no column slots: f64 */
        if (!raw) {
            const uint8_t fct = set->slots[1 + spec->field_col].ctype;
            vt = fct == GS_CT_I64 ? 1 : fct == GS_CT_U64 ? 2 : 0;
        }
        hipLaunchKernelGGL(k_vmask, dim3(ng > 65535 ? 65535 : ng), dim3(256),
                           0, ctx->stream, set->d_groups, ng, spec->d_val,
                           d_valid, set->d_sp_start, set->d_sp_cnt,
                           spec->value_pred.op, spec->value_pred.a,
                           spec->value_pred.b, vt, d_mask, set->d_sel_cnt);
        d_counts = set->d_sel_cnt;
    }
    HIP_TRY(hipEventRecord(ev[3], ctx->stream));

    bool want_rows = spec->d_out_ts && spec->d_out_val;
    {
        /* device exclusive scan of selected counts -> offsets + total:
           no host round-trip mid-pipeline (the round-1 version
           synchronized here and prefix-summed on the host) */
        int nblocks = (ng + SCAN_BLOCK * SCAN_ITEMS - 1) /
                      (SCAN_BLOCK * SCAN_ITEMS);
        if (nblocks > 2048)
            return fail(GS_ERR, "too many groups for the span scan");
        hipLaunchKernelGGL(k_scan_partials, dim3(nblocks), dim3(SCAN_BLOCK),
                           0, ctx->stream, d_counts, ng, set->d_out_off,
                           set->d_blocksums);
        hipLaunchKernelGGL(k_scan_fixup, dim3(1), dim3(64), 0, ctx->stream,
                           ng, set->d_out_off, set->d_blocksums, nblocks);
        hipLaunchKernelGGL(k_scan_add, dim3(grid_for(ng, 256)), dim3(256), 0,
                           ctx->stream, ng, set->d_out_off, set->d_blocksums);
        if (want_rows) {
            if (d_mask)
                hipLaunchKernelGGL(k_compact_masked,
                                   dim3(ng > 2048 ? 2048 : ng), dim3(256), 0,
                                   ctx->stream, set->d_groups, ng, spec->d_ts,
                                   spec->d_val, d_mask, set->d_sp_start,
                                   set->d_sp_cnt, set->d_out_off,
                                   spec->d_out_ts, spec->d_out_val);
            else
                hipLaunchKernelGGL(k_compact, dim3(ng > 2048 ? 2048 : ng),
                                   dim3(256), 0, ctx->stream, set->d_groups,
                                   ng, spec->d_ts, spec->d_val,
                                   set->d_sp_start, set->d_sp_cnt,
                                   set->d_out_off, spec->d_out_ts,
                                   spec->d_out_val);
        }
    }
    HIP_TRY(hipEventRecord(ev[4], ctx->stream));

    if (spec->n_buckets > 0) {
        if (!spec->d_agg_max || !spec->d_agg_sum || !spec->d_agg_count)
            return fail(GS_ERR, "agg outputs missing");
        int nsg = set->nsgroups;
        size_t cells = size_t(nsg) * size_t(spec->n_buckets);
        if (set->partials_cap < cells) {
            if (set->d_pmax) hipFree(set->d_pmax);
            if (set->d_psum) hipFree(set->d_psum);
            if (set->d_pcnt) hipFree(set->d_pcnt);
            if (hipMalloc(&set->d_pmax, cells * 8) != hipSuccess ||
                hipMalloc(&set->d_psum, cells * 8) != hipSuccess ||
                hipMalloc(&set->d_pcnt, cells * 8) != hipSuccess)
                return fail(GS_ERR, "hipMalloc agg partials failed");
            set->partials_cap = cells;
        }
        size_t agg_shm = spec->n_buckets <= 8192
                             ? (size_t(spec->n_buckets) + 1) * 4 : 0;
        hipLaunchKernelGGL(k_agg_partial, dim3(nsg > 65535 ? 65535 : nsg),
                           dim3(256), agg_shm, ctx->stream, set->d_sgroups, nsg,
                           spec->d_ts, spec->d_val,
                           d_mask ? d_mask : d_valid,
                           spec->range.min_ts, spec->range.max_ts,
                           spec->t0, spec->bucket_ns,
                           spec->n_buckets, set->d_pmax, set->d_psum,
                           set->d_pcnt);
        int mb = (spec->n_buckets + 3) / 4;
        hipLaunchKernelGGL(k_agg_merge, dim3(mb > 2048 ? 2048 : mb), dim3(256),
                           0, ctx->stream, nsg, spec->n_buckets, set->d_pmax,
                           set->d_psum, set->d_pcnt, spec->d_agg_max,
                           spec->d_agg_sum, spec->d_agg_count);
    }
    HIP_TRY(hipEventRecord(ev[5], ctx->stream));

    HIP_TRY(hipStreamSynchronize(ctx->stream));
    st = check_dev_err(ctx);
    if (st != GS_OK) return st;
    float ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[0], ev[1]));
    result->ms_decode_ts = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[1], ev[2]));
    result->ms_decode_val = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[2], ev[3]));
    result->ms_filter = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[3], ev[4]));
    result->ms_compact = ms;
    HIP_TRY(hipEventElapsedTime(&ms, ev[4], ev[5]));
    result->ms_agg = ms;

    int64_t out_rows = 0;
    HIP_TRY(hipMemcpy(&out_rows, set->d_out_off + set->ngroups,
                      sizeof(int64_t), hipMemcpyDeviceToHost));
    result->out_rows = out_rows;
    result->decoded_rows = set->total_rows;
    return GS_OK;
}

GsStatus gs_encode_pages_dev(GsCtx *ctx, int32_t kind, const void *d_vals,
                             const uint8_t *d_valid, const int64_t *h_row_off,
                             const int32_t *h_rows, int32_t npages,
                             uint8_t *d_out, int64_t cap_per_page,
                             int64_t *h_lens) {
    if (!ctx || !d_vals || !h_row_off || !h_rows || npages <= 0 || !d_out ||
        !h_lens || kind < 0 || kind > 2)
        return fail(GS_ERR, "bad args to gs_encode_pages_dev");
    int32_t maxr = 0;
    for (int32_t p = 0; p < npages; p++) if (h_rows[p] > maxr) maxr = h_rows[p];
    /* worst case: ints -> uncompressed 8 B/val; f64 -> ~77 bits/val */
    int64_t need = 16 + (int64_t(maxr) + 7) / 8 +
                   (kind == 2 ? (2 + 8 + (int64_t(maxr) + 1) * 10 + 16)
                              : (2 + 8 * int64_t(maxr) + 32));
    if (cap_per_page < need)
        return fail(GS_ERR_CAP, "cap_per_page below worst-case encoded size");
    HIP_TRY(hipSetDevice(ctx->device));
    std::vector<EncPageSpec> specs(npages);
    for (int32_t p = 0; p < npages; p++) {
        specs[p].row_off = h_row_off[p];
        specs[p].nrows = h_rows[p];
        specs[p].pad = 0;
    }
    EncPageSpec *d_specs;
    int64_t *d_lens;
    HIP_TRY(hipMalloc(&d_specs, size_t(npages) * sizeof(EncPageSpec)));
    if (hipMalloc(&d_lens, size_t(npages) * 8) != hipSuccess) {
        hipFree(d_specs);
        return fail(GS_ERR, "hipMalloc lens failed");
    }
    hipMemcpyAsync(d_specs, specs.data(), size_t(npages) * sizeof(EncPageSpec),
                   hipMemcpyHostToDevice, ctx->stream);
    hipLaunchKernelGGL(k_encode_pages, dim3(grid_for(npages, 256)), dim3(256),
                       0, ctx->stream, kind, d_vals, d_valid, d_specs, npages,
                       d_out, cap_per_page, d_lens, ctx->d_err);
    GsStatus st = GS_OK;
    if (hipStreamSynchronize(ctx->stream) != hipSuccess)
        st = fail(GS_ERR, "encode kernel failed");
    hipMemcpy(h_lens, d_lens, size_t(npages) * 8, hipMemcpyDeviceToHost);
    hipFree(d_specs);
    hipFree(d_lens);
    if (st != GS_OK) return st;
    return check_dev_err(ctx);
}

GsStatus gs_compact_merge(GsCtx *ctx, GsGroupSet *const *sets, int32_t nsets,
                          const int64_t *const *d_ts,
                          const double *const *d_val,
                          const uint8_t *const *d_valid, int64_t *d_out_ts,
                          double *d_out_val, uint8_t *d_out_valid,
                          int64_t *h_out_offsets, int64_t *out_rows) {
    if (!ctx || !sets || nsets < 1 || nsets > GS_MAX_STREAMS || !d_ts ||
        !d_val || !d_out_ts || !d_out_val)
        return fail(GS_ERR, "bad args to gs_compact_merge");
    int nseries = sets[0]->nsgroups;
    for (int f = 0; f < nsets; f++)
        if (sets[f]->nsgroups != nseries)
            return fail(GS_ERR, "all streams must cover the same series list");
    HIP_TRY(hipSetDevice(ctx->device));

    CompactArgs a;
    memset(&a, 0, sizeof(a));
    std::vector<void *> scratch;
    auto cleanup = [&]() { for (void *p : scratch) hipFree(p); };
    for (int f = 0; f < nsets; f++) {
        a.ts[f] = d_ts[f];
        a.val[f] = d_val[f];
        a.valid[f] = d_valid ? d_valid[f] : nullptr;
        a.groups[f] = sets[f]->d_sgroups;
        size_t rows = size_t(sets[f]->total_rows);
        uint8_t *fl;
        int32_t *pf;
        if (hipMalloc(&fl, rows ? rows : 1) != hipSuccess ||
            hipMalloc(&pf, (rows ? rows : 1) * 4) != hipSuccess) {
            cleanup();
            return fail(GS_ERR, "hipMalloc compact scratch failed");
        }
        scratch.push_back(fl);
        scratch.push_back(pf);
        a.flags[f] = fl;
        a.prefix[f] = pf;
    }
    int64_t *d_counts, *d_ooff;
    if (hipMalloc(&d_counts, size_t(nsets) * nseries * 8) != hipSuccess ||
        hipMalloc(&d_ooff, size_t(nseries) * 8) != hipSuccess) {
        cleanup();
        return fail(GS_ERR, "hipMalloc compact tables failed");
    }
    scratch.push_back(d_counts);
    scratch.push_back(d_ooff);
    a.counts = d_counts;
    a.out_off = d_ooff;

    int gx = nseries > 2048 ? 2048 : nseries;
    hipLaunchKernelGGL(k_cm_flags, dim3(gx, nsets), dim3(256), 0, ctx->stream,
                       a, nsets, nseries, ctx->d_err);
    int total = nsets * nseries;
    hipLaunchKernelGGL(k_cm_prefix,
                       dim3(total > 65535 ? 65535 : total), dim3(256), 0,
                       ctx->stream, a, nsets, nseries);
    std::vector<int64_t> counts(size_t(nsets) * nseries), ooff(nseries + 1);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    HIP_TRY(hipMemcpy(counts.data(), d_counts, counts.size() * 8,
                      hipMemcpyDeviceToHost));
    int64_t acc = 0;
    for (int s = 0; s < nseries; s++) {
        ooff[s] = acc;
        for (int f = 0; f < nsets; f++) acc += counts[size_t(f) * nseries + s];
    }
    ooff[nseries] = acc;
    HIP_TRY(hipMemcpyAsync(d_ooff, ooff.data(), size_t(nseries) * 8,
                           hipMemcpyHostToDevice, ctx->stream));
    hipLaunchKernelGGL(k_cm_scatter, dim3(gx, nsets), dim3(256), 0,
                       ctx->stream, a, nsets, nseries, d_out_ts, d_out_val,
                       d_out_valid);
    HIP_TRY(hipStreamSynchronize(ctx->stream));
    cleanup();
    GsStatus st = check_dev_err(ctx);
    if (st != GS_OK) return st;
    if (h_out_offsets)
        memcpy(h_out_offsets, ooff.data(), size_t(nseries + 1) * 8);
    if (out_rows) *out_rows = acc;
    return GS_OK;
}

/* ---- Arrow C Data Interface export ---- */
namespace {
struct ExportPriv {
    void *data;
    uint8_t *bitmap;
    const void *bufs[2];
};
void export_release(GsArrowArray *a) {
    if (!a || !a->private_data) return;
    ExportPriv *p = (ExportPriv *)a->private_data;
    free(p->data);
    free(p->bitmap);
    free(p);
    a->release = nullptr;
    a->private_data = nullptr;
}
} // namespace

GsStatus gs_export_group_column(GsCtx *ctx, GsGroupSet *set, int64_t group,
                                const void *d_col, int32_t elem_size,
                                const uint8_t *d_valid, GsArrowArray *out) {
    if (!ctx || !set || group < 0 || size_t(group) >= set->ngroups || !d_col ||
        !out || (elem_size != 8 && elem_size != 1))
        return fail(GS_ERR, "bad args to gs_export_group_column");
    HIP_TRY(hipSetDevice(ctx->device));
    int64_t row0 = set->row_offsets[group];
    int64_t rows = (size_t(group) + 1 < set->ngroups
                        ? set->row_offsets[group + 1]
                        : set->total_rows) - row0;
    ExportPriv *p = (ExportPriv *)calloc(1, sizeof(ExportPriv));
    if (!p) return fail(GS_ERR, "oom");
    p->data = malloc(size_t(rows) * elem_size ? size_t(rows) * elem_size : 1);
    if (!p->data) { free(p); return fail(GS_ERR, "oom"); }
    if (hipMemcpy(p->data, (const uint8_t *)d_col + row0 * elem_size,
                  size_t(rows) * elem_size, hipMemcpyDeviceToHost) !=
        hipSuccess) {
        free(p->data); free(p);
        return fail(GS_ERR, "export D2H failed");
    }
    int64_t null_count = 0;
    if (d_valid) {
        std::vector<uint8_t> vbytes(rows);
        if (hipMemcpy(vbytes.data(), d_valid + row0, size_t(rows),
                      hipMemcpyDeviceToHost) != hipSuccess) {
            free(p->data); free(p);
            return fail(GS_ERR, "export validity D2H failed");
        }
        size_t nb = size_t(rows + 7) / 8;
        p->bitmap = (uint8_t *)calloc(nb ? nb : 1, 1);
        if (!p->bitmap) { free(p->data); free(p); return fail(GS_ERR, "oom"); }
        for (int64_t r = 0; r < rows; r++) {
            if (vbytes[r]) p->bitmap[r >> 3] |= uint8_t(1u << (r & 7));
            else null_count++;
        }
    }
    p->bufs[0] = p->bitmap; /* NULL when no validity buffer */
    p->bufs[1] = p->data;
    out->length = rows;
    out->null_count = d_valid ? null_count : 0;
    out->offset = 0;
    out->n_buffers = 2;
    out->n_children = 0;
    out->buffers = p->bufs;
    out->children = nullptr;
    out->dictionary = nullptr;
    out->release = export_release;
    out->private_data = p;
    return GS_OK;
}

} // extern "C"
