/* C++ host-side mirror of the reference's ColumnGroupReader seam
 * (tskv/src/reader/column_group/mod.rs:33-70,195-243 + decode_pages,
 * tsm/reader.rs:494-560), standing where the Rust shim of
 * INTEGRATION.md would stand: same construction arguments, same read()
 * semantics (one batch of the column group's rows, filtered by the
 * pushed time range), same error behaviour (status + message).
 *
 * This is a self-contained end-to-end exercise of the C ABI from C++
 * (no Python): build TSM pages with the host encoders, upload, scan
 * with a closed time range, verify counts and payloads, print OK.
 * Compiled by __graft_entry__.build() and run by the GPU test suite. */
#include <cinttypes>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include <hip/hip_runtime.h>

#include "../include/cnosdb_gs.h"

extern "C" {
int64_t gs_encode_ts(const int64_t *src, size_t n, uint8_t *dst, size_t cap);
int64_t gs_encode_f64(const double *src, size_t n, uint8_t *dst, size_t cap);
uint32_t gs_crc32(const uint8_t *data, size_t len);
}

#define CHECK(cond, msg)                                                   \
    do {                                                                   \
        if (!(cond)) {                                                     \
            fprintf(stderr, "FAIL: %s (%s)\n", msg, gs_last_error());      \
            return 1;                                                      \
        }                                                                  \
    } while (0)

/* tsm/page.rs:32-38 layout, as the shim's page assembly would emit */
static std::vector<uint8_t> build_page(const uint8_t *data, size_t len,
                                       uint32_t nrows) {
    const uint32_t nb = (nrows + 7) / 8;
    std::vector<uint8_t> p(16 + nb + len);
    const uint32_t crc = gs_crc32(data, len);
    /* the u64 header field is the ROW COUNT (the reference names it
       data_len but stores num_values — page.rs:32-38) */
    const uint64_t dl = nrows;
    for (int i = 0; i < 4; i++) p[i] = uint8_t(nb >> (24 - 8 * i));
    for (int i = 0; i < 8; i++) p[4 + i] = uint8_t(dl >> (56 - 8 * i));
    for (int i = 0; i < 4; i++) p[12 + i] = uint8_t(crc >> (24 - 8 * i));
    for (uint32_t i = 0; i < nb; i++) p[16 + i] = 0xFF; /* all valid */
    if (nrows % 8) p[16 + nb - 1] = uint8_t((1u << (nrows % 8)) - 1);
    memcpy(p.data() + 16 + nb, data, len);
    return p;
}

/* mirror of ColumnGroupReader: try_new(series, pages, projection) +
 * read(time_range) -> one batch */
struct ColumnGroupReaderShim {
    GsCtx *ctx = nullptr;
    GsGroupSet *set = nullptr;
    int64_t rows = 0;

    /* pages[0] = time page (projection pushes the time column first,
       column_group/mod.rs:195-215) */
    GsStatus try_new(uint32_t series_id,
                     const std::vector<std::vector<uint8_t>> &pages,
                     const std::vector<uint8_t> &ctypes_,
                     const std::vector<uint32_t> &nvals) {
        ctx = gs_ctx_create(0);
        if (!ctx) return GS_ERR;
        std::vector<GsPageSpec> specs(pages.size());
        for (size_t i = 0; i < pages.size(); i++) {
            specs[i].bytes = pages[i].data();
            specs[i].len = pages[i].size();
            specs[i].num_values = nvals[i];
            specs[i].ctype = ctypes_[i];
        }
        GsColumnGroupDesc g;
        g.pages = specs.data();
        g.npages = uint32_t(specs.size());
        g.series_id = series_id;
        set = gs_groups_upload(ctx, &g, 1, 1);
        if (!set) return GS_ERR;
        rows = gs_set_rows(set);
        return GS_OK;
    }

    /* read with an Exact pure-time-range pushdown
       (data_source/batch/tskv.rs:351-371) */
    GsStatus read(int64_t min_ts, int64_t max_ts, int64_t *d_ts,
                  double *d_val, int64_t *d_out_ts, double *d_out_val,
                  GsScanResult *res) {
        GsScanSpec spec;
        memset(&spec, 0, sizeof(spec));
        spec.range.min_ts = min_ts;
        spec.range.max_ts = max_ts;
        spec.d_ts = d_ts;
        spec.d_val = d_val;
        spec.d_out_ts = d_out_ts;
        spec.d_out_val = d_out_val;
        return gs_scan(ctx, set, &spec, res);
    }

    ~ColumnGroupReaderShim() {
        if (set) gs_groups_free(set);
        if (ctx) gs_ctx_destroy(ctx);
    }
};

int main() {
    const uint32_t N = 100000; /* reference-shaped page */
    const int64_t T0 = 1700000000000000000LL, NS = 1000000000LL;
    std::vector<int64_t> ts(N);
    std::vector<double> vals(N);
    for (uint32_t i = 0; i < N; i++) {
        ts[i] = T0 + int64_t(i) * NS;
        vals[i] = double((i * 7) % 1000) / 10.0;
    }
    std::vector<uint8_t> buf(size_t(N) * 12 + 64);
    int64_t tlen = gs_encode_ts(ts.data(), N, buf.data(), buf.size());
    CHECK(tlen > 0, "ts encode");
    auto tpage = build_page(buf.data(), size_t(tlen), N);
    int64_t vlen = gs_encode_f64(vals.data(), N, buf.data(), buf.size());
    CHECK(vlen > 0, "f64 encode");
    auto vpage = build_page(buf.data(), size_t(vlen), N);

    ColumnGroupReaderShim rd;
    CHECK(rd.try_new(42, {tpage, vpage}, {GS_CT_TIME, GS_CT_F64}, {N, N}) ==
              GS_OK,
          "try_new/upload");
    CHECK(rd.rows == N, "row count");

    int64_t *d_ts, *d_ots;
    double *d_val, *d_oval;
    CHECK(hipMalloc(&d_ts, size_t(N) * 8) == hipSuccess, "alloc");
    CHECK(hipMalloc(&d_val, size_t(N) * 8) == hipSuccess, "alloc");
    CHECK(hipMalloc(&d_ots, size_t(N) * 8) == hipSuccess, "alloc");
    CHECK(hipMalloc(&d_oval, size_t(N) * 8) == hipSuccess, "alloc");

    /* closed interval selecting rows 1000..=70000 */
    const int64_t lo = T0 + 1000 * NS, hi = T0 + 70000 * NS;
    GsScanResult res;
    CHECK(rd.read(lo, hi, d_ts, d_val, d_ots, d_oval, &res) == GS_OK,
          "read/scan");
    CHECK(res.out_rows == 69001, "selected count (closed interval)");

    std::vector<int64_t> hts(size_t(res.out_rows));
    std::vector<double> hval(size_t(res.out_rows));
    CHECK(hipMemcpy(hts.data(), d_ots, hts.size() * 8,
                    hipMemcpyDeviceToHost) == hipSuccess, "copy");
    CHECK(hipMemcpy(hval.data(), d_oval, hval.size() * 8,
                    hipMemcpyDeviceToHost) == hipSuccess, "copy");
    for (int64_t i = 0; i < res.out_rows; i++) {
        const int64_t row = 1000 + i;
        if (hts[i] != ts[row] || hval[i] != vals[row]) {
            fprintf(stderr, "FAIL: row %" PRId64 " mismatch\n", i);
            return 1;
        }
    }
    /* error behaviour: malformed page must surface a message */
    {
        auto bad = vpage;
        bad[bad.size() - 1] ^= 0xFF; /* corrupt data -> CRC mismatch */
        ColumnGroupReaderShim rd2;
        GsStatus st = rd2.try_new(7, {tpage, bad},
                                  {GS_CT_TIME, GS_CT_F64}, {N, N});
        CHECK(st != GS_OK, "corrupt page must be rejected");
        CHECK(strlen(gs_last_error()) > 0, "error message populated");
    }
    hipFree(d_ts);
    hipFree(d_val);
    hipFree(d_ots);
    hipFree(d_oval);
    printf("shim_host OK: %" PRId64 " rows scanned, %" PRId64 " selected\n",
           int64_t(N), res.out_rows);
    return 0;
}
