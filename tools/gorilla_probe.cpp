/* Standalone ablation probe for the Gorilla decode kernel on gfx950.
 * Variants isolate compute vs store vs load cost. Build:
 *   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/gorilla_probe.cpp \
 *         cnosdb_amd/csrc/gs_encode.cpp -o tools/gorilla_probe
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <vector>
#include <cmath>
#include <random>

extern "C" int64_t gs_encode_f64(const double *, size_t, uint8_t *, size_t);

#define SENT 0x7ff8000000000ffULL

__device__ __forceinline__ uint64_t dbe64(const uint8_t *p) {
    uint64_t v;
    __builtin_memcpy(&v, p, 8);
    return __builtin_bswap64(v);
}

struct PD { uint64_t off; uint32_t len; uint64_t row; };

template <int MODE> /* 0=full 1=nostore 2=2pages-interleaved */
__global__ void k_gor(const uint8_t *__restrict__ blob,
                      const PD *__restrict__ pages, int npages,
                      double *__restrict__ out, unsigned *__restrict__ err) {
    for (int p0 = blockIdx.x * blockDim.x + threadIdx.x; p0 < npages;
         p0 += gridDim.x * blockDim.x) {
        PD pg = pages[p0];
        const uint8_t *data = blob + pg.off;
        double *o = out + pg.row;
        const uint8_t *s = data + 1;
        uint32_t slen = pg.len - 1;
        uint64_t val = dbe64(s + 1);
        const uint8_t *p = s + 9;
        int64_t budget = int64_t(slen - 9) * 8;
        uint64_t hi = 0, lo = 0;
        int nb = 0;
        uint32_t trailing = 0, meaningful = 64;
        uint32_t r = 0;
        uint64_t acc = 0;
        uint64_t nextw = dbe64(p);
        p += 8;
        auto topup = [&]() {
            uint64_t x = nextw;
            nextw = dbe64(p);
            p += 8;
            if (nb == 0) { hi = x; lo = 0; }
            else { hi |= x >> nb; lo = x << (64 - nb); }
            nb += 64;
        };
        auto consume = [&](unsigned k) {
            hi = (k == 64) ? lo : ((hi << k) | (lo >> (64 - k)));
            lo = (k == 64) ? 0 : (lo << k);
            nb -= int(k);
            budget -= int64_t(k);
        };
        auto emit = [&](uint64_t v) {
            if (MODE == 0) o[r++] = __longlong_as_double((long long)v);
            else acc ^= v + r++;
        };
        emit(val);
        for (;;) {
            if (nb < 64) topup();
            if (budget <= 0) { atomicOr(err, 2u); break; }
            uint32_t top13 = uint32_t(hi >> 51);
            if (!(top13 & 0x1000)) {
                consume(1);
            } else {
                if (top13 & 0x0800) {
                    uint32_t lead = (top13 >> 6) & 0x1f;
                    meaningful = top13 & 0x3f;
                    if (meaningful > 0) trailing = 64 - lead - meaningful;
                    else { trailing = 0; meaningful = 64; }
                    consume(13);
                } else consume(2);
                while (nb < int(meaningful)) topup();
                uint64_t sb = (meaningful == 64) ? hi : (hi >> (64 - meaningful));
                consume(meaningful);
                if (budget < 0) { atomicOr(err, 2u); break; }
                val ^= sb << trailing;
                if (val == SENT) break;
            }
            emit(val);
        }
        if (MODE != 0 && acc == 0xdeadbeef) o[pg.row % 64] = 1.0; /* keep acc */
    }
}

/* MODE 0 decode with LDS ring staging: each lane accumulates RING decoded
 * values in LDS; when all lanes' rings fill (lanes run in lockstep, one
 * value per iteration), the wave flushes lane-by-lane with page-contiguous
 * coalesced stores. */
template <int RING>
__global__ void k_gor_lds(const uint8_t *__restrict__ blob,
                          const PD *__restrict__ pages, int npages,
                          double *__restrict__ out,
                          unsigned *__restrict__ err) {
    __shared__ double ring[256 / 64][RING][64 + 1]; /* [wave][slot][lane] */
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    auto rslot = ring[wv];
    int stride = gridDim.x * blockDim.x;
    int base_id = blockIdx.x * blockDim.x + threadIdx.x;
    int rounds = (npages + stride - 1) / stride;
    for (int rd = 0; rd < rounds; rd++) {
        int p0 = base_id + rd * stride;
        bool have = p0 < npages;
        PD pg = pages[have ? p0 : 0];
        const uint8_t *data = blob + pg.off;
        double *o = out + pg.row;
        const uint8_t *s = data + 1;
        uint32_t slen = pg.len - 1;
        uint64_t val = dbe64(s + 1);
        const uint8_t *p = s + 9;
        int64_t budget = int64_t(slen - 9) * 8;
        uint64_t hi = 0, lo = 0;
        int nb = 0;
        uint32_t trailing = 0, meaningful = 64;
        int r = 0;     /* rows emitted (incl. staged) */
        int rfill = 0; /* staged in ring */
        bool done = !have;
        uint64_t nextw = dbe64(p);
        p += 8;
        auto topup = [&]() {
            uint64_t x = nextw;
            nextw = dbe64(p);
            p += 8;
            if (nb == 0) { hi = x; lo = 0; }
            else { hi |= x >> nb; lo = x << (64 - nb); }
            nb += 64;
        };
        auto consume = [&](unsigned k) {
            hi = (k == 64) ? lo : ((hi << k) | (lo >> (64 - k)));
            lo = (k == 64) ? 0 : (lo << k);
            nb -= int(k);
            budget -= int64_t(k);
        };
        auto flush = [&]() {
            /* cooperative: store each source lane's staged run as a
               page-contiguous coalesced store (all 64 lanes alive here) */
            for (int sl = 0; sl < 64; sl++) {
                unsigned long long ob =
                    __shfl((unsigned long long)(uintptr_t)o, sl, 64);
                int cnt = __shfl(rfill, sl, 64);
                int row0 = __shfl(r, sl, 64) - cnt;
                if (lane < cnt)
                    ((double *)(uintptr_t)ob)[row0 + lane] = rslot[lane][sl];
            }
            rfill = 0;
        };
        /* stage the first value */
        if (!done) {
            rslot[rfill][lane] = __longlong_as_double((long long)val);
            rfill++;
            r++;
        }
        while (!__all(done)) {
            if (!done) {
                if (nb < 64) topup();
                if (budget <= 0) { atomicOr(err, 2u); done = true; }
            }
            if (!done) {
                uint32_t top13 = uint32_t(hi >> 51);
                bool stage = true;
                if (!(top13 & 0x1000)) {
                    consume(1);
                } else {
                    if (top13 & 0x0800) {
                        uint32_t lead = (top13 >> 6) & 0x1f;
                        meaningful = top13 & 0x3f;
                        if (meaningful > 0) trailing = 64 - lead - meaningful;
                        else { trailing = 0; meaningful = 64; }
                        consume(13);
                    } else consume(2);
                    while (nb < int(meaningful)) topup();
                    uint64_t sb =
                        (meaningful == 64) ? hi : (hi >> (64 - meaningful));
                    consume(meaningful);
                    if (budget < 0) { atomicOr(err, 2u); done = true; stage = false; }
                    else {
                        val ^= sb << trailing;
                        if (val == SENT) { done = true; stage = false; }
                    }
                }
                if (stage) {
                    rslot[rfill][lane] = __longlong_as_double((long long)val);
                    rfill++;
                    r++;
                }
            }
            if (__any(rfill == RING)) flush();
        }
        flush();
    }
}


/* two pages interleaved per thread: duplicated decode state, shared flush.
 * Tests whether the residual waits are intra-chain latency (ILP helps) or
 * shared-resource contention (it will not). */
template <int RING>
__global__ void k_gor_ilp2(const uint8_t *__restrict__ blob,
                           const PD *__restrict__ pages, int npages,
                           double *__restrict__ out,
                           unsigned *__restrict__ err) {
    __shared__ double ring[256 / 64][RING][128 + 2];
    const int lane = threadIdx.x & 63;
    const int wv = threadIdx.x >> 6;
    auto rslot = ring[wv];
    int stride = gridDim.x * blockDim.x;
    int base_id = blockIdx.x * blockDim.x + threadIdx.x;
    int pairs = (npages + 1) / 2;
    int rounds = (pairs + stride - 1) / stride;
    for (int rd = 0; rd < rounds; rd++) {
        int pr = base_id + rd * stride;
        struct St {
            const uint8_t *p;
            double *o;
            uint64_t hi, lo, val, nextw;
            int64_t budget;
            int nb, r, rfill;
            uint32_t trailing, meaningful;
            bool done;
        } st[2];
        for (int q = 0; q < 2; q++) {
            int p0 = pr * 2 + q;
            bool have = p0 < npages;
            PD pg = pages[have ? p0 : 0];
            const uint8_t *data = blob + pg.off;
            const uint8_t *sx = data + 1;
            st[q].o = out + pg.row;
            st[q].val = dbe64(sx + 1);
            st[q].p = sx + 9;
            st[q].budget = int64_t(pg.len - 1 - 9) * 8;
            st[q].hi = st[q].lo = 0;
            st[q].nb = 0;
            st[q].r = 0;
            st[q].rfill = 0;
            st[q].trailing = 0;
            st[q].meaningful = 64;
            st[q].done = !have;
            st[q].nextw = dbe64(st[q].p);
            st[q].p += 8;
        }
        auto flush = [&]() {
            /* broadcast both runs' descriptors through shuffles (probe
               simplicity; the engine uses LDS descriptors) */
            for (int q = 0; q < 2; q++) {
                for (int sl = 0; sl < 64; sl++) {
                    unsigned long long ob = __shfl(
                        (unsigned long long)(uintptr_t)st[q].o, sl, 64);
                    int cnt = __shfl(st[q].rfill, sl, 64);
                    int row0 = __shfl(st[q].r, sl, 64) - cnt;
                    if (lane < cnt)
                        ((double *)(uintptr_t)ob)[row0 + lane] =
                            rslot[lane][sl * 2 + q];
                }
                st[q].rfill = 0;
            }
        };
        auto step = [&](St &z) {
            if (z.done) return;
            if (z.nb < 64) {
                uint64_t x = z.nextw;
                z.nextw = dbe64(z.p);
                z.p += 8;
                if (z.nb == 0) { z.hi = x; z.lo = 0; }
                else { z.hi |= x >> z.nb; z.lo = x << (64 - z.nb); }
                z.nb += 64;
            }
            if (z.budget <= 0) { atomicOr(err, 2u); z.done = true; return; }
            auto consume = [&](unsigned k) {
                z.hi = (k == 64) ? z.lo : ((z.hi << k) | (z.lo >> (64 - k)));
                z.lo = (k == 64) ? 0 : (z.lo << k);
                z.nb -= int(k);
                z.budget -= int64_t(k);
            };
            uint32_t top13 = uint32_t(z.hi >> 51);
            bool stg = true;
            if (!(top13 & 0x1000)) {
                consume(1);
            } else {
                if (top13 & 0x0800) {
                    uint32_t lead = (top13 >> 6) & 0x1f;
                    z.meaningful = top13 & 0x3f;
                    if (z.meaningful > 0) z.trailing = 64 - lead - z.meaningful;
                    else { z.trailing = 0; z.meaningful = 64; }
                    consume(13);
                } else consume(2);
                while (z.nb < int(z.meaningful)) {
                    uint64_t x = z.nextw;
                    z.nextw = dbe64(z.p);
                    z.p += 8;
                    if (z.nb == 0) { z.hi = x; z.lo = 0; }
                    else { z.hi |= x >> z.nb; z.lo = x << (64 - z.nb); }
                    z.nb += 64;
                }
                uint64_t sb = (z.meaningful == 64) ? z.hi
                                                  : (z.hi >> (64 - z.meaningful));
                consume(z.meaningful);
                if (z.budget < 0) { atomicOr(err, 2u); z.done = true; stg = false; }
                else {
                    z.val ^= sb << z.trailing;
                    if (z.val == SENT) { z.done = true; stg = false; }
                }
            }
            if (stg) {
                int q = (&z == &st[1]) ? 1 : 0;
                rslot[z.rfill][lane * 2 + q] =
                    __longlong_as_double((long long)z.val);
                z.rfill++;
                z.r++;
            }
        };
        /* stage first values */
        for (int q = 0; q < 2; q++)
            if (!st[q].done) {
                rslot[st[q].rfill][lane * 2 + q] =
                    __longlong_as_double((long long)st[q].val);
                st[q].rfill++;
                st[q].r++;
            }
        while (!__all(st[0].done && st[1].done)) {
            step(st[0]);
            step(st[1]);
            if (__any(st[0].rfill == RING || st[1].rfill == RING)) flush();
        }
        flush();
    }
}

/* store-only: same store pattern, no decode */
__global__ void k_store(const PD *__restrict__ pages, int npages, int rows,
                        double *__restrict__ out) {
    for (int p0 = blockIdx.x * blockDim.x + threadIdx.x; p0 < npages;
         p0 += gridDim.x * blockDim.x) {
        double *o = out + pages[p0].row;
        for (int r = 0; r < rows; r++) o[r] = double(r);
    }
}

/* load-only: stream the compressed bytes, no decode */
__global__ void k_load(const uint8_t *__restrict__ blob,
                       const PD *__restrict__ pages, int npages,
                       double *__restrict__ out) {
    for (int p0 = blockIdx.x * blockDim.x + threadIdx.x; p0 < npages;
         p0 += gridDim.x * blockDim.x) {
        PD pg = pages[p0];
        const uint8_t *p = blob + pg.off;
        uint64_t acc = 0;
        for (uint32_t i = 0; i + 8 <= pg.len; i += 8) acc ^= dbe64(p + i);
        if (acc == 0xdeadbeef) out[pg.row % 64] = 1.0;
    }
}

#define CHK(x) do { auto e=(x); if (e!=hipSuccess){printf("ERR %s %s\n",#x,hipGetErrorString(e)); return 1;} } while(0)

int main(int argc, char **argv) {
    int npages = argc > 1 ? atoi(argv[1]) : 160000;
    int rows = argc > 2 ? atoi(argv[2]) : 15625;
    int block = argc > 3 ? atoi(argv[3]) : 256;
    printf("npages=%d rows=%d block=%d\n", npages, rows, block);

    /* host: encode a handful of unique pages, tile them */
    std::mt19937_64 rng(231);
    std::normal_distribution<double> nd(0, 0.5);
    int uniq = 64;
    std::vector<std::vector<uint8_t>> enc(uniq);
    std::vector<double> v(rows);
    for (int u = 0; u < uniq; u++) {
        double w = 50;
        for (int i = 0; i < rows; i++) {
            w += nd(rng);
            if (w < 0) w = 0; if (w > 100) w = 100;
            v[i] = std::round(w * 10) / 10;
        }
        enc[u].resize(rows * 12 + 64);
        int64_t n = gs_encode_f64(v.data(), rows, enc[u].data(), enc[u].size());
        enc[u].resize(n);
    }
    size_t per = 0;
    for (auto &e : enc) per = std::max(per, e.size());
    per = (per + 63) & ~size_t(63);
    size_t blobsz = size_t(npages) * per + 64;
    uint8_t *d_blob;
    double *d_out;
    PD *d_pd;
    unsigned *d_err;
    CHK(hipMalloc(&d_blob, blobsz));
    CHK(hipMalloc(&d_out, size_t(npages) * rows * 8));
    CHK(hipMalloc(&d_pd, npages * sizeof(PD)));
    CHK(hipMalloc(&d_err, 4));
    CHK(hipMemset(d_err, 0, 4));
    std::vector<PD> pd(npages);
    std::vector<uint8_t> hb(blobsz);
    double avg_len = 0;
    for (int i = 0; i < npages; i++) {
        auto &e = enc[i % uniq];
        memcpy(hb.data() + size_t(i) * per, e.data(), e.size());
        pd[i] = {size_t(i) * per, uint32_t(e.size()), uint64_t(i) * rows};
        avg_len += e.size();
    }
    printf("bits/val %.2f\n", avg_len / npages * 8 / rows);
    CHK(hipMemcpy(d_blob, hb.data(), blobsz, hipMemcpyHostToDevice));
    CHK(hipMemcpy(d_pd, pd.data(), npages * sizeof(PD), hipMemcpyDeviceToHost == 99 ? hipMemcpyHostToDevice : hipMemcpyHostToDevice));

    int grid = std::min((npages + block - 1) / block, 16384);
    hipEvent_t a, b;
    hipEventCreate(&a);
    hipEventCreate(&b);
    double vals = double(npages) * rows;
    auto run = [&](const char *name, auto fn) {
        fn(); /* warmup */
        CHK(hipDeviceSynchronize());
        hipEventRecord(a);
        for (int it = 0; it < 3; it++) fn();
        hipEventRecord(b);
        CHK(hipDeviceSynchronize());
        float ms;
        hipEventElapsedTime(&ms, a, b);
        ms /= 3;
        printf("%-12s %8.2f ms  %7.1f Gval/s  %7.1f GB/s(out8B)\n", name, ms,
               vals / ms / 1e6, vals * 8 / ms / 1e6);
        return 0;
    };
    run("full", [&] { k_gor<0><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    run("nostore", [&] { k_gor<1><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    run("storeonly", [&] { k_store<<<grid, block>>>(d_pd, npages, rows, d_out); });
    run("loadonly", [&] { k_load<<<grid, block>>>(d_blob, d_pd, npages, d_out); });
    run("lds16", [&] { k_gor_lds<16><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    run("lds32", [&] { k_gor_lds<32><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    run("ilp2r8", [&] { k_gor_ilp2<8><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    run("ilp2r16", [&] { k_gor_ilp2<16><<<grid, block>>>(d_blob, d_pd, npages, d_out, d_err); });
    /* verify lds16 output matches full */
    {
        std::vector<double> a(rows), c(rows);
        hipMemcpy(a.data(), d_out, rows * 8, hipMemcpyDeviceToHost);
        hipLaunchKernelGGL((k_gor<0>), dim3(grid), dim3(block), 0, 0, d_blob, d_pd, npages, d_out, d_err);
        hipDeviceSynchronize();
        hipMemcpy(c.data(), d_out, rows * 8, hipMemcpyDeviceToHost);
        int bad = 0;
        for (int i = 0; i < rows; i++) if (a[i] != c[i]) bad++;
        printf("lds-vs-full mismatches(page0): %d\n", bad);
    }
    unsigned derr;
    CHK(hipMemcpy(&derr, d_err, 4, hipMemcpyDeviceToHost));
    printf("err=%u (expect 0)\n", derr);
    /* verify full output of page 0 */
    std::vector<double> o(rows);
    CHK(hipMemcpy(o.data(), d_out, rows * 8, hipMemcpyDeviceToHost));
    printf("out[0..3]=%.1f %.1f %.1f %.1f\n", o[0], o[1], o[2], o[3]);
    return 0;
}
