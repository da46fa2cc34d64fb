#!/usr/bin/env python3
"""bench.py — TSBS-devops scan benchmark for the cnosdb_gs MI355X engine.

Measures BASELINE.json's metric ("decoded+filtered values/sec & HBM GB/s,
TSBS devops scan") on BASELINE.json configs[1] — the north-star workload:

  10,000 series x 1,000,000 points per GPU; timestamps at 1 s spacing
  (DeltaTs/RLE pages); f64 gauge values = clipped random walk in [0,100],
  80% of series quantized to 1/4 (measurement precision) and 20%
  full-precision (~8 B/val) — blended ~1.5 B/val, the 12-20
  bits/val TSBS-devops regime SURVEY.md §8d assumes; pages of 4,000 rows;
  time-range filter ts in [25%,75%] of the range (50% selectivity); fused
  per-5-minute max/sum/count buckets.

One STEP = one full pass of the hot path over the resident page set:
decode ts + f64 -> closed-interval time filter -> compacted row output
+ per-bucket aggregates.  Inputs (raw TSM pages) are resident in HBM
before the timed region.  Multi-GPU: series sharded by series_id % N
(weak scaling — each rank holds its own 10k-series shard), one RCCL
all-reduce of the per-bucket sum/count (+max) per step, exactly the one
collective the path needs.

`value` = field values decoded+filtered per second, whole-job across all
ranks.  A "value" is one point of the measured f64 field column (1e7*N
series-points per step... see config).  Roofline + cpu_baseline objects
per the driver contract; the roofline covers the dominant kernel
(k_gor_lds_filtered, the fused Gorilla page decoder).
"""
import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

NS = 1_000_000_000
T0 = 1_700_000_000_000_000_000
BUCKET_NS = 300 * NS  # 5 minutes


def _host_threads():
    """OMP threads per rank: divide the host cores among co-located ranks
    (the round-end 8-GPU run launches 8 ranks on one node)."""
    lw = int(os.environ.get("LOCAL_WORLD_SIZE", "1"))
    return max(1, (os.cpu_count() or 8) // max(1, lw))


def build_workload(nseries, npts, page_rows, unique, sub_batches, nfields=1, seed=231):
    """Generate encoded TSM pages. `unique` distinct value series are
    generated and replicated across series (device copies are distinct, so
    HBM traffic is real); ts pages are identical across series (TSBS
    devops: all hosts share the epoch grid)."""
    import cnosdb_amd as gs
    rng = np.random.default_rng(seed)
    npages = npts // page_rows
    ts_pages = []
    for p in range(npages):
        ts = T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS
        ts_pages.append(gs.page_of(ts, gs.CT_TIME))

    # unique value patterns, encoded with the OMP batch encoder in chunks
    lib = gs.PageLib().lib
    import ctypes
    lib.gs_encode_f64_pages_omp.restype = ctypes.c_int32
    cap = page_rows * 12 + 64
    chunk = max(1, min(unique, 512 // npages or 1))
    val_pages = []  # [unique][npages] full page bytes
    u = 0
    while u < unique:
        cu = min(chunk, unique - u)
        vals = np.empty((cu, npts))
        for k in range(cu):
            walk = np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50, 0, 100)
            if (u + k) % 5 == 0:
                vals[k] = walk  # full-precision gauge (hard pages, ~8 B/val)
            else:
                # measurement-precision gauge: 1/4 quantization; blended
                # ~11-14 bits/val, the 12-20 bits/val TSBS-devops regime
                # SURVEY.md §8d assumes
                vals[k] = np.round(walk * 4) / 4
        flat = np.ascontiguousarray(vals.reshape(-1))
        total_pages = cu * npages
        enc = np.zeros(total_pages * cap, dtype=np.uint8)
        lens = np.zeros(total_pages, dtype=np.int64)
        st = lib.gs_encode_f64_pages_omp(
            flat.ctypes.data_as(ctypes.c_void_p), page_rows, total_pages,
            enc.ctypes.data_as(ctypes.c_void_p), cap,
            lens.ctypes.data_as(ctypes.c_void_p), _host_threads())
        assert st == 0, st
        for k in range(cu):
            row = []
            for p in range(npages):
                i = k * npages + p
                data = enc[i * cap:i * cap + lens[i]].tobytes()
                row.append(gs.build_page(data, page_rows))
            val_pages.append(row)
        u += cu

    # pack pages into one contiguous buffer per sub-batch (group-major,
    # [ts, val] per group; groups series-major) for the vectorized upload
    pattern = rng.integers(0, unique, nseries)
    raw_bytes = 0
    sub = []
    per_sb = nseries // sub_batches
    for sb in range(sub_batches):
        parts, lens, cts = [], [], []
        sids = np.repeat(np.arange(sb * per_sb, (sb + 1) * per_sb,
                                   dtype=np.uint32), npages)
        for s in range(sb * per_sb, (sb + 1) * per_sb):
            for p in range(npages):
                parts.append(ts_pages[p])
                for f in range(nfields):
                    parts.append(val_pages[(pattern[s] + f) % unique][p])
        buf = b"".join(parts)
        lens = np.array([len(x) for x in parts], dtype=np.int64)
        offs = np.zeros(lens.size, dtype=np.int64)
        np.cumsum(lens[:-1], out=offs[1:])
        nvals = np.full(lens.size, page_rows, dtype=np.int64)
        cts = np.tile(np.array([gs.CT_TIME] + [gs.CT_F64] * nfields,
                               dtype=np.uint8), lens.size // (1 + nfields))
        raw_bytes += buf.__len__()
        sub.append((buf, offs, lens, nvals, cts, sids))
    return sub, raw_bytes, npages


def cpu_baseline_leg(nseries_sample, npts, page_rows, lo, hi, seed=231):
    """Oracle (C restatement, OpenMP) timed on host cores over a bounded
    sample of the same workload. kind="port" (the reference is Rust and
    cannot be compiled here)."""
    import ctypes
    from oracle import pyoracle as orc
    import cnosdb_amd as gs

    rng = np.random.default_rng(seed)
    npages = npts // page_rows
    # build sample pages (encoded data buffers + bitsets), same generator
    # mix as build_workload
    ts_datas, val_datas = [], []
    for p in range(npages):
        ts = T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS
        ts_datas.append(gs.encode_ts(ts))
    for s in range(nseries_sample):
        walk = np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50, 0, 100)
        v = walk if s % 5 == 0 else np.round(walk * 4) / 4
        for p in range(npages):
            val_datas.append(gs.encode_f64(v[p * page_rows:(p + 1) * page_rows]))

    o = orc.Oracle().lib

    class PD(ctypes.Structure):
        _fields_ = [("data", ctypes.c_void_p), ("data_len", ctypes.c_uint64),
                    ("bitset", ctypes.c_void_p), ("nrows", ctypes.c_int64),
                    ("out_off", ctypes.c_uint64), ("ctype", ctypes.c_uint8)]

    nb = (page_rows + 7) // 8
    bitset = np.full(nb, 0xFF, dtype=np.uint8)
    total_pages = nseries_sample * npages
    descs_v = (PD * total_pages)()
    keep = []
    for i, d in enumerate(val_datas):
        a = np.frombuffer(d, dtype=np.uint8)
        keep.append(a)
        descs_v[i] = PD(a.ctypes.data, a.size, bitset.ctypes.data, page_rows,
                        i * page_rows, 1)
    descs_t = (PD * total_pages)()
    for s in range(nseries_sample):
        for p in range(npages):
            i = s * npages + p
            a = np.frombuffer(ts_datas[p], dtype=np.uint8)
            keep.append(a)
            descs_t[i] = PD(a.ctypes.data, a.size, bitset.ctypes.data,
                            page_rows, i * page_rows, 0)
    rows = total_pages * page_rows
    out_ts = np.zeros(rows, dtype=np.int64)
    out_v = np.zeros(rows, dtype=np.float64)
    cores = _host_threads()
    t = time.perf_counter()
    st = o.orc_decode_pages_omp(descs_t, total_pages,
                                out_ts.ctypes.data_as(ctypes.c_void_p), cores)
    assert st == 0
    st = o.orc_decode_pages_omp(descs_v, total_pages,
                                out_v.ctypes.data_as(ctypes.c_void_p), cores)
    assert st == 0
    # per-series time filter + compaction (numpy leg of the CPU path)
    sel_parts = []
    ots = out_ts.reshape(nseries_sample, npts)
    ov = out_v.reshape(nseries_sample, npts)
    for s in range(nseries_sample):
        a = np.searchsorted(ots[s], lo, side="left")
        b = np.searchsorted(ots[s], hi, side="right")
        sel_parts.append(ov[s][a:b])
    np.concatenate(sel_parts)
    dt = time.perf_counter() - t
    values = nseries_sample * npts
    return {"value": values / dt, "unit": "values/s", "cores": cores,
            "kind": "port",
            "sample": f"{nseries_sample} series x {npts} pts "
                      f"(decode ts+f64, filter, compact; oracle C+OpenMP)"}


def bench_compact(args):
    """BASELINE config #5: per series k=8 overlapping L0 column groups,
    GPU merge + dedup-by-ts (newest non-null wins) + re-encode.  One step =
    merge + re-encode of the whole resident set.  Secondary benchmark line
    (the driver's headline line is the scan mode)."""
    import torch
    import cnosdb_amd as gs
    rng = np.random.default_rng(231)
    k, nseries, grid_n = 8, args.series, args.npts
    eng = gs.Engine(0)
    grid = T0 + np.arange(grid_n, dtype=np.int64) * NS
    # per-stream ts layout shared across series (20% collisions via
    # overlapping random subsets, config #5)
    t_setup = time.perf_counter()
    stream_ts = []
    for f in range(k):
        take = rng.random(grid_n) < 0.25
        stream_ts.append(grid[take])
    uniq = min(64, nseries)
    gsets, tss, vls = [], [], []
    total = 0
    for f in range(k):
        ts = stream_ts[f]
        tpage = gs.page_of(ts, gs.CT_TIME)
        vpages = [gs.page_of(np.round(np.clip(
            np.cumsum(rng.normal(0, 1, ts.size)) + 50, 0, 100), 2), gs.CT_F64)
            for _ in range(uniq)]
        groups = [(s, [(tpage, gs.CT_TIME), (vpages[s % uniq], gs.CT_F64)])
                  for s in range(nseries)]
        gset = eng.upload(groups, validate_crc=False)
        d_ts = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
        eng.decode(gset, 0, d_ts)
        eng.decode(gset, 1, d_val)
        gsets.append(gset)
        tss.append(d_ts)
        vls.append(d_val)
        total += gset.rows
    d_ots = torch.zeros(total, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(total, dtype=torch.float64, device="cuda")
    block_rows = 1000  # re-encode block size: plenty of pages for the
    # thread-per-page encoder (reference caps blocks by bytes, not rows)
    cap = block_rows * 12 + 128
    setup_s = time.perf_counter() - t_setup

    def step():
        out_rows, offs = eng.compact_merge(gsets, tss, vls, [None] * k,
                                           d_ots, d_oval)
        # vectorized per-series block split (a python loop here cost more
        # than the kernels at small block sizes)
        counts = offs[1:] - offs[:-1]
        npg_per = -(-counts // block_rows)
        series_base = np.repeat(offs[:-1], npg_per)
        series_cnt = np.repeat(counts, npg_per)
        intra = np.concatenate([np.arange(n) for n in npg_per]) * block_rows
        row_off = series_base + intra
        rows_arr = np.minimum(block_rows, series_cnt - intra)
        npg = len(row_off)
        d_enc_ts = torch.zeros(npg * cap, dtype=torch.uint8, device="cuda")
        d_enc_v = torch.zeros(npg * cap, dtype=torch.uint8, device="cuda")
        eng.encode_pages_dev(0, d_ots, np.array(row_off), np.array(rows_arr),
                             d_enc_ts, cap)
        lens = eng.encode_pages_dev(2, d_oval, np.array(row_off),
                                    np.array(rows_arr), d_enc_v, cap)
        return out_rows, int(lens.sum())

    for _ in range(args.warmup):
        out_rows, enc_bytes = step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out_rows, enc_bytes = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    value = total * args.steps / dt

    # CPU baseline: oracle C k-way merge (OpenMP over series) + host
    # re-encode on a bounded sample
    cpu_baseline = None
    if not args.skip_cpu_baseline:
        import ctypes
        from oracle import pyoracle as orc

        class OrcStream(ctypes.Structure):
            _fields_ = [("ts", ctypes.c_void_p), ("val", ctypes.c_void_p),
                        ("valid", ctypes.c_void_p), ("n", ctypes.c_int64)]

        o = orc.Oracle().lib
        o.orc_merge_dedup_many.restype = ctypes.c_int32
        sample_series = min(64, nseries)
        host = []  # [series][k] -> (ts, vals)
        total_in = 0
        for s_ in range(sample_series):
            per = []
            for f in range(k):
                ts = stream_ts[f]
                vals = np.round(np.clip(
                    np.cumsum(rng.normal(0, 1, ts.size)) + 50, 0, 100), 2)
                per.append((ts, vals))
                total_in += ts.size
            host.append(per)
        streams = (OrcStream * (sample_series * k))()
        caps = np.zeros(sample_series + 1, dtype=np.int64)
        for s_ in range(sample_series):
            cap = 0
            for f in range(k):
                ts, vals = host[s_][f]
                streams[s_ * k + f] = OrcStream(
                    ts.ctypes.data, vals.ctypes.data, None, ts.size)
                cap += ts.size
            caps[s_ + 1] = caps[s_] + cap
        out_ts_h = np.zeros(caps[-1], dtype=np.int64)
        out_v_h = np.zeros(caps[-1], dtype=np.float64)
        cnts = np.zeros(sample_series, dtype=np.int64)
        cores = _host_threads()
        tcb = time.perf_counter()
        st_ = o.orc_merge_dedup_many(
            streams, k, sample_series,
            caps[:-1].ctypes.data_as(ctypes.c_void_p),
            out_ts_h.ctypes.data_as(ctypes.c_void_p),
            out_v_h.ctypes.data_as(ctypes.c_void_p),
            cnts.ctypes.data_as(ctypes.c_void_p), cores)
        assert st_ == 0
        for s_ in range(sample_series):  # re-encode (serial C per call)
            a, b = caps[s_], caps[s_] + cnts[s_]
            r = a
            while r < b:
                nblk = min(block_rows, b - r)
                gs.encode_ts(out_ts_h[r:r + nblk])
                gs.encode_f64(out_v_h[r:r + nblk])
                r += nblk
        dt_cb = time.perf_counter() - tcb
        cpu_baseline = {"value": total_in / dt_cb, "unit": "rows/s",
                        "cores": cores, "kind": "port",
                        "sample": f"{sample_series} series x k={k} streams "
                                  "(oracle C k-way merge, OpenMP; host "
                                  "re-encode single-thread)"}
    print(json.dumps({
        "metric": "compaction merge rows/sec (config #5: k=8 overlapping L0 groups, dedup-by-ts, re-encode)",
        "value": value,
        "unit": "rows/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "f64",
        "data": "synthetic",
        "config": {"workload": "compaction-merge (BASELINE configs[4])",
                   "k": k, "series": nseries,
                   "rows_in": int(total), "rows_out": int(out_rows),
                   "encoded_out_bytes": enc_bytes,
                   "block_rows": block_rows, "setup_s": round(setup_s, 1)},
        "cpu_baseline": cpu_baseline,
    }))
    for g in gsets:
        g.free()
    eng.close()


def bench_strings(args):
    """string/tag column decode (snappy blocks ->
    Arrow varbinary layout).  Secondary benchmark line; the driver's
    headline is the scan mode."""
    import torch
    import cnosdb_amd as gs
    rng = np.random.default_rng(231)
    page_rows = args.page_rows
    npages = args.series if args.series != 10000 else 60000
    per_set = min(npages, 60000)  # 268M-row scan limit at 4000 rows/page
    uniq = min(args.unique, 256)
    tagpool = [b"hostname=host_%04d,region=region_%02d,rack=%02d"
               % (i, i % 16, i % 64) for i in range(100)]
    eng = gs.Engine(0)
    t_setup = time.perf_counter()
    tpage = gs.page_of(T0 + np.arange(page_rows, dtype=np.int64) * NS,
                       gs.CT_TIME)
    blocks, str_pages, payloads = [], [], []
    for u in range(uniq):
        idx = rng.integers(0, len(tagpool), page_rows)
        strs = [tagpool[i] for i in idx]
        blk = gs.encode_str(strs)
        blocks.append(blk)
        payloads.append(sum(len(s) for s in strs))
        str_pages.append(gs.build_page(blk, page_rows))
    # one contiguous buffer: [ts page][uniq str pages]; page offsets repeat
    buf = np.frombuffer(b"".join([tpage] + str_pages), dtype=np.uint8)
    str_off = np.zeros(uniq, dtype=np.int64)
    str_len = np.array([len(p) for p in str_pages], dtype=np.int64)
    str_off[0] = len(tpage)
    np.cumsum(str_len[:-1], out=str_off[1:])
    str_off[1:] += len(tpage)
    sets = []
    total_rows = 0
    total_payload = 0
    nsets = (npages + per_set - 1) // per_set
    pay = np.array(payloads, dtype=np.int64)
    for s in range(nsets):
        cnt = min(per_set, npages - s * per_set)
        u_idx = (np.arange(cnt, dtype=np.int64) + s * per_set) % uniq
        page_off = np.empty(cnt * 2, dtype=np.int64)
        page_len = np.empty(cnt * 2, dtype=np.int64)
        page_off[0::2] = 0
        page_len[0::2] = len(tpage)
        page_off[1::2] = str_off[u_idx]
        page_len[1::2] = str_len[u_idx]
        nv = np.full(cnt * 2, page_rows, dtype=np.int64)
        ct = np.tile(np.array([gs.CT_TIME, gs.CT_STR], dtype=np.uint8), cnt)
        gset = eng.upload_packed(buf, page_off, page_len, nv, ct,
                                 np.arange(cnt, dtype=np.int64), 2,
                                 validate_crc=False)
        rows = gset.rows
        total_rows += rows
        total_payload += int(pay[u_idx].sum())
        sets.append((gset,
                     torch.zeros(rows + 1, dtype=torch.int64, device="cuda"),
                     torch.zeros(rows * 64, dtype=torch.uint8, device="cuda")))
    setup_s = time.perf_counter() - t_setup

    def step():
        t = 0
        for gset, d_off, d_bytes in sets:
            t += eng.decode_str(gset, 1, d_off, d_bytes)
        return t

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        got = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert got == total_payload
    value = total_rows * args.steps / dt

    cpu_baseline = None
    if not args.skip_cpu_baseline:
        import ctypes
        from oracle import pyoracle as orc
        o = orc.Oracle().lib
        o.orc_str_decode_pages_omp.restype = ctypes.c_int64
        cores = _host_threads()
        sample_pages = min(2048, npages)
        bufs = [np.frombuffer(blocks[p % uniq], dtype=np.uint8)
                for p in range(sample_pages)]
        ptrs = (ctypes.c_void_p * sample_pages)(*[b.ctypes.data for b in bufs])
        lens_c = (ctypes.c_size_t * sample_pages)(
            *[b.size for b in bufs])
        tcb = time.perf_counter()
        w = o.orc_str_decode_pages_omp(ptrs, lens_c, sample_pages, page_rows,
                                       max(payloads) + 4096, cores)
        dt_cb = time.perf_counter() - tcb
        assert w > 0
        cpu_baseline = {"value": sample_pages * page_rows / dt_cb,
                        "unit": "strings/s", "cores": cores, "kind": "port",
                        "sample": f"{sample_pages} pages x {page_rows} rows "
                                  "(oracle C snappy decode, OpenMP)"}
    print(json.dumps({
        "metric": "string/tag column decode strings/sec (snappy blocks -> Arrow varbinary)",
        "value": value,
        "unit": "strings/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "u8",
        "data": "synthetic",
        "config": {"workload": "tag-column decode (SURVEY §8f row 1)",
                   "pages": npages, "page_rows": page_rows,
                   "tag_cardinality": len(tagpool),
                   "payload_bytes_per_step": int(total_payload),
                   "payload_GBps": total_payload * args.steps / dt / 1e9,
                   "setup_s": round(setup_s, 1)},
        "cpu_baseline": cpu_baseline,
    }))
    for gset, _, _ in sets:
        gset.free()
    eng.close()



def cpu_baseline_general(nseries_sample, npts, page_rows, lo, hi, tstones,
                         pred_a, seed=231):
    """Oracle C decode (real null bitsets) + numpy filter/tombstone/pred/
    compact legs over a bounded general-shape sample (kind="port")."""
    import ctypes
    from oracle import pyoracle as orc
    import cnosdb_amd as gs

    rng = np.random.default_rng(seed)
    npages = npts // page_rows
    nb = (page_rows + 7) // 8
    full = np.full(nb, 0xFF, dtype=np.uint8)
    ts_datas = []
    for p in range(npages):
        ts = T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS
        ts_datas.append(gs.encode_ts(ts))
    val_datas, bitsets = [], []
    for s in range(nseries_sample):
        walk = np.round(np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50,
                                0, 100), 1)
        valid = rng.random(npts) > 0.10
        for p in range(npages):
            sl = slice(p * page_rows, (p + 1) * page_rows)
            v, m = walk[sl], valid[sl]
            val_datas.append(gs.encode_f64(v[m]) if m.any() else b"")
            bitsets.append(np.packbits(m, bitorder="little"))

    o = orc.Oracle().lib

    class PD(ctypes.Structure):
        _fields_ = [("data", ctypes.c_void_p), ("data_len", ctypes.c_uint64),
                    ("bitset", ctypes.c_void_p), ("nrows", ctypes.c_int64),
                    ("out_off", ctypes.c_uint64), ("ctype", ctypes.c_uint8)]

    total_pages = nseries_sample * npages
    descs_v = (PD * total_pages)()
    keep = []
    for i, d in enumerate(val_datas):
        a = np.frombuffer(d, dtype=np.uint8)
        keep.append(a)
        descs_v[i] = PD(a.ctypes.data if a.size else None, a.size,
                        bitsets[i].ctypes.data, page_rows, i * page_rows, 1)
    descs_t = (PD * total_pages)()
    for s in range(nseries_sample):
        for p in range(npages):
            i = s * npages + p
            a = np.frombuffer(ts_datas[p], dtype=np.uint8)
            keep.append(a)
            descs_t[i] = PD(a.ctypes.data, a.size, full.ctypes.data,
                            page_rows, i * page_rows, 0)
    rows = total_pages * page_rows
    out_ts = np.zeros(rows, dtype=np.int64)
    out_v = np.zeros(rows, dtype=np.float64)
    cores = _host_threads()
    t = time.perf_counter()
    st = o.orc_decode_pages_omp(descs_t, total_pages,
                                out_ts.ctypes.data_as(ctypes.c_void_p), cores)
    assert st == 0
    st = o.orc_decode_pages_omp(descs_v, total_pages,
                                out_v.ctypes.data_as(ctypes.c_void_p), cores)
    assert st == 0
    # numpy legs: per-series time span + tombstone clear + value pred +
    # compact + 5-min bucket agg (reader.rs:634-656 / filter.rs:91-142)
    ots = out_ts.reshape(nseries_sample, npts)
    ov = out_v.reshape(nseries_sample, npts)
    valid_all = np.concatenate(
        [np.unpackbits(b, bitorder="little")[:page_rows] for b in bitsets]
    ).reshape(nseries_sample, npts).astype(bool)
    nbuckets = int(npts * NS // BUCKET_NS) + 1
    for s in range(nseries_sample):
        a = np.searchsorted(ots[s], lo, side="left")
        b = np.searchsorted(ots[s], hi, side="right")
        tsl, vsl, msl = ots[s][a:b], ov[s][a:b], valid_all[s][a:b].copy()
        for t0_, t1_ in tstones:
            x = np.searchsorted(tsl, t0_, side="left")
            y = np.searchsorted(tsl, t1_, side="right")
            msl[x:y] = False
        sel = msl & (vsl > pred_a)
        ts_f, v_f = tsl[sel], vsl[sel]
        bi = ((ts_f - T0) // BUCKET_NS).astype(np.int64)
        np.bincount(bi, weights=v_f, minlength=nbuckets)      # sum leg
        cnts = np.bincount(bi, minlength=nbuckets)            # count leg
        if v_f.size:  # max leg: bi is nondecreasing (ts sorted)
            offs_ = np.searchsorted(bi, np.flatnonzero(cnts))
            np.maximum.reduceat(v_f, offs_)
    dt = time.perf_counter() - t
    values = nseries_sample * npts
    return {"value": values / dt, "unit": "values/s", "cores": cores,
            "kind": "port",
            "sample": f"{nseries_sample} series x {npts} pts general shape "
                      f"(10% nulls, 2 tombstones, pred; oracle C+OpenMP + numpy legs)"}


def bench_general(args):
    """General-shape scan: 10% nulls + 2 tombstone
    ranges + value predicate v>50 — the non-TSBS shape that takes the
    general gs_scan path (sequential null-scatter decode, decoded-ts span
    search, value mask, masked compact, k_agg_partial).  Secondary
    benchmark line."""
    import torch
    import cnosdb_amd as gs
    rng = np.random.default_rng(231)
    nseries, npts, page_rows = args.series, args.npts, args.page_rows
    npages = npts // page_rows
    lo = T0 + int(0.25 * npts) * NS
    hi = T0 + int(0.75 * npts) * NS - 1
    nbuckets = int(npts * NS // BUCKET_NS) + 1
    tstones = [(T0 + int(0.30 * npts) * NS, T0 + int(0.33 * npts) * NS - 1),
               (T0 + int(0.60 * npts) * NS, T0 + int(0.63 * npts) * NS - 1)]
    pred = ("gt", 50.0)
    eng = gs.Engine(0)
    t_setup = time.perf_counter()
    uniq = min(args.unique, nseries)
    tpages = [gs.page_of(
        T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS,
        gs.CT_TIME) for p in range(npages)]
    vpool = []
    for _ in range(uniq):
        walk = np.round(np.clip(
            np.cumsum(rng.normal(0, 0.5, page_rows)) + 50, 0, 100), 1)
        valid = rng.random(page_rows) > 0.10
        vpool.append(gs.page_of(walk, gs.CT_F64, valid))
    groups = []
    for s in range(nseries):
        for p in range(npages):
            groups.append((s, [(tpages[p], gs.CT_TIME),
                               (vpool[(s * npages + p) % uniq], gs.CT_F64)]))
    gset = eng.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    agg = dict(bucket_ns=BUCKET_NS, t0=T0, n_buckets=nbuckets,
               d_max=torch.full((nbuckets,), -np.inf, dtype=torch.float64,
                                device="cuda"),
               d_sum=torch.zeros(nbuckets, dtype=torch.float64, device="cuda"),
               d_count=torch.zeros(nbuckets, dtype=torch.int64, device="cuda"))
    setup_s = time.perf_counter() - t_setup

    def step():
        return eng.scan(gset, d_ts, d_val, time_range=(lo, hi),
                        tombstones=tstones, d_out_ts=d_ots, d_out_val=d_oval,
                        agg=agg, value_pred=pred)

    for _ in range(args.warmup):
        r = step()
    torch.cuda.synchronize()
    t0_ = time.perf_counter()
    for _ in range(args.steps):
        r = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0_
    value = rows * args.steps / dt
    cpu_baseline = None
    if not args.skip_cpu_baseline:
        cpu_baseline = cpu_baseline_general(16, npts, page_rows, lo, hi,
                                            tstones, pred[1])
    line = {
        "metric": "decoded+filtered values/sec & HBM GB/s, TSBS devops scan, 1/2/4/8 GPU",
        "value": value, "unit": "values/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000,
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "f64", "data": "synthetic",
        "config": {
            "workload": "general-shape scan (10% nulls + 2 tombstone ranges "
                        "+ value pred v>50; general path, 1 stream)",
            "series_per_gpu": nseries, "points_per_series": npts,
            "page_rows": page_rows, "out_rows_per_step": int(r.out_rows),
            "setup_s": round(setup_s, 1),
            "phase_ms": {"decode_ts": r.ms_decode_ts,
                         "decode_f64": r.ms_decode_val,
                         "filter": r.ms_filter, "compact": r.ms_compact,
                         "agg": r.ms_agg},
        },
        "roofline": None,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(line))



def bench_groupby(args):
    """GROUP BY tag leg (SURVEY 8f continuation): fused aggregate scan +
    gs_groupby_tag keyed on the decoded varbinary tag column (tags are
    SeriesKey members, constant per series).  One step = scan (decode +
    filter + per-series 5-min buckets) + per-(tag, bucket) merge.  The
    tag column itself is decoded once at setup (its decode rate is the
    strings benchmark line)."""
    import torch
    import cnosdb_amd as gs
    rng = np.random.default_rng(231)
    nseries, npts, page_rows = args.series, args.npts, args.page_rows
    npages = npts // page_rows
    uniq_tags = 256
    tagpool = [b"host_%06d" % i for i in range(uniq_tags)]
    t_setup = time.perf_counter()
    tpages = [gs.page_of(
        T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS,
        gs.CT_TIME) for p in range(npages)]
    uniq_f = min(64, nseries)
    fpool = [gs.page_of(np.round(np.clip(
        np.cumsum(rng.normal(0, 0.5, page_rows)) + 50, 0, 100), 1),
        gs.CT_F64) for _ in range(uniq_f)]
    spool = {}
    groups = []
    for s_ in range(nseries):
        tag = tagpool[s_ % uniq_tags]
        if tag not in spool:
            spool[tag] = gs.str_page_of([tag] * page_rows)
        for p in range(npages):
            groups.append((s_, [(tpages[p], gs.CT_TIME),
                                (fpool[(s_ + p) % uniq_f], gs.CT_F64),
                                (spool[tag], gs.CT_STR)]))
    eng = gs.Engine(0)
    gset = eng.upload(groups)
    rows = gset.rows
    lo = T0 + int(0.25 * npts) * NS
    hi = T0 + int(0.75 * npts) * NS - 1
    nb = int(npts * NS // BUCKET_NS) + 1
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    agg = dict(bucket_ns=BUCKET_NS, t0=T0, n_buckets=nb,
               d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                device="cuda"),
               d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
               d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
    d_soff = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_sbytes = torch.zeros(rows * 12, dtype=torch.uint8, device="cuda")
    eng.decode_str(gset, 2, d_soff, d_sbytes)  # setup: tag column resident
    cap = 512
    g_max = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_sum = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_cnt = torch.zeros(cap * nb, dtype=torch.int64, device="cuda")
    setup_s = time.perf_counter() - t_setup

    def step():
        eng.scan(gset, d_ts, d_val, time_range=(lo, hi),
                 d_out_ts=d_ots, d_out_val=d_oval, agg=agg)
        return gs.groupby_tag(eng, gset, nb, g_max, g_sum, g_cnt, cap)

    for _ in range(args.warmup):
        ngids, _ = step()
    torch.cuda.synchronize()
    t0_ = time.perf_counter()
    for _ in range(args.steps):
        ngids, _ = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0_
    # isolate the group-by increment
    torch.cuda.synchronize()
    t1_ = time.perf_counter()
    for _ in range(args.steps):
        gs.groupby_tag(eng, gset, nb, g_max, g_sum, g_cnt, cap)
    torch.cuda.synchronize()
    gb_ms = (time.perf_counter() - t1_) / args.steps * 1000
    value = rows * args.steps / dt
    line = {
        "metric": "decoded+filtered values/sec & HBM GB/s, TSBS devops scan, 1/2/4/8 GPU",
        "value": value, "unit": "values/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000,
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "f64", "data": "synthetic",
        "config": {
            "workload": "TSBS GROUP-BY-hostname: fused agg scan + "
                        "per-(tag,bucket) merge over the decoded tag column "
                        "(256 hostnames)",
            "series_per_gpu": nseries, "points_per_series": npts,
            "page_rows": page_rows, "n_tags": int(ngids),
            "n_buckets": nb, "groupby_ms_per_step": gb_ms,
            "setup_s": round(setup_s, 1),
        },
        "roofline": None, "cpu_baseline": None,
    }
    print(json.dumps(line))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=["scan", "compact", "strings", "general", "groupby"], default="scan")
    ap.add_argument("--gpus", type=int, default=1,
                    help="driver contract flag; the actual world size comes "
                         "from the torchrun environment (WORLD_SIZE)")
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--series", type=int, default=10000)
    ap.add_argument("--npts", type=int, default=1_000_000)
    ap.add_argument("--page-rows", type=int, default=125000,
                help="rows per page; default = reference-shaped (1M pts / 8 pages,\n                max_datablock_size ~100KiB rows, comapcting_block_meta_group.rs:87)")
    ap.add_argument("--sub-batches", type=int, default=2)
    ap.add_argument("--unique", type=int, default=256)
    ap.add_argument("--fields", type=int, default=1,
                    help="f64 field pages per group (8 = TSBS cpu-max-all-8, config #3)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    if args.mode == "compact":
        if args.series == 10000:
            args.series = 512
        if args.npts == 1_000_000:
            args.npts = 100_000
        bench_compact(args)
        return
    if args.mode == "strings":
        if args.page_rows == 125000:
            args.page_rows = 4000  # round-1 strings shape: 60k x 4000-row pages
        bench_strings(args)
        return
    if args.mode == "general":
        if args.series == 10000:
            args.series = 2500  # one resident set (ts+val+outs+masks ~34 B/row)
        if args.page_rows == 125000:
            args.page_rows = 4000  # null pages: chunked GORN path, smaller pages
        bench_general(args)
        return
    if args.mode == "groupby":
        if args.series == 10000:
            args.series = 1000  # decode_str per-set row cap is 268M
        if args.npts == 1_000_000:
            args.npts = 250_000
        bench_groupby(args)
        return

    import torch
    import cnosdb_amd as gs

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    # GS_BENCH_BACKEND=gloo lets the multi-rank path be validated with
    # several ranks sharing one GPU (NCCL requires a device per rank)
    backend = os.environ.get("GS_BENCH_BACKEND", "nccl")
    ndev = torch.cuda.device_count()
    dev_idx = local_rank % max(1, ndev)
    device = torch.device(f"cuda:{dev_idx}")
    torch.cuda.set_device(device)
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        if backend == "nccl":
            dist.init_process_group("nccl", device_id=device)
        else:
            dist.init_process_group(backend)

    nseries, npts, page_rows = args.series, args.npts, args.page_rows
    assert npts % page_rows == 0
    assert nseries % args.sub_batches == 0
    lo = T0 + int(0.25 * npts) * NS
    hi = T0 + int(0.75 * npts) * NS - 1  # closed interval, 50% of rows
    nbuckets = int(npts * NS // BUCKET_NS) + 1

    # ---- build + upload (untimed setup) ----
    t_setup = time.perf_counter()
    sub, raw_bytes, npages = build_workload(
        nseries, npts, page_rows, min(args.unique, nseries),
        args.sub_batches, nfields=args.fields, seed=231 + rank)
    # two engine contexts = two HIP streams on the one device: sub-batches
    # alternate streams so the ALU-bound Gorilla decode of one overlaps the
    # HBM-bound aggregate/ts phases of the other
    engines = [gs.Engine(dev_idx), gs.Engine(dev_idx)]
    sets, eng_of = [], []
    raw_f64_bytes_sb0 = None
    for sb, (buf, offs, lens, nvals, cts, sids) in enumerate(sub):
        e = engines[sb % 2]
        sets.append(e.upload_packed(buf, offs, lens, nvals, cts, sids,
                                    pages_per_group=1 + args.fields,
                                    validate_crc=False))
        eng_of.append(e)
        if sb == 0:
            m = np.ones(lens.size, dtype=bool)
            m[::1 + args.fields] = False  # drop ts pages
            raw_f64_bytes_sb0 = int(lens[m].sum()) // args.fields
    del sub
    sb_rows = sets[0].rows
    nf = args.fields
    douts = [(torch.zeros(sb_rows, dtype=torch.int64, device=device),
              torch.zeros(nf * sb_rows, dtype=torch.float64, device=device))
             for _ in range(2)]
    aggs = []
    for _ in range(2):
        aggs.append(dict(
            bucket_ns=BUCKET_NS, t0=T0, n_buckets=nbuckets,
            d_max=torch.full((nf * nbuckets,), -np.inf, dtype=torch.float64,
                             device=device),
            d_sum=torch.zeros(nf * nbuckets, dtype=torch.float64,
                              device=device),
            d_count=torch.zeros(nf * nbuckets, dtype=torch.int64,
                                device=device)))
    d_sum = torch.zeros(nf * nbuckets, dtype=torch.float64, device=device)
    d_cnt = torch.zeros(nf * nbuckets, dtype=torch.int64, device=device)
    d_max = torch.zeros(nf * nbuckets, dtype=torch.float64, device=device)
    setup_s = time.perf_counter() - t_setup

    fields_list = list(range(nf))
    dummy = [(torch.zeros(1, dtype=torch.int64, device=device),
              torch.zeros(1, dtype=torch.float64, device=device))
             for _ in range(2)]

    def step():
        out_rows = 0
        phase_ms = np.zeros(5)
        # one span/ts pass + all fields per set (gs_scan_fields): the
        # round-1 per-field loop paid the span search + 8 B/row ts
        # generation once per field
        for i, st_ in enumerate(sets):
            gs.scan_fields_async(eng_of[i], st_, fields_list,
                                 dummy[i % 2][0], dummy[i % 2][1],
                                 (lo, hi), douts[i % 2][0],
                                 douts[i % 2][1], aggs[i % 2])
        for i, st_ in enumerate(sets):
            r = eng_of[i].scan_wait(st_)
            out_rows += r.out_rows * nf
            phase_ms += [r.ms_decode_ts, r.ms_decode_val, r.ms_filter,
                         r.ms_compact, r.ms_agg]
        # combine the per-stream bucket partials (tiny)
        torch.add(aggs[0]["d_sum"], aggs[1]["d_sum"], out=d_sum)
        torch.add(aggs[0]["d_count"], aggs[1]["d_count"], out=d_cnt)
        torch.maximum(aggs[0]["d_max"], aggs[1]["d_max"], out=d_max)
        if dist is not None:
            dist.all_reduce(d_sum)
            dist.all_reduce(d_cnt)
            dist.all_reduce(d_max, op=dist.ReduceOp.MAX)
        return out_rows, phase_ms

    # ---- warmup ----
    for _ in range(args.warmup):
        out_rows, _ = step()
    if dist is not None:
        dist.barrier()
    torch.cuda.synchronize()

    # ---- timed steps ----
    t0 = time.perf_counter()
    phases = np.zeros(5)
    for _ in range(args.steps):
        out_rows, pm = step()
        phases += pm
    if dist is not None:
        dist.barrier()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    # MAX over ranks
    if dist is not None:
        t_t = torch.tensor([dt], device=device)
        dist.all_reduce(t_t, op=dist.ReduceOp.MAX)
        dt = float(t_t.item())

    # clean (unoverlapped) per-kernel timings for the roofline: one extra
    # scan on a single stream, untimed region (stream-distance HIP events
    # are inflated by cross-stream overlap during the timed steps)
    engines[0].scan_async(sets[0], douts[0][0], douts[0][1],
                          time_range=(lo, hi), agg=aggs[0], field_col=0)
    r_cal = engines[0].scan_wait(sets[0])

    values_per_step = nseries * npts * args.fields  # field values decoded+filtered per rank
    total_values = values_per_step * world * args.steps
    value = total_values / dt
    ms_per_step = dt / args.steps * 1000

    # roofline: dominant kernel = k_gor_lds_filtered (fused Gorilla decode),
    # 1 launch per sub-batch. algorithmic bytes per launch = compressed f64
    # bytes read + 8 B per SELECTED row written (the fused kernel skips the
    # stores of filtered-out rows; SURVEY.md §8d: count bytes actually
    # moved).  The unfused number (8 B x all rows) is stated in DESIGN.md.
    raw_f64_bytes = raw_f64_bytes_sb0 - \
        (16 + (page_rows + 7) // 8) * (sb_rows // page_rows)
    sel_rows_sb = int(out_rows) // args.sub_batches
    alg_bytes_launch = raw_f64_bytes + 8 * sel_rows_sb
    ms_gorilla_launch = r_cal.ms_decode_val
    achieved = alg_bytes_launch / (ms_gorilla_launch / 1000) if ms_gorilla_launch > 0 else 0
    peak = 8.0e12
    traffic = None
    pmc_path = os.path.join(REPO, "profiles", "pmc_traffic.json")
    if os.path.exists(pmc_path):
        try:
            pmc = json.load(open(pmc_path))
            shape = pmc.get("launch_shape", {})
            # attach only when the PMC collection ran the same launch
            # shape (a stale half-size traffic number
            # was attached to a full-size run)
            if (pmc.get("kernel") == "k_gor_chunks_filtered"
                    and shape.get("rows_per_launch") == sb_rows
                    and shape.get("page_rows") == page_rows
                    and shape.get("sub_batches") == args.sub_batches
                    and shape.get("series") == nseries):
                traffic = pmc.get("bytes_per_launch")
        except Exception:
            pass
    roofline = {"bound": "hbm", "achieved": achieved, "peak": peak,
                "unit": "B/s", "frac": achieved / peak, "traffic": traffic}

    cpu_baseline = None
    if rank == 0 and not args.skip_cpu_baseline:
        cpu_baseline = cpu_baseline_leg(32, npts, page_rows, lo, hi)

    if rank == 0:
        line = {
            "metric": "decoded+filtered values/sec & HBM GB/s, TSBS devops scan, 1/2/4/8 GPU",
            "value": value,
            "unit": "values/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": "tsbs-devops-scan (BASELINE configs[1]: 10k series x 1M pts, reference-shaped 125k-row pages, delta-ts + Gorilla-f64 decode, ts-range filter 50%, fused 5-min max/sum/count)",
                "series_per_gpu": nseries,
                "fields": args.fields,
                "points_per_series": npts,
                "page_rows": page_rows,
                "selectivity": 0.5,
                "n_buckets": nbuckets,
                "raw_page_bytes_per_gpu": raw_bytes,
                "rows_per_step_per_gpu": values_per_step,
                "out_rows_per_step_per_gpu": int(out_rows),
                "sub_batches": args.sub_batches,
                "setup_s": round(setup_s, 1),
                "phase_ms_per_step": {
                    "_note": "stream-distance sums; inflated by dual-stream overlap",
                    "decode_ts": phases[0] / args.steps,
                    "decode_f64": phases[1] / args.steps,
                    "filter": phases[2] / args.steps,
                    "compact": phases[3] / args.steps,
                    "agg": phases[4] / args.steps,
                },
                "phase_ms_unoverlapped_per_subbatch": {
                    "decode_ts": r_cal.ms_decode_ts,
                    "decode_f64": r_cal.ms_decode_val,
                    "filter": r_cal.ms_filter,
                    "agg": r_cal.ms_agg,
                },
                "effective_GBps": value * (raw_bytes / (nseries * npts) + 8) / 1e9,
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line))

    for st_ in sets:
        st_.free()
    for e in engines:
        e.close()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
