"""Gorilla chunked decode (reference-shaped pages): parity + error surface.

The reference's compaction emits DataBlocks of up to max_datablock_size =
102,400 rows (compaction/comapcting_block_meta_group.rs:87,
config/src/tskv/storage_config.rs:136-138); the Gorilla bitstream is
strictly sequential per page (codec/float.rs:445-463), so the engine
records parser sync points every GOR_CHUNK values at upload and decodes
chunk-parallel.  These tests pin that path bit-exactly against the oracle
on pages around and far beyond the chunk boundary, including truncation
and value-repeat/NaN content, and pin the sub-page pruning semantics of
the fused scan."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

rng = np.random.default_rng(20250915)

GOR_CHUNK = 2048  # keep in sync with gs_internal.h


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


def _walk(n, r=rng, repeat_frac=0.3):
    """TSBS-like walk with long exact-repeat runs (ctrl-bit 0 coverage)."""
    v = np.round(np.clip(np.cumsum(r.normal(0, 0.5, n)) + 50, 0, 100), 1)
    reps = r.random(n) < repeat_frac
    for i in range(1, n):
        if reps[i]:
            v[i] = v[i - 1]
    return v


def _upload_f64(engine, pages_vals):
    groups = []
    for i, vals in enumerate(pages_vals):
        n = len(vals)
        ts = np.arange(n, dtype=np.int64) * 1000
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(np.asarray(vals), gs.CT_F64),
                            gs.CT_F64)]))
    return engine.upload(groups)


def test_chunked_decode_bit_exact(engine):
    """Pages straddling every chunk-boundary edge case, plus NaN payloads
    mid-stream (which must NOT terminate: only the sentinel bit pattern
    does, float.rs:16,546-554)."""
    sizes = [GOR_CHUNK - 1, GOR_CHUNK, GOR_CHUNK + 1, 2 * GOR_CHUNK,
             3 * GOR_CHUNK + 7, 100_000, 125_000]
    cases = [_walk(n) for n in sizes]
    # NaN payloads + infinities sprinkled into a big multi-chunk page
    big = _walk(50_000)
    idx = rng.integers(0, big.size, 500)
    big[idx[:200]] = np.inf
    big[idx[200:400]] = -np.inf
    nanpat = np.frombuffer(np.uint64(0x7FF8000000000001).tobytes(),
                           dtype=np.float64)[0]
    big[idx[400:]] = nanpat
    cases.append(big)
    # all-constant page: maximal repeat-ctrl density, ~1 bit/value
    cases.append(np.full(60_000, 42.5))
    gset = _upload_f64(engine, cases)
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    engine.decode(gset, 1, out)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, vals in enumerate(cases):
        got = host[offs[i]:offs[i] + len(vals)]
        exp = orc.decode_f64(gs.encode_f64(vals), len(vals))
        assert got.view(np.uint64).tolist() == exp.view(np.uint64).tolist(), \
            f"chunked f64 case {i} (n={len(vals)})"
    gset.free()


def test_chunked_fused_scan_large_pages(engine):
    """Fused scan over reference-shaped 100k-row pages: span boundaries
    land mid-chunk; whole chunks outside the span are pruned; compacted
    output and aggregates must still match the oracle exactly."""
    npts = 100_000
    nseries = 6
    t0 = 1_700_000_000_000_000_000
    groups, truth = [], []
    for s in range(nseries):
        ts = t0 + np.arange(npts, dtype=np.int64) * 1_000_000_000
        vals = _walk(npts, np.random.default_rng(100 + s))
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        truth.append((ts, vals))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    # span [~9k, ~73k) rows: starts mid-chunk-2, ends mid-chunk-17
    lo = t0 + 9_123 * 1_000_000_000
    hi = t0 + 73_321 * 1_000_000_000
    bucket_ns = 300_000_000_000
    nb = int(npts * 1_000_000_000 // bucket_ns) + 1
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval,
                      agg=dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                               d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    exp_ts, exp_val = [], []
    for ts, vals in truth:
        s0, c = orc.time_span(ts, lo, hi)
        exp_ts.append(ts[s0:s0 + c])
        exp_val.append(vals[s0:s0 + c])
    exp_ts = np.concatenate(exp_ts)
    exp_val = np.concatenate(exp_val)
    assert res.out_rows == exp_ts.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts).all()
    got_val = d_oval[:res.out_rows].cpu().numpy()
    assert got_val.view(np.uint64).tolist() == exp_val.view(np.uint64).tolist()
    emx, esm, ect = orc.bucket_agg(exp_ts, exp_val, None, t0, bucket_ns, nb)
    assert (d_cnt.cpu().numpy() == ect).all()
    assert (d_max.cpu().numpy()[ect > 0] == emx[ect > 0]).all()
    assert np.allclose(d_sum.cpu().numpy()[ect > 0], esm[ect > 0], rtol=1e-12)
    gset.free()


def test_chunked_vs_small_pages_identical(engine):
    """The same series uploaded as one 96k-row page vs 24 4k-row pages
    must produce identical fused-scan output."""
    npts = 96_000
    t0 = 1_700_000_000_000_000_000
    ts = t0 + np.arange(npts, dtype=np.int64) * 1_000_000_000
    vals = _walk(npts, np.random.default_rng(7))
    lo = t0 + 10_000 * 10**9
    hi = t0 + 80_000 * 10**9
    outs = []
    for page_rows in (npts, 4000):
        groups = []
        npages = npts // page_rows
        for p in range(npages):
            sl = slice(p * page_rows, (p + 1) * page_rows)
            groups.append((0, [(gs.page_of(ts[sl], gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals[sl], gs.CT_F64), gs.CT_F64)]))
        gset = engine.upload(groups)
        rows = gset.rows
        d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
        res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                          d_out_ts=d_ots, d_out_val=d_oval)
        outs.append((res.out_rows,
                     d_ots[:res.out_rows].cpu().numpy().copy(),
                     d_oval[:res.out_rows].cpu().numpy().copy()))
        gset.free()
    (n0, t0a, v0), (n1, t1a, v1) = outs
    assert n0 == n1
    assert (t0a == t1a).all()
    assert v0.view(np.uint64).tolist() == v1.view(np.uint64).tolist()


def test_chunked_truncated_large_page_errors(engine):
    """A 100k-row page cut mid-stream must fail decode exactly like the
    sequential path (float.rs:462 "unexpected end of block"): the upload
    pre-pass poisons the unreachable chunks and decode raises."""
    npts = 100_000
    vals = _walk(npts)
    data = gs.encode_f64(vals)
    cut = data[:len(data) // 3]
    page = gs.build_page(cut, npts)
    ts = np.arange(npts, dtype=np.int64) * 1000
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_F64)])])
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode(gset, 1, out)
    gset.free()


def test_chunked_truncation_inside_span_errors(engine):
    """Fused scan whose span covers the truncated region must error; the
    sub-page pruning may only skip work OUTSIDE the span (truncation in a
    never-selected chunk is skippable, like the reference's page pruning
    skipping a corrupt page it never reads, reader/chunk.rs:12-49)."""
    npts = 100_000
    t0 = 1_700_000_000_000_000_000
    ts = t0 + np.arange(npts, dtype=np.int64) * 1_000_000_000
    vals = _walk(npts)
    data = gs.encode_f64(vals)
    cut = data[:int(len(data) * 0.6)]  # truncates somewhere past halfway
    page = gs.build_page(cut, npts)
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_F64)])])
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    # span over the whole page -> must surface the truncation
    with pytest.raises(RuntimeError):
        engine.scan(gset, d_ts, d_val,
                    time_range=(t0, t0 + npts * 10**9),
                    d_out_ts=d_ots, d_out_val=d_oval)
    gset.free()


def _upload_f64_nulls(engine, cases):
    """cases: list of (values, valid_mask)."""
    groups = []
    for i, (vals, valid) in enumerate(cases):
        n = len(vals)
        ts = np.arange(n, dtype=np.int64) * 1000
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(np.asarray(vals), gs.CT_F64, valid),
                            gs.CT_F64)]))
    return engine.upload(groups)


def test_chunked_null_pages_bit_exact(engine):
    """Null-carrying Gorilla pages (PC_GORN): the chunk-parallel
    pending-value pipeline must scatter values to set bits exactly like
    the reference's bitset-driven builder (tsm/reader.rs:763-825), with
    k_valid_expand producing the validity bytes."""
    cases = []
    for n in (100, GOR_CHUNK + 3, 5_000, 100_000):
        vals = _walk(n)
        valid = rng.random(n) > 0.10
        if not valid.any():
            valid[0] = True
        cases.append((vals, valid))
    # trailing-null page: values only in the first 30% (the sentinel is
    # consumed long before the last rows -> GORF_SENT_SEEN tail chunks)
    n = 60_000
    vals = _walk(n)
    valid = np.zeros(n, dtype=bool)
    valid[:n * 3 // 10] = rng.random(n * 3 // 10) > 0.05
    valid[0] = True
    cases.append((vals, valid))
    # leading-null page
    valid2 = np.zeros(n, dtype=bool)
    valid2[n // 2:] = True
    cases.append((vals, valid2))
    gset = _upload_f64_nulls(engine, cases)
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    dv = torch.zeros(gset.rows, dtype=torch.uint8, device="cuda")
    engine.decode(gset, 1, out, dv)
    offs = gset.row_offsets()
    host, hv = out.cpu().numpy(), dv.cpu().numpy()
    for i, (vals, valid) in enumerate(cases):
        n = len(vals)
        present = vals[valid]
        data = gs.encode_f64(present) if present.size else b""
        exp = orc.decode_f64(data, n, valid)
        got = host[offs[i]:offs[i] + n]
        assert got.view(np.uint64).tolist() == exp.view(np.uint64).tolist(), \
            f"null chunked case {i}"
        assert (hv[offs[i]:offs[i] + n] == valid.astype(np.uint8)).all(), \
            f"valid bytes case {i}"
    gset.free()


def test_chunked_null_truncated_errors(engine):
    """Truncated null-carrying multi-chunk page must raise at decode."""
    n = 80_000
    vals = _walk(n)
    valid = rng.random(n) > 0.10
    valid[0] = True
    data = gs.encode_f64(vals[valid])
    cut = data[:len(data) // 3]
    bitset = np.packbits(valid, bitorder="little")
    page = gs.build_page(cut, n, bitset)
    ts = np.arange(n, dtype=np.int64) * 1000
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_F64)])])
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode(gset, 1, out)
    gset.free()
