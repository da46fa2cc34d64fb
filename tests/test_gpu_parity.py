"""GPU parity: the HIP decode/scan path vs the oracle, bit-exact.
All tests need a real MI355X (marked gpu). Mirrors the reference test
strategy: codec round trips incl. adversarial values,
null-bitset scatter, tombstone masking, filter + aggregate semantics."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

rng = np.random.default_rng(231)


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


def _upload_single_col(engine, pages_vals, ctype, valids=None):
    """Each entry one group: time page (trivial) + the column under test?
    For pure column decode we upload the column as slot 0 replacement —
    but slot 0 must be a time page, so build groups with the test column
    in slot 1 and a matching trivial time page in slot 0."""
    groups = []
    for i, vals in enumerate(pages_vals):
        n = len(vals)
        ts = np.arange(n, dtype=np.int64) * 1000
        tpage = gs.page_of(ts, gs.CT_TIME)
        valid = valids[i] if valids else None
        fpage = gs.page_of(np.asarray(vals), ctype, valid)
        groups.append((i, [(tpage, gs.CT_TIME), (fpage, ctype)]))
    return engine.upload(groups)


def _decode_col(engine, gset, ctype, with_valid=False):
    rows = gset.rows
    if ctype in (gs.CT_TIME, gs.CT_I64):
        out = torch.zeros(rows, dtype=torch.int64, device="cuda")
    elif ctype == gs.CT_F64:
        out = torch.zeros(rows, dtype=torch.float64, device="cuda")
    else:
        out = torch.zeros(rows, dtype=torch.uint8, device="cuda")
    dv = torch.zeros(rows, dtype=torch.uint8, device="cuda") if with_valid else None
    engine.decode(gset, 1, out, dv)
    return out, dv


def _i64_corpus():
    cases = [
        np.full(509, 809201799168, dtype=np.int64),          # influx RLE
        np.array([346], dtype=np.int64),                      # influx s8b
        np.arange(1, 13, dtype=np.int64),
        np.array([0], dtype=np.int64),
        np.array([-(2**63), 2**63 - 1, 0], dtype=np.int64),
        np.array([0, 1 << 61, -(1 << 61)], dtype=np.int64),   # uncompressed
        np.full(1000, -345632452354, dtype=np.int64),
    ]
    for _ in range(10):
        n = int(rng.integers(1, 5000))
        cases.append(rng.integers(-2**40, 2**40, n).astype(np.int64))
        cases.append(np.sort(rng.integers(0, 2**50, n)).astype(np.int64))
    # a big page
    cases.append(rng.integers(-2**30, 2**30, 131072).astype(np.int64))
    return cases


def test_decode_i64_all_subencodings(engine):
    cases = _i64_corpus()
    gset = _upload_single_col(engine, cases, gs.CT_I64)
    out, _ = _decode_col(engine, gset, gs.CT_I64)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, vals in enumerate(cases):
        o = offs[i]
        got = host[o:o + len(vals)]
        exp = orc.decode_i64(gs.encode_i64(vals), len(vals))
        assert (got == exp).all(), f"case {i}"
    gset.free()


def test_decode_ts_all_subencodings(engine):
    cases = []
    for _ in range(8):
        n = int(rng.integers(1, 5000))
        start = int(rng.integers(0, 2**50))
        step = int(10 ** rng.integers(0, 10))
        cases.append((start + np.arange(n) * step).astype(np.int64))  # RLE
        cases.append(np.sort(rng.integers(0, 2**50, n)).astype(np.int64))  # s8b
    cases.append(np.array([7], dtype=np.int64))  # scaler-12 single value
    cases.append(np.array([0, 1 << 61], dtype=np.int64))  # n==2 RLE huge delta
    groups = []
    for i, ts in enumerate(cases):
        tpage = gs.page_of(ts, gs.CT_TIME)
        vpage = gs.page_of(np.zeros(len(ts)), gs.CT_F64)
        groups.append((i, [(tpage, gs.CT_TIME), (vpage, gs.CT_F64)]))
    gset = engine.upload(groups)
    out = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
    engine.decode(gset, 0, out)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, ts in enumerate(cases):
        got = host[offs[i]:offs[i] + len(ts)]
        exp = orc.decode_i64(gs.encode_ts(ts), len(ts))
        assert (got == exp).all(), f"ts case {i}"
        assert (got == ts).all(), f"ts case {i} (vs source)"
    gset.free()


def test_decode_gorilla_bit_exact(engine):
    import json, os, struct
    gold = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                       "golden_vectors.json")))
    special = []
    for v in gold["float_special_values"]["values"]:
        if v.startswith("bits:"):
            special.append(struct.unpack("<d", struct.pack("<Q", int(v[5:], 16)))[0])
        else:
            special.append(float(v))
    cases = [
        np.array(special, dtype=np.float64),
        np.array(gold["float_paper"]["values"], dtype=np.float64),
        np.array([1.5], dtype=np.float64),
        np.zeros(1000, dtype=np.float64),
    ]
    for _ in range(6):
        n = int(rng.integers(1, 20000))
        cases.append(np.cumsum(rng.normal(0, 0.5, n)))
        cases.append(np.round(np.clip(np.cumsum(rng.normal(0, 0.5, n)) + 50, 0, 100), 1))
    cases.append(np.round(np.clip(np.cumsum(rng.normal(0, 0.5, 131072)) + 50, 0, 100), 1))
    gset = _upload_single_col(engine, cases, gs.CT_F64)
    out, _ = _decode_col(engine, gset, gs.CT_F64)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, vals in enumerate(cases):
        vals = vals.astype(np.float64)
        got = host[offs[i]:offs[i] + len(vals)]
        assert got.view(np.uint64).tolist() == vals.view(np.uint64).tolist(), f"f64 case {i}"
    gset.free()


def test_decode_bool(engine):
    cases = [rng.integers(0, 2, int(rng.integers(1, 3000))).astype(np.uint8)
             for _ in range(10)]
    gset = _upload_single_col(engine, cases, gs.CT_BOOL)
    out, _ = _decode_col(engine, gset, gs.CT_BOOL)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, vals in enumerate(cases):
        got = host[offs[i]:offs[i] + len(vals)]
        exp = orc.decode_bool(gs.encode_bool(vals), len(vals))
        assert (got == exp).all(), f"bool case {i}"
    gset.free()


def test_decode_bool_with_nulls(engine):
    """null-carrying bool pages take the sequential scatter path"""
    cases, valids = [], []
    for _ in range(6):
        n = int(rng.integers(2, 2000))
        cases.append(rng.integers(0, 2, n).astype(np.uint8))
        valids.append(rng.random(n) > 0.3)
    gset = _upload_single_col(engine, cases, gs.CT_BOOL, valids)
    out, dv = _decode_col(engine, gset, gs.CT_BOOL, with_valid=True)
    offs = gset.row_offsets()
    host, hv = out.cpu().numpy(), dv.cpu().numpy()
    for i, (vals, valid) in enumerate(zip(cases, valids)):
        n = len(vals)
        present = vals[valid]
        data = gs.encode_bool(present) if present.size else b""
        exp = orc.decode_bool(data, n, valid)
        got = host[offs[i]:offs[i] + n]
        assert (got == exp).all(), f"bool-null case {i}"
        assert (hv[offs[i]:offs[i] + n] == valid.astype(np.uint8)).all()
    gset.free()


def test_decode_with_nulls_scatter(engine):
    cases, valids = [], []
    for _ in range(8):
        n = int(rng.integers(3, 4000))
        vals = rng.integers(-2**40, 2**40, n).astype(np.int64)
        valid = rng.random(n) > 0.3
        valid[0] = True
        cases.append(vals)
        valids.append(valid)
    # all-null page
    cases.append(np.array([1, 2, 3], dtype=np.int64))
    valids.append(np.zeros(3, bool))
    gset = _upload_single_col(engine, cases, gs.CT_I64, valids)
    out, dv = _decode_col(engine, gset, gs.CT_I64, with_valid=True)
    offs = gset.row_offsets()
    host, hv = out.cpu().numpy(), dv.cpu().numpy()
    for i, (vals, valid) in enumerate(zip(cases, valids)):
        n = len(vals)
        present = vals[valid]
        data = gs.encode_i64(present) if present.size else b""
        exp = orc.decode_i64(data, n, valid)
        got = host[offs[i]:offs[i] + n]
        assert (got == exp).all(), f"null case {i}"
        assert (hv[offs[i]:offs[i] + n] == valid.astype(np.uint8)).all(), f"valid bytes {i}"
    gset.free()


def test_gorilla_nulls(engine):
    cases, valids = [], []
    for _ in range(5):
        n = int(rng.integers(3, 3000))
        vals = np.cumsum(rng.normal(0, 1, n))
        valid = rng.random(n) > 0.25
        cases.append(vals)
        valids.append(valid)
    gset = _upload_single_col(engine, cases, gs.CT_F64, valids)
    out, dv = _decode_col(engine, gset, gs.CT_F64, with_valid=True)
    offs = gset.row_offsets()
    host, hv = out.cpu().numpy(), dv.cpu().numpy()
    for i, (vals, valid) in enumerate(zip(cases, valids)):
        n = len(vals)
        present = vals[valid].astype(np.float64)
        data = gs.encode_f64(present) if present.size else b""
        exp = orc.decode_f64(data, n, valid)
        got = host[offs[i]:offs[i] + n]
        assert got.view(np.uint64).tolist() == exp.view(np.uint64).tolist(), f"g-null {i}"
        assert (hv[offs[i]:offs[i] + n] == valid.astype(np.uint8)).all()
    gset.free()


def _mk_scan_set(engine, nseries=32, npts=4096, seed=7):
    r = np.random.default_rng(seed)
    groups, truth = [], []
    t0 = 1_700_000_000_000_000_000
    for s in range(nseries):
        ts = t0 + np.arange(npts, dtype=np.int64) * 1_000_000_000
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, npts)) + 50, 0, 100), 1)
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        truth.append((ts, vals))
    return engine.upload(groups), truth, t0


def test_scan_filter_compact(engine):
    gset, truth, t0 = _mk_scan_set(engine)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo = t0 + 1000 * 1_000_000_000
    hi = t0 + 3000 * 1_000_000_000
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval)
    # oracle restatement
    exp_ts, exp_val = [], []
    for ts, vals in truth:
        s, c = orc.time_span(ts, lo, hi)
        exp_ts.append(ts[s:s + c])
        exp_val.append(vals[s:s + c])
    exp_ts = np.concatenate(exp_ts)
    exp_val = np.concatenate(exp_val)
    assert res.out_rows == exp_ts.size
    got_ts = d_ots[:res.out_rows].cpu().numpy()
    got_val = d_oval[:res.out_rows].cpu().numpy()
    assert (got_ts == exp_ts).all()
    assert got_val.view(np.uint64).tolist() == exp_val.view(np.uint64).tolist()
    gset.free()


def test_scan_agg_5min_buckets(engine):
    gset, truth, t0 = _mk_scan_set(engine, nseries=16, npts=8192)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    bucket_ns = 300_000_000_000
    nb = int(8192 * 1_000_000_000 // bucket_ns) + 1
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    res = engine.scan(gset, d_ts, d_val,
                      agg=dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                               d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    assert res.out_rows == rows
    all_ts = np.concatenate([t for t, _ in truth])
    all_v = np.concatenate([v for _, v in truth])
    emx, esm, ect = orc.bucket_agg(all_ts, all_v, None, t0, bucket_ns, nb)
    assert (d_cnt.cpu().numpy() == ect).all()
    gmx = d_max.cpu().numpy()
    assert (gmx[ect > 0] == emx[ect > 0]).all()  # max is order-safe, exact
    gsm = d_sum.cpu().numpy()
    ok = np.isclose(gsm[ect > 0], esm[ect > 0], rtol=1e-12, atol=0)
    assert ok.all()  # tolerance per BASELINE config #3
    gset.free()


def test_scan_tombstones(engine):
    gset, truth, t0 = _mk_scan_set(engine, nseries=8, npts=2048)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    dead = [(t0 + 100 * 10**9, t0 + 200 * 10**9),
            (t0 + 1500 * 10**9, t0 + 1600 * 10**9)]
    bucket_ns = 300_000_000_000
    nb = 8
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    engine.scan(gset, d_ts, d_val, tombstones=dead,
                agg=dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                         d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    ect = np.zeros(nb, dtype=np.int64)
    emx = np.full(nb, -np.inf)
    for ts, vals in truth:
        valid = orc.update_nullbits(ts, dead, np.ones(ts.size, bool))
        m, s, c = orc.bucket_agg(ts, vals, valid, t0, bucket_ns, nb)
        emx = np.maximum(emx, m)
        ect += c
    assert (d_cnt.cpu().numpy() == ect).all()
    assert (d_max.cpu().numpy()[ect > 0] == emx[ect > 0]).all()
    gset.free()


def test_decode_u64_bitcast(engine):
    """u64 fields are bit-cast to i64 and use the Delta codec
    (unsigned.rs:20-45); decode must round-trip the u64 bit patterns."""
    cases = []
    for _ in range(6):
        n = int(rng.integers(1, 3000))
        cases.append(rng.integers(0, 2**64, n, dtype=np.uint64))
    groups = []
    for i, vals in enumerate(cases):
        ts = np.arange(len(vals), dtype=np.int64) * 1000
        fpage = gs.page_of(vals.view(np.int64), gs.CT_I64)  # encode as i64
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (fpage, gs.CT_U64)]))
    gset = engine.upload(groups)
    out = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
    engine.decode(gset, 1, out)
    offs = gset.row_offsets()
    host = out.cpu().numpy().view(np.uint64)
    for i, vals in enumerate(cases):
        got = host[offs[i]:offs[i] + len(vals)]
        assert (got == vals).all(), f"u64 case {i}"
    # COUNT pushdown from metadata
    assert gset.count_pushdown(1) == sum(len(v) for v in cases)
    gset.free()


def test_scan_filter_compact_irregular_ts(engine):
    """Irregular (simple8b) timestamps force the general scan path (the
    fused path requires RLE ts pages); outputs must still match oracle."""
    r = np.random.default_rng(13)
    t0 = 1_700_000_000_000_000_000
    groups, truth = [], []
    for s in range(16):
        ts = t0 + np.sort(r.choice(np.arange(20000, dtype=np.int64),
                                   4096, replace=False)) * 1_000_000_000
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, 4096)) + 50, 0, 100), 1)
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        truth.append((ts, vals))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo = t0 + 2000 * 10**9
    hi = t0 + 15000 * 10**9
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval)
    exp_ts, exp_val = [], []
    for ts, vals in truth:
        s0, c = orc.time_span(ts, lo, hi)
        exp_ts.append(ts[s0:s0 + c])
        exp_val.append(vals[s0:s0 + c])
    exp_ts = np.concatenate(exp_ts)
    exp_val = np.concatenate(exp_val).astype(np.float64)
    assert res.out_rows == exp_ts.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts).all()
    got = d_oval[:res.out_rows].cpu().numpy()
    assert got.view(np.uint64).tolist() == exp_val.view(np.uint64).tolist()
    gset.free()


def test_scan_fused_agg_equals_general(engine):
    """The fused path's aggregates must equal the general path's (the
    general path is forced by omitting compacted outputs)."""
    gset, truth, t0 = _mk_scan_set(engine, nseries=12, npts=4096, seed=3)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    bucket_ns = 300_000_000_000
    nb = 16
    lo = t0 + 500 * 10**9
    hi = t0 + 3900 * 10**9
    results = []
    for with_compact in (False, True):
        d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
        d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
        d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
        kw = {}
        if with_compact:
            kw["d_out_ts"] = torch.zeros(rows, dtype=torch.int64, device="cuda")
            kw["d_out_val"] = torch.zeros(rows, dtype=torch.float64, device="cuda")
        engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                    agg=dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                             d_max=d_max, d_sum=d_sum, d_count=d_cnt), **kw)
        results.append((d_max.cpu().numpy(), d_sum.cpu().numpy(),
                        d_cnt.cpu().numpy()))
    (m0, s0, c0), (m1, s1, c1) = results
    assert (c0 == c1).all()
    assert (m0[c0 > 0] == m1[c0 > 0]).all()
    assert np.allclose(s0, s1, rtol=1e-12)
    gset.free()


def test_decode_null_encoding_raw_pages(engine):
    """Encoding::Null pages (CODEC(NULL) columns): raw BE values, no
    compression (timestamp.rs:301-323, float.rs:387-413)."""
    cases = []
    for _ in range(5):
        n = int(rng.integers(1, 3000))
        cases.append(rng.integers(-2**60, 2**60, n).astype(np.int64))
    groups = []
    for i, vals in enumerate(cases):
        ts = np.arange(len(vals), dtype=np.int64) * 1000
        data = bytes([1]) + vals.astype(">i8").tobytes()  # ENC_NULL + BE
        fpage = gs.build_page(data, len(vals))
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (fpage, gs.CT_I64)]))
    gset = engine.upload(groups)
    out = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
    engine.decode(gset, 1, out)
    offs = gset.row_offsets()
    host = out.cpu().numpy()
    for i, vals in enumerate(cases):
        got = host[offs[i]:offs[i] + len(vals)]
        exp = orc.decode_i64(bytes([1]) + vals.astype(">i8").tobytes(), len(vals))
        assert (got == exp).all() and (got == vals).all(), f"null-enc {i}"
    gset.free()


def test_scan_multi_field(engine):
    """Multi-metric scan (TSBS cpu-max-all style): one set with several f64
    field pages, scanned per field via spec.field_col."""
    r = np.random.default_rng(21)
    t0 = 1_700_000_000_000_000_000
    nseries, npts, nfields = 8, 4096, 3
    groups, truth = [], []
    for s in range(nseries):
        ts = t0 + np.arange(npts, dtype=np.int64) * 10**9
        fields = [np.round(np.clip(np.cumsum(r.normal(0, 0.5, npts)) + 50, 0, 100), 1)
                  for _ in range(nfields)]
        pages = [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME)]
        pages += [(gs.page_of(f, gs.CT_F64), gs.CT_F64) for f in fields]
        groups.append((s, pages))
        truth.append((ts, fields))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo, hi = t0 + 1000 * 10**9, t0 + 3000 * 10**9
    for fc in range(nfields):
        engine.scan_async(gset, d_ots, d_oval, time_range=(lo, hi),
                          field_col=fc)
        res = engine.scan_wait(gset)
        exp_v = []
        for ts, fields in truth:
            s0, c = orc.time_span(ts, lo, hi)
            exp_v.append(fields[fc][s0:s0 + c])
        exp_v = np.concatenate(exp_v).astype(np.float64)
        assert res.out_rows == exp_v.size
        got = d_oval[:res.out_rows].cpu().numpy()
        assert got.view(np.uint64).tolist() == exp_v.view(np.uint64).tolist(), f"field {fc}"
    gset.free()


def test_crc_validation_rejects_corruption(engine):
    ts = np.arange(100, dtype=np.int64)
    page = bytearray(gs.page_of(ts, gs.CT_TIME))
    page[-1] ^= 0xFF  # corrupt data region
    with pytest.raises(RuntimeError):
        engine.upload([(0, [(bytes(page), gs.CT_TIME),
                            (gs.page_of(np.zeros(100), gs.CT_F64), gs.CT_F64)])])


def test_scan_value_predicate(engine):
    """DataFilter value predicates (reader/filter.rs:91-142): rows kept iff
    time range AND pred(value); null field values fail the predicate."""
    r = np.random.default_rng(31)
    t0 = 1_700_000_000_000_000_000
    groups, truth = [], []
    for s in range(10):
        n = 4096
        ts = t0 + np.arange(n, dtype=np.int64) * 10**9
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, n)) + 50, 0, 100), 1)
        valid = r.random(n) > 0.2
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64, valid), gs.CT_F64)]))
        truth.append((ts, vals, valid))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo, hi = t0 + 500 * 10**9, t0 + 3500 * 10**9
    for pred, npred in [(("gt", 50.0), lambda v: v > 50.0),
                        (("le", 49.0), lambda v: v <= 49.0),
                        (("between", 40.0, 60.0), lambda v: (v >= 40) & (v <= 60))]:
        res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                          d_out_ts=d_ots, d_out_val=d_oval, value_pred=pred)
        exp_ts, exp_val = [], []
        for ts, vals, valid in truth:
            s0, c = orc.time_span(ts, lo, hi)
            sel = np.zeros(ts.size, bool)
            sel[s0:s0 + c] = True
            sel &= valid & npred(np.where(valid, vals, np.nan))
            exp_ts.append(ts[sel])
            exp_val.append(vals[sel])
        exp_ts = np.concatenate(exp_ts)
        exp_val = np.concatenate(exp_val).astype(np.float64)
        assert res.out_rows == exp_ts.size, pred
        assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts).all(), pred
        got = d_oval[:res.out_rows].cpu().numpy()
        assert got.view(np.uint64).tolist() == exp_val.view(np.uint64).tolist(), pred
    gset.free()


def test_arrow_export(engine):
    """Arrow C data interface export: group rows + packed validity bitmap
    (the SURVEY §8b output form for the Rust shim)."""
    cases, valids = [], []
    for _ in range(4):
        n = int(rng.integers(2, 1500))
        cases.append(rng.integers(-2**40, 2**40, n).astype(np.int64))
        valids.append(rng.random(n) > 0.3)
    gset = _upload_single_col(engine, cases, gs.CT_I64, valids)
    out, dv = _decode_col(engine, gset, gs.CT_I64, with_valid=True)
    for g, (vals, valid) in enumerate(zip(cases, valids)):
        arr, data, bitmap = engine.export_group_column(gset, g, out, 8, dv)
        assert arr.length == len(vals)
        assert arr.null_count == int((~valid).sum())
        got = data.view(np.int64)
        exp = np.where(valid, vals, 0)
        assert (got == exp).all()
        bits = np.unpackbits(bitmap, bitorder="little")[:len(vals)]
        assert (bits.astype(bool) == valid).all()
    gset.free()


def test_scan_tombstones_plus_value_pred(engine):
    """Tombstone masking composes with the value predicate: deleted rows
    are nulls, and nulls fail the predicate."""
    r = np.random.default_rng(41)
    t0 = 1_700_000_000_000_000_000
    groups, truth = [], []
    for s in range(6):
        n = 3000
        ts = t0 + np.arange(n, dtype=np.int64) * 10**9
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, n)) + 50, 0, 100), 1)
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        truth.append((ts, vals))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    dead = [(t0 + 200 * 10**9, t0 + 400 * 10**9)]
    lo, hi = t0 + 100 * 10**9, t0 + 2500 * 10**9
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi), tombstones=dead,
                      d_out_ts=d_ots, d_out_val=d_oval, value_pred=("gt", 50.0))
    exp_ts, exp_val = [], []
    for ts, vals in truth:
        valid = orc.update_nullbits(ts, dead, np.ones(ts.size, bool))
        sel = (ts >= lo) & (ts <= hi) & valid & (vals > 50.0)
        exp_ts.append(ts[sel])
        exp_val.append(vals[sel])
    exp_ts = np.concatenate(exp_ts)
    exp_val = np.concatenate(exp_val).astype(np.float64)
    assert res.out_rows == exp_ts.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts).all()
    got = d_oval[:res.out_rows].cpu().numpy()
    assert got.view(np.uint64).tolist() == exp_val.view(np.uint64).tolist()
    gset.free()


def test_truncated_gorilla_stream_errors(engine):
    """A stream cut before its sentinel must fail decode (the reference's
    "unexpected end of block", float.rs:462)."""
    vals = np.cumsum(rng.normal(0, 1, 2000))
    data = gs.encode_f64(vals)
    cut = data[:len(data) // 2]
    page = gs.build_page(cut, 2000)
    ts = np.arange(2000, dtype=np.int64) * 1000
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_F64)])])
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode(gset, 1, out)
    gset.free()


def test_scan_fused_agg_multigroup(engine):
    """Fused aggregate with several page-groups per series (merged into
    one series-group): per-group RLE deltas differ, groups are separated
    by time gaps, the scan range cuts mid-group and leaves whole groups
    empty — exercises the closed-form boundary table of
    k_agg_partial_rle (page-group search + per-group division)."""
    r = np.random.default_rng(53)
    t0 = 1_700_000_000_000_000_000
    ns = 10**9
    groups, truth = [], []
    for s in range(7):
        cur = t0 if s < 6 else t0 - 50_000 * ns  # series 6: fully below lo
        sts, svals = [], []
        for _ in range(7):
            n = int(r.integers(100, 900))
            step = int(r.choice([1, 2, 5, 10])) * ns
            ts = cur + np.arange(n, dtype=np.int64) * step
            vals = np.round(r.normal(50, 10, n), 2)
            groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
            sts.append(ts)
            svals.append(vals)
            cur = ts[-1] + int(r.integers(1, 1000)) * ns  # gap between groups
        truth.append((np.concatenate(sts), np.concatenate(svals)))
    gset = engine.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo, hi = t0 + 700 * ns, t0 + 9000 * ns
    bucket_ns = 300 * ns
    nb = 40
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval,
                      agg=dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                               d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    exp_ts, exp_val = [], []
    emx = np.full(nb, -np.inf)
    esm = np.zeros(nb)
    ect = np.zeros(nb, dtype=np.int64)
    for ts, vals in truth:
        sel = (ts >= lo) & (ts <= hi)
        exp_ts.append(ts[sel])
        exp_val.append(vals[sel])
        m, su, c = orc.bucket_agg(ts[sel], vals[sel], None, t0, bucket_ns, nb)
        emx = np.maximum(emx, m)
        esm += su
        ect += c
    exp_ts_c = np.concatenate(exp_ts)
    exp_val_c = np.concatenate(exp_val).astype(np.float64)
    assert res.out_rows == exp_ts_c.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts_c).all()
    got = d_oval[:res.out_rows].cpu().numpy()
    assert got.view(np.uint64).tolist() == exp_val_c.view(np.uint64).tolist()
    assert (d_cnt.cpu().numpy() == ect).all()
    gmx = d_max.cpu().numpy()
    assert (gmx[ect > 0] == emx[ect > 0]).all()
    assert np.allclose(d_sum.cpu().numpy(), esm, rtol=1e-12)
    gset.free()


def test_upload_packed_matches_upload(engine):
    """The vectorized packed upload (one contiguous buffer + offset
    arrays, used by bench.py) must produce the same decode and scan
    results as the per-page upload path."""
    r = np.random.default_rng(71)
    t0 = 1_700_000_000_000_000_000
    nser, npts = 12, 4096
    pages = []
    for s in range(nser):
        ts = t0 + np.arange(npts, dtype=np.int64) * 10**9
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, npts)) + 50, 0, 100), 1)
        pages.append((gs.page_of(ts, gs.CT_TIME), gs.page_of(vals, gs.CT_F64)))
    groups = [(s, [(tp, gs.CT_TIME), (vp, gs.CT_F64)])
              for s, (tp, vp) in enumerate(pages)]
    g1 = engine.upload(groups)
    buf = b"".join(tp + vp for tp, vp in pages)
    off, ln = [], []
    cur = 0
    for tp, vp in pages:
        off += [cur, cur + len(tp)]
        ln += [len(tp), len(vp)]
        cur += len(tp) + len(vp)
    g2 = engine.upload_packed(
        np.frombuffer(buf, dtype=np.uint8),
        np.array(off, dtype=np.int64), np.array(ln, dtype=np.int64),
        np.full(nser * 2, npts, dtype=np.int64),
        np.tile(np.array([gs.CT_TIME, gs.CT_F64], dtype=np.uint8), nser),
        np.arange(nser, dtype=np.int64), 2)
    assert g1.rows == g2.rows
    outs = []
    for g in (g1, g2):
        rows = g.rows
        d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
        nb = 16
        d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
        d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
        d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
        res = engine.scan(g, d_ts, d_val,
                          time_range=(t0 + 100 * 10**9, t0 + 4000 * 10**9),
                          d_out_ts=d_ots, d_out_val=d_oval,
                          agg=dict(bucket_ns=300 * 10**9, t0=t0, n_buckets=nb,
                                   d_max=d_max, d_sum=d_sum, d_count=d_cnt))
        outs.append((res.out_rows, d_ots[:res.out_rows].cpu().numpy(),
                     d_oval[:res.out_rows].cpu().numpy(),
                     d_max.cpu().numpy(), d_sum.cpu().numpy(),
                     d_cnt.cpu().numpy()))
        g.free()
    (n1, ts1, v1, m1, s1, c1), (n2, ts2, v2, m2, s2, c2) = outs
    assert n1 == n2
    assert (ts1 == ts2).all()
    assert v1.view(np.uint64).tolist() == v2.view(np.uint64).tolist()
    assert (c1 == c2).all()
    assert m1.tolist() == m2.tolist()
    assert s1.tolist() == s2.tolist()


def test_nonmonotonic_ts_page_errors(engine):
    """a CRC-valid but unsorted time page must fail decode
    loudly (the span/tombstone binary searches assume ascending ts; the
    reference's row-wise filtering cannot mis-select)."""
    ts = np.array([1000, 500, 2000, 1500], dtype=np.int64)  # unsorted
    data = bytes([1]) + ts.astype(">i8").tobytes()  # Null encoding, raw BE
    tpage = gs.build_page(data, 4)
    vals = np.array([1.0, 2.0, 3.0, 4.0])
    gset = engine.upload([(0, [(tpage, gs.CT_TIME),
                               (gs.page_of(vals, gs.CT_F64), gs.CT_F64)])])
    out = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
    with pytest.raises(RuntimeError, match="non-monotonic"):
        engine.decode(gset, 0, out)
    gset.free()


def test_truncated_string_payload_errors(engine):
    """a string page whose payload ends while valid rows
    remain must error (the reference emits a shorter array there —
    silently nulling would be an undetectable divergence)."""
    import struct
    # Null-encoded string block ([1][u64 BE len][bytes]..., string.rs:169-183)
    # holding only 3 strings while the bitset marks 4 valid rows
    payload = b"".join(struct.pack(">Q", 6) + b"abcdef" for _ in range(3))
    page = gs.build_page(bytes([1]) + payload, 4)
    ts = np.arange(4, dtype=np.int64) * 1000
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_STR)])])
    rows = gset.rows
    d_off = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(rows * 64, dtype=torch.uint8, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode_str(gset, 1, d_off, d_bytes)
    gset.free()


def test_scan_fields_equals_per_field(engine):
    """gs_scan_fields (one span/ts pass, N fields) must equal N separate
    fused scans — compacted values and aggregates, bit-exact."""
    nseries, npts, nf = 10, 8192, 3
    r = np.random.default_rng(21)
    t0 = 1_700_000_000_000_000_000
    groups = []
    per_field_vals = [[] for _ in range(nf)]
    for s in range(nseries):
        ts = t0 + np.arange(npts, dtype=np.int64) * 1_000_000_000
        cols = [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME)]
        for f in range(nf):
            v = np.round(np.clip(np.cumsum(r.normal(0, 0.5, npts)) + 50,
                                 0, 100), 1)
            cols.append((gs.page_of(v, gs.CT_F64), gs.CT_F64))
            per_field_vals[f].append(v)
        groups.append((s, cols))
    gset = engine.upload(groups)
    rows = gset.rows
    lo = t0 + 1000 * 10**9
    hi = t0 + 7000 * 10**9
    bucket_ns = 300_000_000_000
    nb = int(npts * 10**9 // bucket_ns) + 1
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    # multi-field call
    d_oval_m = torch.zeros(nf * rows, dtype=torch.float64, device="cuda")
    agg_m = dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                 d_max=torch.full((nf * nb,), -np.inf, dtype=torch.float64,
                                  device="cuda"),
                 d_sum=torch.zeros(nf * nb, dtype=torch.float64,
                                   device="cuda"),
                 d_count=torch.zeros(nf * nb, dtype=torch.int64,
                                     device="cuda"))
    res_m = gs.scan_fields(engine, gset, list(range(nf)), d_ts, d_val,
                           (lo, hi), d_ots, d_oval_m, agg_m)
    # per-field reference calls
    for f in range(nf):
        d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_ots2 = torch.zeros(rows, dtype=torch.int64, device="cuda")
        agg = dict(bucket_ns=bucket_ns, t0=t0, n_buckets=nb,
                   d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                    device="cuda"),
                   d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
                   d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
        res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                          d_out_ts=d_ots2, d_out_val=d_oval, agg=agg,
                          field_col=f)
        assert res.out_rows == res_m.out_rows
        n = res.out_rows
        assert bool((d_ots[:n] == d_ots2[:n]).all())
        assert bool((d_oval_m[f * rows:f * rows + n] == d_oval[:n]).all())
        assert bool((agg_m["d_count"][f * nb:(f + 1) * nb] ==
                     agg["d_count"]).all())
        nz = agg["d_count"] > 0
        assert bool((agg_m["d_max"][f * nb:(f + 1) * nb][nz] ==
                     agg["d_max"][nz]).all())
        assert bool(torch.allclose(agg_m["d_sum"][f * nb:(f + 1) * nb],
                                   agg["d_sum"], rtol=1e-12))
    gset.free()


def test_scan_i64_field_with_pred(engine):
    """Integer fields flow through the general scan path: decode + time
    filter + typed value predicate + compact (DataFilter evaluates the
    pushed expr in the column's own type, reader/filter.rs:91-142).
    Aggregates stay f64-only (TSBS path) and are not requested here."""
    t0 = 1_700_000_000_000_000_000
    r = np.random.default_rng(31)
    groups, truth = [], []
    for s in range(6):
        n = 5000
        ts = t0 + np.arange(n, dtype=np.int64) * 1_000_000_000
        vals = r.integers(-1000, 1000, n).astype(np.int64)
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_I64), gs.CT_I64)]))
        truth.append((ts, vals))
    gset = engine.upload(groups)
    rows = gset.rows
    lo = t0 + 500 * 10**9
    hi = t0 + 4500 * 10**9
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval,
                      value_pred=("gt", 250.0))
    exp_ts, exp_val = [], []
    for ts, vals in truth:
        sel = (ts >= lo) & (ts <= hi) & (vals > 250)
        exp_ts.append(ts[sel])
        exp_val.append(vals[sel])
    exp_ts = np.concatenate(exp_ts)
    exp_val = np.concatenate(exp_val)
    assert res.out_rows == exp_ts.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts).all()
    got = d_oval[:res.out_rows].cpu().numpy().view(np.int64)
    assert (got == exp_val).all()
    gset.free()


def test_scan_u64_field_unsigned_compare(engine):
    """u64 fields (bit-cast i64 slot, unsigned.rs:20-45): the predicate
    must compare UNSIGNED — a value with the top bit set is larger than
    any positive literal, not negative."""
    t0 = 1_700_000_000_000_000_000
    n = 1000
    ts = t0 + np.arange(n, dtype=np.int64) * 1_000_000_000
    vals = np.arange(n, dtype=np.uint64)
    vals[::10] += np.uint64(2**63)  # huge unsigned values
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals.view(np.int64), gs.CT_I64),
                                gs.CT_U64)])])
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    res = engine.scan(gset, d_ts, d_val, time_range=(int(ts[0]), int(ts[-1])),
                      d_out_ts=d_ots, d_out_val=d_oval,
                      value_pred=("gt", 500.0))
    sel = vals > np.uint64(500)
    assert res.out_rows == int(sel.sum())
    got = d_oval[:res.out_rows].cpu().numpy().view(np.uint64)
    assert (got == vals[sel]).all()
    gset.free()
