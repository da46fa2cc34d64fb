"""GROUP BY tag over the decoded varbinary tag column:
dictionary in first-occurrence order + deterministic per-(tag, bucket)
merge of the per-series aggregate partials.  Mirrors the reference shape
where tags are SeriesKey members (constant per series,
tskv/src/reader/series.rs:23-100) and the group-by above TskvExec is
DataFusion hash-agg."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000
BUCKET = 300 * NS


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


def test_groupby_tag_parity(engine):
    rng = np.random.default_rng(11)
    nseries, npts = 20, 4096
    tags = [b"host_%d" % (s % 4) for s in range(nseries)]
    groups, truth = [], []
    for s in range(nseries):
        ts = T0 + np.arange(npts, dtype=np.int64) * NS
        vals = np.round(np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50,
                                0, 100), 1)
        spage = gs.str_page_of([tags[s]] * npts)
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64),
                           (spage, gs.CT_STR)]))
        truth.append((ts, vals))
    gset = engine.upload(groups)
    rows = gset.rows
    lo = T0 + 1000 * NS
    hi = T0 + 3000 * NS
    nb = int(npts * NS // BUCKET) + 1
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    agg = dict(bucket_ns=BUCKET, t0=T0, n_buckets=nb,
               d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                device="cuda"),
               d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
               d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
    engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                d_out_ts=d_ots, d_out_val=d_oval, agg=agg)
    # decode the tag column (fills the per-row string pos/sz the group-by
    # dictionary hashes)
    d_off = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(rows * 16, dtype=torch.uint8, device="cuda")
    engine.decode_str(gset, 2, d_off, d_bytes)
    cap = 16
    g_max = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_sum = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_cnt = torch.zeros(cap * nb, dtype=torch.int64, device="cuda")
    ngids, rep = gs.groupby_tag(engine, gset, nb, g_max, g_sum, g_cnt, cap)
    assert ngids == 4
    # first-occurrence order: host_0, host_1, host_2, host_3
    offs_h = d_off.cpu().numpy()
    bytes_h = d_bytes.cpu().numpy()
    got_tags = [bytes(bytes_h[offs_h[r]:offs_h[r + 1]].tobytes())
                for r in rep]
    assert got_tags == [b"host_0", b"host_1", b"host_2", b"host_3"]
    gm = g_max.cpu().numpy().reshape(cap, nb)[:ngids]
    gsm = g_sum.cpu().numpy().reshape(cap, nb)[:ngids]
    gc = g_cnt.cpu().numpy().reshape(cap, nb)[:ngids]
    # numpy oracle composition
    for t in range(4):
        sel_ts, sel_v = [], []
        for s in range(nseries):
            if s % 4 != t:
                continue
            ts, vals = truth[s]
            m = (ts >= lo) & (ts <= hi)
            sel_ts.append(ts[m])
            sel_v.append(vals[m])
        ts_all = np.concatenate(sel_ts)
        v_all = np.concatenate(sel_v)
        bi = ((ts_all - T0) // BUCKET).astype(np.int64)
        ec = np.bincount(bi, minlength=nb)
        es = np.bincount(bi, weights=v_all, minlength=nb)
        assert (gc[t] == ec).all()
        nz = ec > 0
        emx = np.full(nb, -np.inf)
        order = np.argsort(bi, kind="stable")
        off2 = np.searchsorted(bi[order], np.flatnonzero(nz))
        emx[nz] = np.maximum.reduceat(v_all[order], off2)
        assert (gm[t][nz] == emx[nz]).all()
        assert np.allclose(gsm[t][nz], es[nz], rtol=1e-12)
    gset.free()


def test_groupby_tag_high_cardinality(engine):
    """Every series its own tag: ngids == nseries, per-tag result equals
    the series' own partials."""
    rng = np.random.default_rng(12)
    nseries, npts = 64, 1024
    groups = []
    for s in range(nseries):
        ts = T0 + np.arange(npts, dtype=np.int64) * NS
        vals = np.full(npts, float(s))
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64),
                           (gs.str_page_of([b"u%05d" % s] * npts),
                            gs.CT_STR)]))
    gset = engine.upload(groups)
    rows = gset.rows
    nb = int(npts * NS // BUCKET) + 1
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    agg = dict(bucket_ns=BUCKET, t0=T0, n_buckets=nb,
               d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                device="cuda"),
               d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
               d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
    engine.scan(gset, d_ts, d_val, agg=agg)
    d_off = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(rows * 16, dtype=torch.uint8, device="cuda")
    engine.decode_str(gset, 2, d_off, d_bytes)
    cap = 128
    g_max = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_sum = torch.zeros(cap * nb, dtype=torch.float64, device="cuda")
    g_cnt = torch.zeros(cap * nb, dtype=torch.int64, device="cuda")
    ngids, rep = gs.groupby_tag(engine, gset, nb, g_max, g_sum, g_cnt, cap)
    assert ngids == nseries
    gm = g_max.cpu().numpy().reshape(cap, nb)[:ngids]
    gc = g_cnt.cpu().numpy().reshape(cap, nb)[:ngids]
    for s in range(nseries):
        nz = gc[s] > 0
        assert gc[s].sum() == npts
        assert (gm[s][nz] == float(s)).all()
    gset.free()


def test_groupby_error_paths(engine):
    """gs_groupby_tag preconditions fail loudly: no prior aggregate scan,
    no decoded string column, cap too small."""
    rng = np.random.default_rng(3)
    n = 1024
    ts = T0 + np.arange(n, dtype=np.int64) * NS
    vals = rng.normal(50, 5, n)
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals, gs.CT_F64), gs.CT_F64),
                               (gs.str_page_of([b"h"] * n), gs.CT_STR)])])
    rows = gset.rows
    nb = 4
    g_max = torch.zeros(nb, dtype=torch.float64, device="cuda")
    g_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    g_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    with pytest.raises(RuntimeError, match="prior aggregate scan"):
        gs.groupby_tag(engine, gset, nb, g_max, g_sum, g_cnt, 4)
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    agg = dict(bucket_ns=BUCKET, t0=T0, n_buckets=nb,
               d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                device="cuda"),
               d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
               d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
    engine.scan(gset, d_ts, d_val, agg=agg)
    with pytest.raises(RuntimeError, match="decoded string column"):
        gs.groupby_tag(engine, gset, nb, g_max, g_sum, g_cnt, 4)
    gset.free()


def test_scan_fields_rejects_unfusable(engine):
    """gs_scan_fields is fused-only: null-carrying fields must error."""
    rng = np.random.default_rng(4)
    n = 1024
    ts = T0 + np.arange(n, dtype=np.int64) * NS
    vals = rng.normal(50, 5, n)
    valid = rng.random(n) > 0.5
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals, gs.CT_F64, valid),
                                gs.CT_F64)])])
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    with pytest.raises(RuntimeError, match="fused-capable"):
        gs.scan_fields(engine, gset, [0], d_ts, d_val,
                       (int(ts[0]), int(ts[-1])), d_ots, d_oval)
    gset.free()
