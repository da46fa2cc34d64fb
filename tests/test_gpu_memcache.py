"""Memcache (hot-rows) scan leg, SURVEY §8f row 3: unflushed in-memory
rows get the same series structure as a page set (gs_raw_set) and run
through the SAME filter/aggregate scan — and they dedup against TSM
streams via gs_compact_merge (newest wins), mirroring
MemCacheReader + DataMerger (reader/memcache_reader.rs,
reader/sort_merge.rs:152-343)."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
NS = 10**9


@pytest.fixture(scope="module")
def engine():
    eng = gs.Engine(0)
    yield eng
    eng.close()


def _mk_mem_rows(rng, nseries, lo_n=500, hi_n=4000):
    per = []
    for _ in range(nseries):
        n = int(rng.integers(lo_n, hi_n))
        ts = T0 + np.sort(rng.choice(np.arange(6 * n, dtype=np.int64), n,
                                     replace=False)) * NS
        vals = np.round(np.clip(np.cumsum(rng.normal(0, 0.6, n)) + 50,
                                0, 100), 1)
        per.append((ts, vals))
    return per


def test_memcache_raw_scan(engine):
    """Raw rows: time range + value predicate + 5-min aggregate, vs the
    numpy oracle composition."""
    r = np.random.default_rng(61)
    per = _mk_mem_rows(r, 20)
    counts = np.array([t.size for t, _ in per], dtype=np.int64)
    rset = engine.raw_set(counts)
    d_ts = torch.from_numpy(np.concatenate([t for t, _ in per])).cuda()
    d_val = torch.from_numpy(np.concatenate([v for _, v in per])).cuda()
    rows = int(counts.sum())
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo, hi = T0 + 300 * NS, T0 + 15000 * NS
    bucket_ns = 300 * NS
    nb = 64
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    res = engine.scan(rset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval,
                      value_pred=("ge", 45.0),
                      agg=dict(bucket_ns=bucket_ns, t0=T0, n_buckets=nb,
                               d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    exp_ts, exp_val = [], []
    emx = np.full(nb, -np.inf)
    esm = np.zeros(nb)
    ect = np.zeros(nb, dtype=np.int64)
    for ts, vals in per:
        span = (ts >= lo) & (ts <= hi)
        sel = span & (vals >= 45.0)
        exp_ts.append(ts[sel])
        exp_val.append(vals[sel])
        m, su, c = orc.bucket_agg(ts[span], vals[span], sel[span], T0,
                                  bucket_ns, nb)
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
    rset.free()


def test_memcache_tombstones(engine):
    """Tombstone masking applies to raw rows too (deleted closed ranges
    straddling hot data)."""
    r = np.random.default_rng(62)
    per = _mk_mem_rows(r, 6, 300, 1200)
    counts = np.array([t.size for t, _ in per], dtype=np.int64)
    rset = engine.raw_set(counts)
    d_ts = torch.from_numpy(np.concatenate([t for t, _ in per])).cuda()
    d_val = torch.from_numpy(np.concatenate([v for _, v in per])).cuda()
    rows = int(counts.sum())
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    dead = [(T0 + 500 * NS, T0 + 900 * NS)]
    lo, hi = T0, T0 + 4000 * NS
    res = engine.scan(rset, d_ts, d_val, time_range=(lo, hi),
                      tombstones=dead, d_out_ts=d_ots, d_out_val=d_oval,
                      value_pred=("gt", 0.0))
    exp_ts = []
    for ts, vals in per:
        valid = orc.update_nullbits(ts, dead, np.ones(ts.size, bool))
        sel = (ts >= lo) & (ts <= hi) & valid & (vals > 0.0)
        exp_ts.append(ts[sel])
    exp_ts_c = np.concatenate(exp_ts)
    assert res.out_rows == exp_ts_c.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts_c).all()
    rset.free()


def test_memcache_over_tsm_merge(engine):
    """Hot + cold: TSM pages decode, memcache raw rows overlap them, the
    two streams merge-dedup (memcache = newest wins at equal ts), and
    the merged stream scans via a raw set — vs oracle merge_dedup +
    bucket_agg composition."""
    r = np.random.default_rng(63)
    nseries = 8
    grid_n = 3000
    grid = T0 + np.arange(grid_n, dtype=np.int64) * NS
    tsm_groups, tsm_truth = [], []
    for s in range(nseries):
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.5, grid_n)) + 50,
                                0, 100), 1)
        tsm_groups.append((s, [(gs.page_of(grid, gs.CT_TIME), gs.CT_TIME),
                               (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        tsm_truth.append(vals)
    tsm = engine.upload(tsm_groups)
    d_tts = torch.zeros(tsm.rows, dtype=torch.int64, device="cuda")
    d_tval = torch.zeros(tsm.rows, dtype=torch.float64, device="cuda")
    engine.decode(tsm, 0, d_tts)
    engine.decode(tsm, 1, d_tval)
    # memcache: tail overlap (rewrites last 500 grid points) + new points
    mem = []
    for s in range(nseries):
        over = grid[-500:]
        new = grid[-1] + np.arange(1, 301, dtype=np.int64) * NS
        mts = np.concatenate([over, new])
        mvals = np.round(r.normal(70, 5, mts.size), 1)
        mem.append((mts, mvals))
    mcounts = np.array([t.size for t, _ in mem], dtype=np.int64)
    mset = engine.raw_set(mcounts)
    d_mts = torch.from_numpy(np.concatenate([t for t, _ in mem])).cuda()
    d_mval = torch.from_numpy(np.concatenate([v for _, v in mem])).cuda()
    cap = tsm.rows + int(mcounts.sum())
    d_ots = torch.zeros(cap, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(cap, dtype=torch.float64, device="cuda")
    out_rows, offs = engine.compact_merge(
        [tsm, mset], [d_tts, d_mts], [d_tval, d_mval], [None, None],
        d_ots, d_oval)
    # scan the merged stream through a raw set
    mg_counts = np.diff(offs)
    mgset = engine.raw_set(mg_counts)
    lo, hi = T0 + 1000 * NS, grid[-1] + 300 * NS
    bucket_ns = 300 * NS
    nb = 16
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    t0a = T0 + 2000 * NS
    res = engine.scan(mgset, d_ots[:out_rows], d_oval[:out_rows],
                      time_range=(lo, hi),
                      agg=dict(bucket_ns=bucket_ns, t0=t0a, n_buckets=nb,
                               d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    emx = np.full(nb, -np.inf)
    esm = np.zeros(nb)
    ect = np.zeros(nb, dtype=np.int64)
    n_exp = 0
    for s in range(nseries):
        uts, uval, uvalid = orc.merge_dedup(
            [(grid, tsm_truth[s], None), (mem[s][0], mem[s][1], None)])
        a, b = offs[s], offs[s + 1]
        assert b - a == uts.size, s
        got_ts = d_ots[a:b].cpu().numpy()
        assert (got_ts == uts).all(), s
        got_v = d_oval[a:b].cpu().numpy()
        assert (got_v == uval).all(), s  # memcache value wins on overlap
        span = (uts >= lo) & (uts <= hi)
        n_exp += int(span.sum())
        m, su, c = orc.bucket_agg(uts[span], uval[span], None, t0a,
                                  bucket_ns, nb)
        emx = np.maximum(emx, m)
        esm += su
        ect += c
    assert res.decoded_rows == out_rows
    assert res.out_rows == n_exp
    assert (d_cnt.cpu().numpy() == ect).all()
    gmx = d_max.cpu().numpy()
    assert (gmx[ect > 0] == emx[ect > 0]).all()
    assert np.allclose(d_sum.cpu().numpy(), esm, rtol=1e-12)
    tsm.free()
    mset.free()
    mgset.free()
