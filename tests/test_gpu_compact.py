"""GPU compaction merge (BASELINE config #5) vs the oracle restatement of
sort_merge + BatchMergeBuilder dedup (reader/batch_builder.rs:106-155)."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

rng = np.random.default_rng(77)
T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000


def _mk_streams(nseries, k, npts, collide=0.2, nulls=0.0):
    """k overlapping L0 streams per series with `collide` fraction of
    timestamp collisions across streams (exercises the dedup rule)."""
    streams = []  # [k][series] -> (ts, vals, valid)
    base_grid = T0 + np.arange(npts * 4, dtype=np.int64) * NS
    for f in range(k):
        per = []
        for s in range(nseries):
            # each stream picks a sorted subset of the shared grid; shared
            # grid ensures collisions between streams
            take = rng.random(base_grid.size) < (0.25 + 0.05 * f)
            ts = base_grid[take][:npts]
            vals = np.round(np.clip(np.cumsum(rng.normal(0, 1, ts.size)) + 50, 0, 100), 1)
            valid = None
            if nulls > 0:
                valid = rng.random(ts.size) > nulls
            per.append((ts, vals, valid))
        streams.append(per)
    return streams


def _upload_stream(engine, per_series):
    groups = []
    for s, (ts, vals, valid) in enumerate(per_series):
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64, valid), gs.CT_F64)]))
    return engine.upload(groups)


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


@pytest.mark.parametrize("nulls", [0.0, 0.3])
def test_compact_merge_dedup(engine, nulls):
    nseries, k, npts = 6, 8, 2000
    streams = _mk_streams(nseries, k, npts, nulls=nulls)
    gsets, tss, vls, vds = [], [], [], []
    total = 0
    for f in range(k):
        gset = _upload_stream(engine, streams[f])
        rows = gset.rows
        d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_vd = torch.zeros(rows, dtype=torch.uint8, device="cuda")
        engine.decode(gset, 0, d_ts)
        engine.decode(gset, 1, d_val, d_vd)
        gsets.append(gset)
        tss.append(d_ts)
        vls.append(d_val)
        vds.append(d_vd)
        total += rows
    d_ots = torch.zeros(total, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(total, dtype=torch.float64, device="cuda")
    d_ovd = torch.zeros(total, dtype=torch.uint8, device="cuda")
    out_rows, offs = engine.compact_merge(gsets, tss, vls, vds,
                                          d_ots, d_oval, d_ovd)
    got_ts = d_ots[:out_rows].cpu().numpy()
    got_val = d_oval[:out_rows].cpu().numpy()
    got_vd = d_ovd[:out_rows].cpu().numpy()
    # oracle per series
    for s in range(nseries):
        per = [streams[f][s] for f in range(k)]
        ets, ev, evd = orc.merge_dedup(per)
        lo, hi = offs[s], offs[s + 1]
        assert hi - lo == ets.size, f"series {s}"
        assert (got_ts[lo:hi] == ets).all()
        assert (got_vd[lo:hi].astype(bool) == evd).all()
        gv = got_val[lo:hi]
        assert gv[evd].view(np.uint64).tolist() == ev[evd].view(np.uint64).tolist()
        assert (gv[~evd] == 0).all()
    for g in gsets:
        g.free()


def test_compact_merge_disjoint_streams(engine):
    """non-overlapping streams: merge is a pure interleave"""
    nseries, k = 4, 3
    streams = []
    for f in range(k):
        per = []
        for s in range(nseries):
            ts = T0 + (np.arange(500, dtype=np.int64) * k + f) * NS
            vals = rng.normal(0, 1, 500)
            per.append((ts, vals, None))
        streams.append(per)
    gsets, tss, vls = [], [], []
    total = 0
    for f in range(k):
        gset = _upload_stream(engine, streams[f])
        d_ts = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
        engine.decode(gset, 0, d_ts)
        engine.decode(gset, 1, d_val)
        gsets.append(gset)
        tss.append(d_ts)
        vls.append(d_val)
        total += gset.rows
    d_ots = torch.zeros(total, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(total, dtype=torch.float64, device="cuda")
    out_rows, offs = engine.compact_merge(gsets, tss, vls, [None] * k,
                                          d_ots, d_oval)
    assert out_rows == total
    got_ts = d_ots[:out_rows].cpu().numpy()
    for s in range(nseries):
        per = [streams[f][s] for f in range(k)]
        ets, ev, _ = orc.merge_dedup(per)
        lo, hi = offs[s], offs[s + 1]
        assert (got_ts[lo:hi] == ets).all()
        gv = d_oval[lo:hi].cpu().numpy()
        assert gv.view(np.uint64).tolist() == ev.view(np.uint64).tolist()
    for g in gsets:
        g.free()


def test_compact_merge_fuzz(engine):
    """Randomized merge configurations stressing the LDS-window kernels:
    k up to 16 (GS_MAX_STREAMS), wildly different stream densities (empty
    windows / windows much larger than a tile), tiny and tile-boundary
    series sizes, heavy collisions, nulls."""
    r = np.random.default_rng(13)
    for trial in range(10):
        k = int(r.choice([2, 3, 5, 8, 16]))
        nseries = int(r.integers(1, 4))
        sizes = [int(r.choice([1, 7, 100, 2047, 2048, 2049, 5000]))
                 for _ in range(k)]
        nulls = float(r.choice([0.0, 0.3]))
        grid = T0 + np.arange(20000, dtype=np.int64) * NS
        streams = []
        for f in range(k):
            per = []
            for s in range(nseries):
                take = np.sort(r.choice(grid.size, size=min(sizes[f],
                                                            grid.size),
                                        replace=False))
                ts = grid[take]
                vals = np.round(r.normal(50, 10, ts.size), 1)
                valid = (r.random(ts.size) > nulls) if nulls else None
                per.append((ts, vals, valid))
            streams.append(per)
        gsets, tss, vls, vds = [], [], [], []
        total = 0
        for f in range(k):
            gset = _upload_stream(engine, streams[f])
            d_ts = torch.zeros(gset.rows, dtype=torch.int64, device="cuda")
            d_val = torch.zeros(gset.rows, dtype=torch.float64,
                                device="cuda")
            d_vd = torch.zeros(gset.rows, dtype=torch.uint8, device="cuda")
            engine.decode(gset, 0, d_ts)
            engine.decode(gset, 1, d_val, d_vd)
            gsets.append(gset)
            tss.append(d_ts)
            vls.append(d_val)
            vds.append(d_vd)
            total += gset.rows
        d_ots = torch.zeros(total, dtype=torch.int64, device="cuda")
        d_oval = torch.zeros(total, dtype=torch.float64, device="cuda")
        d_ovd = torch.zeros(total, dtype=torch.uint8, device="cuda")
        out_rows, offs = engine.compact_merge(gsets, tss, vls, vds,
                                              d_ots, d_oval, d_ovd)
        got_ts = d_ots[:out_rows].cpu().numpy()
        got_val = d_oval[:out_rows].cpu().numpy()
        got_vd = d_ovd[:out_rows].cpu().numpy()
        for s in range(nseries):
            per = [streams[f][s] for f in range(k)]
            ets, ev, evd = orc.merge_dedup(per)
            lo, hi = offs[s], offs[s + 1]
            assert hi - lo == ets.size, f"trial {trial} series {s}"
            assert (got_ts[lo:hi] == ets).all(), f"trial {trial} ts"
            assert (got_vd[lo:hi].astype(bool) == evd).all(), \
                f"trial {trial} vd"
            gv = got_val[lo:hi]
            assert gv[evd].view(np.uint64).tolist() == \
                ev[evd].view(np.uint64).tolist(), f"trial {trial} val"
        for g in gsets:
            g.free()
