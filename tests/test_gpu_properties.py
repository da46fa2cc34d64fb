"""Size-independent properties at near-BASELINE scale + randomized scan
fuzzing vs the oracle composition.

The tier contract asks for full-size validation through properties the
domain offers (sortedness, closed-form counts/sums, cross-path equality)
since the oracle only runs at small sizes.  The regular ts grid makes
exact integer closed forms available: span counts, per-group arithmetic-
series sums of the selected timestamps, and per-bucket row counts."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000
BUCKET = 300 * NS


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


def test_fullsize_scan_properties(engine):
    """2.5e8 rows (250 series x 1M pts, reference-shaped 125k-row pages):
    every verifiable closed-form invariant of the fused scan, verified on
    device (no host transfer of the 2 GB outputs)."""
    nseries, npts, page_rows = 250, 1_000_000, 125_000
    npages = npts // page_rows
    rng = np.random.default_rng(99)
    uniq = 16
    tpages = [gs.page_of(
        T0 + (np.arange(page_rows, dtype=np.int64) + p * page_rows) * NS,
        gs.CT_TIME) for p in range(npages)]
    vpool = [gs.page_of(np.round(np.clip(
        np.cumsum(rng.normal(0, 0.5, page_rows)) + 50, 0, 100), 1),
        gs.CT_F64) for _ in range(uniq)]
    groups = []
    for s in range(nseries):
        for p in range(npages):
            groups.append((s, [(tpages[p], gs.CT_TIME),
                               (vpool[(s + p) % uniq], gs.CT_F64)]))
    gset = engine.upload(groups)
    rows = gset.rows
    assert rows == nseries * npts
    # closed-interval range cutting mid-page and mid-chunk
    lo = T0 + 250_123 * NS
    hi = T0 + 750_789 * NS
    sel_per_series = 750_789 - 250_123 + 1
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
    res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                      d_out_ts=d_ots, d_out_val=d_oval, agg=agg)
    # (1) exact selected-row count, closed form
    assert res.out_rows == nseries * sel_per_series
    n = res.out_rows
    ots = d_ots[:n].view(nseries, sel_per_series)
    # (2) per-series sortedness AND exact endpoints (on device)
    assert bool((ots[:, 0] == lo).all())
    assert bool((ots[:, -1] == hi).all())
    assert bool((ots[:, 1:] > ots[:, :-1]).all())
    # (3) exact arithmetic-series sum of selected ts per series (closed
    # form, integer-exact modulo 2^64 — compute both sides in uint64)
    k = np.uint64(sel_per_series)
    s_lo = np.uint64(lo)
    ssum = (k * s_lo + np.uint64(NS) * (k * (k - np.uint64(1)) //
                                        np.uint64(2)))
    got = ots.view(torch.int64).sum(dim=1)  # wraps like int64
    assert bool((got.cpu().numpy().astype(np.uint64) == ssum).all())
    # (4) bucket counts: every bucket fully inside the range holds
    # nseries*300 rows; totals match out_rows
    cnt = agg["d_count"].cpu().numpy()
    assert cnt.sum() == n
    b_lo = (250_123 * NS) // BUCKET
    b_hi = (750_789 * NS) // BUCKET
    assert (cnt[b_lo + 1:b_hi] == nseries * 300).all()
    assert cnt[:b_lo].sum() == 0 and cnt[b_hi + 1:].sum() == 0
    # (5) cross-path equality at size: the general path (forced by
    # omitting compacted outputs = no fused precondition) must produce
    # identical aggregates
    agg2 = dict(bucket_ns=BUCKET, t0=T0, n_buckets=nb,
                d_max=torch.full((nb,), -np.inf, dtype=torch.float64,
                                 device="cuda"),
                d_sum=torch.zeros(nb, dtype=torch.float64, device="cuda"),
                d_count=torch.zeros(nb, dtype=torch.int64, device="cuda"))
    engine.scan(gset, d_ts, d_val, time_range=(lo, hi), agg=agg2)
    assert bool((agg2["d_count"] == agg["d_count"]).all())
    nzm = agg["d_count"] > 0
    assert bool((agg2["d_max"][nzm] == agg["d_max"][nzm]).all())
    assert bool(torch.allclose(agg2["d_sum"], agg["d_sum"], rtol=1e-12))
    # (6) max invariants: within the value clip range
    mx = agg["d_max"].cpu().numpy()
    assert (mx[cnt > 0] >= 0).all() and (mx[cnt > 0] <= 100).all()
    gset.free()


def test_scan_fuzz_vs_oracle(engine):
    """20 random scan configurations (sizes, page shapes straddling chunk
    boundaries, nulls, tombstones, value predicates, degenerate ranges)
    compared against the numpy/oracle composition."""
    rng = np.random.default_rng(20250915)
    for trial in range(20):
        nseries = int(rng.integers(1, 6))
        page_rows = int(rng.choice([7, 100, 2047, 2048, 2049, 5000]))
        npages = int(rng.integers(1, 4))
        npts = page_rows * npages
        null_frac = float(rng.choice([0.0, 0.1, 0.4]))
        use_pred = bool(rng.integers(0, 2))
        use_tomb = bool(rng.integers(0, 2))
        groups, truth = [], []
        for s in range(nseries):
            ts = T0 + np.arange(npts, dtype=np.int64) * NS
            vals = np.round(np.clip(
                np.cumsum(rng.normal(0, 0.5, npts)) + 50, 0, 100), 1)
            valid = (rng.random(npts) > null_frac)
            if not valid.any():
                valid[0] = True
            pages = []
            for p in range(npages):
                sl = slice(p * page_rows, (p + 1) * page_rows)
                pages.append((gs.page_of(ts[sl], gs.CT_TIME), gs.CT_TIME))
                pages.append((gs.page_of(vals[sl], gs.CT_F64,
                                         None if null_frac == 0.0
                                         else valid[sl]), gs.CT_F64))
            for p in range(npages):
                groups.append((s, [pages[2 * p], pages[2 * p + 1]]))
            truth.append((ts, vals, valid if null_frac else
                          np.ones(npts, dtype=bool)))
        gset = engine.upload(groups)
        rows = gset.rows
        a = int(rng.integers(0, npts))
        b = int(rng.integers(0, npts))
        lo = T0 + min(a, b) * NS
        hi = T0 + max(a, b) * NS
        tombs = None
        if use_tomb:
            c = int(rng.integers(0, npts))
            d = c + int(rng.integers(0, npts // 2 + 1))
            tombs = [(T0 + c * NS, T0 + d * NS)]
        pred = ("gt", 50.0) if use_pred else None
        d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
        res = engine.scan(gset, d_ts, d_val, time_range=(lo, hi),
                          tombstones=tombs, d_out_ts=d_ots,
                          d_out_val=d_oval, value_pred=pred)
        exp_ts, exp_val = [], []
        for ts, vals, valid in truth:
            m = (ts >= lo) & (ts <= hi)
            vmask = valid.copy()  # decode validity (nulls)
            if tombs:
                tm = (ts >= tombs[0][0]) & (ts <= tombs[0][1])
                vmask = vmask & ~tm  # tombstone clears validity
            if pred:
                # DataFilter semantics: null/tombstoned rows fail the
                # predicate; selection = span & validity & comparison
                sel = m & vmask & (vals > pred[1])
                ev = vals[sel].copy()
            else:
                # pure time filter: compaction keeps every span row; null
                # rows DECODE as 0.0 (arrow append_null buffer), while
                # tombstoned rows keep their decoded value (the tombstone
                # clears only the validity bits, reader.rs:634-656)
                sel = m
                ev = vals[sel].copy()
                ev[~valid[sel]] = 0.0
            exp_ts.append(ts[sel])
            exp_val.append(ev)
        exp_ts = np.concatenate(exp_ts) if exp_ts else np.array([], np.int64)
        exp_val = np.concatenate(exp_val) if exp_val else np.array([])
        assert res.out_rows == exp_ts.size, \
            f"trial {trial}: rows {res.out_rows} != {exp_ts.size}"
        got_ts = d_ots[:res.out_rows].cpu().numpy()
        got_val = d_oval[:res.out_rows].cpu().numpy()
        assert (got_ts == exp_ts).all(), f"trial {trial} ts"
        assert got_val.view(np.uint64).tolist() == \
            exp_val.view(np.uint64).tolist(), f"trial {trial} val"
        gset.free()
