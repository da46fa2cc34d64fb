"""Column-group pruning at the seam: host mirror of filter_column_groups
(tskv/src/reader/chunk.rs:12-49) + end-to-end proof that pruned pages are
never uploaded or decoded."""
import numpy as np
import pytest

import cnosdb_amd as gs


def test_prune_time_range():
    stats = [(0, 99, 0.0, 1.0), (100, 199, 0.0, 1.0), (200, 299, 0.0, 1.0)]
    # closed interval: group touching the bound stays
    assert gs.prune_column_groups(stats, time_range=(100, 200)) == \
        [False, True, True]
    assert gs.prune_column_groups(stats, time_range=(99, 99)) == \
        [True, False, False]


def test_prune_value_pred_keep_if_maybe():
    stats = [(0, 9, 10.0, 20.0), (0, 9, 30.0, 40.0)]
    assert gs.prune_column_groups(stats, value_pred=("gt", 25.0)) == \
        [False, True]
    assert gs.prune_column_groups(stats, value_pred=("le", 15.0)) == \
        [True, False]
    assert gs.prune_column_groups(stats, value_pred=("eq", 35.0)) == \
        [False, True]
    assert gs.prune_column_groups(stats, value_pred=("between", 21.0, 29.0)) \
        == [False, False]
    # ne prunes only a constant group equal to the literal
    assert gs.prune_column_groups([(0, 9, 5.0, 5.0)],
                                  value_pred=("ne", 5.0)) == [False]
    # missing value stats: never prunable by the value predicate
    assert gs.prune_column_groups([(0, 9, None, None)],
                                  value_pred=("gt", 1e9)) == [True]


def test_prune_combined():
    stats = [(0, 99, 0.0, 1.0), (100, 199, 50.0, 60.0)]
    keep = gs.prune_column_groups(stats, time_range=(50, 150),
                                  value_pred=("gt", 10.0))
    assert keep == [False, True]


@pytest.mark.gpu
def test_pruned_scan_equals_full_scan():
    """Scanning only the kept groups gives the same result as scanning
    everything — and the pruned pages were never uploaded (smaller set)."""
    import torch
    e = gs.Engine(0)
    t0 = 1_700_000_000_000_000_000
    ns = 1_000_000_000
    npts = 4096
    rng = np.random.default_rng(5)
    groups, stats = [], []
    for s in range(24):
        ts = t0 + (np.arange(npts, dtype=np.int64) + s * npts) * ns
        vals = np.round(np.clip(np.cumsum(rng.normal(0, 0.5, npts)) + 50,
                                0, 100), 1)
        groups.append((0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64), gs.CT_F64)]))
        stats.append((int(ts[0]), int(ts[-1]), float(vals.min()),
                      float(vals.max())))
    lo = t0 + int(6.5 * npts) * ns
    hi = t0 + int(14.2 * npts) * ns
    keep = gs.prune_column_groups(stats, time_range=(lo, hi))
    assert 0 < sum(keep) < len(keep)
    outs = []
    for subset in (groups, [g for g, k in zip(groups, keep) if k]):
        gset = e.upload(subset)
        rows = gset.rows
        d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
        d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
        d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
        r = e.scan(gset, d_ts, d_val, time_range=(lo, hi),
                   d_out_ts=d_ots, d_out_val=d_oval)
        outs.append((rows, r.out_rows,
                     d_ots[:r.out_rows].cpu().numpy().copy(),
                     d_oval[:r.out_rows].cpu().numpy().copy()))
        gset.free()
    (full_rows, n0, ts0, v0), (kept_rows, n1, ts1, v1) = outs
    assert kept_rows < full_rows  # pruned pages never uploaded
    assert n0 == n1
    assert (ts0 == ts1).all()
    assert v0.view(np.uint64).tolist() == v1.view(np.uint64).tolist()
    e.close()
