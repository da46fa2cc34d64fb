"""One mixed-everything integration scan on the general path: groups with
heterogeneous ts encodings (RLE grid / simple8b irregular / sparse scaled
deltas), null-carrying Gorilla field pages, tombstones, a value predicate,
compacted outputs AND aggregates in a single gs_scan — validated against a
composed numpy/oracle restatement.  This is the closest single test to the
reference's scan-level sqllogictest cases."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000


def test_mixed_everything_scan():
    eng = gs.Engine(0)
    r = np.random.default_rng(97)
    groups, truth = [], []
    for s in range(24):
        kind = s % 3
        n = int(r.integers(500, 5000))
        if kind == 0:  # regular grid -> RLE ts
            ts = T0 + np.arange(n, dtype=np.int64) * NS
        elif kind == 1:  # irregular -> simple8b ts
            ts = T0 + np.sort(r.choice(np.arange(4 * n, dtype=np.int64), n,
                                       replace=False)) * NS
        else:  # sparse large steps
            ts = T0 + np.sort(r.choice(np.arange(20 * n, dtype=np.int64), n,
                                       replace=False)) * (7 * NS)
        vals = np.round(np.clip(np.cumsum(r.normal(0, 0.7, n)) + 50, 0, 100), 1)
        valid = r.random(n) > 0.15
        groups.append((s, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(vals, gs.CT_F64, valid), gs.CT_F64)]))
        truth.append((ts, vals, valid))
    gset = eng.upload(groups)
    rows = gset.rows
    d_ts = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_val = torch.zeros(rows, dtype=torch.float64, device="cuda")
    d_ots = torch.zeros(rows, dtype=torch.int64, device="cuda")
    d_oval = torch.zeros(rows, dtype=torch.float64, device="cuda")
    lo = T0 + 300 * NS
    hi = T0 + 30000 * NS
    dead = [(T0 + 1000 * NS, T0 + 1500 * NS), (T0 + 9000 * NS, T0 + 9100 * NS)]
    bucket_ns = 300 * NS
    nb = 120
    d_max = torch.full((nb,), -np.inf, dtype=torch.float64, device="cuda")
    d_sum = torch.zeros(nb, dtype=torch.float64, device="cuda")
    d_cnt = torch.zeros(nb, dtype=torch.int64, device="cuda")
    res = eng.scan(gset, d_ts, d_val, time_range=(lo, hi), tombstones=dead,
                   d_out_ts=d_ots, d_out_val=d_oval,
                   value_pred=("between", 30.0, 70.0),
                   agg=dict(bucket_ns=bucket_ns, t0=T0, n_buckets=nb,
                            d_max=d_max, d_sum=d_sum, d_count=d_cnt))
    # composed oracle
    exp_ts, exp_val = [], []
    emx = np.full(nb, -np.inf)
    esm = np.zeros(nb)
    ect = np.zeros(nb, dtype=np.int64)
    for ts, vals, valid in truth:
        v2 = orc.update_nullbits(ts, dead, valid)
        sel = (ts >= lo) & (ts <= hi) & v2 & \
            (np.where(v2, vals, np.nan) >= 30.0) & \
            (np.where(v2, vals, np.nan) <= 70.0)
        exp_ts.append(ts[sel])
        exp_val.append(vals[sel])
        # aggregate over filter-passing rows (mask = sel within time span)
        span = (ts >= lo) & (ts <= hi)
        m, su, c = orc.bucket_agg(ts[span], vals[span], sel[span],
                                  T0, bucket_ns, nb)
        emx = np.maximum(emx, m)
        esm += su
        ect += c
    exp_ts_c = np.concatenate(exp_ts)  # engine output is in group order
    exp_val_c = np.concatenate(exp_val).astype(np.float64)
    assert res.out_rows == exp_ts_c.size
    assert (d_ots[:res.out_rows].cpu().numpy() == exp_ts_c).all()
    got_v = d_oval[:res.out_rows].cpu().numpy()
    assert got_v.view(np.uint64).tolist() == exp_val_c.view(np.uint64).tolist()
    assert (d_cnt.cpu().numpy() == ect).all()
    gmx = d_max.cpu().numpy()
    assert (gmx[ect > 0] == emx[ect > 0]).all()
    assert np.allclose(d_sum.cpu().numpy(), esm, rtol=1e-12)
    gset.free()
    eng.close()


def test_cpp_shim_host_end_to_end():
    """The C++ host-side mirror of ColumnGroupReader (examples/
    shim_host.cpp) exercises the C ABI with no Python in the loop:
    build pages -> upload -> scan -> verify -> error behaviour."""
    import subprocess
    import os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exe = os.path.join(repo, "examples", "shim_host")
    assert os.path.exists(exe), "built by __graft_entry__.build()"
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    assert "shim_host OK" in out.stdout
