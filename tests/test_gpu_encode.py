"""GPU page re-encode vs the host encoder (byte-exact), plus the full
config #5 pipeline: decode k streams -> merge+dedup -> re-encode ->
decode again -> compare with oracle merge."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

rng = np.random.default_rng(55)
T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000


@pytest.fixture(scope="module")
def engine():
    e = gs.Engine(0)
    yield e
    e.close()


def test_encode_pages_byte_exact_f64(engine):
    rows = 4096
    npages = 32
    vals = np.concatenate([
        np.round(np.clip(np.cumsum(rng.normal(0, 0.5, rows)) + 50, 0, 100), 1)
        for _ in range(npages)])
    d_vals = torch.from_numpy(vals).cuda()
    cap = rows * 12 + 128
    d_out = torch.zeros(npages * cap, dtype=torch.uint8, device="cuda")
    lens = engine.encode_pages_dev(
        2, d_vals, np.arange(npages, dtype=np.int64) * rows,
        np.full(npages, rows, dtype=np.int32), d_out, cap)
    host = d_out.cpu().numpy()
    for p in range(npages):
        got = host[p * cap:p * cap + lens[p]].tobytes()
        exp = gs.page_of(vals[p * rows:(p + 1) * rows], gs.CT_F64)
        assert got == exp, f"page {p}"


def test_encode_pages_byte_exact_ts_i64(engine):
    rows = 3000
    cases = [
        T0 + np.arange(rows, dtype=np.int64) * NS,                   # ts RLE
        np.sort(rng.integers(0, 2**50, rows)).astype(np.int64),      # ts s8b
        rng.integers(-2**40, 2**40, rows).astype(np.int64),          # i64 s8b
        np.full(rows, 42, dtype=np.int64),                           # i64 RLE
        np.concatenate([[0, 1 << 61], rng.integers(0, 100, rows - 2)]).astype(np.int64),  # uncompressed
    ]
    kinds = [0, 0, 1, 1, 1]
    vals = np.concatenate(cases)
    d_vals = torch.from_numpy(vals).cuda()
    cap = rows * 10 + 128
    d_out = torch.zeros(len(cases) * cap, dtype=torch.uint8, device="cuda")
    for i, (c, kind) in enumerate(zip(cases, kinds)):
        lens = engine.encode_pages_dev(
            kind, d_vals, np.array([i * rows], dtype=np.int64),
            np.array([rows], dtype=np.int32), d_out[i * cap:], cap)
        got = d_out[i * cap:i * cap + lens[0]].cpu().numpy().tobytes()
        exp = gs.page_of(c, gs.CT_TIME if kind == 0 else gs.CT_I64)
        assert got == exp, f"case {i}"


def test_encode_gorilla_with_nulls(engine):
    rows = 2048
    vals = np.cumsum(rng.normal(0, 1, rows))
    valid = rng.random(rows) > 0.3
    d_vals = torch.from_numpy(vals).cuda()
    d_valid = torch.from_numpy(valid.astype(np.uint8)).cuda()
    cap = rows * 12 + 128
    d_out = torch.zeros(cap, dtype=torch.uint8, device="cuda")
    lens = engine.encode_pages_dev(
        2, d_vals, np.array([0], dtype=np.int64),
        np.array([rows], dtype=np.int32), d_out, cap, d_valid=d_valid)
    got = d_out[:lens[0]].cpu().numpy().tobytes()
    exp = gs.page_of(vals, gs.CT_F64, valid)
    assert got == exp


def test_config5_end_to_end(engine):
    """decode k=8 overlapping L0 streams -> GPU merge+dedup -> GPU
    re-encode at a block cap -> decode the new pages -> must equal the
    oracle merge (BASELINE config #5)."""
    nseries, k = 4, 8
    grid = T0 + np.arange(8000, dtype=np.int64) * NS
    streams = []
    for f in range(k):
        per = []
        for s in range(nseries):
            take = rng.random(grid.size) < 0.3
            ts = grid[take]
            vals = np.round(np.clip(np.cumsum(rng.normal(0, 1, ts.size)) + 50, 0, 100), 2)
            per.append((ts, vals, None))
        streams.append(per)
    gsets, tss, vls = [], [], []
    total = 0
    for f in range(k):
        groups = [(s, [(gs.page_of(t, gs.CT_TIME), gs.CT_TIME),
                       (gs.page_of(v, gs.CT_F64), gs.CT_F64)])
                  for s, (t, v, _) in enumerate(streams[f])]
        gset = engine.upload(groups)
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
    # re-encode per series at a max-datablock-style row cap
    block_rows = 1500
    row_off, rows_arr, page_series = [], [], []
    for s in range(nseries):
        r = offs[s]
        while r < offs[s + 1]:
            n = min(block_rows, offs[s + 1] - r)
            row_off.append(r)
            rows_arr.append(n)
            page_series.append(s)
            r += n
    cap = block_rows * 12 + 128
    d_enc_ts = torch.zeros(len(row_off) * cap, dtype=torch.uint8, device="cuda")
    d_enc_v = torch.zeros(len(row_off) * cap, dtype=torch.uint8, device="cuda")
    lens_ts = engine.encode_pages_dev(0, d_ots, np.array(row_off), np.array(rows_arr), d_enc_ts, cap)
    lens_v = engine.encode_pages_dev(2, d_oval, np.array(row_off), np.array(rows_arr), d_enc_v, cap)
    # decode the re-encoded pages and compare with the oracle merge
    henc_ts = d_enc_ts.cpu().numpy()
    henc_v = d_enc_v.cpu().numpy()
    for s in range(nseries):
        ets, ev, _ = orc.merge_dedup(streams[s] if False else [streams[f][s] for f in range(k)])
        got_ts, got_v = [], []
        for i, ps in enumerate(page_series):
            if ps != s:
                continue
            pts = henc_ts[i * cap:i * cap + lens_ts[i]].tobytes()
            pv = henc_v[i * cap:i * cap + lens_v[i]].tobytes()
            nb_t = int.from_bytes(pts[0:4], "big")
            nb_v = int.from_bytes(pv[0:4], "big")
            nrows = int.from_bytes(pts[4:12], "big")
            got_ts.append(orc.decode_i64(pts[16 + nb_t:], nrows))
            got_v.append(orc.decode_f64(pv[16 + nb_v:], nrows))
        got_ts = np.concatenate(got_ts)
        got_v = np.concatenate(got_v)
        assert (got_ts == ets).all(), f"series {s} ts"
        assert got_v.view(np.uint64).tolist() == ev.view(np.uint64).tolist()
    for g in gsets:
        g.free()
