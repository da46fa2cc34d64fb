"""GPU string column decode (gs_decode_str) vs the oracle: snappy and
uncompressed blocks, nulls, empty strings, all-null pages, tag-like
low-cardinality corpora and >64 KiB payload pages.  Boundary:
str_snappy_decode_to_array (codec/string.rs:226-276) reached through
data_buf_to_arrow_array (tsm/reader.rs:658-731)."""
import numpy as np
import pytest
import torch

import cnosdb_amd as gs
from oracle import pyoracle as orc

pytestmark = pytest.mark.gpu

rng = np.random.default_rng(29)


@pytest.fixture(scope="module")
def engine():
    eng = gs.Engine(0)
    yield eng
    eng.close()


def _mk_groups(cases):
    """cases: list of (strings, valid-or-None). One group per case, cols =
    [time, str]."""
    groups = []
    for i, (strs, valid) in enumerate(cases):
        n = len(strs)
        ts = np.arange(n, dtype=np.int64) * 10**9
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.str_page_of(strs, valid), gs.CT_STR)]))
    return groups


def _check(engine, cases):
    gset = engine.upload(_mk_groups(cases))
    rows = gset.rows
    cap = sum(sum(len(s) for s in strs) for strs, _ in cases) + 16
    d_off = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(cap, dtype=torch.uint8, device="cuda")
    d_valid = torch.zeros(rows, dtype=torch.uint8, device="cuda")
    total = engine.decode_str(gset, 1, d_off, d_bytes, d_valid)
    off = d_off.cpu().numpy()
    data = d_bytes[:total].cpu().numpy().tobytes()
    vd = d_valid.cpu().numpy()
    offs = gset.row_offsets()
    for i, (strs, valid) in enumerate(cases):
        n = len(strs)
        page = gs.str_page_of(strs, valid)
        blk = bytes(page[16 + (n + 7) // 8:])
        exp = orc.decode_str(blk, n, valid)
        base = offs[i]
        for r in range(n):
            got = data[off[base + r]:off[base + r + 1]]
            if exp[r] is None:
                assert vd[base + r] == 0 and got == b"", (i, r)
            else:
                assert vd[base + r] == 1 and got == exp[r], (i, r)
    gset.free()


def test_str_decode_snappy_corpora(engine):
    tags = [b"hostname=host_%04d" % i for i in range(64)]
    cases = []
    for t in range(8):
        n = int(rng.integers(1, 3000))
        strs = [tags[int(rng.integers(0, 64))] if rng.random() < 0.6
                else bytes(rng.integers(0, 256, int(rng.integers(0, 60))).astype(np.uint8))
                for _ in range(n)]
        valid = rng.random(n) > 0.25 if t % 2 else None
        cases.append((strs, valid))
    cases.append(([b""] * 37, None))                        # all-empty strings
    cases.append(([b"x" * 300] * 5, np.zeros(5, bool)))     # all-null page
    big = [bytes(rng.integers(97, 123, 60).astype(np.uint8)) for _ in range(2500)]
    cases.append((big, None))                               # >64 KiB payload
    _check(engine, cases)


def test_str_decode_uncompressed_blocks(engine):
    """Encoding::Null string blocks ([u64 BE len][bytes],
    string.rs:169-183) share the decode path."""
    strs = [b"alpha", b"", b"\x00\xff weird", b"tail"]
    valid = np.array([True, True, False, True])
    present = [s for s, v in zip(strs, valid) if v]
    blk = bytes([1]) + b"".join(len(s).to_bytes(8, "big") + s for s in present)
    n = len(strs)
    ts = np.arange(n, dtype=np.int64) * 10**9
    page = gs.build_page(blk, n, np.packbits(valid, bitorder="little"))
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_STR)])])
    d_off = torch.zeros(n + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(256, dtype=torch.uint8, device="cuda")
    d_valid = torch.zeros(n, dtype=torch.uint8, device="cuda")
    total = engine.decode_str(gset, 1, d_off, d_bytes, d_valid)
    exp = orc.decode_str(blk, n, valid)
    off = d_off.cpu().numpy()
    data = d_bytes[:total].cpu().numpy().tobytes()
    for r in range(n):
        got = data[off[r]:off[r + 1]]
        assert got == (exp[r] or b""), r
    assert (d_valid.cpu().numpy() == valid.astype(np.uint8)).all()
    gset.free()


def test_str_decode_rejects_numeric_api(engine):
    """gs_decode on a string column and gs_decode_str on a numeric column
    both fail loudly."""
    strs = [b"a", b"bb"]
    ts = np.arange(2, dtype=np.int64)
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (gs.str_page_of(strs), gs.CT_STR)])])
    out = torch.zeros(2, dtype=torch.int64, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode(gset, 1, out)
    d_off = torch.zeros(3, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(8, dtype=torch.uint8, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode_str(gset, 0, d_off, d_bytes)
    gset.free()


def test_str_decode_corrupt_block_errors(engine):
    """A snappy stream cut short must fail decode (decoder error path,
    string.rs:197)."""
    strs = [b"hello world, this is a longer string"] * 50
    blk = gs.encode_str(strs)
    cut = blk[:len(blk) // 2]
    n = len(strs)
    ts = np.arange(n, dtype=np.int64)
    page = gs.build_page(cut, n)
    gset = engine.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                               (page, gs.CT_STR)])])
    d_off = torch.zeros(n + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(4096, dtype=torch.uint8, device="cuda")
    with pytest.raises(RuntimeError):
        engine.decode_str(gset, 1, d_off, d_bytes)
    gset.free()


def test_strings_scale_parity(engine):
    """Large-set parity for the word-wide snappy copies: 4096 pages deep
    into a multi-GB scratch slab must decode byte-identically to the
    oracle (guards the unaligned-access hazard found on the GORN path)."""
    rng2 = np.random.default_rng(77)
    page_rows = 4000
    npages = 4096
    pool = []
    tagpool = [b"host_%06d" % i for i in range(64)]
    for u in range(32):
        idx = rng2.integers(0, len(tagpool), page_rows)
        blk = [tagpool[i] for i in idx]
        pool.append((gs.str_page_of(blk), blk))
    groups = []
    expect = []
    for p in range(npages):
        spage, blk = pool[p % 32]
        ts = np.arange(page_rows, dtype=np.int64) * 1000
        groups.append((p, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (spage, gs.CT_STR)]))
        expect.append(blk)
    gset = engine.upload(groups)
    rows = gset.rows
    d_off = torch.zeros(rows + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(rows * 16, dtype=torch.uint8, device="cuda")
    engine.decode_str(gset, 1, d_off, d_bytes)
    offs = d_off.cpu().numpy()
    data = d_bytes.cpu().numpy()
    # verify pages spread across the whole set, incl. the deepest ones
    for p in (0, 1, npages // 2, npages - 2, npages - 1):
        base = p * page_rows
        for r in (0, 1, page_rows // 2, page_rows - 1):
            got = data[offs[base + r]:offs[base + r + 1]].tobytes()
            assert got == expect[p][r], (p, r)
    # whole-set checksum vs expectation
    total = sum(len(b) for blk in expect for b in blk)
    assert int(offs[rows]) == total
    gset.free()
