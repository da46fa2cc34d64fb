"""Property-based round trips (hypothesis): product encoder -> oracle
decoder must reproduce the input bit-exactly for every codec, on
adversarial shapes the fixed corpus may miss."""
import numpy as np
from hypothesis import given, settings, strategies as st

import cnosdb_amd as gs
from oracle import pyoracle as orc

I64 = st.integers(min_value=-(2**63), max_value=2**63 - 1)


@settings(max_examples=60, deadline=None)
@given(st.lists(I64, min_size=1, max_size=400))
def test_i64_roundtrip(vals):
    a = np.array(vals, dtype=np.int64)
    enc = gs.encode_i64(a)
    assert enc == orc.encode_i64(a)
    assert (orc.decode_i64(enc, a.size) == a).all()


@settings(max_examples=60, deadline=None)
@given(st.lists(I64, min_size=1, max_size=400))
def test_ts_roundtrip(vals):
    a = np.array(vals, dtype=np.int64)
    enc = gs.encode_ts(a)
    assert enc == orc.encode_ts(a)
    assert (orc.decode_i64(enc, a.size) == a).all()


@settings(max_examples=60, deadline=None)
@given(st.lists(st.floats(allow_nan=False, allow_infinity=True, width=64),
                min_size=1, max_size=300))
def test_f64_roundtrip(vals):
    a = np.array(vals, dtype=np.float64)
    enc = gs.encode_f64(a)
    assert enc == orc.encode_f64(a)
    dec = orc.decode_f64(enc, a.size)
    assert dec.view(np.uint64).tolist() == a.view(np.uint64).tolist()


@settings(max_examples=40, deadline=None)
@given(st.lists(I64, min_size=2, max_size=200), st.data())
def test_i64_roundtrip_with_nulls(vals, data):
    a = np.array(vals, dtype=np.int64)
    valid = np.array(data.draw(st.lists(st.booleans(), min_size=a.size,
                                        max_size=a.size)), dtype=bool)
    present = a[valid]
    enc = gs.encode_i64(present) if present.size else b""
    dec = orc.decode_i64(enc, a.size, valid)
    exp = np.where(valid, a, 0)
    assert (dec == exp).all()


@settings(max_examples=60, deadline=None)
@given(st.lists(st.binary(min_size=0, max_size=120), min_size=1,
                max_size=300))
def test_str_roundtrip(strs):
    """product snappy encoder == oracle byte-for-byte, and the oracle
    decoder restores the strings (codec/string.rs snappy block)."""
    enc = gs.encode_str(strs)
    assert enc == orc.encode_str(strs)
    assert orc.decode_str(enc, len(strs)) == strs


@settings(max_examples=30, deadline=None)
@given(st.lists(st.sampled_from([b"north", b"south", b"east", b"west",
                                 b"", b"x" * 90]),
                min_size=1, max_size=500), st.data())
def test_str_roundtrip_with_nulls(strs, data):
    """repetitive tag-like corpora (snappy copies) with a validity mask:
    null rows consume nothing from the payload (string.rs:226-276)."""
    valid = data.draw(st.lists(st.booleans(), min_size=len(strs),
                               max_size=len(strs)))
    present = [s for s, v in zip(strs, valid) if v]
    enc = gs.encode_str(present)
    got = orc.decode_str(enc, len(strs), np.array(valid, dtype=bool))
    exp = [s if v else None for s, v in zip(strs, valid)]
    assert got == exp
