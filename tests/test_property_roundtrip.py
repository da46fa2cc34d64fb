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
