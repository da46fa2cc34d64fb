"""ORACLE — TEST INFRASTRUCTURE ONLY (see tsm_oracle.c header).

ctypes bindings for liboracle.so plus the numpy restatement of the
scan-level semantics (filter/aggregate/tombstone) used to check the GPU
product path.  May be imported ONLY from tests/, __graft_entry__.smoke()
and bench.py's cpu_baseline leg.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def _ensure_built():
    if not os.path.exists(_SO):
        subprocess.check_call(["make", "-C", _DIR])


class Oracle:
    _inst = None

    def __new__(cls):
        if cls._inst is None:
            cls._inst = super().__new__(cls)
            cls._inst._load()
        return cls._inst

    def _load(self):
        _ensure_built()
        lib = ctypes.CDLL(_SO)
        self.lib = lib
        for nm in ("orc_ts_encode", "orc_i64_encode", "orc_f64_encode",
                   "orc_bool_encode", "orc_null_encode_i64", "orc_build_page"):
            getattr(lib, nm).restype = ctypes.c_int64
        lib.orc_crc32.restype = ctypes.c_uint32
        for nm in ("orc_decode_i64", "orc_decode_f64", "orc_decode_bool",
                   "orc_page_check", "orc_decode_pages_omp"):
            getattr(lib, nm).restype = ctypes.c_int32


def _ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def _bitset(nrows, valid=None):
    if valid is None:
        nb = (nrows + 7) // 8
        bs = np.full(nb, 0xFF, dtype=np.uint8)
        if nrows % 8:
            bs[-1] = (1 << (nrows % 8)) - 1
        return bs
    return np.packbits(np.asarray(valid, dtype=bool), bitorder="little")


def encode_ts(values):
    o = Oracle().lib
    v = np.ascontiguousarray(values, dtype=np.int64)
    buf = np.zeros(v.size * 9 + 64, dtype=np.uint8)
    n = o.orc_ts_encode(_ptr(v), v.size, _ptr(buf), buf.size)
    assert n >= 0, n
    return buf[:n].tobytes()


def encode_i64(values):
    o = Oracle().lib
    v = np.ascontiguousarray(values, dtype=np.int64)
    buf = np.zeros(v.size * 9 + 64, dtype=np.uint8)
    n = o.orc_i64_encode(_ptr(v), v.size, _ptr(buf), buf.size)
    assert n >= 0, n
    return buf[:n].tobytes()


def encode_f64(values):
    o = Oracle().lib
    v = np.ascontiguousarray(values, dtype=np.float64)
    buf = np.zeros(v.size * 12 + 64, dtype=np.uint8)
    n = o.orc_f64_encode(_ptr(v), v.size, _ptr(buf), buf.size)
    assert n >= 0, n
    return buf[:n].tobytes()


def encode_bool(values):
    o = Oracle().lib
    v = np.ascontiguousarray(values, dtype=np.uint8)
    buf = np.zeros(v.size + 64, dtype=np.uint8)
    n = o.orc_bool_encode(_ptr(v), v.size, _ptr(buf), buf.size)
    assert n >= 0, n
    return buf[:n].tobytes()


def decode_i64(data, nrows, valid=None):
    """Mirror of *_decode_to_array for ts/i64 pages: returns int64 array of
    nrows with null slots 0 (arrow builder semantics)."""
    o = Oracle().lib
    d = np.frombuffer(data, dtype=np.uint8)
    bs = _bitset(nrows, valid)
    out = np.zeros(nrows, dtype=np.int64)
    st = o.orc_decode_i64(_ptr(d), d.size, _ptr(bs), nrows, _ptr(out))
    if st != 0:
        raise RuntimeError(f"orc_decode_i64: {st}")
    return out


def decode_f64(data, nrows, valid=None):
    o = Oracle().lib
    d = np.frombuffer(data, dtype=np.uint8)
    bs = _bitset(nrows, valid)
    out = np.zeros(nrows, dtype=np.float64)
    st = o.orc_decode_f64(_ptr(d), d.size, _ptr(bs), nrows, _ptr(out))
    if st != 0:
        raise RuntimeError(f"orc_decode_f64: {st}")
    return out


def decode_bool(data, nrows, valid=None):
    o = Oracle().lib
    d = np.frombuffer(data, dtype=np.uint8)
    bs = _bitset(nrows, valid)
    out = np.zeros(nrows, dtype=np.uint8)
    st = o.orc_decode_bool(_ptr(d), d.size, _ptr(bs), nrows, _ptr(out))
    if st != 0:
        raise RuntimeError(f"orc_decode_bool: {st}")
    return out


def crc32(data):
    d = np.frombuffer(data, dtype=np.uint8)
    return Oracle().lib.orc_crc32(_ptr(d), d.size)


# ------------- scan-level restatement (numpy; cites tskv sources) ----------

def update_nullbits(ts, ranges, valid):
    """tsm/reader.rs:634-656: clear validity for rows whose ts lies in a
    deleted CLOSED range. ts sorted; valid is a bool array (modified copy
    returned)."""
    valid = np.asarray(valid, dtype=bool).copy()
    ts = np.asarray(ts)
    for mn, mx in ranges:
        start = np.searchsorted(ts, mn, side="left")
        i = np.searchsorted(ts, mx, side="left")
        end = i + 1 if i < ts.size and ts[i] == mx else i
        valid[start:end] = False
    return valid


def time_span(ts, mn, mx):
    """closed-interval [mn,mx] span on sorted ts (TimeRange semantics,
    common/models/src/predicate/domain.rs:36-44)."""
    s = np.searchsorted(ts, mn, side="left")
    e = np.searchsorted(ts, mx, side="right")
    return s, e - s


def merge_dedup(streams):
    """sort_merge + BatchMergeBuilder dedup (reader/sort_merge.rs:152-343,
    reader/batch_builder.rs:106-155): k time-sorted streams ordered
    oldest -> newest (grouped chunks sorted by file_id, iterator.rs:488;
    loser-tree ties break toward the lower stream index,
    sort_merge.rs:305-312).  Equal-ts rows collapse to one; per column the
    value comes from the NEWEST stream containing that ts with a non-null
    value, else the row is null (value slot 0).  streams: list of
    (ts, val, valid-or-None)."""
    all_ts = np.concatenate([np.asarray(t) for t, _, _ in streams])
    uts = np.unique(all_ts)
    out_val = np.zeros(uts.size, dtype=np.float64)
    out_valid = np.zeros(uts.size, dtype=bool)
    for t, v, vd in streams:  # oldest -> newest: later valid rows overwrite
        idx = np.searchsorted(uts, np.asarray(t))
        if vd is None:
            out_val[idx] = v
            out_valid[idx] = True
        else:
            m = np.asarray(vd, dtype=bool)
            out_val[idx[m]] = np.asarray(v)[m]
            out_valid[idx[m]] = True
    return uts, out_val, out_valid


def bucket_agg(ts, vals, valid, t0, bucket_ns, n_buckets):
    """stock-DataFusion-style per-bucket max/sum/count over non-null rows
    (the downsampling aggregate run above TskvExec)."""
    ts = np.asarray(ts)
    vals = np.asarray(vals)
    m = np.asarray(valid, dtype=bool) if valid is not None else np.ones(ts.size, bool)
    b = (ts - t0) // bucket_ns
    m = m & (b >= 0) & (b < n_buckets)
    b = b[m]
    v = vals[m]
    mx = np.full(n_buckets, -np.inf)
    sm = np.zeros(n_buckets)
    ct = np.zeros(n_buckets, dtype=np.int64)
    np.maximum.at(mx, b, v)
    np.add.at(sm, b, v)
    np.add.at(ct, b, 1)
    return mx, sm, ct


def encode_str(strings):
    """string.rs:32-88 — Snappy string block [7][0x10][snappy raw]."""
    lib = Oracle().lib
    lib.orc_str_encode.restype = ctypes.c_int64
    concat = b"".join(strings)
    lens = np.array([len(s) for s in strings], dtype=np.uint64)
    src = np.frombuffer(concat, dtype=np.uint8) if concat else np.zeros(1, np.uint8)
    cap = 2 + 32 + len(concat) + len(concat) // 6 + lens.size * 10 + 64
    dst = np.zeros(cap, dtype=np.uint8)
    n = lib.orc_str_encode(_ptr(src), _ptr(lens), len(strings), _ptr(dst),
                           dst.size)
    if n < 0:
        raise RuntimeError("orc_str_encode failed")
    return dst[:n].tobytes()


def decode_str(data, nrows, valid=None):
    """str_snappy_decode_to_array semantics (string.rs:226-276): returns
    (list of bytes-or-None per row).  Handles Snappy=7 and Null=1."""
    lib = Oracle().lib
    lib.orc_str_decode.restype = ctypes.c_int64
    src = np.frombuffer(data, dtype=np.uint8) if data else np.zeros(1, np.uint8)
    if valid is not None:
        bs = np.packbits(np.asarray(valid, dtype=bool), bitorder="little")
        bsp = _ptr(bs)
    else:
        bsp = None
    bytes_out = np.zeros(len(data) * 64 + (1 << 20), dtype=np.uint8)
    lens = np.zeros(max(1, nrows), dtype=np.int64)
    n = lib.orc_str_decode(_ptr(src), len(data), bsp, nrows, _ptr(bytes_out),
                           bytes_out.size, _ptr(lens))
    if n < 0:
        raise RuntimeError("orc_str_decode failed")
    out, off = [], 0
    for r in range(nrows):
        if lens[r] < 0:
            out.append(None)
        else:
            out.append(bytes_out[off:off + lens[r]].tobytes())
            off += lens[r]
    return out
