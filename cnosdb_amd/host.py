"""ctypes host bindings for libcnosdb_gs.so (see include/cnosdb_gs.h).

The binding mirrors the seam the reference's thin Rust shim would call
(SURVEY.md §8b): page upload standing where `ColumnGroupReader::read`
stands, `Engine.decode` standing where `decode_pages` stands
(tskv/src/tsm/reader.rs:494-560), `Engine.scan` standing where the
pure-time-range `DataFilter` + downsampling aggregate stand.
"""
import ctypes
import os

import numpy as np

CT_TIME, CT_I64, CT_F64, CT_BOOL, CT_U64, CT_STR = 0, 1, 2, 3, 4, 5

_DIR = os.path.dirname(os.path.abspath(__file__))


def lib_path():
    return os.path.join(_DIR, "libcnosdb_gs.so")


class GsTimeRange(ctypes.Structure):
    _fields_ = [("min_ts", ctypes.c_int64), ("max_ts", ctypes.c_int64)]


class GsPageSpec(ctypes.Structure):
    _fields_ = [
        ("bytes", ctypes.c_void_p),
        ("len", ctypes.c_uint64),
        ("num_values", ctypes.c_uint32),
        ("ctype", ctypes.c_uint8),
        ("column_id", ctypes.c_uint32),
    ]


class GsColumnGroupDesc(ctypes.Structure):
    _fields_ = [
        ("pages", ctypes.POINTER(GsPageSpec)),
        ("npages", ctypes.c_uint32),
        ("series_id", ctypes.c_uint32),
    ]


PRED_OPS = {"gt": 1, "ge": 2, "lt": 3, "le": 4, "eq": 5, "ne": 6,
            "between": 7}


class GsValuePred(ctypes.Structure):
    _fields_ = [("op", ctypes.c_int32), ("a", ctypes.c_double),
                ("b", ctypes.c_double)]


class GsScanSpec(ctypes.Structure):
    _fields_ = [
        ("range", GsTimeRange),
        ("field_col", ctypes.c_int32),
        ("tombstones", ctypes.POINTER(GsTimeRange)),
        ("n_tombstones", ctypes.c_size_t),
        ("bucket_ns", ctypes.c_int64),
        ("t0", ctypes.c_int64),
        ("n_buckets", ctypes.c_int32),
        ("d_agg_max", ctypes.c_void_p),
        ("d_agg_sum", ctypes.c_void_p),
        ("d_agg_count", ctypes.c_void_p),
        ("d_out_ts", ctypes.c_void_p),
        ("d_out_val", ctypes.c_void_p),
        ("d_ts", ctypes.c_void_p),
        ("d_val", ctypes.c_void_p),
        ("value_pred", GsValuePred),
    ]


class GsArrowArray(ctypes.Structure):
    pass


GsArrowArray._fields_ = [
    ("length", ctypes.c_int64), ("null_count", ctypes.c_int64),
    ("offset", ctypes.c_int64), ("n_buffers", ctypes.c_int64),
    ("n_children", ctypes.c_int64),
    ("buffers", ctypes.POINTER(ctypes.c_void_p)),
    ("children", ctypes.POINTER(ctypes.POINTER(GsArrowArray))),
    ("dictionary", ctypes.POINTER(GsArrowArray)),
    ("release", ctypes.c_void_p), ("private_data", ctypes.c_void_p),
]


class GsScanResult(ctypes.Structure):
    _fields_ = [
        ("out_rows", ctypes.c_int64),
        ("decoded_rows", ctypes.c_int64),
        ("ms_decode_ts", ctypes.c_double),
        ("ms_decode_val", ctypes.c_double),
        ("ms_filter", ctypes.c_double),
        ("ms_compact", ctypes.c_double),
        ("ms_agg", ctypes.c_double),
    ]


class PageLib:
    """Loads libcnosdb_gs.so and declares signatures. Singleton per process."""

    _inst = None

    def __new__(cls):
        if cls._inst is None:
            cls._inst = super().__new__(cls)
            cls._inst._load()
        return cls._inst

    def _load(self):
        path = lib_path()
        if not os.path.exists(path):
            raise RuntimeError(
                f"cnosdb_gs HIP extension not built: {path} missing. "
                "Run __graft_entry__.build() — the product path has no "
                "CPU fallback.")
        lib = ctypes.CDLL(path)
        self.lib = lib
        for nm in ("gs_encode_ts", "gs_encode_i64", "gs_encode_f64",
                   "gs_encode_bool", "gs_encode_str", "gs_build_page"):
            getattr(lib, nm).restype = ctypes.c_int64
        lib.gs_decode_str.restype = ctypes.c_int32
        lib.gs_decode_str.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint32,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64,
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64)]
        lib.gs_crc32.restype = ctypes.c_uint32
        lib.gs_version.restype = ctypes.c_char_p
        lib.gs_last_error.restype = ctypes.c_char_p
        lib.gs_device_count.restype = ctypes.c_int32
        lib.gs_ctx_create.restype = ctypes.c_void_p
        lib.gs_ctx_create.argtypes = [ctypes.c_int32]
        lib.gs_ctx_destroy.argtypes = [ctypes.c_void_p]
        lib.gs_groups_upload.restype = ctypes.c_void_p
        lib.gs_groups_upload.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(GsColumnGroupDesc),
            ctypes.c_size_t, ctypes.c_int32]
        lib.gs_groups_free.argtypes = [ctypes.c_void_p]
        lib.gs_set_rows.restype = ctypes.c_int64
        lib.gs_set_rows.argtypes = [ctypes.c_void_p]
        lib.gs_set_groups.restype = ctypes.c_int64
        lib.gs_set_groups.argtypes = [ctypes.c_void_p]
        lib.gs_set_series.restype = ctypes.c_int64
        lib.gs_set_series.argtypes = [ctypes.c_void_p]
        lib.gs_count_pushdown.restype = ctypes.c_int64
        lib.gs_count_pushdown.argtypes = [ctypes.c_void_p, ctypes.c_uint32]
        lib.gs_set_row_offsets.restype = ctypes.c_int32
        lib.gs_set_row_offsets.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        lib.gs_decode.restype = ctypes.c_int32
        lib.gs_decode.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_uint32, ctypes.c_void_p,
                                  ctypes.c_void_p]
        lib.gs_apply_tombstone.restype = ctypes.c_int32
        lib.gs_apply_tombstone.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.POINTER(GsTimeRange), ctypes.c_size_t]
        lib.gs_scan.restype = ctypes.c_int32
        lib.gs_scan.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                ctypes.POINTER(GsScanSpec),
                                ctypes.POINTER(GsScanResult)]
        lib.gs_scan_async.restype = ctypes.c_int32
        lib.gs_scan_async.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.POINTER(GsScanSpec)]
        lib.gs_scan_wait.restype = ctypes.c_int32
        lib.gs_scan_wait.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                     ctypes.POINTER(GsScanResult)]
        lib.gs_encode_pages_dev.restype = ctypes.c_int32
        lib.gs_encode_pages_dev.argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p,
            ctypes.c_int64, ctypes.c_void_p]
        lib.gs_export_group_column.restype = ctypes.c_int32
        lib.gs_export_group_column.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p,
            ctypes.c_int32, ctypes.c_void_p, ctypes.POINTER(GsArrowArray)]
        lib.gs_compact_merge.restype = ctypes.c_int32
        lib.gs_compact_merge.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p), ctypes.c_int32,
            ctypes.POINTER(ctypes.c_void_p), ctypes.POINTER(ctypes.c_void_p),
            ctypes.POINTER(ctypes.c_void_p), ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64)]

    def err(self):
        return self.lib.gs_last_error().decode()


def _np_ptr(arr):
    return arr.ctypes.data_as(ctypes.c_void_p)


# ---- host-side encoders (product write path; CPU like the reference's) ----

def encode_ts(values):
    values = np.ascontiguousarray(values, dtype=np.int64)
    lib = PageLib().lib
    buf = np.zeros(values.size * 9 + 64, dtype=np.uint8)
    n = lib.gs_encode_ts(_np_ptr(values), values.size, _np_ptr(buf), buf.size)
    if n < 0:
        raise RuntimeError(f"gs_encode_ts failed: {n}")
    return buf[:n].tobytes()


def encode_i64(values):
    values = np.ascontiguousarray(values, dtype=np.int64)
    lib = PageLib().lib
    buf = np.zeros(values.size * 9 + 64, dtype=np.uint8)
    n = lib.gs_encode_i64(_np_ptr(values), values.size, _np_ptr(buf), buf.size)
    if n < 0:
        raise RuntimeError(f"gs_encode_i64 failed: {n}")
    return buf[:n].tobytes()


def encode_f64(values):
    values = np.ascontiguousarray(values, dtype=np.float64)
    lib = PageLib().lib
    buf = np.zeros(values.size * 12 + 64, dtype=np.uint8)
    n = lib.gs_encode_f64(_np_ptr(values), values.size, _np_ptr(buf), buf.size)
    if n < 0:
        raise RuntimeError(f"gs_encode_f64 failed: {n}")
    return buf[:n].tobytes()


def encode_bool(values):
    values = np.ascontiguousarray(values, dtype=np.uint8)
    lib = PageLib().lib
    buf = np.zeros(values.size + 64, dtype=np.uint8)
    n = lib.gs_encode_bool(_np_ptr(values), values.size, _np_ptr(buf), buf.size)
    if n < 0:
        raise RuntimeError(f"gs_encode_bool failed: {n}")
    return buf[:n].tobytes()


def encode_str(strings):
    """Snappy string block (codec/string.rs:32-88): strings = list of
    bytes (non-null values only)."""
    lib = PageLib().lib
    concat = b"".join(strings)
    lens = np.array([len(s) for s in strings], dtype=np.uint64)
    src = np.frombuffer(concat, dtype=np.uint8) if concat else np.zeros(1, np.uint8)
    cap = 2 + 32 + len(concat) + len(concat) // 6 + len(strings) * 10 + 64
    buf = np.zeros(cap, dtype=np.uint8)
    n = lib.gs_encode_str(_np_ptr(src), _np_ptr(lens), len(strings),
                          _np_ptr(buf), buf.size)
    if n < 0:
        raise RuntimeError(f"gs_encode_str failed: {n}")
    return buf[:n].tobytes()


def str_page_of(strings, valid=None):
    """Build a string column page. strings = list of bytes per row;
    valid = bool array or None; null rows' strings are not encoded
    (page.rs/A.2 semantics)."""
    nrows = len(strings)
    if valid is None:
        present = strings
        bitset = None
    else:
        valid = np.asarray(valid, dtype=bool)
        present = [s for s, v in zip(strings, valid) if v]
        bitset = np.packbits(valid, bitorder="little")
    return build_page(encode_str(present), nrows, bitset)


def build_page(data, nrows, bitset=None):
    """Assemble full page bytes (tsm/page.rs layout). bitset: packed
    LSB-first validity bytes; None -> all valid."""
    lib = PageLib().lib
    if bitset is None:
        nb = (nrows + 7) // 8
        bitset = np.full(nb, 0xFF, dtype=np.uint8)
        if nrows % 8:
            bitset[-1] = (1 << (nrows % 8)) - 1
    else:
        bitset = np.ascontiguousarray(bitset, dtype=np.uint8)
    data = np.frombuffer(data, dtype=np.uint8) if isinstance(data, (bytes, bytearray)) else np.ascontiguousarray(data, dtype=np.uint8)
    out = np.zeros(16 + bitset.size + data.size, dtype=np.uint8)
    n = lib.gs_build_page(_np_ptr(bitset), nrows, _np_ptr(data), data.size,
                          _np_ptr(out), out.size)
    if n < 0:
        raise RuntimeError(f"gs_build_page failed: {n}")
    return out[:n].tobytes()


def page_of(values, ctype, valid=None):
    """Encode a column + assemble its page. valid: bool array or None.
    The encoded stream holds only non-null values (page.rs/A.2)."""
    values = np.asarray(values)
    nrows = values.size
    if valid is None:
        present = values
        bitset = None
    else:
        valid = np.asarray(valid, dtype=bool)
        present = values[valid]
        bitset = np.packbits(valid, bitorder="little")
    if ctype == CT_TIME:
        data = encode_ts(present.astype(np.int64))
    elif ctype == CT_I64:
        data = encode_i64(present.astype(np.int64))
    elif ctype == CT_F64:
        data = encode_f64(present.astype(np.float64))
    elif ctype == CT_BOOL:
        data = encode_bool(present.astype(np.uint8))
    else:
        raise ValueError(ctype)
    return build_page(data, nrows, bitset)


# ------------------------------ device engine ------------------------------

class GroupSet:
    def __init__(self, engine, handle, nrows, ngroups, keepalive):
        self._engine = engine
        self._h = handle
        self.rows = nrows
        self.ngroups = ngroups
        self._keepalive = keepalive  # page byte buffers must outlive upload

    def count_pushdown(self, col=0):
        """pushed-down COUNT from page metadata (no decode)."""
        return PageLib().lib.gs_count_pushdown(self._h, col)

    def row_offsets(self):
        out = np.zeros(self.ngroups, dtype=np.int64)
        st = PageLib().lib.gs_set_row_offsets(self._h, _np_ptr(out))
        if st != 0:
            raise RuntimeError(PageLib().err())
        return out

    def free(self):
        if self._h:
            PageLib().lib.gs_groups_free(self._h)
            self._h = None
            self._keepalive = None


class Engine:
    """One GPU device context. Raises at construction if no HIP device —
    the product path fails loudly rather than falling back to CPU."""

    def __init__(self, device=0):
        self._pl = PageLib()
        self.lib = self._pl.lib
        ctx = self.lib.gs_ctx_create(device)
        if not ctx:
            raise RuntimeError(f"gs_ctx_create failed: {self._pl.err()}")
        self._ctx = ctypes.c_void_p(ctx)
        self.device = device

    def close(self):
        if self._ctx:
            self.lib.gs_ctx_destroy(self._ctx)
            self._ctx = None

    def upload_packed(self, buf, page_off, page_len, num_values, ctypes_arr,
                      series_ids, pages_per_group, validate_crc=True):
        """Vectorized upload: all pages live in one contiguous host buffer.
        page_off/page_len/num_values/ctypes_arr are per-page numpy arrays in
        group-major order (group g's pages at [g*ppg, (g+1)*ppg), pages[0]
        of each group = time page); series_ids is per-group."""
        buf = np.frombuffer(buf, dtype=np.uint8) if isinstance(buf, (bytes, bytearray)) else buf
        npages = page_off.size
        ngroups = npages // pages_per_group
        spec_dt = np.dtype({
            "names": ["ptr", "len", "nv", "ct", "cid"],
            "formats": ["<u8", "<u8", "<u4", "u1", "<u4"],
            "offsets": [0, 8, 16, 20, 24],
            "itemsize": ctypes.sizeof(GsPageSpec),
        })
        specs = np.zeros(npages, dtype=spec_dt)
        specs["ptr"] = buf.ctypes.data + page_off.astype(np.uint64)
        specs["len"] = page_len.astype(np.uint64)
        specs["nv"] = num_values.astype(np.uint32)
        specs["ct"] = ctypes_arr.astype(np.uint8)
        specs["cid"] = np.tile(np.arange(pages_per_group, dtype=np.uint32), ngroups)
        gdesc_dt = np.dtype({
            "names": ["pages", "npages", "sid"],
            "formats": ["<u8", "<u4", "<u4"],
            "offsets": [0, 8, 12],
            "itemsize": ctypes.sizeof(GsColumnGroupDesc),
        })
        gdescs = np.zeros(ngroups, dtype=gdesc_dt)
        gdescs["pages"] = specs.ctypes.data + \
            (np.arange(ngroups, dtype=np.uint64) * pages_per_group *
             ctypes.sizeof(GsPageSpec))
        gdescs["npages"] = pages_per_group
        gdescs["sid"] = series_ids.astype(np.uint32)
        h = self.lib.gs_groups_upload(
            self._ctx,
            ctypes.cast(gdescs.ctypes.data, ctypes.POINTER(GsColumnGroupDesc)),
            ngroups, 1 if validate_crc else 0)
        if not h:
            raise RuntimeError(f"gs_groups_upload failed: {self._pl.err()}")
        h = ctypes.c_void_p(h)
        rows = self.lib.gs_set_rows(h)
        # gs_groups_upload synchronizes after staging all page bytes, so the
        # host buffer need not be retained (matters at 8 ranks x ~17 GB)
        return GroupSet(self, h, rows, ngroups, None)

    def upload(self, groups, validate_crc=True):
        """groups: list of (series_id, [(page_bytes, ctype), ...]);
        pages[0] must be the time page."""
        keep = []
        gdescs = (GsColumnGroupDesc * len(groups))()
        for gi, (sid, pages) in enumerate(groups):
            specs = (GsPageSpec * len(pages))()
            keep.append(specs)
            for pi, (pb, ctype) in enumerate(pages):
                if isinstance(pb, (bytes, bytearray)):
                    pb = np.frombuffer(pb, dtype=np.uint8)
                keep.append(pb)
                nrows = int.from_bytes(pb[4:12].tobytes(), "big")
                specs[pi].bytes = pb.ctypes.data
                specs[pi].len = pb.size
                specs[pi].num_values = nrows
                specs[pi].ctype = ctype
                specs[pi].column_id = pi
            gdescs[gi].pages = specs
            gdescs[gi].npages = len(pages)
            gdescs[gi].series_id = sid
        h = self.lib.gs_groups_upload(self._ctx, gdescs, len(groups),
                                      1 if validate_crc else 0)
        if not h:
            raise RuntimeError(f"gs_groups_upload failed: {self._pl.err()}")
        h = ctypes.c_void_p(h)
        rows = self.lib.gs_set_rows(h)
        return GroupSet(self, h, rows, len(groups), keep)

    def decode(self, gset, col, d_out, d_valid=None):
        """d_out/d_valid: torch CUDA tensors (int64/float64/uint8)."""
        st = self.lib.gs_decode(self._ctx, gset._h, col,
                                ctypes.c_void_p(d_out.data_ptr()),
                                ctypes.c_void_p(d_valid.data_ptr()) if d_valid is not None else None)
        if st != 0:
            raise RuntimeError(f"gs_decode failed ({st}): {self._pl.err()}")

    def raw_set(self, counts):
        """Series structure over caller-resident device arrays (memcache
        rows, MemCacheReader): counts[i] = rows of series i.  The
        returned set is scanned with Engine.scan where d_ts/d_val hold
        the rows themselves, and composes with compact_merge."""
        counts = np.ascontiguousarray(counts, dtype=np.int64)
        self.lib.gs_raw_set.restype = ctypes.c_void_p
        h = self.lib.gs_raw_set(self._ctx, _np_ptr(counts), counts.size)
        if not h:
            raise RuntimeError(f"gs_raw_set failed: {self._pl.err()}")
        return GroupSet(self, ctypes.c_void_p(h), int(counts.sum()),
                        counts.size, None)

    def decode_str(self, gset, col, d_offsets, d_bytes, d_valid=None):
        """String column decode to Arrow varbinary layout
        (str_snappy_decode_to_array, string.rs:226-276): d_offsets int64
        CUDA tensor of rows+1, d_bytes uint8 CUDA tensor (capacity);
        returns total payload bytes written."""
        total = ctypes.c_int64(0)
        st = self.lib.gs_decode_str(
            self._ctx, gset._h, col,
            ctypes.c_void_p(d_offsets.data_ptr()),
            ctypes.c_void_p(d_bytes.data_ptr()), d_bytes.numel(),
            ctypes.c_void_p(d_valid.data_ptr()) if d_valid is not None else None,
            ctypes.byref(total))
        if st != 0:
            raise RuntimeError(f"gs_decode_str failed ({st}): {self._pl.err()}")
        return total.value

    def apply_tombstone(self, gset, d_ts, d_valid, ranges):
        arr = (GsTimeRange * len(ranges))(*[GsTimeRange(a, b) for a, b in ranges])
        st = self.lib.gs_apply_tombstone(self._ctx, gset._h,
                                         ctypes.c_void_p(d_ts.data_ptr()),
                                         ctypes.c_void_p(d_valid.data_ptr()),
                                         arr, len(ranges))
        if st != 0:
            raise RuntimeError(f"gs_apply_tombstone failed: {self._pl.err()}")

    def export_group_column(self, gset, group, d_col, elem_size=8,
                            d_valid=None):
        """Export one group's rows as an Arrow-C-data-interface array
        (host copy).  Returns (GsArrowArray, np data view, np bitmap or
        None); call arr-release via free_arrow when done."""
        arr = GsArrowArray()
        st = self.lib.gs_export_group_column(
            self._ctx, gset._h, group, ctypes.c_void_p(d_col.data_ptr()),
            elem_size,
            ctypes.c_void_p(d_valid.data_ptr()) if d_valid is not None else None,
            ctypes.byref(arr))
        if st != 0:
            raise RuntimeError(f"gs_export_group_column failed: {self._pl.err()}")
        n = arr.length
        data = np.ctypeslib.as_array(
            ctypes.cast(arr.buffers[1], ctypes.POINTER(ctypes.c_uint8)),
            shape=(n * elem_size,)).copy()
        bitmap = None
        if arr.buffers[0]:
            bitmap = np.ctypeslib.as_array(
                ctypes.cast(arr.buffers[0], ctypes.POINTER(ctypes.c_uint8)),
                shape=((n + 7) // 8,)).copy()
        rel = ctypes.CFUNCTYPE(None, ctypes.POINTER(GsArrowArray))(arr.release)
        rel(ctypes.byref(arr))
        return arr, data, bitmap

    def compact_merge(self, gsets, d_ts_list, d_val_list, d_valid_list,
                      d_out_ts, d_out_val, d_out_valid=None):
        """k-way merge+dedup (BASELINE config #5). gsets oldest->newest;
        d_*_list: per-stream decoded torch tensors (valid entries may be
        None). Returns (out_rows, per-series offsets np.array)."""
        k = len(gsets)
        sets = (ctypes.c_void_p * k)(*[g._h for g in gsets])
        ts_p = (ctypes.c_void_p * k)(*[t.data_ptr() for t in d_ts_list])
        val_p = (ctypes.c_void_p * k)(*[t.data_ptr() for t in d_val_list])
        vd_p = (ctypes.c_void_p * k)(
            *[(t.data_ptr() if t is not None else None) for t in d_valid_list])
        nseries = self.lib.gs_set_series(gsets[0]._h)
        offs = np.zeros(nseries + 1, dtype=np.int64)
        rows = ctypes.c_int64(0)
        st = self.lib.gs_compact_merge(
            self._ctx, sets, k, ts_p, val_p, vd_p,
            ctypes.c_void_p(d_out_ts.data_ptr()),
            ctypes.c_void_p(d_out_val.data_ptr()),
            ctypes.c_void_p(d_out_valid.data_ptr()) if d_out_valid is not None else None,
            _np_ptr(offs), ctypes.byref(rows))
        if st != 0:
            raise RuntimeError(f"gs_compact_merge failed ({st}): {self._pl.err()}")
        return rows.value, offs

    def encode_pages_dev(self, kind, d_vals, row_off, rows, d_out,
                         cap_per_page, d_valid=None):
        """GPU page re-encode. kind: 0=ts 1=i64 2=f64. row_off/rows: host
        numpy arrays; d_out capacity npages*cap_per_page. Returns lens."""
        row_off = np.ascontiguousarray(row_off, dtype=np.int64)
        rows = np.ascontiguousarray(rows, dtype=np.int32)
        npages = row_off.size
        lens = np.zeros(npages, dtype=np.int64)
        st = self.lib.gs_encode_pages_dev(
            self._ctx, kind, ctypes.c_void_p(d_vals.data_ptr()),
            ctypes.c_void_p(d_valid.data_ptr()) if d_valid is not None else None,
            _np_ptr(row_off), _np_ptr(rows), npages,
            ctypes.c_void_p(d_out.data_ptr()), cap_per_page, _np_ptr(lens))
        if st != 0:
            raise RuntimeError(f"gs_encode_pages_dev failed ({st}): {self._pl.err()}")
        return lens

    def _mk_spec(self, d_ts, d_val, time_range, tombstones, d_out_ts,
                 d_out_val, agg, field_col=0):
        spec = GsScanSpec()
        lo, hi = time_range if time_range else (-(2**63), 2**63 - 1)
        spec.range = GsTimeRange(lo, hi)
        spec.field_col = field_col
        if tombstones:
            tarr = (GsTimeRange * len(tombstones))(*[GsTimeRange(a, b) for a, b in tombstones])
            spec.tombstones = tarr
            spec.n_tombstones = len(tombstones)
            spec._keep = tarr
        if d_ts is not None:
            spec.d_ts = d_ts.data_ptr()
            spec.d_val = d_val.data_ptr()
        if d_out_ts is not None:
            spec.d_out_ts = d_out_ts.data_ptr()
            spec.d_out_val = d_out_val.data_ptr()
        if agg:
            spec.bucket_ns = agg["bucket_ns"]
            spec.t0 = agg["t0"]
            spec.n_buckets = agg["n_buckets"]
            spec.d_agg_max = agg["d_max"].data_ptr()
            spec.d_agg_sum = agg["d_sum"].data_ptr()
            spec.d_agg_count = agg["d_count"].data_ptr()
        return spec

    def scan_async(self, gset, d_out_ts, d_out_val, time_range=None, agg=None,
                   field_col=0):
        """Enqueue the fused scan without synchronizing (fused-capable
        shapes only); pair with scan_wait."""
        spec = self._mk_spec(None, None, time_range, None, d_out_ts,
                             d_out_val, agg, field_col)
        st = self.lib.gs_scan_async(self._ctx, gset._h, ctypes.byref(spec))
        if st != 0:
            raise RuntimeError(f"gs_scan_async failed ({st}): {self._pl.err()}")

    def scan_wait(self, gset):
        res = GsScanResult()
        st = self.lib.gs_scan_wait(self._ctx, gset._h, ctypes.byref(res))
        if st != 0:
            raise RuntimeError(f"gs_scan_wait failed ({st}): {self._pl.err()}")
        return res

    def scan(self, gset, d_ts, d_val, time_range=None, tombstones=None,
             d_out_ts=None, d_out_val=None, agg=None, field_col=0,
             value_pred=None):
        """Fused scan. agg: dict(bucket_ns, t0, n_buckets, d_max, d_sum,
        d_count).  value_pred: (op, a) or ("between", a, b) evaluated like
        DataFilter's pushed expr.  Returns GsScanResult."""
        spec = GsScanSpec()
        spec.field_col = field_col
        if value_pred:
            spec.value_pred.op = PRED_OPS[value_pred[0]]
            spec.value_pred.a = float(value_pred[1])
            spec.value_pred.b = float(value_pred[2]) if len(value_pred) > 2 else 0.0
        lo, hi = time_range if time_range else (-(2**63), 2**63 - 1)
        spec.range = GsTimeRange(lo, hi)
        if tombstones:
            tarr = (GsTimeRange * len(tombstones))(*[GsTimeRange(a, b) for a, b in tombstones])
            spec.tombstones = tarr
            spec.n_tombstones = len(tombstones)
        spec.d_ts = d_ts.data_ptr()
        spec.d_val = d_val.data_ptr()
        if d_out_ts is not None:
            spec.d_out_ts = d_out_ts.data_ptr()
            spec.d_out_val = d_out_val.data_ptr()
        if agg:
            spec.bucket_ns = agg["bucket_ns"]
            spec.t0 = agg["t0"]
            spec.n_buckets = agg["n_buckets"]
            spec.d_agg_max = agg["d_max"].data_ptr()
            spec.d_agg_sum = agg["d_sum"].data_ptr()
            spec.d_agg_count = agg["d_count"].data_ptr()
        res = GsScanResult()
        st = self.lib.gs_scan(self._ctx, gset._h, ctypes.byref(spec),
                              ctypes.byref(res))
        if st != 0:
            raise RuntimeError(f"gs_scan failed ({st}): {self._pl.err()}")
        return res


def prune_column_groups(stats, time_range=None, value_pred=None):
    """Host-side mirror of `filter_column_groups` (tskv/src/reader/
    chunk.rs:12-49): the reference evaluates a DataFusion PruningPredicate
    over per-column-group min/max statistics (PageMeta ValueStatistics,
    tsm/page.rs:599-639) and drops groups no row of which can match.  The
    shim runs this BEFORE gs_groups_upload, so pruned pages are never
    read, uploaded, or decoded (a pruned page's corruption is therefore
    never observed — same as the reference never reading it).

    stats: iterable of (min_ts, max_ts, min_val, max_val); min_val/max_val
    may be None when the field column has no stats (never prunable then).
    time_range: closed interval (lo, hi) (TimeRange semantics,
    common/models/src/predicate/domain.rs:36-39).
    value_pred: ("gt"|"ge"|"lt"|"le"|"eq"|"ne"|"between", a[, b]) with
    PruningPredicate keep-if-maybe semantics (nulls fail predicates but
    other rows may pass, so stats ranges decide only certain misses).

    Returns a list of bools (True = keep), like the reference's indices.
    """
    keep = []
    for st in stats:
        min_ts, max_ts, min_v, max_v = st
        k = True
        if time_range is not None:
            lo, hi = time_range
            if max_ts < lo or min_ts > hi:
                k = False
        if k and value_pred is not None and min_v is not None \
                and max_v is not None:
            op, a = value_pred[0], value_pred[1]
            if op == "gt":
                k = max_v > a
            elif op == "ge":
                k = max_v >= a
            elif op == "lt":
                k = min_v < a
            elif op == "le":
                k = min_v <= a
            elif op == "eq":
                k = min_v <= a <= max_v
            elif op == "ne":
                k = not (min_v == max_v == a)
            elif op == "between":
                b = value_pred[2]
                k = max_v >= a and min_v <= b
        keep.append(bool(k))
    return keep


def _bind_groupby(lib):
    import ctypes
    if getattr(lib, "_gb_bound", False):
        return
    lib.gs_groupby_tag.restype = ctypes.c_int
    lib.gs_groupby_tag.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int,
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
        ctypes.c_int, ctypes.POINTER(ctypes.c_int)]
    lib._gb_bound = True


def groupby_tag(engine, gset, n_buckets, d_max, d_sum, d_count,
                cap_gids, tag_col=0):
    """GROUP BY tag over a completed aggregate scan + decoded tag column
    (gs_groupby_tag).  Returns (ngids, tag_rep_rows ndarray)."""
    import ctypes
    _bind_groupby(engine.lib)
    rep = np.zeros(cap_gids, dtype=np.int64)
    ng = ctypes.c_int(0)
    st = engine.lib.gs_groupby_tag(
        engine._ctx, gset._h, tag_col, n_buckets,
        d_max.data_ptr(), d_sum.data_ptr(), d_count.data_ptr(),
        rep.ctypes.data_as(ctypes.c_void_p), cap_gids, ctypes.byref(ng))
    if st != 0:
        raise RuntimeError(f"gs_groupby_tag failed ({st}): {engine._pl.err()}")
    return ng.value, rep[:ng.value]


def _bind_scan_fields(lib):
    import ctypes
    if getattr(lib, "_sf_bound", False):
        return
    lib.gs_scan_fields.restype = ctypes.c_int
    lib.gs_scan_fields.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                   ctypes.c_void_p, ctypes.c_void_p,
                                   ctypes.c_int, ctypes.c_void_p]
    lib._sf_bound = True


def scan_fields(engine, gset, fields, d_ts, d_val, time_range,
                d_out_ts, d_out_val, agg=None):
    """Fused multi-field scan (gs_scan_fields): one span/ts pass, then
    decode+aggregate per field.  d_out_val must hold len(fields)*rows
    doubles (field i at offset i*rows); agg buffers len(fields)*n_buckets."""
    import ctypes
    _bind_scan_fields(engine.lib)
    spec = GsScanSpec()
    spec.field_col = fields[0]
    lo, hi = time_range
    spec.range = GsTimeRange(lo, hi)
    spec.d_ts = d_ts.data_ptr()
    spec.d_val = d_val.data_ptr()
    spec.d_out_ts = d_out_ts.data_ptr()
    spec.d_out_val = d_out_val.data_ptr()
    if agg:
        spec.bucket_ns = agg["bucket_ns"]
        spec.t0 = agg["t0"]
        spec.n_buckets = agg["n_buckets"]
        spec.d_agg_max = agg["d_max"].data_ptr()
        spec.d_agg_sum = agg["d_sum"].data_ptr()
        spec.d_agg_count = agg["d_count"].data_ptr()
    fc = (ctypes.c_int32 * len(fields))(*fields)
    res = GsScanResult()
    st = engine.lib.gs_scan_fields(engine._ctx, gset._h, ctypes.byref(spec),
                                   fc, len(fields), ctypes.byref(res))
    if st != 0:
        raise RuntimeError(
            f"gs_scan_fields failed ({st}): {engine._pl.err()}")
    return res


def scan_fields_async(engine, gset, fields, d_ts, d_val, time_range,
                      d_out_ts, d_out_val, agg=None):
    """Async gs_scan_fields; pair with engine.scan_wait(gset)."""
    import ctypes
    _bind_scan_fields(engine.lib)
    if not getattr(engine.lib, "_sfa_bound", False):
        engine.lib.gs_scan_fields_async.restype = ctypes.c_int
        engine.lib.gs_scan_fields_async.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p,
            ctypes.c_void_p, ctypes.c_int]
        engine.lib._sfa_bound = True
    spec = GsScanSpec()
    spec.field_col = fields[0]
    lo, hi = time_range
    spec.range = GsTimeRange(lo, hi)
    spec.d_ts = d_ts.data_ptr()
    spec.d_val = d_val.data_ptr()
    spec.d_out_ts = d_out_ts.data_ptr()
    spec.d_out_val = d_out_val.data_ptr()
    if agg:
        spec.bucket_ns = agg["bucket_ns"]
        spec.t0 = agg["t0"]
        spec.n_buckets = agg["n_buckets"]
        spec.d_agg_max = agg["d_max"].data_ptr()
        spec.d_agg_sum = agg["d_sum"].data_ptr()
        spec.d_agg_count = agg["d_count"].data_ptr()
    fc = (ctypes.c_int32 * len(fields))(*fields)
    st = engine.lib.gs_scan_fields_async(engine._ctx, gset._h,
                                         ctypes.byref(spec), fc, len(fields))
    if st != 0:
        raise RuntimeError(
            f"gs_scan_fields_async failed ({st}): {engine._pl.err()}")
