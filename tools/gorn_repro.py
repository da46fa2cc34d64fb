"""Repro/bisect helper for the general-shape decode error (PC_GORN).
Builds bench-like null pages, decodes, and on error bisects to a single
failing page, dumping its parameters.  Run on a GPU box."""
import sys
import numpy as np
import torch

sys.path.insert(0, ".")
import cnosdb_amd as gs
from oracle import pyoracle as orc

T0 = 1_700_000_000_000_000_000
NS = 1_000_000_000


def make_pool(rng, page_rows, uniq):
    pool = []
    for _ in range(uniq):
        walk = np.round(np.clip(
            np.cumsum(rng.normal(0, 0.5, page_rows)) + 50, 0, 100), 1)
        valid = rng.random(page_rows) > 0.10
        pool.append((walk, valid))
    return pool


def try_decode(eng, entries, page_rows):
    """entries: list of (walk, valid). One group per entry."""
    groups = []
    for i, (walk, valid) in enumerate(entries):
        ts = T0 + (np.arange(page_rows, dtype=np.int64)) * NS
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(walk, gs.CT_F64, valid), gs.CT_F64)]))
    gset = eng.upload(groups)
    out = torch.zeros(gset.rows, dtype=torch.float64, device="cuda")
    dv = torch.zeros(gset.rows, dtype=torch.uint8, device="cuda")
    err = None
    try:
        eng.decode(gset, 1, out, dv)
        # verify values
        offs = gset.row_offsets()
        host, hv = out.cpu().numpy(), dv.cpu().numpy()
        for i, (walk, valid) in enumerate(entries):
            n = len(walk)
            present = walk[valid]
            data = gs.encode_f64(present) if present.size else b""
            exp = orc.decode_f64(data, n, valid)
            got = host[offs[i]:offs[i] + n]
            if got.view(np.uint64).tolist() != exp.view(np.uint64).tolist():
                err = f"value mismatch entry {i}"
                break
            if not (hv[offs[i]:offs[i] + n] == valid.astype(np.uint8)).all():
                err = f"valid mismatch entry {i}"
                break
    except RuntimeError as e:
        err = str(e)
    gset.free()
    return err


def main():
    page_rows = int(sys.argv[1]) if len(sys.argv) > 1 else 4000
    nentries = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
    eng = gs.Engine(0)
    rng = np.random.default_rng(231)
    pool = make_pool(rng, page_rows, 256)
    entries = [pool[i % 256] for i in range(nentries)]
    err = try_decode(eng, entries, page_rows)
    if err is None:
        print(f"OK: {nentries} pages x {page_rows} rows decode clean")
        return
    print("ERROR on full set:", err)
    # bisect down to a minimal failing subset
    cur = entries
    while len(cur) > 1:
        half = len(cur) // 2
        a, b = cur[:half], cur[half:]
        ea = try_decode(eng, a, page_rows)
        if ea:
            cur = a
            continue
        eb = try_decode(eng, b, page_rows)
        if eb:
            cur = b
            continue
        print(f"neither half of {len(cur)} fails alone -> "
              f"scale/interaction-dependent")
        for k in (2, 4, 8):
            sub = cur[:max(1, len(cur) // k)]
            e = try_decode(eng, sub, page_rows)
            print(f"  first 1/{k}: {'FAIL ' + e if e else 'ok'}")
        return
    walk, valid = cur[0]
    print("single failing page: nulls=", int((~valid).sum()),
          "first-null=", int(np.argmin(valid)),
          "last-valid=", int(np.flatnonzero(valid)[-1]))
    np.save("gpurun_out/fail_walk.npy", walk)
    np.save("gpurun_out/fail_valid.npy", valid)
    print("dumped to gpurun_out/fail_{walk,valid}.npy")


if __name__ == "__main__" and not (len(sys.argv) > 1 and sys.argv[1] == "dump"):
    main()


def dump_states(nentries=98304, page_rows=4000):
    """Upload a failing-size set and diff chunk states of identical pool
    pages at low vs high page index."""
    import ctypes

    class DevGorChunk(ctypes.Structure):
        _fields_ = [("data_off", ctypes.c_uint64),
                    ("bitset_off", ctypes.c_uint64),
                    ("bitpos", ctypes.c_uint64),
                    ("val", ctypes.c_uint64),
                    ("row_off", ctypes.c_int64),
                    ("grp", ctypes.c_uint32),
                    ("row0", ctypes.c_uint32),
                    ("cnt", ctypes.c_uint32),
                    ("data_len", ctypes.c_uint32),
                    ("trailing", ctypes.c_uint8),
                    ("meaningful", ctypes.c_uint8),
                    ("last", ctypes.c_uint8),
                    ("flags", ctypes.c_uint8)]

    eng = gs.Engine(0)
    rng = np.random.default_rng(231)
    pool = make_pool(rng, page_rows, 256)
    entries = [pool[i % 256] for i in range(nentries)]
    groups = []
    for i, (walk, valid) in enumerate(entries):
        ts = T0 + (np.arange(page_rows, dtype=np.int64)) * NS
        groups.append((i, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME),
                           (gs.page_of(walk, gs.CT_F64, valid), gs.CT_F64)]))
    gset = eng.upload(groups)
    lib = eng.lib
    lib.gs_debug_gorn_table.restype = ctypes.c_int64
    lib.gs_debug_gorn_table.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_uint32, ctypes.c_void_p,
                                        ctypes.c_int64]
    n = lib.gs_debug_gorn_table(eng._ctx, gset._h, 1, None, 0)
    print("gorn chunks:", n)
    arr = (DevGorChunk * n)()
    r = lib.gs_debug_gorn_table(eng._ctx, gset._h, 1, arr, n)
    assert r == n
    per_page = n // nentries
    print("chunks/page:", per_page)
    bad = 0
    for p in range(nentries):
        ref_p = p % 256  # pool twin at low index
        for k in range(per_page):
            a, b = arr[ref_p * per_page + k], arr[p * per_page + k]
            if (a.bitpos, a.val, a.trailing, a.meaningful, a.flags) != \
               (b.bitpos, b.val, b.trailing, b.meaningful, b.flags):
                if bad < 8:
                    print(f"page {p} chunk {k}: bitpos {b.bitpos} vs "
                          f"{a.bitpos}, val {b.val:#x} vs {a.val:#x}, "
                          f"fl {b.flags} vs {a.flags}, dlen {b.data_len}")
                bad += 1
    print("mismatched chunk states:", bad)
    skel_bad = 0
    for p in range(nentries):
        for k in range(per_page):
            b = arr[p * per_page + k]
            exp_row0 = k * 2048
            exp_cnt = min(2048, page_rows - exp_row0)
            a = arr[(p % 256) * per_page + k]
            if (b.row0 != exp_row0 or b.cnt != exp_cnt or
                    b.grp != p or b.data_len != a.data_len or
                    b.last != (1 if k == per_page - 1 else 0)):
                if skel_bad < 5:
                    print(f"SKEL page {p} chunk {k}: row0={b.row0} cnt={b.cnt} "
                          f"grp={b.grp} dlen={b.data_len} last={b.last}")
                skel_bad += 1
    print("skeleton-corrupt entries:", skel_bad)
    # kernel-view vs copy-engine-view of the same bytes
    lib.gs_debug_kernel_crc.restype = ctypes.c_int64
    lib.gs_debug_kernel_crc.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        ctypes.c_uint64, ctypes.c_uint64,
                                        ctypes.c_void_p]

    def kcrc(off, ln):
        o = ctypes.c_uint64(0)
        assert lib.gs_debug_kernel_crc(eng._ctx, gset._h, off, ln,
                                       ctypes.byref(o)) == 0
        return o.value

    def hcrc(off, ln):
        buf = np.zeros(ln, dtype=np.uint8)
        lib.gs_debug_read_blob.restype = ctypes.c_int64
        lib.gs_debug_read_blob.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                           ctypes.c_uint64, ctypes.c_uint64,
                                           ctypes.c_void_p]
        assert lib.gs_debug_read_blob(eng._ctx, gset._h, off, ln,
                                      buf.ctypes.data_as(ctypes.c_void_p)) == ln
        b = buf.astype(np.uint64)
        idx = np.arange(ln, dtype=np.uint64)
        return int(((b + 1) * (idx + np.uint64(0x9E3779B97F4A7C15))).sum(
            dtype=np.uint64))

    for p in (0, 60000, 65536, 70000, nentries - 1):
        if p >= nentries:
            continue
        c = arr[p * per_page]
        kv = kcrc(c.data_off, c.data_len)
        hv = hcrc(c.data_off, c.data_len)
        print(f"page {p} data_off={c.data_off}: kernel-crc "
              f"{'==' if kv == hv else '!='} copy-crc")
    # offsets of the first mismatching page vs its twin + blob byte diff
    lib.gs_debug_read_blob.restype = ctypes.c_int64
    lib.gs_debug_read_blob.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                       ctypes.c_uint64, ctypes.c_uint64,
                                       ctypes.c_void_p]

    def blob_bytes(off, ln):
        buf = np.zeros(ln, dtype=np.uint8)
        r2 = lib.gs_debug_read_blob(eng._ctx, gset._h, off, ln,
                                    buf.ctypes.data_as(ctypes.c_void_p))
        assert r2 == ln, (off, ln, r2)
        return buf

    for p in (65535, 65536, 65537, 90000, 98000):
        if p >= nentries:
            continue
        ref_p = p % 256
        a, b = arr[ref_p * per_page], arr[p * per_page]
        print(f"page {p}: data_off {b.data_off} bitset_off {b.bitset_off} "
              f"dlen {b.data_len}; twin data_off {a.data_off} dlen {a.data_len}")
        da = blob_bytes(a.data_off, a.data_len)
        db = blob_bytes(b.data_off, b.data_len)
        same_data = (da == db).all() if a.data_len == b.data_len else False
        ba = blob_bytes(a.bitset_off, 500)
        bb = blob_bytes(b.bitset_off, 500)
        print(f"  data identical: {same_data}; bitset identical: "
              f"{(ba == bb).all()}")
        if not same_data and a.data_len == b.data_len:
            d = np.flatnonzero(da != db)
            print(f"  first byte diffs at {d[:6].tolist()} of {a.data_len}")
    # also scan for poisoned states
    pois = sum(1 for c in arr if c.row0 and c.bitpos >= c.data_len * 8)
    print("poisoned chunks:", pois)
    # rerun the sync kernel post-upload: if states become twin-consistent,
    # something later in upload stomped the table
    lib.gs_debug_rerun_gorn_sync.restype = ctypes.c_int64
    lib.gs_debug_rerun_gorn_sync.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                             ctypes.c_uint32]
    lib.gs_debug_rerun_gorn_sync(eng._ctx, gset._h, 1)
    r = lib.gs_debug_gorn_table(eng._ctx, gset._h, 1, arr, n)
    bad2 = 0
    for p in range(nentries):
        ref_p = p % 256
        for k in range(per_page):
            a, b = arr[ref_p * per_page + k], arr[p * per_page + k]
            if (a.bitpos, a.val, a.trailing, a.meaningful, a.flags) != \
               (b.bitpos, b.val, b.trailing, b.meaningful, b.flags):
                bad2 += 1
    print("mismatched after sync rerun:", bad2)
    gset.free()


if len(sys.argv) > 1 and sys.argv[1] == "dump":
    dump_states(int(sys.argv[2]) if len(sys.argv) > 2 else 98304,
                int(sys.argv[3]) if len(sys.argv) > 3 else 4000)
