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


if __name__ == "__main__":
    main()
