"""Keep bench.py's CPU-side code paths green without a GPU: the workload
builder (packed pages must parse) and the cpu_baseline leg (oracle+OpenMP)."""
import numpy as np

from oracle import pyoracle as orc


def test_build_workload_pages_parse():
    import bench
    sub, raw, npages = bench.build_workload(8, 16000, 4000, 4, 2, nfields=2)
    assert len(sub) == 2 and npages == 4
    buf, offs, lens, nvals, cts, sids = sub[0]
    assert offs[-1] + lens[-1] == len(buf)
    assert cts[:6].tolist() == [0, 2, 2, 0, 2, 2]
    b = np.frombuffer(buf, np.uint8)
    # every page header must parse and carry the right row count + crc
    for i in range(min(12, lens.size)):
        pg = b[offs[i]:offs[i] + lens[i]]
        nb = int.from_bytes(pg[0:4].tobytes(), "big")
        rows = int.from_bytes(pg[4:12].tobytes(), "big")
        crc = int.from_bytes(pg[12:16].tobytes(), "big")
        assert rows == 4000
        assert crc == orc.crc32(pg[16 + nb:].tobytes())
        # decode the data region with the oracle
        data = pg[16 + nb:].tobytes()
        if cts[i] == 0:
            ts = orc.decode_i64(data, rows)
            assert (np.diff(ts) == 10**9).all()
        else:
            orc.decode_f64(data, rows)


def test_cpu_baseline_leg_small():
    import bench
    r = bench.cpu_baseline_leg(2, 16000, 4000, bench.T0 + 4000 * bench.NS,
                               bench.T0 + 12000 * bench.NS - 1)
    assert r["value"] > 0 and r["unit"] == "values/s" and r["kind"] == "port"
