import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, ctypes
import cnosdb_amd as gs
eng = gs.Engine(0)
def run(tag, blk, n, valid):
    ts = np.arange(n, dtype=np.int64) * 10**9
    bitset = np.packbits(valid, bitorder="little") if valid is not None else None
    page = gs.build_page(blk, n, bitset)
    gset = eng.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME), (page, gs.CT_STR)])])
    d_off = torch.zeros(n + 1, dtype=torch.int64, device="cuda")
    d_bytes = torch.zeros(1024, dtype=torch.uint8, device="cuda")
    d_valid = torch.zeros(n, dtype=torch.uint8, device="cuda")
    try:
        total = eng.decode_str(gset, 1, d_off, d_bytes, d_valid)
        print(tag, "total", total, "off", d_off.cpu().tolist(), "valid", d_valid.cpu().tolist(),
              "bytes", d_bytes[:total].cpu().numpy().tobytes())
    except Exception as e:
        print(tag, "ERR", e)
    gset.free()
strs = [b"alpha", b"", b"tail"]
blk_all = bytes([1]) + b"".join(len(s).to_bytes(8, "big") + s for s in strs)
run("nullenc-allvalid", blk_all, 3, None)
valid = np.array([True, True, False, True])
run("nullenc-bitset  ", blk_all, 4, valid)
blk_sn = gs.encode_str(strs)
run("snappy-allvalid ", blk_sn, 3, None)
run("snappy-bitset   ", blk_sn, 4, valid)
eng.close()
