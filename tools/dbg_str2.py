import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
import cnosdb_amd as gs
eng = gs.Engine(0)
strs = [b"alpha", b"", b"tail", b"a", b"bb", b"ccc", b"d", b"ee", b"f"]
n = 9
blk = bytes([1]) + b"".join(len(s).to_bytes(8, "big") + s for s in strs)
ts = np.arange(n, dtype=np.int64) * 10**9
page = gs.build_page(blk, n)
gset = eng.upload([(0, [(gs.page_of(ts, gs.CT_TIME), gs.CT_TIME), (page, gs.CT_STR)])])
d_off = torch.zeros(n + 1, dtype=torch.int64, device="cuda")
d_bytes = torch.zeros(4096, dtype=torch.uint8, device="cuda")
total = eng.decode_str(gset, 1, d_off, d_bytes)
off = d_off.cpu().numpy()
d = np.diff(off)
print("DIAG enc=%d data_len=%d s0=%d s1=%d s8=%d allvalid=%d data_off=%d scr=%d" % tuple(d[:8].tolist()))
gset.free(); eng.close()
