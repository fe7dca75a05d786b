import sys, pathlib
sys.path.insert(0, "/root/repo")
import numpy as np, torch
import presto_amd as P
from presto_amd.tpcds import DsGen, ds_q17, ds_q72
gen = DsGen("/root/repo/oracle/liboracle.so")

def dev(a): return torch.from_numpy(np.ascontiguousarray(a)).cuda()
# fact date columns store raw day indexes, so the date
    # dimension is keyed by day index too (day 0 never occurs
    # in facts)
    sks = np.arange(0, 73049, dtype=np.int64)
d_year, d_qname, _ = gen.date_dim()

# ---- q72 at SF1: find the 12 missing groups ----
sf = 1.0
cs = gen.catalog_sales(sf, want_all=True)
cr = gen.catalog_returns(sf)
inv = gen.inventory(sf)
cs_p = P.Page({k: dev(cs[k]) for k in ("sold","ship","item","order","qty","cdemo","hdemo","promo")})
cr_p = P.Page({k: dev(cr[k]) for k in ("item","order")})
inv_p = P.Page({k: dev(inv[k]) for k in ("date","item","wh","qoh")})
date_y = P.Page({"sk": sks, "year": d_year})
cdemo = P.Page({"sk": np.arange(1,1920801,dtype=np.int64), "m": gen.cdemo_marital()})
hdemo = P.Page({"sk": np.arange(1,7201,dtype=np.int64), "b": gen.hdemo_buypot()})
got = ds_q72(gen, sf, cs_p, [inv_p], cr_p, date_y, cdemo, hdemo, 1999, 2, 3)
exp = gen.q72(sf, 1999, 2, 3)
gk = {g[:3]: g[3:] for g in got}
ek = {e[:3]: e[3:] for e in exp}
miss = [k for k in ek if k not in gk]
extra = [k for k in gk if k not in ek]
diff = [(k, gk[k], ek[k]) for k in ek if k in gk and gk[k] != ek[k]]
print("q72 missing:", miss[:6])
print("q72 extra:", extra[:6])
print("q72 valdiff:", diff[:4])
# inspect one missing group's source: which week/item?
if miss:
    it_id, wh, wk = miss[0]
    print("missing group:", it_id, wh, wk)
