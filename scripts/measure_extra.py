#!/usr/bin/env python3
"""Measurement harness for the beyond-contract query pipelines
(q8/q9/q12/q14/q17/q18/q19/q21/q22) at a given scale factor.

TEST/BENCH INFRASTRUCTURE: generates inputs with the oracle's dbgen
restatement (like bench.py), stages the hot columns into HBM as torch
tensors, times the GPU pipeline (best of reps, wall clock around the
operator graph), and verifies the result against the CPU oracle run on
the host cores.  Writes one JSON line per query to stdout and
gpurun_out/extra_q.json.

Usage: python scripts/measure_extra.py [--sf 30] [--reps 3]
       [--queries q9,q12,...]
"""
import argparse
import json
import os
import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from tests.oracle_binding import OracleLib  # noqa: E402
import presto_amd as P  # noqa: E402
from presto_amd.pipelines import Q22_CODE_NATIONS  # noqa: E402

REPO = pathlib.Path(__file__).resolve().parent.parent


def dev(a):
    return torch.from_numpy(np.ascontiguousarray(a)).cuda()


def page(cols):
    return P.Page({k: (v if isinstance(v, P.Varbin) else dev(v))
                   for k, v in cols.items()})


def run(name, fn, reps):
    best = None
    out = None
    for _ in range(reps):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = fn()
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        best = dt if best is None else min(best, dt)
    return out, best


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sf", type=float, default=30.0)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--queries", default="q8,q9,q12,q14,q17,q18,q19,q21,q22")
    args = ap.parse_args()
    sf = args.sf
    orc = OracleLib(str(REPO / "oracle" / "liboracle.so"))
    want = set(args.queries.split(","))

    print(f"generating SF{sf} inputs on host ...", file=sys.stderr)
    li = orc.gen_lineitem2(sf)
    lpk = orc.gen_lineitem_partkey(sf)
    orders = orc.gen_orders(sf)
    cust = orc.gen_customer2(sf)
    supp = orc.gen_supplier(sf)
    n_li = len(li["orderkey"])
    results = []

    def record(q, secs, ok, extra=None):
        r = {"query": q, "sf": sf, "ms": round(secs * 1e3, 3),
             "grows_per_s": round(n_li / secs / 1e9, 3), "exact": bool(ok),
             "n_lineitem": n_li}
        if extra:
            r.update(extra)
        results.append(r)
        print(json.dumps(r), flush=True)

    if "q8" in want:
        ptype = orc.gen_part_type(sf)
        pages = (page({"custkey": cust["custkey"],
                       "nationkey": cust["nationkey"]}),
                 page({k: orders[k] for k in ("orderkey", "custkey",
                                              "orderdate")}),
                 page({"suppkey": supp["suppkey"],
                       "nationkey": supp["nationkey"]}),
                 page({"partkey": np.arange(1, len(ptype) + 1,
                                            dtype=np.int64),
                       "type_id": ptype}),
                 page({"orderkey": li["orderkey"], "suppkey": li["suppkey"],
                       "partkey": lpk, "extendedprice": li["extendedprice"],
                       "discount": li["discount"]}))
        got, secs = run("q8", lambda: P.pipelines.q8(*pages), args.reps)
        exp = orc.q8(cust, orders, li, lpk, supp, ptype)
        record("q8", secs, tuple(got[0]) == tuple(exp[0]) and
               tuple(got[1]) == tuple(exp[1]))
        del pages, ptype

    if "q9" in want:
        ps = orc.gen_partsupp(sf)
        words = orc.gen_part_name_words(sf)
        names = [orc.color_name(i) for i in range(92)]
        strings = [" ".join(names[w] for w in row).encode() for row in words]
        n_part = len(strings)
        pages = (P.Page({"partkey": dev(np.arange(1, n_part + 1,
                                                  dtype=np.int64)),
                         "name": P.Varbin(strings)}),
                 page({"suppkey": supp["suppkey"],
                       "nationkey": supp["nationkey"]}),
                 page({"orderkey": orders["orderkey"],
                       "orderdate": orders["orderdate"]}),
                 page({"partkey": ps["partkey"], "suppkey": ps["suppkey"],
                       "supplycost": ps["supplycost_cents"] / 100.0}),
                 page({"partkey": lpk, "suppkey": li["suppkey"],
                       "orderkey": li["orderkey"],
                       "quantity": li["quantity"],
                       "extendedprice": li["extendedprice"],
                       "discount": li["discount"]}))
        got, secs = run("q9", lambda: P.pipelines.q9(*pages), args.reps)
        gid = orc.color_id("green")
        p_match = (words == gid).any(axis=1).astype(np.uint8)
        exp = orc.q9(li, lpk, orders, supp, ps, p_match)
        record("q9", secs, np.array_equal(np.array(got, np.int64), exp))
        del pages, ps, words, strings

    if "q12" in want:
        pri = orc.gen_orders_priority(sf)
        lid = orc.gen_lineitem_dates(sf)
        smode = orc.gen_lineitem_shipmode(sf)
        pages = (page({"orderkey": orders["orderkey"], "priority": pri}),
                 page({"orderkey": li["orderkey"], "shipmode": smode,
                       "shipdate": li["shipdate"],
                       "commitdate": lid["commitdate"],
                       "receiptdate": lid["receiptdate"]}))
        got, secs = run("q12", lambda: P.pipelines.q12(*pages), args.reps)
        hi, lo = orc.q12(orders, pri, li, lid, smode)
        record("q12", secs,
               got == {4: (hi[4], lo[4]), 6: (hi[6], lo[6])})
        del pages

    if "q14" in want:
        ptype = orc.gen_part_type(sf)
        pages = (page({"partkey": np.arange(1, len(ptype) + 1,
                                            dtype=np.int64),
                       "type_id": ptype}),
                 page({"partkey": lpk, "extendedprice": li["extendedprice"],
                       "discount": li["discount"],
                       "shipdate": li["shipdate"]}))
        got, secs = run("q14", lambda: P.pipelines.q14(*pages), args.reps)
        exp = orc.q14(li, lpk, ptype)
        record("q14", secs, tuple(got) == tuple(exp))
        del pages, ptype

    if "q17" in want:
        part2 = orc.gen_part2(sf)
        n_part = len(part2["brand"])
        pages = (page({"partkey": np.arange(1, n_part + 1, dtype=np.int64),
                       "brand": part2["brand"],
                       "container": part2["container"]}),
                 page({"partkey": lpk, "quantity": li["quantity"],
                       "extendedprice": li["extendedprice"]}))
        got, secs = run("q17", lambda: P.pipelines.q17(*pages), args.reps)
        exp = orc.q17(li, lpk, part2)
        record("q17", secs, got == exp)
        del pages, part2

    if "q18" in want:
        tp = orc.gen_orders_totalprice(sf)
        li1 = orc.gen_lineitem(sf)
        pages = (page({"orderkey": orders["orderkey"],
                       "custkey": orders["custkey"],
                       "orderdate": orders["orderdate"],
                       "totalprice": tp}),
                 page({"orderkey": li1["orderkey"],
                       "quantity": li1["quantity"]}))
        got, secs = run("q18", lambda: P.pipelines.q18(*pages), args.reps)
        exp = orc.q18(orders, tp, li1)
        record("q18", secs, got == exp)
        del pages, tp, li1

    if "q19" in want:
        part3 = orc.gen_part3(sf)
        smode = orc.gen_lineitem_shipmode(sf)
        sinst = orc.gen_lineitem_shipinstruct(sf)
        n_part = len(part3["brand"])
        pages = (page({"partkey": np.arange(1, n_part + 1, dtype=np.int64),
                       "brand": part3["brand"],
                       "container": part3["container"],
                       "size": part3["size"]}),
                 page({"partkey": lpk, "quantity": li["quantity"],
                       "extendedprice": li["extendedprice"],
                       "discount": li["discount"], "shipmode": smode,
                       "shipinstruct": sinst}))
        got, secs = run("q19", lambda: P.pipelines.q19(*pages), args.reps)
        exp = orc.q19(li, lpk, smode, sinst, part3)
        record("q19", secs, got == exp)
        del pages, part3

    if "q21" in want:
        lid = orc.gen_lineitem_dates(sf)
        pages = (page({"suppkey": supp["suppkey"],
                       "nationkey": supp["nationkey"]}),
                 page({"orderkey": orders["orderkey"]}),
                 page({"orderkey": li["orderkey"], "suppkey": li["suppkey"],
                       "linestatus": li["linestatus"],
                       "commitdate": lid["commitdate"],
                       "receiptdate": lid["receiptdate"]}))
        got, secs = run("q21", lambda: P.pipelines.q21(*pages), args.reps)
        exp = orc.q21(supp, li, lid)
        record("q21", secs, got == exp)
        del pages

    if "q22" in want:
        abal = orc.gen_customer_acctbal(sf)
        pages = (page({"custkey": cust["custkey"],
                       "nationkey": cust["nationkey"], "acctbal": abal}),
                 page({"custkey": orders["custkey"]}))
        got, secs = run("q22", lambda: P.pipelines.q22(*pages), args.reps)
        exp = orc.q22(cust, abal, orders)
        record("q22", secs, list(got[0]) == exp[0] and
               list(got[1]) == exp[1])
        del pages

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/extra_q.json", "w") as f:
        for r in results:
            f.write(json.dumps(r) + "\n")
    bad = [r["query"] for r in results if not r["exact"]]
    print(f"done: {len(results)} queries, inexact: {bad or 'none'}",
          file=sys.stderr)
    return 1 if bad else 0


if __name__ == "__main__":
    sys.exit(main())
