"""ctypes binding for oracle/liboracle.so — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg use this
module; the product package (presto_amd) never imports it.
"""
import ctypes as C

import numpy as np


class Q1Group(C.Structure):
    _fields_ = [
        ("returnflag", C.c_uint8), ("linestatus", C.c_uint8),
        ("count_order", C.c_int64), ("sum_qty_units", C.c_int64),
        ("sum_base_cents", C.c_int64), ("sum_disc_1e4", C.c_int64),
        ("sum_charge_1e6_hi", C.c_int64), ("sum_charge_1e6_lo", C.c_uint64),
        ("sum_disc_cents", C.c_int64),
        ("f64_sum_qty", C.c_double), ("f64_sum_base", C.c_double),
        ("f64_sum_disc_price", C.c_double), ("f64_sum_charge", C.c_double),
        ("f64_sum_disc", C.c_double),
    ]


class Q3Row(C.Structure):
    _fields_ = [
        ("orderkey", C.c_int64), ("revenue_1e4", C.c_int64),
        ("orderdate", C.c_int32), ("shippriority", C.c_int32),
        ("f64_revenue", C.c_double),
    ]


def _p(a):
    return a.ctypes.data_as(C.c_void_p)


class OracleLib:
    def __init__(self, path):
        self.lib = C.CDLL(path)
        L = self.lib
        L.tpch_customer_count.restype = C.c_int64
        L.tpch_customer_count.argtypes = [C.c_double]
        L.tpch_orders_count.restype = C.c_int64
        L.tpch_orders_count.argtypes = [C.c_double]
        L.tpch_lineitem_count.restype = C.c_int64
        L.tpch_lineitem_count.argtypes = [C.c_double]
        L.tpch_lineitem_offset.restype = C.c_int64
        L.tpch_lineitem_offset.argtypes = [C.c_double, C.c_int64]
        L.tpch_gen_lineitem.restype = C.c_int64
        L.oracle_q1.restype = C.c_int32
        L.oracle_q3.restype = C.c_int32
        L.oracle_murmur3_finalize.restype = C.c_uint64
        L.oracle_murmur3_finalize.argtypes = [C.c_uint64]
        L.oracle_bigint_hash.restype = C.c_uint64
        L.oracle_bigint_hash.argtypes = [C.c_int64]
        L.oracle_partition.restype = C.c_int32
        L.oracle_partition.argtypes = [C.c_uint64, C.c_int32]
        L.oracle_bigint_group_by.restype = C.c_int64
        L.oracle_join_bigint.restype = C.c_int64

    # ---- generator ----
    def lineitem_count(self, sf):
        return self.lib.tpch_lineitem_count(C.c_double(sf))

    def gen_lineitem(self, sf):
        n = self.lineitem_count(sf)
        n_ord = self.lib.tpch_orders_count(C.c_double(sf))
        cols = dict(
            orderkey=np.empty(n, np.int64), quantity=np.empty(n, np.float64),
            extendedprice=np.empty(n, np.float64),
            discount=np.empty(n, np.float64), tax=np.empty(n, np.float64),
            shipdate=np.empty(n, np.int32), returnflag=np.empty(n, np.uint8),
            linestatus=np.empty(n, np.uint8))
        w = self.lib.tpch_gen_lineitem(
            C.c_double(sf), C.c_int64(0), C.c_int64(n_ord),
            _p(cols["orderkey"]), _p(cols["quantity"]),
            _p(cols["extendedprice"]), _p(cols["discount"]), _p(cols["tax"]),
            _p(cols["shipdate"]), _p(cols["returnflag"]),
            _p(cols["linestatus"]))
        assert w == n
        return cols

    def gen_orders(self, sf):
        n = self.lib.tpch_orders_count(C.c_double(sf))
        cols = dict(orderkey=np.empty(n, np.int64),
                    custkey=np.empty(n, np.int64),
                    orderdate=np.empty(n, np.int32))
        self.lib.tpch_gen_orders(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                                 _p(cols["orderkey"]), _p(cols["custkey"]),
                                 _p(cols["orderdate"]), None)
        return cols

    def gen_customer(self, sf):
        n = self.lib.tpch_customer_count(C.c_double(sf))
        cols = dict(custkey=np.empty(n, np.int64),
                    mktseg=np.empty(n, np.uint8))
        self.lib.tpch_gen_customer(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                                   _p(cols["custkey"]), _p(cols["mktseg"]))
        return cols

    # ---- oracles ----
    def q1(self, li):
        groups = (Q1Group * 6)()
        ng = self.lib.oracle_q1(
            C.c_int64(len(li["quantity"])), _p(li["quantity"]),
            _p(li["extendedprice"]), _p(li["discount"]), _p(li["tax"]),
            _p(li["shipdate"]), _p(li["returnflag"]), _p(li["linestatus"]),
            groups)
        return [groups[i] for i in range(ng)]

    def q3(self, cust, orders, li, limit=10):
        rows = (Q3Row * limit)()
        nr = self.lib.oracle_q3(
            C.c_int64(len(cust["custkey"])), _p(cust["custkey"]),
            _p(cust["mktseg"]), C.c_int64(len(orders["orderkey"])),
            _p(orders["orderkey"]), _p(orders["custkey"]),
            _p(orders["orderdate"]), C.c_int64(len(li["orderkey"])),
            _p(li["orderkey"]), _p(li["extendedprice"]), _p(li["discount"]),
            _p(li["shipdate"]), C.c_int32(limit), rows)
        return [rows[i] for i in range(nr)]

    def group_by(self, keys):
        keys = np.ascontiguousarray(keys, np.int64)
        gids = np.empty(len(keys), np.int32)
        ng = self.lib.oracle_bigint_group_by(C.c_int64(len(keys)), _p(keys),
                                             _p(gids))
        return ng, gids

    def join(self, build_keys, probe_keys, cap=None):
        b = np.ascontiguousarray(build_keys, np.int64)
        p = np.ascontiguousarray(probe_keys, np.int64)
        if cap is None:
            cap = 4 * (len(b) + len(p)) + 16
        while True:
            op = np.empty(cap, np.int64)
            ob = np.empty(cap, np.int64)
            n = self.lib.oracle_join_bigint(C.c_int64(len(b)), _p(b),
                                            C.c_int64(len(p)), _p(p), _p(op),
                                            _p(ob), C.c_int64(cap))
            if n <= cap:
                return op[:n], ob[:n]
            cap = n  # function reports the true total; retry sized


class Q5Row(C.Structure):
    _fields_ = [("nationkey", C.c_uint8), ("name", C.c_char * 32),
                ("revenue_1e4", C.c_int64)]


def _bind_q5(self):
    self.lib.oracle_q5.restype = C.c_int32


def gen_lineitem2(self, sf):
    n = self.lineitem_count(sf)
    n_ord = self.lib.tpch_orders_count(C.c_double(sf))
    cols = dict(
        orderkey=np.empty(n, np.int64), quantity=np.empty(n, np.float64),
        extendedprice=np.empty(n, np.float64),
        discount=np.empty(n, np.float64), tax=np.empty(n, np.float64),
        shipdate=np.empty(n, np.int32), returnflag=np.empty(n, np.uint8),
        linestatus=np.empty(n, np.uint8), suppkey=np.empty(n, np.int64))
    w = self.lib.tpch_gen_lineitem2(
        C.c_double(sf), C.c_int64(0), C.c_int64(n_ord),
        _p(cols["orderkey"]), _p(cols["quantity"]),
        _p(cols["extendedprice"]), _p(cols["discount"]), _p(cols["tax"]),
        _p(cols["shipdate"]), _p(cols["returnflag"]), _p(cols["linestatus"]),
        _p(cols["suppkey"]))
    assert w == n
    return cols


def gen_customer2(self, sf):
    n = self.lib.tpch_customer_count(C.c_double(sf))
    cols = dict(custkey=np.empty(n, np.int64), mktseg=np.empty(n, np.uint8),
                nationkey=np.empty(n, np.uint8))
    self.lib.tpch_gen_customer2(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                                _p(cols["custkey"]), _p(cols["mktseg"]),
                                _p(cols["nationkey"]))
    return cols


def gen_supplier(self, sf):
    n = self.lib.tpch_supplier_count(C.c_double(sf))
    cols = dict(suppkey=np.empty(n, np.int64),
                nationkey=np.empty(n, np.uint8))
    self.lib.tpch_gen_supplier(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                               _p(cols["suppkey"]), _p(cols["nationkey"]))
    return cols


def q5(self, cust, orders, li, supp):
    self.lib.tpch_gen_lineitem2.restype = C.c_int64
    rows = (Q5Row * 25)()
    self.lib.oracle_q5.restype = C.c_int32
    nr = self.lib.oracle_q5(
        C.c_int64(len(cust["custkey"])), _p(cust["custkey"]),
        _p(cust["nationkey"]), C.c_int64(len(orders["orderkey"])),
        _p(orders["orderkey"]), _p(orders["custkey"]),
        _p(orders["orderdate"]), C.c_int64(len(li["orderkey"])),
        _p(li["orderkey"]), _p(li["suppkey"]), _p(li["extendedprice"]),
        _p(li["discount"]), C.c_int64(len(supp["suppkey"])),
        _p(supp["nationkey"]), rows)
    return [rows[i] for i in range(nr)]


OracleLib.gen_lineitem2 = gen_lineitem2
OracleLib.gen_customer2 = gen_customer2
OracleLib.gen_supplier = gen_supplier
OracleLib.q5 = q5


def q6(self, li):
    rev = C.c_int64()
    cnt = C.c_int64()
    self.lib.oracle_q6(C.c_int64(len(li["quantity"])), _p(li["quantity"]),
                       _p(li["extendedprice"]), _p(li["discount"]),
                       _p(li["shipdate"]), C.byref(rev), C.byref(cnt))
    return rev.value, cnt.value


OracleLib.q6 = q6


class Q7Row(C.Structure):
    _fields_ = [("supp_nation", C.c_uint8), ("cust_nation", C.c_uint8),
                ("year", C.c_int32), ("revenue_1e4", C.c_int64)]


def q7(self, cust, orders, li, supp):
    rows = (Q7Row * 4)()
    self.lib.oracle_q7.restype = C.c_int32
    nr = self.lib.oracle_q7(
        C.c_int64(len(cust["custkey"])), _p(cust["custkey"]),
        _p(cust["nationkey"]), C.c_int64(len(orders["orderkey"])),
        _p(orders["orderkey"]), _p(orders["custkey"]),
        C.c_int64(len(li["orderkey"])), _p(li["orderkey"]),
        _p(li["suppkey"]), _p(li["extendedprice"]), _p(li["discount"]),
        _p(li["shipdate"]), C.c_int64(len(supp["suppkey"])),
        _p(supp["nationkey"]), rows)
    return [rows[i] for i in range(nr)]


OracleLib.q7 = q7


def gen_orders_priority(self, sf):
    n = self.lib.tpch_orders_count(C.c_double(sf))
    pri = np.empty(n, np.uint8)
    self.lib.tpch_gen_orders_priority(C.c_double(sf), C.c_int64(0),
                                      C.c_int64(n), _p(pri))
    return pri


def gen_lineitem_dates(self, sf):
    n = self.lineitem_count(sf)
    n_ord = self.lib.tpch_orders_count(C.c_double(sf))
    ok = np.empty(n, np.int64)
    cd = np.empty(n, np.int32)
    rd = np.empty(n, np.int32)
    self.lib.tpch_gen_lineitem_dates.restype = C.c_int64
    w = self.lib.tpch_gen_lineitem_dates(C.c_double(sf), C.c_int64(0),
                                         C.c_int64(n_ord), _p(ok), _p(cd),
                                         _p(rd))
    assert w == n
    return dict(orderkey=ok, commitdate=cd, receiptdate=rd)


def q4(self, orders, pri, lid):
    counts = (C.c_int64 * 5)()
    self.lib.oracle_q4(
        C.c_int64(len(orders["orderkey"])), _p(orders["orderkey"]),
        _p(orders["orderdate"]), _p(pri),
        C.c_int64(len(lid["orderkey"])), _p(lid["orderkey"]),
        _p(lid["commitdate"]), _p(lid["receiptdate"]), counts)
    return list(counts)


OracleLib.gen_orders_priority = gen_orders_priority
OracleLib.gen_lineitem_dates = gen_lineitem_dates
OracleLib.q4 = q4


def gen_part_type(self, sf):
    n = int(200000 * sf)
    t = np.empty(n, np.uint8)
    self.lib.tpch_gen_part_type(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                                _p(t))
    return t


def gen_lineitem_partkey(self, sf):
    n = self.lineitem_count(sf)
    n_ord = self.lib.tpch_orders_count(C.c_double(sf))
    pk = np.empty(n, np.int64)
    self.lib.tpch_gen_lineitem_partkey.restype = C.c_int64
    w = self.lib.tpch_gen_lineitem_partkey(C.c_double(sf), C.c_int64(0),
                                           C.c_int64(n_ord), _p(pk))
    assert w == n
    return pk


def q8(self, cust, orders, li, lpk, supp, ptype):
    br = (C.c_int64 * 2)()
    tt = (C.c_int64 * 2)()
    self.lib.oracle_q8(
        C.c_int64(len(cust["custkey"])), _p(cust["custkey"]),
        _p(cust["nationkey"]), C.c_int64(len(orders["orderkey"])),
        _p(orders["orderkey"]), _p(orders["custkey"]),
        _p(orders["orderdate"]), C.c_int64(len(li["orderkey"])),
        _p(li["orderkey"]), _p(li["suppkey"]), _p(lpk),
        _p(li["extendedprice"]), _p(li["discount"]),
        C.c_int64(len(supp["suppkey"])), _p(supp["nationkey"]),
        C.c_int64(len(ptype)), _p(ptype), br, tt)
    return list(br), list(tt)


OracleLib.gen_part_type = gen_part_type
OracleLib.gen_lineitem_partkey = gen_lineitem_partkey
OracleLib.q8 = q8


def gen_lineitem_shipmode(self, sf):
    n = self.lineitem_count(sf)
    n_ord = self.lib.tpch_orders_count(C.c_double(sf))
    sm = np.empty(n, np.uint8)
    self.lib.tpch_gen_lineitem_shipmode.restype = C.c_int64
    w = self.lib.tpch_gen_lineitem_shipmode(C.c_double(sf), C.c_int64(0),
                                            C.c_int64(n_ord), _p(sm))
    assert w == n
    return sm


def gen_part2(self, sf):
    n = int(200000 * sf)
    mfgr = np.empty(n, np.uint8)
    brand = np.empty(n, np.uint8)
    cntr = np.empty(n, np.uint8)
    self.lib.tpch_gen_part2(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                            _p(mfgr), _p(brand), _p(cntr))
    return {"mfgr": mfgr, "brand": brand, "container": cntr}


def gen_partsupp(self, sf):
    n_part = int(200000 * sf)
    n = n_part * 4
    pk = np.empty(n, np.int64)
    sk = np.empty(n, np.int64)
    aq = np.empty(n, np.int32)
    cost = np.empty(n, np.int64)
    self.lib.tpch_gen_partsupp(C.c_double(sf), C.c_int64(0),
                               C.c_int64(n_part), _p(pk), _p(sk), _p(aq),
                               _p(cost))
    return {"partkey": pk, "suppkey": sk, "availqty": aq,
            "supplycost_cents": cost}


def q14(self, li, lpk, ptype):
    promo = C.c_int64()
    total = C.c_int64()
    self.lib.oracle_q14(C.c_int64(len(li["orderkey"])),
                        _p(li["extendedprice"]), _p(li["discount"]),
                        _p(li["shipdate"]), _p(lpk), C.c_int64(len(ptype)),
                        _p(ptype), C.byref(promo), C.byref(total))
    return promo.value, total.value


def q12(self, orders, pri, li, lid, smode):
    hi = (C.c_int64 * 7)()
    lo = (C.c_int64 * 7)()
    self.lib.oracle_q12(C.c_int64(len(orders["orderkey"])),
                        _p(orders["orderkey"]), _p(pri),
                        C.c_int64(len(li["orderkey"])), _p(li["orderkey"]),
                        _p(smode), _p(li["shipdate"]), _p(lid["commitdate"]),
                        _p(lid["receiptdate"]), hi, lo)
    return list(hi), list(lo)


def q17(self, li, lpk, part2):
    out = C.c_int64()
    self.lib.oracle_q17(C.c_int64(len(li["orderkey"])), _p(lpk),
                        _p(li["quantity"]), _p(li["extendedprice"]),
                        C.c_int64(len(part2["brand"])), _p(part2["brand"]),
                        _p(part2["container"]), C.byref(out))
    return out.value


def q11(self, ps, supp, n_part):
    out_pk = np.empty(n_part, np.int64)
    out_val = np.empty(n_part, np.int64)
    self.lib.oracle_q11.restype = C.c_int64
    n = self.lib.oracle_q11(C.c_int64(len(ps["partkey"])), _p(ps["partkey"]),
                            _p(ps["suppkey"]), _p(ps["availqty"]),
                            _p(ps["supplycost_cents"]),
                            C.c_int64(len(supp["suppkey"])),
                            _p(supp["nationkey"]), C.c_int64(n_part),
                            _p(out_pk), _p(out_val))
    return out_pk[:n], out_val[:n]


OracleLib.gen_lineitem_shipmode = gen_lineitem_shipmode
OracleLib.gen_part2 = gen_part2
OracleLib.gen_partsupp = gen_partsupp
OracleLib.q14 = q14
OracleLib.q12 = q12
OracleLib.q17 = q17
OracleLib.q11 = q11


def gen_orders_totalprice(self, sf):
    n = self.lib.tpch_orders_count(C.c_double(sf))
    tp = np.empty(n, np.int64)
    self.lib.tpch_gen_orders_totalprice(C.c_double(sf), C.c_int64(0),
                                        C.c_int64(n), _p(tp))
    return tp


def q18(self, orders, tp, li, limit=100):
    n_ord = len(orders["orderkey"])
    ck = np.empty(limit, np.int64)
    ok = np.empty(limit, np.int64)
    od = np.empty(limit, np.int32)
    otp = np.empty(limit, np.int64)
    qt = np.empty(limit, np.int64)
    self.lib.oracle_q18.restype = C.c_int64
    n = self.lib.oracle_q18(C.c_int64(n_ord), _p(orders["orderkey"]),
                            _p(orders["custkey"]), _p(orders["orderdate"]),
                            _p(tp), C.c_int64(len(li["orderkey"])),
                            _p(li["orderkey"]), _p(li["quantity"]),
                            C.c_int32(limit), _p(ck), _p(ok), _p(od),
                            _p(otp), _p(qt))
    return [(int(ck[i]), int(ok[i]), int(od[i]), int(otp[i]), int(qt[i]))
            for i in range(n)]


OracleLib.gen_orders_totalprice = gen_orders_totalprice
OracleLib.q18 = q18


def gen_customer_acctbal(self, sf):
    n = self.lib.tpch_customer_count(C.c_double(sf))
    ab = np.empty(n, np.int64)
    self.lib.tpch_gen_customer_acctbal(C.c_double(sf), C.c_int64(0),
                                       C.c_int64(n), _p(ab))
    return ab


def q21(self, supp, li, lid, limit=100):
    n_supp = len(supp["suppkey"])
    out_sk = np.empty(limit, np.int64)
    out_cnt = np.empty(limit, np.int64)
    self.lib.oracle_q21.restype = C.c_int64
    n = self.lib.oracle_q21(C.c_int64(n_supp), _p(supp["nationkey"]),
                            C.c_int64(len(li["orderkey"])),
                            _p(li["orderkey"]), _p(li["suppkey"]),
                            _p(li["linestatus"]), _p(lid["commitdate"]),
                            _p(lid["receiptdate"]), C.c_int32(limit),
                            _p(out_sk), _p(out_cnt))
    return [(int(out_sk[i]), int(out_cnt[i])) for i in range(n)]


Q22_CODE_NATIONS = np.array([3, 7, 8, 13, 19, 20, 21], np.uint8)  # codes
# '13','17','18','23','29','30','31' ascending


def q22(self, cust, abal, orders):
    cn = Q22_CODE_NATIONS
    out_cnt = np.empty(len(cn), np.int64)
    out_sum = np.empty(len(cn), np.int64)
    self.lib.oracle_q22(C.c_int64(len(cust["custkey"])), _p(cust["custkey"]),
                        _p(cust["nationkey"]), _p(abal),
                        C.c_int64(len(orders["orderkey"])),
                        _p(orders["custkey"]), C.c_int32(len(cn)), _p(cn),
                        _p(out_cnt), _p(out_sum))
    return out_cnt.tolist(), out_sum.tolist()


OracleLib.gen_customer_acctbal = gen_customer_acctbal
OracleLib.q21 = q21
OracleLib.q22 = q22


def gen_part3(self, sf):
    n = int(200000 * sf)
    mfgr = np.empty(n, np.uint8)
    brand = np.empty(n, np.uint8)
    cntr = np.empty(n, np.uint8)
    size = np.empty(n, np.uint8)
    self.lib.tpch_gen_part3(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                            _p(mfgr), _p(brand), _p(cntr), _p(size))
    return {"mfgr": mfgr, "brand": brand, "container": cntr, "size": size}


def gen_lineitem_shipinstruct(self, sf):
    n = self.lineitem_count(sf)
    n_ord = self.lib.tpch_orders_count(C.c_double(sf))
    si = np.empty(n, np.uint8)
    self.lib.tpch_gen_lineitem_shipinstruct.restype = C.c_int64
    w = self.lib.tpch_gen_lineitem_shipinstruct(C.c_double(sf), C.c_int64(0),
                                                C.c_int64(n_ord), _p(si))
    assert w == n
    return si


def q19(self, li, lpk, smode, sinst, part3):
    out = C.c_int64()
    self.lib.oracle_q19(C.c_int64(len(li["orderkey"])), _p(lpk),
                        _p(li["quantity"]), _p(li["extendedprice"]),
                        _p(li["discount"]), _p(smode), _p(sinst),
                        C.c_int64(len(part3["brand"])), _p(part3["brand"]),
                        _p(part3["container"]), _p(part3["size"]),
                        C.byref(out))
    return out.value


OracleLib.gen_part3 = gen_part3
OracleLib.gen_lineitem_shipinstruct = gen_lineitem_shipinstruct
OracleLib.q19 = q19


def gen_part_name_words(self, sf):
    n = int(200000 * sf)
    w = np.empty(n * 5, np.uint8)
    self.lib.tpch_gen_part_name_words(C.c_double(sf), C.c_int64(n), _p(w))
    return w.reshape(-1, 5)


def color_id(self, word):
    return int(self.lib.tpch_color_id(word.encode()))


def q9(self, li, lpk, orders, supp, ps, p_match):
    out = np.empty(25 * 7, np.int64)
    self.lib.oracle_q9(C.c_int64(len(li["orderkey"])), _p(lpk),
                       _p(li["suppkey"]), _p(li["quantity"]),
                       _p(li["extendedprice"]), _p(li["discount"]),
                       _p(li["orderkey"]), C.c_int64(len(orders["orderkey"])),
                       _p(orders["orderkey"]), _p(orders["orderdate"]),
                       C.c_int64(len(supp["suppkey"])), _p(supp["nationkey"]),
                       C.c_int64(len(p_match)), _p(p_match),
                       _p(ps["suppkey"]), _p(ps["supplycost_cents"]),
                       _p(out))
    return out.reshape(25, 7)


def nation_name(self, nk):
    buf = C.create_string_buffer(32)
    n = self.lib.tpch_nation_name(C.c_int32(nk), buf)
    return buf.value.decode()


OracleLib.gen_part_name_words = gen_part_name_words
OracleLib.color_id = color_id
OracleLib.q9 = q9
OracleLib.nation_name = nation_name


def color_name(self, cid):
    buf = C.create_string_buffer(16)
    self.lib.tpch_color_name(C.c_int32(cid), buf)
    return buf.value.decode()


OracleLib.color_name = color_name


def text_pool(self):
    self.lib.tpch_text_pool.restype = C.c_void_p
    self.lib.tpch_text_pool_size.restype = C.c_int64
    p = self.lib.tpch_text_pool()
    n = self.lib.tpch_text_pool_size()
    return C.cast(p, C.POINTER(C.c_char * n)).contents.raw


def gen_orders_comment(self, sf):
    n = self.lib.tpch_orders_count(C.c_double(sf))
    off = np.empty(n, np.int64)
    ln = np.empty(n, np.int32)
    self.lib.tpch_gen_orders_comment(C.c_double(sf), C.c_int64(0),
                                     C.c_int64(n), _p(off), _p(ln))
    return off, ln


def gen_supplier_bbb(self, sf):
    n = self.lib.tpch_supplier_count(C.c_double(sf))
    b = np.empty(n, np.uint8)
    self.lib.tpch_gen_supplier_bbb(C.c_double(sf), C.c_int64(0),
                                   C.c_int64(n), _p(b))
    return b


def q13(self, sf, orders, cap=128):
    n_cust = self.lib.tpch_customer_count(C.c_double(sf))
    off, ln = self.gen_orders_comment(sf)
    self.lib.tpch_text_pool.restype = C.c_void_p
    pool = self.lib.tpch_text_pool()
    out_c = np.empty(cap, np.int64)
    out_d = np.empty(cap, np.int64)
    self.lib.oracle_q13.restype = C.c_int64
    n = self.lib.oracle_q13(C.c_int64(n_cust),
                            C.c_int64(len(orders["custkey"])),
                            _p(orders["custkey"]), _p(off), _p(ln),
                            C.c_void_p(pool), _p(out_c), _p(out_d),
                            C.c_int64(cap))
    return [(int(out_c[i]), int(out_d[i])) for i in range(n)]


def part_type_name(self, tid):
    buf = C.create_string_buffer(32)
    self.lib.tpch_part_type_name(C.c_int32(tid), buf)
    return buf.value.decode()


def q16(self, part3, ptype, ps, bbb, cap=32768):
    n_part = len(part3["brand"])
    ob = np.empty(cap, np.uint8)
    ot = np.empty(cap, np.uint8)
    oz = np.empty(cap, np.uint8)
    oc = np.empty(cap, np.int32)
    self.lib.oracle_q16.restype = C.c_int64
    n = self.lib.oracle_q16(C.c_int64(n_part), _p(part3["brand"]),
                            _p(ptype), _p(part3["size"]),
                            C.c_int64(len(ps["partkey"])), _p(ps["partkey"]),
                            _p(ps["suppkey"]), C.c_int64(len(bbb)), _p(bbb),
                            _p(ob), _p(ot), _p(oz), _p(oc), C.c_int64(cap))
    return [(int(ob[i]), int(ot[i]), int(oz[i]), int(oc[i]))
            for i in range(n)]


OracleLib.text_pool = text_pool
OracleLib.gen_orders_comment = gen_orders_comment
OracleLib.gen_supplier_bbb = gen_supplier_bbb
OracleLib.q13 = q13
OracleLib.part_type_name = part_type_name
OracleLib.q16 = q16


def gen_orders_comment_varbin(self, sf):
    """Materialized o_comment bytes as (data u8, offsets i32) arrays —
    builds the page column a GPU scan would receive."""
    off, ln = self.gen_orders_comment(sf)
    pool = np.frombuffer(self.text_pool(), np.uint8)
    n = len(off)
    outoffs = np.zeros(n + 1, np.int32)
    np.cumsum(ln, out=outoffs[1:])
    total = int(outoffs[-1])
    starts = np.repeat(off - (outoffs[:-1].astype(np.int64)), ln)
    idx = starts + np.arange(total)
    data = pool[idx]
    return data, outoffs


OracleLib.gen_orders_comment_varbin = gen_orders_comment_varbin


def gen_supplier_acctbal(self, sf):
    n = self.lib.tpch_supplier_count(C.c_double(sf))
    ab = np.empty(n, np.int64)
    self.lib.tpch_gen_supplier_acctbal(C.c_double(sf), C.c_int64(0),
                                       C.c_int64(n), _p(ab))
    return ab


def _phones(self, fn, n):
    a = np.empty(n, np.int32); b = np.empty(n, np.int32)
    c = np.empty(n, np.int32)
    fn(C.c_double(0), C.c_int64(0), C.c_int64(n), _p(a), _p(b), _p(c))
    return a, b, c


def supplier_phone(self, sf, nat):
    n = self.lib.tpch_supplier_count(C.c_double(sf))
    a, b, c = _phones(self, self.lib.tpch_gen_supplier_phone, n)
    return [f"{int(nat[i])+10}-{a[i]}-{b[i]}-{c[i]}" for i in range(n)]


def customer_phone(self, sf, nat):
    n = self.lib.tpch_customer_count(C.c_double(sf))
    a, b, c = _phones(self, self.lib.tpch_gen_customer_phone, n)
    return [f"{int(nat[i])+10}-{a[i]}-{b[i]}-{c[i]}" for i in range(n)]


def gen_customer_comment(self, sf):
    n = self.lib.tpch_customer_count(C.c_double(sf))
    off = np.empty(n, np.int64); ln = np.empty(n, np.int32)
    self.lib.tpch_gen_customer_comment(C.c_double(sf), C.c_int64(0),
                                       C.c_int64(n), _p(off), _p(ln))
    return off, ln


def gen_supplier_comment(self, sf):
    n = self.lib.tpch_supplier_count(C.c_double(sf))
    off = np.empty(n, np.int64); ln = np.empty(n, np.int32)
    self.lib.tpch_gen_supplier_comment(C.c_double(sf), C.c_int64(0),
                                       C.c_int64(n), _p(off), _p(ln))
    return off, ln


def q10(self, orders, li, n_cust, limit=20):
    ck = np.empty(limit, np.int64); rv = np.empty(limit, np.int64)
    self.lib.oracle_q10.restype = C.c_int64
    n = self.lib.oracle_q10(C.c_int64(len(orders["orderkey"])),
                            _p(orders["orderkey"]), _p(orders["custkey"]),
                            _p(orders["orderdate"]),
                            C.c_int64(len(li["orderkey"])),
                            _p(li["orderkey"]), _p(li["returnflag"]),
                            _p(li["extendedprice"]), _p(li["discount"]),
                            C.c_int64(n_cust), C.c_int32(limit), _p(ck),
                            _p(rv))
    return [(int(ck[i]), int(rv[i])) for i in range(n)]


def q15(self, li, n_supp, cap=16):
    sk = np.empty(cap, np.int64); rv = np.empty(cap, np.int64)
    self.lib.oracle_q15.restype = C.c_int64
    n = self.lib.oracle_q15(C.c_int64(len(li["orderkey"])),
                            _p(li["suppkey"]), _p(li["extendedprice"]),
                            _p(li["discount"]), _p(li["shipdate"]),
                            C.c_int64(n_supp), _p(sk), _p(rv),
                            C.c_int64(cap))
    return [(int(sk[i]), int(rv[i])) for i in range(n)]


def q20(self, words, ps, li, lpk, supp, cap=None):
    if cap is None:
        cap = len(supp["suppkey"]) + 1
    sk = np.empty(cap, np.int64)
    fid = self.color_id("forest")
    self.lib.oracle_q20.restype = C.c_int64
    n = self.lib.oracle_q20(C.c_int64(len(words)), _p(words.reshape(-1)),
                            C.c_int32(fid), C.c_int64(len(ps["partkey"])),
                            _p(ps["partkey"]), _p(ps["suppkey"]),
                            _p(ps["availqty"]),
                            C.c_int64(len(li["orderkey"])), _p(lpk),
                            _p(li["suppkey"]), _p(li["quantity"]),
                            _p(li["shipdate"]),
                            C.c_int64(len(supp["suppkey"])),
                            _p(supp["nationkey"]), _p(sk), C.c_int64(cap))
    return [int(sk[i]) for i in range(n)]


def q2(self, part3, ptype, ps, supp, s_abal, limit=100):
    sk = np.empty(limit, np.int64); pk = np.empty(limit, np.int64)
    self.lib.oracle_q2.restype = C.c_int64
    n = self.lib.oracle_q2(C.c_int64(len(ptype)), _p(ptype),
                           _p(part3["size"]),
                           C.c_int64(len(ps["partkey"])), _p(ps["partkey"]),
                           _p(ps["suppkey"]), _p(ps["supplycost_cents"]),
                           C.c_int64(len(supp["suppkey"])),
                           _p(supp["nationkey"]), _p(s_abal),
                           C.c_int32(limit), _p(sk), _p(pk))
    return [(int(sk[i]), int(pk[i])) for i in range(n)]


OracleLib.gen_supplier_acctbal = gen_supplier_acctbal
OracleLib.supplier_phone = supplier_phone
OracleLib.customer_phone = customer_phone
OracleLib.gen_customer_comment = gen_customer_comment
OracleLib.gen_supplier_comment = gen_supplier_comment
OracleLib.q10 = q10
OracleLib.q15 = q15
OracleLib.q20 = q20
OracleLib.q2 = q2
