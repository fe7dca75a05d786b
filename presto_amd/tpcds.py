"""TPC-DS config-5 pipelines (BASELINE.json configs[4]): Q17 and Q72 —
decimal arithmetic + deep multi-join — composed from the same GPU
operator set as TPC-H (filter/semijoin, chained + agg hash builds,
emit-mode joins, the general multi-channel GroupBy).

Data comes from oracle/tpcds.c's dsdgen restatement; parity is pinned
with the oracle as the single data source (see oracle/tpcds.h: the
reference vendors neither the Teradata generator source nor any TPC-DS
golden vectors, so raw dsdgen parity is unpinned in-repo — stated
openly, as SURVEY.md §8c prescribes).  The oracle restates the
query17.tpl / query72.tpl semantics; GPU results must match it
integer-exactly.

Reference anchors for the operator shapes: HashBuilderOperator.java:55,
LookupJoinOperator.java:481-604, MultiChannelGroupByHash.java:300-380,
PageProcessor.java:299-343 (semijoin pushdown = dynamic-filter analog).
"""
import ctypes as C

import numpy as np

from .engine import (
    Operator, Page, PlanFilterProject, PlanHashBuild, PlanLookupJoin,
    PlanGroupBy, Pred, Proj, Agg,
    OP_FILTER_PROJECT, OP_HASH_BUILD, OP_LOOKUP_JOIN, OP_GROUPBY_MULTI,
    CMP_LT, CMP_GT, CMP_GE, CMP_LE, CMP_EQ, CMP_NE,
    PROJ_IDENT, PROJ_MUL, PROJ_KEYSHL, PROJ_SUBDIV, PROJ_KEYSHL_DIV,
    AGG_COUNT, AGG_SUM_I64, lib,
)

DATE_COUNT = 73049


class DsGen:
    """ctypes view of the tpcds.c generator + oracle."""

    def __init__(self, lib_path):
        L = C.CDLL(str(lib_path))
        for f in ("dsgen_store_sales_count", "dsgen_store_returns_count",
                  "dsgen_catalog_sales_count", "dsgen_catalog_returns_count",
                  "dsgen_inventory_count", "dsgen_item_count",
                  "dsgen_store_count", "dsgen_warehouse_count",
                  "dsgen_customer_count", "dsgen_promotion_count"):
            getattr(L, f).restype = C.c_int64
            getattr(L, f).argtypes = [C.c_double]
        L.oracle_ds_q17.restype = C.c_int64
        L.oracle_ds_q72.restype = C.c_int64
        self.L = L

    def _p(self, a):
        return C.c_void_p(0 if a is None else a.ctypes.data)

    def date_dim(self):
        year = np.zeros(DATE_COUNT, np.int32)
        qname = np.zeros(DATE_COUNT, np.int32)
        week = np.zeros(DATE_COUNT, np.int32)
        self.L.dsgen_date_dim(self._p(year), self._p(qname), self._p(week))
        return year, qname, week

    def store_sales(self, sf):
        n = self.L.dsgen_store_sales_count(C.c_double(sf))
        d = np.zeros(n, np.int32)
        it = np.zeros(n, np.int64)
        cu = np.zeros(n, np.int64)
        st = np.zeros(n, np.int64)
        tk = np.zeros(n, np.int64)
        q = np.zeros(n, np.int32)
        self.L.dsgen_store_sales(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                                 self._p(d), self._p(it), self._p(cu),
                                 self._p(st), self._p(tk), self._p(q))
        return dict(date=d, item=it, cust=cu, store=st, ticket=tk, qty=q)

    def store_returns(self, sf):
        n = self.L.dsgen_store_returns_count(C.c_double(sf))
        d = np.zeros(n, np.int32)
        it = np.zeros(n, np.int64)
        cu = np.zeros(n, np.int64)
        tk = np.zeros(n, np.int64)
        q = np.zeros(n, np.int32)
        self.L.dsgen_store_returns(C.c_double(sf), C.c_int64(0),
                                   C.c_int64(n), self._p(d), self._p(it),
                                   self._p(cu), self._p(tk), self._p(q))
        return dict(date=d, item=it, cust=cu, ticket=tk, qty=q)

    def catalog_sales(self, sf, want_all=False):
        n = self.L.dsgen_catalog_sales_count(C.c_double(sf))
        sold = np.zeros(n, np.int32)
        it = np.zeros(n, np.int64)
        cu = np.zeros(n, np.int64)
        q = np.zeros(n, np.int32)
        ship = np.zeros(n, np.int32) if want_all else None
        on = np.zeros(n, np.int64) if want_all else None
        cd = np.zeros(n, np.int64) if want_all else None
        hd = np.zeros(n, np.int64) if want_all else None
        pr = np.zeros(n, np.int64) if want_all else None
        self.L.dsgen_catalog_sales(
            C.c_double(sf), C.c_int64(0), C.c_int64(n), self._p(sold),
            self._p(ship), self._p(it), self._p(cu), self._p(on),
            self._p(q), self._p(cd), self._p(hd), self._p(pr))
        out = dict(sold=sold, item=it, cust=cu, qty=q)
        if want_all:
            out.update(ship=ship, order=on, cdemo=cd, hdemo=hd, promo=pr)
        return out

    def catalog_returns(self, sf):
        n = self.L.dsgen_catalog_returns_count(C.c_double(sf))
        it = np.zeros(n, np.int64)
        on = np.zeros(n, np.int64)
        self.L.dsgen_catalog_returns(C.c_double(sf), C.c_int64(0),
                                     C.c_int64(n), self._p(it), self._p(on))
        return dict(item=it, order=on)

    def inventory(self, sf):
        n = self.L.dsgen_inventory_count(C.c_double(sf))
        d = np.zeros(n, np.int32)
        it = np.zeros(n, np.int64)
        wh = np.zeros(n, np.int64)
        q = np.zeros(n, np.int32)
        self.L.dsgen_inventory(C.c_double(sf), C.c_int64(0), C.c_int64(n),
                               self._p(d), self._p(it), self._p(wh),
                               self._p(q))
        return dict(date=d, item=it, wh=wh, qoh=q)

    def cdemo_marital(self):
        n = 1920800
        m = np.zeros(n, np.uint8)
        self.L.dsgen_cdemo(self._p(m))
        return m

    def hdemo_buypot(self):
        n = 7200
        b = np.zeros(n, np.uint8)
        self.L.dsgen_hdemo(self._p(b))
        return b

    def store_state(self, sf):
        n = self.L.dsgen_store_count(C.c_double(sf))
        s = np.zeros(n, np.uint8)
        self.L.dsgen_store(C.c_double(sf), self._p(s))
        return s

    def q17(self, sf, q0):
        N = 1 << 20
        gi = np.zeros(N, np.int64)
        gs = np.zeros(N, np.int32)
        a = [np.zeros(N, np.int64) for _ in range(9)]
        n = self.L.oracle_ds_q17(
            C.c_double(sf), C.c_int32(q0), C.c_int64(N), self._p(gi),
            self._p(gs), *(self._p(x) for x in a))
        return [(int(gi[i]), int(gs[i])) + tuple(int(x[i]) for x in a)
                for i in range(n)]

    def q72(self, sf, year, marital, buypot):
        N = 1 << 23
        gi = np.zeros(N, np.int64)
        gw = np.zeros(N, np.int64)
        gk = np.zeros(N, np.int32)
        b = [np.zeros(N, np.int64) for _ in range(3)]
        n = self.L.oracle_ds_q72(
            C.c_double(sf), C.c_int32(year), C.c_int32(marital),
            C.c_int32(buypot), C.c_int64(N), self._p(gi), self._p(gw),
            self._p(gk), *(self._p(x) for x in b))
        return [(int(gi[i]), int(gw[i]), int(gk[i])) +
                tuple(int(x[i]) for x in b) for i in range(n)]


def _filter(page, preds=(), projs=(), semi=None, semi_col=0, raw=False):
    fp = PlanFilterProject()
    fp.n_preds = len(preds)
    for i, p in enumerate(preds):
        fp.preds[i] = p
    fp.n_proj = len(projs)
    for i, p in enumerate(projs):
        fp.proj[i] = p
    if semi is not None:
        fp.semijoin_table = semi
        fp.semijoin_col = semi_col
    else:
        fp.semijoin_table = 0
    f = Operator(OP_FILTER_PROJECT, fp)
    if raw:
        f.add_input_raw(page)
    else:
        f.add_input(page)
    return f, f.get_output_raw()


def _date_set(date_page, lo, hi):
    """Dense key set of date_sks whose qname/year column c1 is in
    [lo, hi] — the dynamic-filter analog of the date_dim dimension
    joins."""
    f, out = _filter(date_page,
                     preds=(Pred(1, CMP_GE, lo, 0.0),
                            Pred(1, CMP_LE, hi, 0.0)),
                     projs=(Proj(PROJ_IDENT, 0, 0, 0),))
    bp = PlanHashBuild()
    bp.key_col = 0
    bp.semijoin_table = -1
    bp.capacity_hint = DATE_COUNT + 1
    bp.key_set_only = 1
    bp.dense_array = 1
    b = Operator(OP_HASH_BUILD, bp)
    b.add_input_raw(out)
    b.finish()
    f.destroy()
    return b


def _chain_build(page_raw, key_col, payload_cols, hint):
    bp = PlanHashBuild()
    bp.key_col = key_col
    bp.semijoin_table = -1
    bp.n_payload = len(payload_cols)
    for i, c in enumerate(payload_cols):
        bp.payload_col[i] = c
    bp.capacity_hint = max(hint, 16)
    b = Operator(OP_HASH_BUILD, bp)
    b.add_input_raw(page_raw)
    b.finish()
    return b


def _emit_join(page_raw, table, key_col, emit_cols):
    jp = PlanLookupJoin()
    jp.table = table
    jp.key_col = key_col
    jp.mode = 0
    jp.n_emit = len(emit_cols)
    for i, c in enumerate(emit_cols):
        jp.emit_probe_cols[i] = c
    j = Operator(OP_LOOKUP_JOIN, jp)
    j.add_input_raw(page_raw)
    return j, j.get_output_raw()


def ds_q17(gen, sf, ss_page, sr_page, cs_page, date_page, q0):
    """TPC-DS Q17 (query17.tpl): store sale in quarter q0, its return in
    q0..q0+2 (joined on customer, item, ticket), a catalog re-purchase
    in q0..q0+2 (joined on customer, item); grouped by (item, store):
    count/sum/sum-of-squares of the three quantities, integer-exact.
    Returns host rows [(item_id, state, c,s,q x3)] sorted, after the
    display-side item_id/state fold."""
    ops, tables = [], []

    set_q0 = _date_set(date_page, q0, q0)
    set_q02 = _date_set(date_page, q0, q0 + 2)
    tables += [set_q0, set_q02]

    # store_sales in q0 -> chained table keyed by ticket
    f_ss = _filter(ss_page,
                   projs=(Proj(PROJ_IDENT, 4, 0, 0),   # ticket
                          Proj(PROJ_IDENT, 2, 0, 0),   # cust
                          Proj(PROJ_IDENT, 1, 0, 0),   # item
                          Proj(PROJ_IDENT, 5, 0, 0),   # qty
                          Proj(PROJ_IDENT, 3, 0, 0)),  # store
                   semi=set_q0.table(), semi_col=0)
    ops.append(f_ss[0])
    b1 = _chain_build(f_ss[1], 0, (1, 2, 3, 4), f_ss[1].n_rows)
    tables.append(b1)

    # store_returns in q0..q0+2 probe on ticket, then (cust,item) equality
    f_sr = _filter(sr_page,
                   projs=(Proj(PROJ_IDENT, 3, 0, 0),   # ticket
                          Proj(PROJ_IDENT, 2, 0, 0),   # cust
                          Proj(PROJ_IDENT, 1, 0, 0),   # item
                          Proj(PROJ_IDENT, 4, 0, 0)),  # rqty
                   semi=set_q02.table(), semi_col=0)
    ops.append(f_sr[0])
    j1, p1 = _emit_join(f_sr[1], b1.table(), 0, (1, 2, 3))
    ops.append(j1)
    # p1: [r_cust, r_item, rqty, s_cust, s_item, s_qty, s_store]
    eq1 = Pred(0, CMP_EQ, 0, 0.0)
    eq1.rhs_col = 3 + 1
    eq2 = Pred(1, CMP_EQ, 0, 0.0)
    eq2.rhs_col = 4 + 1
    f_eq = _filter(p1, preds=(eq1, eq2),
                   projs=(Proj(PROJ_KEYSHL, 0, 1, 20),  # (cust<<20)|item
                          Proj(PROJ_IDENT, 5, 0, 0),    # ssqty
                          Proj(PROJ_IDENT, 2, 0, 0),    # rqty
                          Proj(PROJ_IDENT, 6, 0, 0),    # store
                          Proj(PROJ_IDENT, 1, 0, 0)),   # item
                   raw=True)
    ops.append(f_eq[0])
    b2 = _chain_build(f_eq[1], 0, (1, 2, 3, 4), f_eq[1].n_rows)
    tables.append(b2)

    # catalog re-purchases in q0..q0+2 probe on (cust, item)
    f_cs = _filter(cs_page,
                   projs=(Proj(PROJ_KEYSHL, 2, 1, 20),
                          Proj(PROJ_IDENT, 3, 0, 0)),   # csqty
                   semi=set_q02.table(), semi_col=0)
    ops.append(f_cs[0])
    j2, p2 = _emit_join(f_cs[1], b2.table(), 0, (1,))
    ops.append(j2)
    # p2: [csqty, ssqty, rqty, store, item]

    g = PlanGroupBy()
    g.n_keys = 2
    g.key_col[0] = 4  # item
    g.key_col[1] = 3  # store
    g.capacity_hint = max(p2.n_rows * 2, 4096)
    g.n_aggs = 6
    specs = ((1, False), (1, True), (2, False), (2, True), (0, False),
             (0, True))
    for i, (ch, sq) in enumerate(specs):
        g.aggs[i] = Agg(AGG_SUM_I64,
                        Proj(PROJ_MUL, ch, ch, 0) if sq
                        else Proj(PROJ_IDENT, ch, 0, 0), 0)
        g.agg_filter[i] = -1
    gop = Operator(OP_GROUPBY_MULTI, g)
    gop.add_input_raw(p2)
    gop.finish()
    out = gop.get_output(["item", "store", "s_ss", "q_ss", "s_sr", "q_sr",
                          "s_cs", "q_cs", "cnt"])
    gop.destroy()
    for o in ops:
        o.destroy()
    for t in tables:
        lib().c.pg_table_destroy(t.table())
        t.destroy()

    # display-side fold: item_sk -> i_item_id (pairs share an id),
    # store_sk -> s_state; equal (id, state) groups merge
    state = gen.store_state(sf)
    agg = {}
    for i in range(len(out["item"])):
        key = ((int(out["item"][i]) + 1) // 2,
               int(state[int(out["store"][i]) - 1]))
        cur = agg.setdefault(key, [0] * 9)
        cnt = int(out["cnt"][i])
        vals = (cnt, int(out["s_ss"][i]), int(out["q_ss"][i]),
                cnt, int(out["s_sr"][i]), int(out["q_sr"][i]),
                cnt, int(out["s_cs"][i]), int(out["q_cs"][i]))
        for j in range(9):
            cur[j] += vals[j]
    return [k + tuple(v) for k, v in sorted(agg.items())]


def ds_q72(gen, sf, cs_page, inv_pages, cr_page, date_page, cdemo_page,
           hdemo_page, year, marital, buypot):
    """TPC-DS Q72 (query72.tpl): promotional-item inventory shortfalls —
    catalog sales in `year` with the demographics filters and
    ship > sold + 5, joined to the weekly inventory snapshot
    (same item, sold week, any warehouse) where on-hand < ordered;
    LEFT JOINs to promotion (promo/no_promo split) and catalog_returns
    (row multiplicity).  Grouped by (item, warehouse, week_seq).
    Returns host rows [(item_id, wh, week, no_promo, promo, total)]."""
    ops, tables = [], []

    set_year = _date_set(date_page, year, year)
    tables.append(set_year)

    # demographics dynamic-filter sets (dense sks)
    def demo_set(page, code, cap):
        f, out = _filter(page, preds=(Pred(1, CMP_EQ, code, 0.0),),
                         projs=(Proj(PROJ_IDENT, 0, 0, 0),))
        bp = PlanHashBuild()
        bp.key_col = 0
        bp.semijoin_table = -1
        bp.capacity_hint = cap + 1
        bp.key_set_only = 1
        bp.dense_array = 1
        b = Operator(OP_HASH_BUILD, bp)
        b.add_input_raw(out)
        b.finish()
        f.destroy()
        return b

    cdset = demo_set(cdemo_page, marital, 1920800)
    hdset = demo_set(hdemo_page, buypot, 7200)
    tables += [cdset, hdset]

    # cs: year + ship > sold + 5 + demographics, then (item, week) key
    ship_pred = Pred(1, CMP_GT, 5, 0.0)  # ship > sold + 5
    ship_pred.rhs_col = 0 + 1
    f1 = _filter(cs_page, preds=(ship_pred,),
                 projs=tuple(Proj(PROJ_IDENT, c, 0, 0)
                             for c in (0, 2, 3, 4, 5, 6, 7)),
                 semi=set_year.table(), semi_col=0)
    ops.append(f1[0])
    # f1: [sold, item, cust?, ...] -> cs_page cols are
    # [sold, ship, item, order, qty, cdemo, hdemo, promo]
    # emitted: [sold, item, order, qty, cdemo, hdemo, promo]
    f2 = _filter(f1[1], projs=tuple(Proj(PROJ_IDENT, c, 0, 0)
                                    for c in (0, 1, 2, 3, 5, 6)),
                 semi=cdset.table(), semi_col=4, raw=True)
    ops.append(f2[0])
    # [sold, item, order, qty, hdemo, promo]
    f3 = _filter(f2[1], projs=(Proj(PROJ_SUBDIV, 0, 0, 7),  # week
                               Proj(PROJ_IDENT, 1, 0, 0),
                               Proj(PROJ_IDENT, 2, 0, 0),
                               Proj(PROJ_IDENT, 3, 0, 0),
                               Proj(PROJ_IDENT, 5, 0, 0)),
                 semi=hdset.table(), semi_col=4, raw=True)
    ops.append(f3[0])
    # [week, item, order, qty, promo]
    f4 = _filter(f3[1], projs=(Proj(PROJ_KEYSHL, 1, 0, 14),  # item<<14|wk
                               Proj(PROJ_IDENT, 3, 0, 0),    # qty
                               Proj(PROJ_IDENT, 4, 0, 0),    # promo
                               Proj(PROJ_KEYSHL, 1, 2, 25)),  # crkey
                 raw=True)
    ops.append(f4[0])
    b3 = _chain_build(f4[1], 0, (1, 2, 3), f4[1].n_rows)
    tables.append(b3)

    # inventory probes the (item, week) table; on-hand < ordered.
    # ONE pass: the (item<<14 | date/7) key and the week channel are both
    # derived in the same projection set (KEYSHL_DIV / SUBDIV)
    joined_parts = []
    for inv in inv_pages:
        fi2 = _filter(inv,
                      projs=(Proj(PROJ_KEYSHL_DIV, 1, 0, (14 << 16) | 7),
                             Proj(PROJ_IDENT, 1, 0, 0),
                             Proj(PROJ_SUBDIV, 0, 0, 7),
                             Proj(PROJ_IDENT, 2, 0, 0),
                             Proj(PROJ_IDENT, 3, 0, 0)))
        jj, pj = _emit_join(fi2[1], b3.table(), 0, (1, 2, 3, 4))
        fi2[0].destroy()
        # pj: [item, wk, wh, qoh, qty, promo, crkey]
        qlt = Pred(3, CMP_LT, 0, 0.0)
        qlt.rhs_col = 4 + 1
        fq = _filter(pj, preds=(qlt,),
                     projs=tuple(Proj(PROJ_IDENT, c, 0, 0)
                                 for c in (0, 1, 2, 5, 6)), raw=True)
        jj.destroy()
        joined_parts.append(fq)
        # fq out: [item, wk, wh, promo, crkey]

    # base counts per (item, wk, wh) with the promo split
    g1 = PlanGroupBy()
    g1.n_keys = 3
    for i in range(3):
        g1.key_col[i] = i
    g1.capacity_hint = max(sum(f[1].n_rows for f in joined_parts) * 2,
                           4096)
    g1.n_preds = 0
    g1.preds[0] = Pred(3, CMP_EQ, 0, 0.0)
    g1.preds[1] = Pred(3, CMP_NE, 0, 0.0)
    g1.n_aggs = 2
    g1.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    g1.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    g1.agg_filter[0] = 0
    g1.agg_filter[1] = 1
    gop1 = Operator(OP_GROUPBY_MULTI, g1)
    for f in joined_parts:
        gop1.add_input_raw(f[1])
    gop1.finish()
    base = gop1.get_output(["item", "wk", "wh", "no_promo", "promo",
                            "cnt"])
    gop1.destroy()

    # catalog_returns multiplicity: LEFT JOIN row expansion =
    # base + (matches - 1) for matched rows
    f_cr = _filter(cr_page, projs=(Proj(PROJ_KEYSHL, 0, 1, 25),))
    ops.append(f_cr[0])
    gcr = PlanGroupBy()
    gcr.n_keys = 1
    gcr.key_col[0] = 0
    gcr.capacity_hint = max(f_cr[1].n_rows, 1024)
    gcr.n_aggs = 1
    gcr.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    gcr.agg_filter[0] = -1
    gopc = Operator(OP_GROUPBY_MULTI, gcr)
    gopc.add_input_raw(f_cr[1])
    gopc.finish()
    crg = gopc.get_output_raw()  # [crkey, cnt, cnt]

    bcr = PlanHashBuild()
    bcr.key_col = 0
    bcr.semijoin_table = -1
    bcr.n_payload = 1
    bcr.payload_col[0] = 1
    bcr.capacity_hint = max(crg.n_rows, 16)
    bcr.agg_table = 1
    ocr = Operator(OP_HASH_BUILD, bcr)
    ocr.add_input_raw(crg)
    ocr.finish()
    gopc.destroy()
    tables.append(ocr)

    g2 = PlanGroupBy()
    g2.n_keys = 3
    for i in range(3):
        g2.key_col[i] = i
    g2.capacity_hint = max(len(base["item"]) * 2, 4096)
    g2.n_preds = 0
    g2.preds[0] = Pred(3, CMP_EQ, 0, 0.0)
    g2.preds[1] = Pred(3, CMP_NE, 0, 0.0)
    g2.n_aggs = 3
    g2.aggs[0] = Agg(AGG_SUM_I64, Proj(PROJ_SUBDIV, 4, 1, 1), 0)
    g2.aggs[1] = Agg(AGG_SUM_I64, Proj(PROJ_SUBDIV, 4, 1, 1), 0)
    g2.aggs[2] = Agg(AGG_SUM_I64, Proj(PROJ_SUBDIV, 4, 1, 1), 0)
    g2.agg_filter[0] = 0
    g2.agg_filter[1] = 1
    g2.agg_filter[2] = -1
    gop2 = Operator(OP_GROUPBY_MULTI, g2)
    any_extra = False
    for f in joined_parts:
        jm, pm = _emit_join(f[1], ocr.table(), 4, (0, 1, 2, 3))
        # pm: [item, wk, wh, promo, cr_cnt]
        if pm.n_rows:
            gop2.add_input_raw(pm)
            any_extra = True
        jm.destroy()
    gop2.finish()
    extra = gop2.get_output(["item", "wk", "wh", "e_np", "e_p", "e_t",
                             "cnt"]) if any_extra else None
    gop2.destroy()
    for f in joined_parts:
        f[0].destroy()
    for o in ops:
        o.destroy()
    for t in tables:
        lib().c.pg_table_destroy(t.table())
        t.destroy()

    # host fold (display side), vectorized: item_sk -> item_id, merge
    # base + extra group rows, sort by (item_id, wh, week)
    def cols(d, names):
        return [np.asarray(d[n]) for n in names]

    b_item, b_wh, b_wk, b_np_, b_p, b_c = cols(
        base, ("item", "wh", "wk", "no_promo", "promo", "cnt"))
    parts = [((b_item + 1) // 2, b_wh, b_wk, b_np_, b_p, b_c)]
    if extra is not None:
        e_item, e_wh, e_wk, e_np_, e_p, e_t = cols(
            extra, ("item", "wh", "wk", "e_np", "e_p", "e_t"))
        parts.append(((e_item + 1) // 2, e_wh, e_wk, e_np_, e_p, e_t))
    item = np.concatenate([p[0] for p in parts])
    wh = np.concatenate([p[1] for p in parts])
    wk = np.concatenate([p[2] for p in parts]).astype(np.int64)
    vnp = np.concatenate([p[3] for p in parts])
    vp = np.concatenate([p[4] for p in parts])
    vt = np.concatenate([p[5] for p in parts])
    key = (item << 32) | (wh << 16) | wk
    ukey, inv_ = np.unique(key, return_inverse=True)
    s_np = np.bincount(inv_, vnp, minlength=len(ukey)).astype(np.int64)
    s_p = np.bincount(inv_, vp, minlength=len(ukey)).astype(np.int64)
    s_t = np.bincount(inv_, vt, minlength=len(ukey)).astype(np.int64)
    return list(zip((ukey >> 32).tolist(),
                    ((ukey >> 16) & 0xffff).tolist(),
                    (ukey & 0xffff).tolist(),
                    s_np.tolist(), s_p.tolist(), s_t.tolist()))
