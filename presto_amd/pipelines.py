"""Query pipelines assembled from the C-ABI operators — the host analog of
LocalExecutionPlanner wiring OperatorFactories into a Driver
(presto-main-base/.../sql/planner/LocalExecutionPlanner.java:1663,2552,3648
for the ScanFilterAndProject / HashBuilder / HashAggregation factories) and
of the hand-wired pipelines in
presto-benchmark/.../HandTpchQuery1.java:30-110.

Date constants: epoch-day literals from the benchmark SQL
(presto-benchto-benchmarks/.../tpch/q01.sql, q03.sql):
  Q1: shipdate <= DATE '1998-12-01' - 90 days = 1998-09-02 = 10471
  Q3: DATE '1995-03-15' = 9204
"""
import ctypes as C

from .engine import (
    Operator, Page, PlanFilterProject, PlanHashAggSmall, PlanHashBuild,
    PlanLookupJoin, PlanTopN, PlanPartition, PlanGroupBy, Pred, Proj, Agg,
    OP_FILTER_PROJECT, OP_GROUPBY_MULTI,
    CMP_LE, CMP_LT, CMP_GT, CMP_GE, CMP_EQ, CMP_NE, CMP_CONTAINS,
    CMP_PREFIX, CMP_CONTAINS2, CMP_NOT_CONTAINS2,
    PROJ_IDENT, PROJ_DISC_PRICE, PROJ_CHARGE, PROJ_MUL, PROJ_DIV,
    PROJ_KEYSHL, PROJ_SHR, AGG_SUM_I64,
    AGG_COUNT, AGG_SUM_F64, AGG_SUM_DEC,
    OP_HASH_AGG_SMALL, OP_HASH_BUILD, OP_LOOKUP_JOIN, OP_TOPN, OP_PARTITION,
)

Q1_SHIP_MAX = 10471
Q3_DATE = 9204


def q1_plan(page: Page, mode="f64"):
    p = PlanHashAggSmall()
    p.n_preds = 1
    p.preds[0] = Pred(page.channel("shipdate"), CMP_LE, Q1_SHIP_MAX, 0.0)
    p.n_keys = 2
    p.key_col[0] = page.channel("returnflag")
    p.key_col[1] = page.channel("linestatus")
    p.n_vals[0] = 3
    p.n_vals[1] = 2
    for j, v in enumerate(b"ANR"):
        p.key_vals[0][j] = v
    for j, v in enumerate(b"FO"):
        p.key_vals[1][j] = v
    qty = page.channel("quantity")
    ep = page.channel("extendedprice")
    dc = page.channel("discount")
    tx = page.channel("tax")
    sums = [
        (Proj(PROJ_IDENT, qty, 0, 0), 0),
        (Proj(PROJ_IDENT, ep, 0, 0), 2),
        (Proj(PROJ_DISC_PRICE, ep, dc, 0), 4),
        (Proj(PROJ_CHARGE, ep, dc, tx), 6),
        (Proj(PROJ_IDENT, dc, 0, 0), 2),
    ]
    func = AGG_SUM_DEC if mode == "dec" else AGG_SUM_F64
    p.n_aggs = 6
    for i, (proj, scale) in enumerate(sums):
        p.aggs[i] = Agg(func, proj, scale)
    p.aggs[5] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    return p


Q1_F64_NAMES = ["returnflag", "linestatus", "sum_qty", "sum_base",
                "sum_disc_price", "sum_charge", "sum_disc", "count"]
Q1_DEC_NAMES = ["returnflag", "linestatus",
                "sum_qty_hi", "sum_qty_lo", "sum_base_hi", "sum_base_lo",
                "sum_disc_price_hi", "sum_disc_price_lo",
                "sum_charge_hi", "sum_charge_lo",
                "sum_disc_hi", "sum_disc_lo", "count"]


def q1(page: Page, mode="f64"):
    """Full Q1 over one lineitem page. Returns dict of numpy group columns
    (groups in (returnflag, linestatus) order)."""
    op = Operator(OP_HASH_AGG_SMALL, q1_plan(page, mode))
    try:
        op.add_input(page)
        op.finish()
        return op.get_output(Q1_F64_NAMES if mode == "f64" else Q1_DEC_NAMES)
    finally:
        op.destroy()


def okey_max(n_orders):
    """Largest o_orderkey for an n_orders-row orders table: dbgen's
    mk_sparse keeps the low 3 index bits and shifts the rest up by 2
    (oracle/tpchgen.c:141-145), which is monotone in the row index."""
    return ((n_orders >> 3) << 5) | (n_orders & 7)


class Q3Pipeline:
    """Q3 operator graph, reusable across inputs (tables freed on close).

    customer(BUILDING) -> key set          [HashBuilderOperator analog]
    orders(date<9204) semijoin set -> tbl  [HashBuilder w/ fused filter]
    lineitem(date>9204) probe tbl          [LookupJoin + fused grouped SUM]
    groups -> TopN 10 (rev desc, odate asc, okey asc)
    """

    def __init__(self, cust: Page, orders: Page, mode="dec", limit=10):
        self.mode = mode
        self.limit = limit
        b1p = PlanHashBuild()
        b1p.n_preds = 1
        b1p.preds[0] = Pred(cust.channel("mktseg"), CMP_EQ, 1, 0.0)
        b1p.key_col = cust.channel("custkey")
        b1p.semijoin_table = -1
        b1p.n_payload = 0
        b1p.capacity_hint = cust.n_rows  # custkeys dense 1..n
        b1p.key_set_only = 1
        b1p.dense_array = 1  # 1-byte membership flags, L2/L3-resident
        self.b1 = Operator(OP_HASH_BUILD, b1p)
        self.b1.add_input(cust)
        self.b1.finish()
        self.set_tbl = self.b1.table()

        b2p = PlanHashBuild()
        b2p.n_preds = 1
        b2p.preds[0] = Pred(orders.channel("orderdate"), CMP_LT, Q3_DATE, 0.0)
        b2p.key_col = orders.channel("orderkey")
        b2p.semijoin_table = self.set_tbl
        b2p.semijoin_col = orders.channel("custkey")
        b2p.n_payload = 1
        b2p.payload_col[0] = orders.channel("orderdate")
        b2p.capacity_hint = max(orders.n_rows // 8, 16)  # ~48.6%*20% pass
        b2p.key_set_only = 0
        b2p.agg_table = 1  # direct single-scan insert (fused-agg probe only)
        b2p.pack_bits = 16  # slot = orderkey<<16 | orderdate: one CAS
                            # carries key AND payload (date < 2^16)
        b2p.bitmap_max_key = okey_max(orders.n_rows)
        # dynamic-filter bitmap: ~91% of probes miss; one (mostly
        # L3-resident at SF100: 75 MB) bit load rejects them before
        # the hash+tag chain
        self.b2 = Operator(OP_HASH_BUILD, b2p)
        self.b2.add_input(orders)
        self.b2.finish()
        self.tbl = self.b2.table()

    def run(self, lineitem: Page):
        jp = PlanLookupJoin()
        jp.table = self.tbl
        jp.n_preds = 1
        jp.preds[0] = Pred(lineitem.channel("shipdate"), CMP_GT, Q3_DATE, 0.0)
        jp.key_col = lineitem.channel("orderkey")
        jp.mode = 1
        jp.proj = Proj(PROJ_DISC_PRICE, lineitem.channel("extendedprice"),
                       lineitem.channel("discount"), 0)
        jp.dec_scale = 4
        # dec mode consumes only the exact tick sums: skip the fx128 legs
        # (halves the per-hit atomic traffic); f64 mode keeps them
        jp.dec_only = 1 if self.mode == "dec" else 0
        j = Operator(OP_LOOKUP_JOIN, jp)
        try:
            j.add_input(lineitem)
            j.finish()
            groups = j.get_output_raw()
            # groups page cols: [orderkey, orderdate, sum_dec, sum_f64, cnt]
            tp = PlanTopN()
            tp.limit = self.limit
            tp.val_col = 2 if self.mode == "dec" else 3
            tp.date_col = 1
            tp.key_col = 0
            t = Operator(OP_TOPN, tp)
            try:
                t.add_input_raw(groups)
                t.finish()
                names = ["orderkey",
                         "revenue_1e4" if self.mode == "dec" else "revenue",
                         "orderdate"]
                return t.get_output(names)
            finally:
                t.destroy()
        finally:
            j.destroy()

    def close(self):
        from .engine import lib
        lib().c.pg_table_destroy(self.set_tbl)
        lib().c.pg_table_destroy(self.tbl)
        self.b1.destroy()
        self.b2.destroy()


def q3(cust: Page, orders: Page, lineitem: Page, mode="dec", limit=10):
    p = Q3Pipeline(cust, orders, mode=mode, limit=limit)
    try:
        return p.run(lineitem)
    finally:
        p.close()


class Q5Pipeline:
    """Q5 operator graph (q05.sql): region 'ASIA', orderdate in
    [1994-01-01, 1995-01-01), local-supplier join c_nationkey=s_nationkey,
    revenue grouped per nation.

    customer -> custkey->nationkey table
    orders(date) probe it (emit orderkey + cust nation) -> orderkey table
    supplier -> suppkey->nationkey table
    lineitem probe orders-table (emit suppkey, price, disc + cust nation)
             probe supplier-table (emit ... + supplier nation)
    small-key agg: key = supplier nation restricted to ASIA nations
    (drop_unlisted = the region membership), pred cnat == snat (col-col).
    """
    Q5_LO, Q5_HI = 8766, 9131
    ASIA = (8, 9, 12, 18, 21)

    def __init__(self, cust: Page, orders: Page, supp: Page):
        b1 = PlanHashBuild()
        b1.key_col = cust.channel("custkey")
        b1.semijoin_table = -1
        b1.n_payload = 1
        b1.payload_col[0] = cust.channel("nationkey")
        b1.capacity_hint = cust.n_rows
        self.b1 = Operator(OP_HASH_BUILD, b1)
        self.b1.add_input(cust)
        self.b1.finish()

        jo = PlanLookupJoin()
        jo.table = self.b1.table()
        jo.n_preds = 2
        jo.preds[0] = Pred(orders.channel("orderdate"), CMP_GE, self.Q5_LO,
                           0.0)
        jo.preds[1] = Pred(orders.channel("orderdate"), CMP_LT, self.Q5_HI,
                           0.0)
        jo.key_col = orders.channel("custkey")
        jo.mode = 0
        jo.n_emit = 1
        jo.emit_probe_cols[0] = orders.channel("orderkey")
        j = Operator(OP_LOOKUP_JOIN, jo)
        j.add_input(orders)
        opage = j.get_output_raw()  # [orderkey, cust_nationkey]

        b2 = PlanHashBuild()
        b2.key_col = 0
        b2.semijoin_table = -1
        b2.n_payload = 1
        b2.payload_col[0] = 1
        b2.capacity_hint = max(opage.n_rows, 16)
        self.b2 = Operator(OP_HASH_BUILD, b2)
        self.b2.add_input_raw(opage)
        self.b2.finish()
        j.destroy()

        b3 = PlanHashBuild()
        b3.key_col = supp.channel("suppkey")
        b3.semijoin_table = -1
        b3.n_payload = 1
        b3.payload_col[0] = supp.channel("nationkey")
        b3.capacity_hint = supp.n_rows
        self.b3 = Operator(OP_HASH_BUILD, b3)
        self.b3.add_input(supp)
        self.b3.finish()

    def run(self, li: Page):
        j1 = PlanLookupJoin()
        j1.table = self.b2.table()
        j1.key_col = li.channel("orderkey")
        j1.mode = 0
        j1.n_emit = 3
        j1.emit_probe_cols[0] = li.channel("suppkey")
        j1.emit_probe_cols[1] = li.channel("extendedprice")
        j1.emit_probe_cols[2] = li.channel("discount")
        ja = Operator(OP_LOOKUP_JOIN, j1)
        ja.add_input(li)
        pa = ja.get_output_raw()  # [suppkey, ep, dc, cnat]

        j2 = PlanLookupJoin()
        j2.table = self.b3.table()
        j2.key_col = 0
        j2.mode = 0
        j2.n_emit = 3
        j2.emit_probe_cols[0] = 1
        j2.emit_probe_cols[1] = 2
        j2.emit_probe_cols[2] = 3
        jb = Operator(OP_LOOKUP_JOIN, j2)
        jb.add_input_raw(pa)
        pb = jb.get_output_raw()  # [ep, dc, cnat, snat]
        ja.destroy()

        agg = PlanHashAggSmall()
        agg.n_preds = 1
        p = Pred(2, CMP_EQ, 0, 0.0)
        p.rhs_col = 3 + 1  # compare channel 2 (cnat) == channel 3 (snat)
        agg.preds[0] = p
        agg.n_keys = 1
        agg.key_col[0] = 3
        agg.n_vals[0] = len(self.ASIA)
        for i, v in enumerate(self.ASIA):
            agg.key_vals[0][i] = v
        agg.n_aggs = 2
        agg.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, 0, 1, 0), 4)
        agg.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
        agg.drop_unlisted_keys = 1
        ao = Operator(OP_HASH_AGG_SMALL, agg)
        ao.add_input_raw(pb)
        ao.finish()
        out = ao.get_output(["nationkey", "rev_hi", "rev_lo", "count"])
        ao.destroy()
        jb.destroy()
        return out

    def close(self):
        from .engine import lib
        for b in (self.b1, self.b2, self.b3):
            lib().c.pg_table_destroy(b.table())
            b.destroy()


def q5_composed(cust: Page, orders: Page, supp: Page, li: Page):
    p = Q5Pipeline(cust, orders, supp)
    try:
        return p.run(li)
    finally:
        p.close()


class Q5PipelineFused:
    """Q5 with the single-pass fused probe (LOOKUP_JOIN mode 2): lineitem
    is scanned ONCE — probe the orderkey->cust-nation agg table, dense
    suppkey->supplier-nation lookup, local-supplier equality, per-nation
    register accumulation.  The codegen-analog specialization of the
    composed graph above (same results, no materialized intermediates)."""

    def __init__(self, cust: Page, orders: Page, supp: Page):
        # customer dimension as an agg table (slot payload = nationkey)
        b1 = PlanHashBuild()
        b1.key_col = cust.channel("custkey")
        b1.semijoin_table = -1
        b1.n_payload = 1
        b1.payload_col[0] = cust.channel("nationkey")
        b1.capacity_hint = cust.n_rows  # custkeys dense 1..n
        b1.dense_array = 1
        self.b1 = Operator(OP_HASH_BUILD, b1)
        self.b1.add_input(cust)
        self.b1.finish()

        # orders build fused with the customer dimension join: one scan of
        # orders, payload = cust_table[o_custkey].nationkey
        b2 = PlanHashBuild()
        b2.n_preds = 2
        b2.preds[0] = Pred(orders.channel("orderdate"), CMP_GE,
                           Q5Pipeline.Q5_LO, 0.0)
        b2.preds[1] = Pred(orders.channel("orderdate"), CMP_LT,
                           Q5Pipeline.Q5_HI, 0.0)
        b2.key_col = orders.channel("orderkey")
        b2.semijoin_table = -1
        b2.n_payload = 1
        b2.payload_col[0] = 0  # sourced through the lookup instead
        b2.payload_lookup_table = self.b1.table()
        b2.payload_lookup_key_col = orders.channel("custkey")
        # the date window passes ~15.2% of orders; //4 keeps fill ~0.18.
        # Measured: smaller tables (fill 0.36 / 0.71) cost 1-17 ms of the
        # 600M-row probe — linear-probe cluster length beats any L3
        # residency gain, so size for SHORT clusters.
        b2.capacity_hint = max(orders.n_rows // 4, 64)
        b2.agg_table = 1
        b2.pack_bits = 8  # slot = orderkey<<8 | cust_nation (u8)
        b2.bitmap_max_key = okey_max(orders.n_rows)
        self.b2 = Operator(OP_HASH_BUILD, b2)
        self.b2.add_input(orders)
        self.b2.finish()

        b3 = PlanHashBuild()
        b3.key_col = supp.channel("suppkey")
        b3.n_payload = 1
        b3.payload_col[0] = supp.channel("nationkey")
        b3.capacity_hint = supp.n_rows
        b3.dense_array = 1
        self.b3 = Operator(OP_HASH_BUILD, b3)
        self.b3.add_input(supp)
        self.b3.finish()

    def run(self, li: Page):
        jp = PlanLookupJoin()
        jp.table = self.b2.table()
        jp.key_col = li.channel("orderkey")
        jp.mode = 2
        jp.proj = Proj(PROJ_DISC_PRICE, li.channel("extendedprice"),
                       li.channel("discount"), 0)
        jp.dec_scale = 4
        jp.table2 = self.b3.table()
        jp.table2_key_col = li.channel("suppkey")
        jp.n_group_vals = len(Q5Pipeline.ASIA)
        for i, v in enumerate(Q5Pipeline.ASIA):
            jp.group_vals[i] = v
        j = Operator(OP_LOOKUP_JOIN, jp)
        try:
            j.add_input(li)
            j.finish()
            return j.get_output(["nationkey", "rev_lo", "rev_f64", "count"])
        finally:
            j.destroy()

    def close(self):
        from .engine import lib
        for b in (self.b1, self.b2, self.b3):
            lib().c.pg_table_destroy(b.table())
            b.destroy()


def q5(cust: Page, orders: Page, supp: Page, li: Page):
    p = Q5PipelineFused(cust, orders, supp)
    try:
        return p.run(li)
    finally:
        p.close()


def q6(li: Page):
    """Q6 scalar aggregate (q06.sql): keyless HASH_AGG_SMALL with the
    BETWEEN predicates fused; exact decimal ticks (scale 4) + count."""
    p = PlanHashAggSmall()
    p.n_preds = 5
    p.preds[0] = Pred(li.channel("shipdate"), CMP_GE, 8766, 0.0)
    p.preds[1] = Pred(li.channel("shipdate"), CMP_LT, 9131, 0.0)
    p.preds[2] = Pred(li.channel("discount"), CMP_GE, 0, 0.05)
    p.preds[3] = Pred(li.channel("discount"), CMP_LE, 0, 0.07)
    p.preds[4] = Pred(li.channel("quantity"), CMP_LT, 0, 24.0)
    p.n_keys = 0
    p.n_aggs = 2
    p.aggs[0] = Agg(AGG_SUM_DEC,
                    Proj(PROJ_MUL, li.channel("extendedprice"),
                         li.channel("discount"), 0), 4)
    p.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    op = Operator(OP_HASH_AGG_SMALL, p)
    try:
        op.add_input(li)
        op.finish()
        return op.get_output(["rev_hi", "rev_lo", "count"])
    finally:
        op.destroy()


def q7(cust: Page, orders: Page, supp: Page, li: Page):
    """Q7 volume shipping (q07.sql): FRANCE(6)<->GERMANY(7) pairs, shipdate
    in [1995-01-01, 1996-12-31], volume grouped by (supp_nation,
    cust_nation, year).  Composed: dense customer dimension, orders
    agg-table with fused dimension lookup (no date filter), supplier hash
    table, two emit joins, then four keyless aggregations (the OR over
    nation pairs and the year split decompose into conjunctive plans).
    Returns list of (supp_nation, cust_nation, year, revenue_1e4)."""
    b1 = PlanHashBuild()
    b1.key_col = cust.channel("custkey")
    b1.semijoin_table = -1
    b1.n_payload = 1
    b1.payload_col[0] = cust.channel("nationkey")
    b1.capacity_hint = cust.n_rows
    b1.dense_array = 1
    o1 = Operator(OP_HASH_BUILD, b1)
    o1.add_input(cust)
    o1.finish()

    # orderkey -> customer nation as a dense u8 array over the orderkey
    # range (one scan + one byte per possible key; nation is fetched
    # through the dense customer dimension during the fill).  Presence =
    # nonzero, so the stored value is nation+1 — downstream predicates
    # are biased accordingly.
    b2 = PlanHashBuild()
    b2.key_col = orders.channel("orderkey")
    b2.semijoin_table = -1
    b2.n_payload = 1
    b2.payload_col[0] = 0
    b2.payload_lookup_table = o1.table()
    b2.payload_lookup_key_col = orders.channel("custkey")
    b2.capacity_hint = okey_max(orders.n_rows)
    b2.dense_array = 1
    b2.dense_payload_bias = 1
    o2 = Operator(OP_HASH_BUILD, b2)
    o2.add_input(orders)
    o2.finish()

    b3 = PlanHashBuild()
    b3.key_col = supp.channel("suppkey")
    b3.semijoin_table = -1
    b3.n_payload = 1
    b3.payload_col[0] = supp.channel("nationkey")
    b3.capacity_hint = supp.n_rows
    o3 = Operator(OP_HASH_BUILD, b3)
    o3.add_input(supp)
    o3.finish()

    j1 = PlanLookupJoin()
    j1.table = o2.table()
    j1.n_preds = 2
    j1.preds[0] = Pred(li.channel("shipdate"), CMP_GE, 9131, 0.0)
    j1.preds[1] = Pred(li.channel("shipdate"), CMP_LE, 9861, 0.0)
    j1.key_col = li.channel("orderkey")
    j1.mode = 0
    j1.n_emit = 3
    j1.emit_probe_cols[0] = li.channel("suppkey")
    j1.emit_probe_cols[1] = li.channel("extendedprice")
    j1.emit_probe_cols[2] = li.channel("discount")
    # NOTE: shipdate needed downstream -> emit it too
    j1.n_emit = 4
    j1.emit_probe_cols[3] = li.channel("shipdate")
    ja = Operator(OP_LOOKUP_JOIN, j1)
    ja.add_input(li)
    pa = ja.get_output_raw()  # [suppkey, ep, dc, sdate, cnat]

    j2 = PlanLookupJoin()
    j2.table = o3.table()
    j2.key_col = 0
    j2.mode = 0
    j2.n_emit = 4
    for i, c in enumerate((1, 2, 3, 4)):
        j2.emit_probe_cols[i] = c
    jb = Operator(OP_LOOKUP_JOIN, j2)
    jb.add_input_raw(pa)
    pb = jb.get_output_raw()  # [ep, dc, sdate, cnat, snat]
    ja.destroy()

    out = []
    for sn, cn in ((6, 7), (7, 6)):
        for ylo, yhi, yr in ((9131, 9495, 1995), (9496, 9861, 1996)):
            p = PlanHashAggSmall()
            p.n_preds = 4
            p.preds[0] = Pred(4, CMP_EQ, sn, 0.0)
            p.preds[1] = Pred(3, CMP_EQ, cn + 1, 0.0)  # dense bias
            p.preds[2] = Pred(2, CMP_GE, ylo, 0.0)
            p.preds[3] = Pred(2, CMP_LE, yhi, 0.0)
            p.n_keys = 0
            p.n_aggs = 2
            p.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, 0, 1, 0), 4)
            p.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
            ao = Operator(OP_HASH_AGG_SMALL, p)
            ao.add_input_raw(pb)
            ao.finish()
            res = ao.get_output(["hi", "lo", "cnt"])
            ao.destroy()
            if len(res["lo"]) and res["cnt"][0] > 0:
                out.append((sn, cn, yr, int(res["lo"][0])))
    jb.destroy()
    from .engine import lib
    for o in (o1, o2, o3):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return out


AMERICA_NATIONS = (1, 2, 3, 17, 24)  # tpch_nation_region(k) == 1 (AMERICA)
BRAZIL = 2


def q8(cust: Page, orders: Page, supp: Page, part: Page, li: Page):
    """Q8 national market share (q08.sql): revenue of 'ECONOMY ANODIZED
    STEEL' (type id 103) parts sold to AMERICA-region customers with
    orderdate in 1995..1996, split into BRAZIL(2)-supplier vs total per
    order year.  Composed from the §8 operators only:

      nation keys -> key set; customer semijoin filter -> custkey flag set
      orders chained build (date range + customer semijoin, payload
        orderdate)                      [HashBuilderOperator analog]
      part(type==103) -> partkey flag set; supplier(BRAZIL) -> flag set
      lineitem semijoin(part) filter -> emit join (payload orderdate)
        -> brazil semijoin filter + 4 keyless year aggregations
    Returns (brazil_1e4[2], total_1e4[2]) exact ticks for (1995, 1996);
    share = brazil/total (golden q08_sf1.result: 0.0344 / 0.0415)."""
    import numpy as np
    from .engine import lib

    # American nation keys as a tiny key set
    natp = Page({"nationkey": np.asarray(AMERICA_NATIONS, dtype=np.int64)})
    bn = PlanHashBuild()
    bn.key_col = 0
    bn.semijoin_table = -1
    bn.capacity_hint = 32
    bn.key_set_only = 1
    on = Operator(OP_HASH_BUILD, bn)
    on.add_input(natp)
    on.finish()

    # AMERICA customers -> dense custkey flag set
    fc = PlanFilterProject()
    fc.n_proj = 1
    fc.proj[0] = Proj(PROJ_IDENT, cust.channel("custkey"), 0, 0)
    fc.semijoin_table = on.table()
    fc.semijoin_col = cust.channel("nationkey")
    f1 = Operator(OP_FILTER_PROJECT, fc)
    f1.add_input(cust)
    cpage = f1.get_output_raw()

    bc = PlanHashBuild()
    bc.key_col = 0
    bc.semijoin_table = -1
    bc.capacity_hint = cust.n_rows  # custkeys dense 1..n
    bc.key_set_only = 1
    bc.dense_array = 1
    oc = Operator(OP_HASH_BUILD, bc)
    oc.add_input_raw(cpage)
    oc.finish()

    # qualifying orders: chained table keyed by orderkey, payload orderdate
    bo = PlanHashBuild()
    bo.n_preds = 2
    bo.preds[0] = Pred(orders.channel("orderdate"), CMP_GE, 9131, 0.0)
    bo.preds[1] = Pred(orders.channel("orderdate"), CMP_LE, 9861, 0.0)
    bo.key_col = orders.channel("orderkey")
    bo.semijoin_table = oc.table()
    bo.semijoin_col = orders.channel("custkey")
    bo.n_payload = 1
    bo.payload_col[0] = orders.channel("orderdate")
    bo.capacity_hint = max(orders.n_rows // 8, 16)
    bo.bitmap_max_key = okey_max(orders.n_rows)
    oo = Operator(OP_HASH_BUILD, bo)
    oo.add_input(orders)
    oo.finish()

    # part type 103 -> dense partkey flag set
    bp = PlanHashBuild()
    bp.n_preds = 1
    bp.preds[0] = Pred(part.channel("type_id"), CMP_EQ, 103, 0.0)
    bp.key_col = part.channel("partkey")
    bp.semijoin_table = -1
    bp.capacity_hint = part.n_rows
    bp.key_set_only = 1
    bp.dense_array = 1
    opart = Operator(OP_HASH_BUILD, bp)
    opart.add_input(part)
    opart.finish()

    # BRAZIL suppliers -> dense suppkey flag set
    bb = PlanHashBuild()
    bb.n_preds = 1
    bb.preds[0] = Pred(supp.channel("nationkey"), CMP_EQ, BRAZIL, 0.0)
    bb.key_col = supp.channel("suppkey")
    bb.semijoin_table = -1
    bb.capacity_hint = supp.n_rows
    bb.key_set_only = 1
    bb.dense_array = 1
    ob = Operator(OP_HASH_BUILD, bb)
    ob.add_input(supp)
    ob.finish()

    # lineitem: keep type-103 parts, project the join/agg columns
    fl = PlanFilterProject()
    fl.n_proj = 4
    fl.proj[0] = Proj(PROJ_IDENT, li.channel("orderkey"), 0, 0)
    fl.proj[1] = Proj(PROJ_IDENT, li.channel("suppkey"), 0, 0)
    fl.proj[2] = Proj(PROJ_IDENT, li.channel("extendedprice"), 0, 0)
    fl.proj[3] = Proj(PROJ_IDENT, li.channel("discount"), 0, 0)
    fl.semijoin_table = opart.table()
    fl.semijoin_col = li.channel("partkey")
    f2 = Operator(OP_FILTER_PROJECT, fl)
    f2.add_input(li)
    lpage = f2.get_output_raw()  # [orderkey, suppkey, ep, dc]

    # emit join against qualifying orders; payload appends orderdate
    jp = PlanLookupJoin()
    jp.table = oo.table()
    jp.key_col = 0
    jp.mode = 0
    jp.n_emit = 3
    jp.emit_probe_cols[0] = 1
    jp.emit_probe_cols[1] = 2
    jp.emit_probe_cols[2] = 3
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input_raw(lpage)
    jpage = jo.get_output_raw()  # [suppkey, ep, dc, orderdate]

    # brazil-supplier subset
    fb = PlanFilterProject()
    fb.n_proj = 3
    fb.proj[0] = Proj(PROJ_IDENT, 1, 0, 0)
    fb.proj[1] = Proj(PROJ_IDENT, 2, 0, 0)
    fb.proj[2] = Proj(PROJ_IDENT, 3, 0, 0)
    fb.semijoin_table = ob.table()
    fb.semijoin_col = 0
    f3 = Operator(OP_FILTER_PROJECT, fb)
    f3.add_input_raw(jpage)
    bpage = f3.get_output_raw()  # [ep, dc, orderdate]

    def rev(page, odcol, epcol, dccol, lo, hi):
        p = PlanHashAggSmall()
        p.n_preds = 2
        p.preds[0] = Pred(odcol, CMP_GE, lo, 0.0)
        p.preds[1] = Pred(odcol, CMP_LE, hi, 0.0)
        p.n_keys = 0
        p.n_aggs = 1
        p.aggs[0] = Agg(AGG_SUM_DEC,
                        Proj(PROJ_DISC_PRICE, epcol, dccol, 0), 4)
        a = Operator(OP_HASH_AGG_SMALL, p)
        try:
            a.add_input_raw(page)
            a.finish()
            r = a.get_output(["hi", "lo"])
            if not len(r["lo"]):
                return 0
            return (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))
        finally:
            a.destroy()

    YEARS = ((9131, 9495), (9496, 9861))  # 1995, 1996 epoch-day ranges
    total = [rev(jpage, 3, 1, 2, lo, hi) for lo, hi in YEARS]
    brazil = [rev(bpage, 2, 0, 1, lo, hi) for lo, hi in YEARS]

    f3.destroy()
    jo.destroy()
    f2.destroy()
    f1.destroy()
    for o in (on, oc, oo, opart, ob):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return brazil, total


def q14(part: Page, li: Page):
    """Q14 promo revenue (q14.sql): shipdate in [1995-09-01, 1995-10-01) =
    [9374, 9404); promo parts = type ids 125..149 ('PROMO*') as a dense
    partkey flag set.  Returns (promo_1e4, total_1e4) exact ticks; the
    result is 100.00*promo/total at scale 6 HALF_UP."""
    import numpy as np
    from .engine import lib

    bp = PlanHashBuild()
    bp.n_preds = 1
    bp.preds[0] = Pred(part.channel("type_id"), CMP_GE, 125, 0.0)
    bp.key_col = part.channel("partkey")
    bp.semijoin_table = -1
    bp.capacity_hint = part.n_rows
    bp.key_set_only = 1
    bp.dense_array = 1
    ob = Operator(OP_HASH_BUILD, bp)
    ob.add_input(part)
    ob.finish()

    def rev(page, semijoin, epc, dcc, sdc, pkc):
        p = PlanHashAggSmall()
        p.n_preds = 2
        p.preds[0] = Pred(sdc, CMP_GE, 9374, 0.0)
        p.preds[1] = Pred(sdc, CMP_LT, 9404, 0.0)
        p.n_keys = 0
        p.n_aggs = 1
        p.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, epc, dcc, 0), 4)
        if semijoin:
            fp = PlanFilterProject()
            fp.n_preds = 2
            fp.preds[0] = Pred(sdc, CMP_GE, 9374, 0.0)
            fp.preds[1] = Pred(sdc, CMP_LT, 9404, 0.0)
            fp.n_proj = 2
            fp.proj[0] = Proj(PROJ_IDENT, epc, 0, 0)
            fp.proj[1] = Proj(PROJ_IDENT, dcc, 0, 0)
            fp.semijoin_table = ob.table()
            fp.semijoin_col = pkc
            f = Operator(OP_FILTER_PROJECT, fp)
            f.add_input(page)
            fpage = f.get_output_raw()
            p.n_preds = 0
            p.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, 0, 1, 0), 4)
            a = Operator(OP_HASH_AGG_SMALL, p)
            a.add_input_raw(fpage)
            a.finish()
            r = a.get_output(["hi", "lo"])
            a.destroy()
            f.destroy()
        else:
            a = Operator(OP_HASH_AGG_SMALL, p)
            a.add_input(page)
            a.finish()
            r = a.get_output(["hi", "lo"])
            a.destroy()
        if not len(r["lo"]):
            return 0
        return (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))

    epc, dcc = li.channel("extendedprice"), li.channel("discount")
    sdc, pkc = li.channel("shipdate"), li.channel("partkey")
    total = rev(li, False, epc, dcc, sdc, pkc)
    promo = rev(li, True, epc, dcc, sdc, pkc)
    lib().c.pg_table_destroy(ob.table())
    ob.destroy()
    return promo, total


def q12(orders: Page, li: Page):
    """Q12 shipmode priority (q12.sql): late-commit lineitems received in
    1994 joined to orders; counts per (shipmode in {MAIL=4, SHIP=6},
    priority class).  Returns {mode_id: (high, low)}."""
    # orderkey -> priority as a dense u8 array over the orderkey range
    # (presence = nonzero, so priority+1 is stored; the class-split
    # predicates below are biased accordingly)
    bo = PlanHashBuild()
    bo.key_col = orders.channel("orderkey")
    bo.semijoin_table = -1
    bo.n_payload = 1
    bo.payload_col[0] = orders.channel("priority")
    bo.capacity_hint = okey_max(orders.n_rows)
    bo.dense_array = 1
    bo.dense_payload_bias = 1
    oo = Operator(OP_HASH_BUILD, bo)
    oo.add_input(orders)
    oo.finish()

    cdc, rdc = li.channel("commitdate"), li.channel("receiptdate")
    sdc = li.channel("shipdate")
    fl = PlanFilterProject()
    fl.n_preds = 4
    p0 = Pred(cdc, CMP_LT, 0, 0.0)
    p0.rhs_col = rdc + 1
    fl.preds[0] = p0
    p1 = Pred(sdc, CMP_LT, 0, 0.0)
    p1.rhs_col = cdc + 1
    fl.preds[1] = p1
    fl.preds[2] = Pred(rdc, CMP_GE, 8766, 0.0)
    fl.preds[3] = Pred(rdc, CMP_LT, 9131, 0.0)
    fl.n_proj = 2
    fl.proj[0] = Proj(PROJ_IDENT, li.channel("orderkey"), 0, 0)
    fl.proj[1] = Proj(PROJ_IDENT, li.channel("shipmode"), 0, 0)
    f = Operator(OP_FILTER_PROJECT, fl)
    f.add_input(li)
    fpage = f.get_output_raw()  # [orderkey, shipmode]

    jp = PlanLookupJoin()
    jp.table = oo.table()
    jp.key_col = 0
    jp.mode = 0
    jp.n_emit = 1
    jp.emit_probe_cols[0] = 1
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input_raw(fpage)
    jpage = jo.get_output_raw()  # [shipmode, priority]

    res = {4: [0, 0], 6: [0, 0]}
    for cls, (op_, val) in enumerate(((CMP_LE, 2), (CMP_GE, 3))):
        # priority stored +1 (dense bias): URGENT/HIGH {0,1} -> {1,2}
        ag = PlanHashAggSmall()
        ag.n_preds = 1
        ag.preds[0] = Pred(1, op_, val, 0.0)  # priority class split
        ag.n_keys = 1
        ag.key_col[0] = 0
        ag.n_vals[0] = 2
        ag.key_vals[0][0] = 4  # MAIL
        ag.key_vals[0][1] = 6  # SHIP
        ag.drop_unlisted_keys = 1
        ag.n_aggs = 1
        ag.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
        a = Operator(OP_HASH_AGG_SMALL, ag)
        a.add_input_raw(jpage)
        a.finish()
        out = a.get_output(["shipmode", "count"])
        a.destroy()
        for i in range(len(out["shipmode"])):
            res[int(out["shipmode"][i])][cls] += int(out["count"][i])
    jo.destroy()
    f.destroy()
    from .engine import lib
    lib().c.pg_table_destroy(oo.table())
    oo.destroy()
    return {m: tuple(v) for m, v in res.items()}


def q17(part: Page, li: Page):
    """Q17 small-quantity-order revenue (q17.sql): Brand#23 'MED BOX'
    (container id 17) parts; rows with quantity < 0.2*avg(part quantity).
    The correlated avg is a fused-agg probe (sum_qty + count per target
    part); the tiny per-part cutoffs (0.2*avg <= 10) come back to the
    host (coordinator-side scalar-subquery plan constants, ~200 rows) and
    drive per-cutoff-class flag-set semijoins.  Returns the exact cents
    sum of extendedprice (result = cents/7.0 at scale 2 HALF_UP)."""
    import numpy as np
    from .engine import lib
    pkc = part.channel("partkey")
    tables = []

    def target_preds(plan):
        plan.n_preds = 2
        plan.preds[0] = Pred(part.channel("brand"), CMP_EQ, 23, 0.0)
        plan.preds[1] = Pred(part.channel("container"), CMP_EQ, 17, 0.0)

    bt = PlanHashBuild()
    target_preds(bt)
    bt.key_col = pkc
    bt.semijoin_table = -1
    bt.capacity_hint = max(part.n_rows // 16, 4096)
    bt.agg_table = 1
    bt.bitmap_max_key = part.n_rows  # ~0.1% of 600M probes hit
    ot = Operator(OP_HASH_BUILD, bt)
    ot.add_input(part)
    ot.finish()
    tables.append(ot)

    bs = PlanHashBuild()
    target_preds(bs)
    bs.key_col = pkc
    bs.semijoin_table = -1
    bs.capacity_hint = part.n_rows
    bs.key_set_only = 1
    bs.dense_array = 1
    os_ = Operator(OP_HASH_BUILD, bs)
    os_.add_input(part)
    os_.finish()
    tables.append(os_)

    # per-target-part sum(quantity) + count over ALL lineitems
    jp = PlanLookupJoin()
    jp.table = ot.table()
    jp.key_col = li.channel("partkey")
    jp.mode = 1
    jp.proj = Proj(PROJ_IDENT, li.channel("quantity"), 0, 0)
    jp.dec_scale = 0
    jp.dec_only = 1
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input(li)
    jo.finish()
    g = jo.get_output(["partkey", "sum_qty", "sum_f64", "cnt"])
    jo.destroy()

    # cutoff class per part: qty < sum/(5*cnt)  =>  qty <= c_p
    classes = {}
    for i in range(len(g["partkey"])):
        s, c = int(g["sum_qty"][i]), int(g["cnt"][i])
        cp = (s + 5 * c - 1) // (5 * c) - 1  # max qty with 5*qty*c < s
        if cp >= 1:
            classes.setdefault(cp, []).append(int(g["partkey"][i]))

    # li restricted to target parts (small page)
    fl = PlanFilterProject()
    fl.n_proj = 3
    fl.proj[0] = Proj(PROJ_IDENT, li.channel("partkey"), 0, 0)
    fl.proj[1] = Proj(PROJ_IDENT, li.channel("quantity"), 0, 0)
    fl.proj[2] = Proj(PROJ_IDENT, li.channel("extendedprice"), 0, 0)
    fl.semijoin_table = os_.table()
    fl.semijoin_col = li.channel("partkey")
    f = Operator(OP_FILTER_PROJECT, fl)
    f.add_input(li)
    fpage = f.get_output_raw()  # [partkey, qty, ep]

    total = 0
    for cp, pks in sorted(classes.items()):
        cpage = Page({"partkey": np.asarray(pks, dtype=np.int64)})
        bc = PlanHashBuild()
        bc.key_col = 0
        bc.semijoin_table = -1
        bc.capacity_hint = part.n_rows
        bc.key_set_only = 1
        bc.dense_array = 1
        oc = Operator(OP_HASH_BUILD, bc)
        oc.add_input(cpage)
        oc.finish()

        fc = PlanFilterProject()
        fc.n_preds = 1
        fc.preds[0] = Pred(1, CMP_LE, 0, float(cp))
        fc.n_proj = 1
        fc.proj[0] = Proj(PROJ_IDENT, 2, 0, 0)
        fc.semijoin_table = oc.table()
        fc.semijoin_col = 0
        ff = Operator(OP_FILTER_PROJECT, fc)
        ff.add_input_raw(fpage)
        cpage2 = ff.get_output_raw()

        ap = PlanHashAggSmall()
        ap.n_keys = 0
        ap.n_aggs = 1
        ap.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_IDENT, 0, 0, 0), 2)
        a = Operator(OP_HASH_AGG_SMALL, ap)
        a.add_input_raw(cpage2)
        a.finish()
        r = a.get_output(["hi", "lo"])
        if len(r["lo"]):
            total += (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))
        a.destroy()
        ff.destroy()
        lib().c.pg_table_destroy(oc.table())
        oc.destroy()

    f.destroy()
    for o in tables:
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return total


def q11(supp: Page, ps: Page, n_part: int):
    """Q11 important stock (q11.sql): GERMANY(7) partsupp grouped by
    partkey, value = sum(supplycost*availqty); HAVING value >
    0.0001*total; ORDER BY value DESC (partkey ASC tiebreak).  The MUL
    decimal projection yields 1e-4 ticks (cents x hundredth-encoded
    qty); the strict HAVING threshold total/10000 becomes a plan
    constant.  Returns (partkeys, value_cents) host arrays; the ~1k-row
    final ORDER BY runs host-side (output-stage sort)."""
    import numpy as np
    from .engine import lib

    bg = PlanHashBuild()
    bg.n_preds = 1
    bg.preds[0] = Pred(supp.channel("nationkey"), CMP_EQ, 7, 0.0)
    bg.key_col = supp.channel("suppkey")
    bg.semijoin_table = -1
    bg.capacity_hint = supp.n_rows
    bg.key_set_only = 1
    bg.dense_array = 1
    og = Operator(OP_HASH_BUILD, bg)
    og.add_input(supp)
    og.finish()

    fl = PlanFilterProject()
    fl.n_proj = 3
    fl.proj[0] = Proj(PROJ_IDENT, ps.channel("partkey"), 0, 0)
    fl.proj[1] = Proj(PROJ_IDENT, ps.channel("supplycost"), 0, 0)
    fl.proj[2] = Proj(PROJ_IDENT, ps.channel("availqty"), 0, 0)
    fl.semijoin_table = og.table()
    fl.semijoin_col = ps.channel("suppkey")
    f = Operator(OP_FILTER_PROJECT, fl)
    f.add_input(ps)
    gpage = f.get_output_raw()  # [partkey, cost, qty]

    ta = PlanHashAggSmall()
    ta.n_keys = 0
    ta.n_aggs = 1
    ta.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_MUL, 1, 2, 0), 4)
    a = Operator(OP_HASH_AGG_SMALL, ta)
    a.add_input_raw(gpage)
    a.finish()
    r = a.get_output(["hi", "lo"])
    total_1e4 = (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))
    a.destroy()

    keys = Page({"partkey": np.arange(1, n_part + 1, dtype=np.int64)})
    bk = PlanHashBuild()
    bk.key_col = 0
    bk.semijoin_table = -1
    bk.capacity_hint = n_part
    bk.agg_table = 1
    ok = Operator(OP_HASH_BUILD, bk)
    ok.add_input(keys)
    ok.finish()

    jp = PlanLookupJoin()
    jp.table = ok.table()
    jp.key_col = 0
    jp.mode = 1
    jp.proj = Proj(PROJ_MUL, 1, 2, 0)
    jp.dec_scale = 4
    jp.dec_only = 1
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input_raw(gpage)
    jo.finish()
    groups = jo.get_output_raw()  # [partkey, sum_1e4, f64, cnt]

    ft = PlanFilterProject()
    ft.n_preds = 1
    ft.preds[0] = Pred(1, CMP_GT, total_1e4 // 10000, 0.0)
    ft.n_proj = 2
    ft.proj[0] = Proj(PROJ_IDENT, 0, 0, 0)
    ft.proj[1] = Proj(PROJ_IDENT, 1, 0, 0)
    ft.semijoin_table = 0
    fo = Operator(OP_FILTER_PROJECT, ft)
    fo.add_input_raw(groups)
    out = fo.get_output(["partkey", "value_1e4"])
    fo.destroy()
    jo.destroy()
    f.destroy()
    for o in (og, ok):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    pk = out["partkey"]
    val = out["value_1e4"] // 100  # exact: ticks are cents*100
    order = np.lexsort((pk, -val))
    return pk[order], val[order]


def q18(orders: Page, li: Page, limit=100):
    """Q18 large-volume customers (q18.sql): per-order quantity sums via
    a fused-agg probe over the full orders key table; HAVING sum > 300
    as a plan-constant filter; the ~1e-5-selectivity survivors join back
    to the orders columns (emit join) and the bounded final ORDER BY
    runs host-side.  Returns [(custkey, orderkey, orderdate,
    totalprice_cents, sum_qty)] sorted (totalprice desc, orderdate asc,
    orderkey asc) LIMIT limit."""
    import numpy as np
    from .engine import lib

    # per-order qty sums take NOTHING from orders: range-group domain
    # over the orderkey range replaces the 150M-row build, and the
    # orderkey-clustered lineitem makes the accumulator atomics
    # near-sequential (see q21)
    bo = PlanHashBuild()
    bo.semijoin_table = -1
    bo.capacity_hint = okey_max(orders.n_rows)
    bo.range_group = 1
    oo = Operator(OP_HASH_BUILD, bo)
    oo.finish()

    jp = PlanLookupJoin()
    jp.table = oo.table()
    jp.key_col = li.channel("orderkey")
    jp.mode = 1
    jp.proj = Proj(PROJ_IDENT, li.channel("quantity"), 0, 0)
    jp.dec_scale = 0
    jp.dec_only = 1
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input(li)
    jo.finish()
    groups = jo.get_output_raw()  # [orderkey, sum_qty, f64, cnt]

    ft = PlanFilterProject()
    ft.n_preds = 1
    ft.preds[0] = Pred(1, CMP_GT, 300, 0.0)
    ft.n_proj = 2
    ft.proj[0] = Proj(PROJ_IDENT, 0, 0, 0)
    ft.proj[1] = Proj(PROJ_IDENT, 1, 0, 0)
    fo = Operator(OP_FILTER_PROJECT, ft)
    fo.add_input_raw(groups)
    big = fo.get_output_raw()  # [orderkey, sum_qty]

    bb = PlanHashBuild()
    bb.key_col = 0
    bb.semijoin_table = -1
    bb.n_payload = 1
    bb.payload_col[0] = 1
    bb.capacity_hint = max(big.n_rows, 16)
    bb.bitmap_max_key = okey_max(orders.n_rows)  # ~6K hits of 150M probes
    ob = Operator(OP_HASH_BUILD, bb)
    ob.add_input_raw(big)
    ob.finish()

    je = PlanLookupJoin()
    je.table = ob.table()
    je.key_col = orders.channel("orderkey")
    je.mode = 0
    je.n_emit = 4
    je.emit_probe_cols[0] = orders.channel("orderkey")
    je.emit_probe_cols[1] = orders.channel("custkey")
    je.emit_probe_cols[2] = orders.channel("orderdate")
    je.emit_probe_cols[3] = orders.channel("totalprice")
    js = Operator(OP_LOOKUP_JOIN, je)
    js.add_input(orders)
    out = js.get_output(["orderkey", "custkey", "orderdate", "totalprice",
                         "sum_qty"])
    js.destroy()
    fo.destroy()
    jo.destroy()
    for o in (oo, ob):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    rows = [(int(out["custkey"][i]), int(out["orderkey"][i]),
             int(out["orderdate"][i]), int(out["totalprice"][i]),
             int(out["sum_qty"][i])) for i in range(len(out["orderkey"]))]
    rows.sort(key=lambda r: (-r[3], r[2], r[1]))
    return rows[:limit]


def q19(part: Page, li: Page):
    """Q19 discounted revenue (q19.sql): part attributes join (chained
    table, 3 u8 payloads) fused with the shipmode/shipinstruct/quantity
    pre-filter, then the three brand/container/size/qty disjuncts as
    twelve conjunctive keyless aggregations over the joined page (the
    OR decomposes per container id).  Exact 1e-4 ticks."""
    import numpy as np
    from .engine import lib

    bp = PlanHashBuild()
    bp.key_col = part.channel("partkey")
    bp.semijoin_table = -1
    bp.n_payload = 3
    bp.payload_col[0] = part.channel("brand")
    bp.payload_col[1] = part.channel("container")
    bp.payload_col[2] = part.channel("size")
    bp.capacity_hint = part.n_rows
    ob = Operator(OP_HASH_BUILD, bp)
    ob.add_input(part)
    ob.finish()

    jp = PlanLookupJoin()
    jp.table = ob.table()
    jp.n_preds = 3
    jp.preds[0] = Pred(li.channel("shipmode"), CMP_EQ, 1, 0.0)   # AIR
    jp.preds[1] = Pred(li.channel("shipinstruct"), CMP_EQ, 0, 0.0)
    jp.preds[2] = Pred(li.channel("quantity"), CMP_LE, 0, 30.0)
    jp.key_col = li.channel("partkey")
    jp.mode = 0
    jp.n_emit = 3
    jp.emit_probe_cols[0] = li.channel("quantity")
    jp.emit_probe_cols[1] = li.channel("extendedprice")
    jp.emit_probe_cols[2] = li.channel("discount")
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input(li)
    jpage = jo.get_output_raw()  # [qty, ep, dc, brand, container, size]

    DISJ = ((12, (0, 1, 5, 4), 1, 11, 5),
            (23, (18, 17, 21, 20), 10, 20, 10),
            (34, (8, 9, 13, 12), 20, 30, 15))
    total = 0
    for bnum, cset, qlo, qhi, szhi in DISJ:
        for c in cset:
            p = PlanHashAggSmall()
            p.n_preds = 6
            p.preds[0] = Pred(3, CMP_EQ, bnum, 0.0)
            p.preds[1] = Pred(4, CMP_EQ, c, 0.0)
            p.preds[2] = Pred(5, CMP_GE, 1, 0.0)
            p.preds[3] = Pred(5, CMP_LE, szhi, 0.0)
            p.preds[4] = Pred(0, CMP_GE, 0, float(qlo))
            p.preds[5] = Pred(0, CMP_LE, 0, float(qhi))
            p.n_keys = 0
            p.n_aggs = 1
            p.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, 1, 2, 0), 4)
            a = Operator(OP_HASH_AGG_SMALL, p)
            a.add_input_raw(jpage)
            a.finish()
            r = a.get_output(["hi", "lo"])
            if len(r["lo"]):
                total += (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))
            a.destroy()
    jo.destroy()
    lib().c.pg_table_destroy(ob.table())
    ob.destroy()
    return total


def q21(supp: Page, orders: Page, li: Page, limit=100):
    """Q21 suppliers who kept orders waiting (q21.sql).  The correlated
    EXISTS / NOT-EXISTS pair decomposes into per-order aggregates; all of
    them come out of ONE multi-accumulator fused-agg probe over lineitem
    (LOOKUP_JOIN mode 1, n_aggs=5 with per-aggregate FILTER predicates —
    the InMemoryHashAggregationBuilder + AggregationNode-mask analog):
      a0 = sum(suppkey)            over all lines of the order
      a1 = count                   where linestatus = 'F'
      a2 = sum(suppkey)            where receiptdate > commitdate (late)
      a3 = count                   where late
      a4 = sum(suppkey*suppkey)    where late  (raw integer product)
      cnt = count of all lines
    An order qualifies when every line is 'F' (a1 == cnt), it has a late
    line (a3 > 0), all late lines share one supplier (zero variance:
    a3*a4 == a2*a2, exact in f64 — both sides < 2^53), and some line has
    a different supplier (a2*cnt != a0*a3); s* = a2/a3 is that supplier.
    numwait(s) = sum of a3 over qualifying orders with s* = s (counting
    the l1 late lines, per q21.sql's count(*)), SAUDI ARABIA semijoin.
    Returns [(suppkey, numwait)] sorted (numwait desc, suppkey asc)
    LIMIT limit."""
    from .engine import lib
    tables = []

    # The probe takes NOTHING from orders (no payload — the table is only
    # the group-by domain, and every lineitem orderkey exists in orders by
    # FK): a RANGE-GROUP table over the orderkey domain replaces the whole
    # 150M-row build, and the orderkey-clustered lineitem makes the
    # accumulator atomics near-sequential instead of hash-scattered.
    b = PlanHashBuild()
    b.semijoin_table = -1
    b.capacity_hint = okey_max(orders.n_rows)
    b.range_group = 1
    otbl = Operator(OP_HASH_BUILD, b)
    otbl.finish()
    tables.append(otbl)

    sk = li.channel("suppkey")
    jp = PlanLookupJoin()
    jp.table = otbl.table()
    jp.key_col = li.channel("orderkey")
    jp.mode = 1
    jp.n_preds = 0  # preds[] holds the per-aggregate FILTER predicates
    jp.preds[0] = Pred(li.channel("linestatus"), CMP_EQ, ord("F"), 0.0)
    late = Pred(li.channel("receiptdate"), CMP_GT, 0, 0.0)
    late.rhs_col = li.channel("commitdate") + 1
    jp.preds[1] = late
    jp.n_aggs = 5
    jp.aggs[0] = Agg(AGG_SUM_I64, Proj(PROJ_IDENT, sk, 0, 0), 0)
    jp.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    jp.aggs[2] = Agg(AGG_SUM_I64, Proj(PROJ_IDENT, sk, 0, 0), 0)
    jp.aggs[3] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    jp.aggs[4] = Agg(AGG_SUM_I64, Proj(PROJ_MUL, sk, sk, 0), 0)
    for i, f in enumerate((-1, 0, 1, 1, 1)):
        jp.agg_filter[i] = f
    # accumulator packing: the probe is atomic-op-rate bound, and every
    # per-order total is bounded by integrity facts — at most 7 lineitems
    # per order (TPC-H PK) and suppkey < supplier count — so a0/a2 (sums
    # of suppkeys) and the three counts share ONE u64 word; only a4
    # (sum of squared suppkeys) needs its own.  One flush = 1-2 atomics
    # on 1-2 adjacent words instead of up to 6.
    wsum = (7 * supp.n_rows).bit_length()
    if 2 * wsum + 12 <= 64:
        jp.acc_pack = 1
        jp.acc_pack_shift[0] = 0           # a0: sum(suppkey) all
        jp.acc_pack_width[0] = wsum
        jp.acc_pack_shift[2] = wsum        # a2: sum(suppkey) late
        jp.acc_pack_width[2] = wsum
        jp.acc_pack_shift[1] = 2 * wsum    # a1: cnt F
        jp.acc_pack_width[1] = 4
        jp.acc_pack_shift[3] = 2 * wsum + 4  # a3: cnt late
        jp.acc_pack_width[3] = 4
        jp.acc_pack_shift[4] = -1          # a4: sum(sk^2) -> own word
        jp.acc_pack_cnt_shift = 2 * wsum + 8
        jp.acc_pack_cnt_width = 4
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input(li)
    jo.finish()
    pg = jo.get_output_raw()
    # pg: [ok(0), sum_all(1), cnt_F(2), sum_l(3), cnt_l(4), sq_l(5),
    #      cnt_all(6)]

    f1p = PlanFilterProject()
    f1p.n_preds = 2
    pf = Pred(2, CMP_EQ, 0, 0.0)   # every line 'F': cnt_F == cnt_all
    pf.rhs_col = 6 + 1
    f1p.preds[0] = pf
    f1p.preds[1] = Pred(4, CMP_GT, 0, 0.0)  # EXISTS late line
    f1p.n_proj = 6
    f1p.proj[0] = Proj(PROJ_DIV, 3, 4, 0)   # s* = sum_l / cnt_l
    f1p.proj[1] = Proj(PROJ_IDENT, 4, 0, 0)  # cnt_l
    f1p.proj[2] = Proj(PROJ_MUL, 4, 5, 0)   # m1 = cnt_l * sq_l  (<2^53)
    f1p.proj[3] = Proj(PROJ_MUL, 3, 3, 0)   # m2 = sum_l^2       (<2^53)
    f1p.proj[4] = Proj(PROJ_MUL, 3, 6, 0)   # m3 = sum_l * cnt_all
    f1p.proj[5] = Proj(PROJ_MUL, 1, 4, 0)   # m4 = sum_all * cnt_l
    f1 = Operator(OP_FILTER_PROJECT, f1p)
    f1.add_input_raw(pg)
    pe = f1.get_output_raw()  # [s*, cnt_l, m1, m2, m3, m4]

    bsa = PlanHashBuild()
    bsa.n_preds = 1
    bsa.preds[0] = Pred(supp.channel("nationkey"), CMP_EQ, 20, 0.0)
    bsa.key_col = supp.channel("suppkey")
    bsa.semijoin_table = -1
    bsa.capacity_hint = supp.n_rows
    bsa.key_set_only = 1
    bsa.dense_array = 1
    osa = Operator(OP_HASH_BUILD, bsa)
    osa.add_input(supp)
    osa.finish()
    tables.append(osa)

    f2p = PlanFilterProject()
    f2p.n_preds = 2
    p1 = Pred(2, CMP_EQ, 0, 0.0)  # zero variance: m1 == m2
    p1.rhs_col = 3 + 1
    f2p.preds[0] = p1
    p2 = Pred(4, CMP_NE, 0, 0.0)  # some different supplier: m3 != m4
    p2.rhs_col = 5 + 1
    f2p.preds[1] = p2
    f2p.n_proj = 2
    f2p.proj[0] = Proj(PROJ_IDENT, 0, 0, 0)
    f2p.proj[1] = Proj(PROJ_IDENT, 1, 0, 0)
    f2p.semijoin_table = osa.table()
    f2p.semijoin_col = 0
    f2 = Operator(OP_FILTER_PROJECT, f2p)
    f2.add_input_raw(pe)
    pf2 = f2.get_output_raw()  # [s*, cnt_l] qualifying orders

    bw = PlanHashBuild()
    bw.n_preds = 1
    bw.preds[0] = Pred(supp.channel("nationkey"), CMP_EQ, 20, 0.0)
    bw.key_col = supp.channel("suppkey")
    bw.semijoin_table = -1
    bw.capacity_hint = supp.n_rows + 64
    bw.agg_table = 1
    ow = Operator(OP_HASH_BUILD, bw)
    ow.add_input(supp)
    ow.finish()
    tables.append(ow)

    jw = PlanLookupJoin()
    jw.table = ow.table()
    jw.key_col = 0
    jw.mode = 1
    jw.proj = Proj(PROJ_IDENT, 1, 0, 0)
    jw.dec_scale = 0
    jw.dec_only = 1
    jn = Operator(OP_LOOKUP_JOIN, jw)
    jn.add_input_raw(pf2)
    jn.finish()
    out = jn.get_output(["suppkey", "numwait", "f64", "cnt"])
    jn.destroy()
    f2.destroy()
    f1.destroy()
    jo.destroy()
    for o in tables:
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    rows = sorted(
        ((int(out["suppkey"][i]), int(out["numwait"][i]))
         for i in range(len(out["suppkey"]))),
        key=lambda r: (-r[1], r[0]))
    return rows[:limit]


Q9_YEAR_BOUNDS = (8035, 8401, 8766, 9131, 9496, 9862, 10227, 10592)


def q9(part: Page, supp: Page, orders: Page, ps: Page, li: Page):
    """Q9 product-type profit (q09.sql): LIKE '%green%' as a VARBIN
    CONTAINS predicate building a dense part flag set; the partsupp
    composite-key (partkey, suppkey) lookup is an emit join on partkey
    followed by the suppkey equality (4 candidate suppliers per part);
    supplier-nation and order-year attach by payload emit joins; the
    (nation, year) grouping runs as per-group conjunctive aggregations
    of the two exact sums (revenue - supplycost*qty).  Returns a 25x7
    list of exact 1e-4 tick profits ([nation][year-1992])."""
    import numpy as np
    from .engine import lib

    bg = PlanHashBuild()
    bg.n_preds = 1
    pgreen = Pred(part.channel("name"), CMP_CONTAINS, 0, 0.0)
    pgreen.sval = b"green"
    pgreen.slen = 5
    bg.preds[0] = pgreen
    bg.key_col = part.channel("partkey")
    bg.semijoin_table = -1
    bg.capacity_hint = part.n_rows
    bg.key_set_only = 1
    bg.dense_array = 1
    og = Operator(OP_HASH_BUILD, bg)
    og.add_input(part)
    og.finish()

    fl = PlanFilterProject()
    fl.n_proj = 6
    for i, name in enumerate(("partkey", "suppkey", "orderkey", "quantity",
                              "extendedprice", "discount")):
        fl.proj[i] = Proj(PROJ_IDENT, li.channel(name), 0, 0)
    fl.semijoin_table = og.table()
    fl.semijoin_col = li.channel("partkey")
    f = Operator(OP_FILTER_PROJECT, fl)
    f.add_input(li)
    gli = f.get_output_raw()  # [pk, sk, ok, qty, ep, dc]

    bp = PlanHashBuild()
    bp.key_col = ps.channel("partkey")
    bp.semijoin_table = -1
    bp.n_payload = 2
    bp.payload_col[0] = ps.channel("suppkey")
    bp.payload_col[1] = ps.channel("supplycost")
    bp.capacity_hint = ps.n_rows
    ops_ = Operator(OP_HASH_BUILD, bp)
    ops_.add_input(ps)
    ops_.finish()

    j1 = PlanLookupJoin()
    j1.table = ops_.table()
    j1.key_col = 0
    j1.mode = 0
    j1.n_emit = 6
    for i in range(6):
        j1.emit_probe_cols[i] = i
    ja = Operator(OP_LOOKUP_JOIN, j1)
    ja.add_input_raw(gli)
    pa = ja.get_output_raw()  # [pk, sk, ok, qty, ep, dc, ps_sk, cost]

    fe = PlanFilterProject()
    fe.n_preds = 1
    pe = Pred(1, CMP_EQ, 0, 0.0)
    pe.rhs_col = 6 + 1
    fe.preds[0] = pe
    fe.n_proj = 6
    for i, c in enumerate((1, 2, 3, 4, 5, 7)):
        fe.proj[i] = Proj(PROJ_IDENT, c, 0, 0)
    f2 = Operator(OP_FILTER_PROJECT, fe)
    f2.add_input_raw(pa)
    pb = f2.get_output_raw()  # [sk, ok, qty, ep, dc, cost]

    bs = PlanHashBuild()
    bs.key_col = supp.channel("suppkey")
    bs.semijoin_table = -1
    bs.n_payload = 1
    bs.payload_col[0] = supp.channel("nationkey")
    bs.capacity_hint = supp.n_rows
    os_ = Operator(OP_HASH_BUILD, bs)
    os_.add_input(supp)
    os_.finish()

    j2 = PlanLookupJoin()
    j2.table = os_.table()
    j2.key_col = 0
    j2.mode = 0
    j2.n_emit = 5
    for i, c in enumerate((1, 2, 3, 4, 5)):
        j2.emit_probe_cols[i] = c
    jb = Operator(OP_LOOKUP_JOIN, j2)
    jb.add_input_raw(pb)
    pc = jb.get_output_raw()  # [ok, qty, ep, dc, cost, nat]

    # orderkey -> orderdate as a dense i32 array over the orderkey
    # range (2.4 GB at SF100): one scan + direct stores replace the
    # 150M-row chained insert, and each probe is one i32 load (presence
    # = nonzero payload; epoch-day dates are always nonzero)
    bo = PlanHashBuild()
    bo.key_col = orders.channel("orderkey")
    bo.semijoin_table = -1
    bo.n_payload = 1
    bo.payload_col[0] = orders.channel("orderdate")
    bo.capacity_hint = okey_max(orders.n_rows)
    bo.dense_array = 1
    oo = Operator(OP_HASH_BUILD, bo)
    oo.add_input(orders)
    oo.finish()

    j3 = PlanLookupJoin()
    j3.table = oo.table()
    j3.key_col = 0
    j3.mode = 0
    j3.n_emit = 5
    for i, c in enumerate((1, 2, 3, 4, 5)):
        j3.emit_probe_cols[i] = c
    jc = Operator(OP_LOOKUP_JOIN, j3)
    jc.add_input_raw(pc)
    pd = jc.get_output_raw()  # [qty, ep, dc, cost, nat, odate]

    profit = [[0] * 7 for _ in range(25)]
    # ONE general multi-channel group-by over (nation, orderdate) — the
    # MultiChannelGroupByHash analog replaces the former 25 nation-split
    # scans x 7 year aggregations (175 launches); the date groups fold
    # into years on the host (exact integer ticks end to end)
    gq = PlanGroupBy()
    gq.n_keys = 2
    gq.key_col[0] = 4  # nat
    gq.key_col[1] = 5  # odate
    gq.capacity_hint = 25 * 2500 + 1024
    gq.n_aggs = 2
    gq.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_DISC_PRICE, 1, 2, 0), 4)
    gq.aggs[1] = Agg(AGG_SUM_DEC, Proj(PROJ_MUL, 3, 0, 0), 4)
    gq.agg_filter[0] = -1
    gq.agg_filter[1] = -1
    gop = Operator(OP_GROUPBY_MULTI, gq)
    gop.add_input_raw(pd)
    gop.finish()
    gout = gop.get_output(["nat", "odate", "rev", "cost", "cnt"])
    gop.destroy()
    years = np.searchsorted(np.asarray(Q9_YEAR_BOUNDS[1:]),
                            np.asarray(gout["odate"]), side="right")
    for i in range(len(gout["nat"])):
        y = int(years[i])
        if y < 7:
            profit[int(gout["nat"][i])][y] += \
                int(gout["rev"][i]) - int(gout["cost"][i])

    jc.destroy()
    jb.destroy()
    f2.destroy()
    ja.destroy()
    f.destroy()
    for o in (og, ops_, os_, oo):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return profit


def q13(n_cust: int, orders, max_count=64):
    """Q13 customer distribution (q13.sql): the NOT LIKE
    '%special%requests%' comment filter runs as the ordered
    two-substring VARBIN predicate (CONTAINS2 negated); per-customer
    order counts come from a fused-agg probe over the full customer key
    table; the count histogram is a second fused-agg probe keyed by the
    count value.  The LEFT OUTER zero bucket is n_cust minus the
    customers with qualifying orders.  Returns [(c_count, custdist)]
    sorted (custdist desc, c_count desc).

    orders: one Page or an iterable of Pages — the o_comment
    VariableWidthBlock has int32 offsets (the reference's Slice cap), so
    beyond ~SF20 the orders table arrives as MULTIPLE pages, exactly as
    Presto's Driver would feed them; every operator here accumulates
    across addInput calls."""
    import numpy as np
    from .engine import lib
    order_pages = [orders] if isinstance(orders, Page) else list(orders)

    # the "table" is the keys 1..n_cust themselves: a range-group
    # domain (no build, no hash; acc index = custkey-1)
    bc = PlanHashBuild()
    bc.semijoin_table = -1
    bc.capacity_hint = n_cust
    bc.range_group = 1
    oc = Operator(OP_HASH_BUILD, bc)
    oc.finish()

    # count-only probe with a packed 1-word accumulator: per-customer
    # order counts are bounded far below 2^16, so count+cnt share one
    # u64 (one atomic per row into a ~120 MB L3-resident array)
    jp = PlanLookupJoin()
    jp.table = oc.table()
    jp.key_col = 0
    jp.mode = 1
    jp.n_aggs = 1
    jp.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    jp.agg_filter[0] = -1
    jp.acc_pack = 1
    jp.acc_pack_shift[0] = 0
    jp.acc_pack_width[0] = 16
    jp.acc_pack_cnt_shift = 16
    jp.acc_pack_cnt_width = 16
    jo = Operator(OP_LOOKUP_JOIN, jp)
    fp = PlanFilterProject()
    fp.n_preds = 1
    pr = Pred(order_pages[0].channel("comment"), CMP_NOT_CONTAINS2, 8, 0.0)
    pr.sval = b"special" + b"requests"
    pr.slen = 7
    fp.preds[0] = pr
    fp.n_proj = 1
    fp.proj[0] = Proj(PROJ_IDENT, order_pages[0].channel("custkey"), 0, 0)
    f = Operator(OP_FILTER_PROJECT, fp)
    for opg in order_pages:
        f.add_input(opg)
        okp = f.get_output_raw()  # [custkey] of qualifying orders
        jo.add_input_raw(okp)
    jo.finish()
    groups = jo.get_output_raw()  # [custkey, n_orders, cnt]
    n_with_orders = groups.n_rows

    bh = PlanHashBuild()
    bh.semijoin_table = -1
    bh.capacity_hint = max_count  # counts beyond max_count: range miss
    bh.range_group = 1
    oh = Operator(OP_HASH_BUILD, bh)
    oh.finish()

    jh = PlanLookupJoin()
    jh.table = oh.table()
    jh.key_col = 1  # the per-customer count
    jh.mode = 1
    jh.n_aggs = 1
    jh.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    jh.agg_filter[0] = -1
    jh.acc_pack = 1
    jh.acc_pack_shift[0] = 0
    jh.acc_pack_width[0] = 32
    jh.acc_pack_cnt_shift = 32
    jh.acc_pack_cnt_width = 32
    jo2 = Operator(OP_LOOKUP_JOIN, jh)
    jo2.add_input_raw(groups)
    jo2.finish()
    hist = jo2.get_output(["c_count", "n_cust", "custdist"])
    rows = [(int(hist["c_count"][i]), int(hist["custdist"][i]))
            for i in range(len(hist["c_count"]))]
    rows.append((0, n_cust - n_with_orders))
    rows.sort(key=lambda r: (-r[1], -r[0]))
    jo2.destroy()
    jo.destroy()
    f.destroy()
    for o in (oc, oh):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return rows


def q16(part: Page, ps: Page, supp: Page, type_name):
    """Q16 parts/supplier relationship (q16.sql).  The disjunctive part
    qualifiers (size IN 8 values, type NOT LIKE 'MEDIUM POLISHED%')
    build one dense flag set through repeated conjunctive fill passes;
    complaint suppliers come from the CONTAINS2 'Customer..Complaints'
    comment predicate (anti-semijoin); count(DISTINCT suppkey) per
    (brand, type, size) runs as composite-key (KEYSHL) fused-agg
    dedup + a second grouping probe.  type_name maps a type id to its
    display string for the final ORDER BY.  Returns
    [(brand, type_id, size, supplier_cnt)] in golden order."""
    import numpy as np
    from .engine import lib

    # qualifying-part flag set: the disjunction (8 sizes x NOT a type
    # range) decomposes into 16 conjunctive filter passes whose partkey
    # outputs feed ONE dense flag-set build (flags OR across add_input)
    SIZES = (49, 14, 23, 45, 19, 3, 36, 9)
    brandc = part.channel("brand")
    typec = part.channel("type_id")
    sizec = part.channel("size")
    qual_pages = []
    fops = []
    for v in SIZES:
        for trange in ((CMP_LE, 64), (CMP_GE, 70)):
            fp = PlanFilterProject()
            fp.n_preds = 3
            fp.preds[0] = Pred(sizec, CMP_EQ, v, 0.0)
            fp.preds[1] = Pred(brandc, CMP_NE, 45, 0.0)
            fp.preds[2] = Pred(typec, trange[0], trange[1], 0.0)
            fp.n_proj = 1
            fp.proj[0] = Proj(PROJ_IDENT, part.channel("partkey"), 0, 0)
            f = Operator(OP_FILTER_PROJECT, fp)
            f.add_input(part)
            qual_pages.append(f.get_output_raw())
            fops.append(f)
    bqual = PlanHashBuild()
    bqual.key_col = 0
    bqual.semijoin_table = -1
    bqual.capacity_hint = part.n_rows
    bqual.key_set_only = 1
    bqual.dense_array = 1
    oqual = Operator(OP_HASH_BUILD, bqual)
    for pg_ in qual_pages:
        oqual.add_input_raw(pg_)
    oqual.finish()
    for f in fops:
        f.destroy()

    # complaint suppliers: CONTAINS2('Customer','Complaints') flag set
    bc = PlanHashBuild()
    bc.n_preds = 1
    pc = Pred(supp.channel("comment"), CMP_CONTAINS2, 10, 0.0)
    pc.sval = b"Customer" + b"Complaints"
    pc.slen = 8
    bc.preds[0] = pc
    bc.key_col = supp.channel("suppkey")
    bc.semijoin_table = -1
    bc.capacity_hint = supp.n_rows
    bc.key_set_only = 1
    bc.dense_array = 1
    ocompl = Operator(OP_HASH_BUILD, bc)
    ocompl.add_input(supp)
    ocompl.finish()

    # part attributes keyed by partkey (qualifying parts only)
    ba = PlanHashBuild()
    ba.key_col = part.channel("partkey")
    ba.semijoin_table = oqual.table()
    ba.semijoin_col = part.channel("partkey")
    ba.bitmap_max_key = part.n_rows
    ba.n_payload = 3
    ba.payload_col[0] = brandc
    ba.payload_col[1] = typec
    ba.payload_col[2] = sizec
    ba.capacity_hint = part.n_rows
    oattr = Operator(OP_HASH_BUILD, ba)
    oattr.add_input(part)
    oattr.finish()

    # ps -> attributes emit join, then composite keys
    j1 = PlanLookupJoin()
    j1.table = oattr.table()
    j1.key_col = ps.channel("partkey")
    j1.mode = 0
    j1.n_emit = 1
    j1.emit_probe_cols[0] = ps.channel("suppkey")
    ja = Operator(OP_LOOKUP_JOIN, j1)
    ja.add_input(ps)
    pa = ja.get_output_raw()  # [sk, brand, type, size]

    f1p = PlanFilterProject()
    f1p.n_proj = 4
    f1p.proj[0] = Proj(PROJ_IDENT, 1, 0, 0)    # brand
    f1p.proj[1] = Proj(PROJ_IDENT, 2, 0, 0)    # type
    f1p.proj[2] = Proj(PROJ_IDENT, 3, 0, 0)    # size
    f1p.proj[3] = Proj(PROJ_IDENT, 0, 0, 0)    # sk
    f1p.semijoin_table = ocompl.table()
    f1p.semijoin_col = 0
    f1p.semijoin_anti = 1
    f1 = Operator(OP_FILTER_PROJECT, f1p)
    f1.add_input_raw(pa)
    pb = f1.get_output_raw()  # [brand, type, size, sk]

    # count(DISTINCT suppkey) per (brand, type, size) as two general
    # multi-channel group-bys (MultiChannelGroupByHash analog):
    # distinct (brand,type,size,sk) rows, then supplier count per group
    g1p = PlanGroupBy()
    g1p.n_keys = 4
    for i in range(4):
        g1p.key_col[i] = i
    g1p.capacity_hint = max(pb.n_rows, 1024)
    g1p.n_aggs = 1
    g1p.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    g1p.agg_filter[0] = -1
    g1 = Operator(OP_GROUPBY_MULTI, g1p)
    g1.add_input_raw(pb)
    g1.finish()
    pdist = g1.get_output_raw()  # [brand, type, size, sk, cnt, cnt]

    g2p = PlanGroupBy()
    g2p.n_keys = 3
    for i in range(3):
        g2p.key_col[i] = i
    g2p.capacity_hint = max(pdist.n_rows, 1024)
    g2p.n_aggs = 1
    g2p.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    g2p.agg_filter[0] = -1
    g2 = Operator(OP_GROUPBY_MULTI, g2p)
    g2.add_input_raw(pdist)
    g2.finish()
    out = g2.get_output(["brand", "type", "size", "scnt", "cnt"])

    rows = []
    for i in range(len(out["brand"])):
        rows.append((int(out["brand"][i]), int(out["type"][i]),
                     int(out["size"][i]), int(out["scnt"][i])))
    rows.sort(key=lambda r: (-r[3], r[0], type_name(r[1]), r[2]))
    for op_ in (g2, g1, f1, ja):
        op_.destroy()
    for o in (oqual, ocompl, oattr):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return rows


def q10(cust_n: int, orders: Page, li: Page, limit=20):
    """Q10 returned items (q10.sql): orders date filter fused into a
    chained build (payload custkey); returned lineitems emit-join to
    their customer; per-customer revenue by fused-agg probe; bounded
    top-`limit` (revenue desc, custkey asc) host-merged from the group
    page.  Returns [(custkey, revenue_1e4)]."""
    import numpy as np
    from .engine import lib

    bo = PlanHashBuild()
    bo.n_preds = 2
    bo.preds[0] = Pred(orders.channel("orderdate"), CMP_GE, 8674, 0.0)
    bo.preds[1] = Pred(orders.channel("orderdate"), CMP_LT, 8766, 0.0)
    bo.key_col = orders.channel("orderkey")
    bo.semijoin_table = -1
    bo.n_payload = 1
    bo.payload_col[0] = orders.channel("custkey")
    bo.capacity_hint = max(orders.n_rows // 8, 16)
    bo.bitmap_max_key = okey_max(orders.n_rows)
    oo = Operator(OP_HASH_BUILD, bo)
    oo.add_input(orders)
    oo.finish()

    # the customer group domain is the keys 1..cust_n themselves
    bc = PlanHashBuild()
    bc.semijoin_table = -1
    bc.capacity_hint = cust_n
    bc.range_group = 1
    oc = Operator(OP_HASH_BUILD, bc)
    oc.finish()

    # one fused pass: probe orders by orderkey, group the revenue by the
    # matched o_custkey payload into the customer table (mode 3)
    ja = PlanLookupJoin()
    ja.table = oo.table()
    ja.table2 = oc.table()
    ja.n_preds = 1
    ja.preds[0] = Pred(li.channel("returnflag"), CMP_EQ, ord("R"), 0.0)
    ja.key_col = li.channel("orderkey")
    ja.mode = 3
    ja.proj = Proj(PROJ_DISC_PRICE, li.channel("extendedprice"),
                   li.channel("discount"), 0)
    ja.dec_scale = 4
    ja.dec_only = 1
    j2 = Operator(OP_LOOKUP_JOIN, ja)
    j2.add_input(li)
    j2.finish()
    g = j2.get_output(["custkey", "rev", "f64", "cnt"])
    ck, rev = g["custkey"], g["rev"]
    nz = rev > 0
    ck, rev = ck[nz], rev[nz]
    if len(rev) > limit:  # O(n) preselect, ties kept for the exact sort
        kth = np.partition(rev, len(rev) - limit)[len(rev) - limit]
        keep = rev >= kth
        ck, rev = ck[keep], rev[keep]
    top = np.lexsort((ck, -rev))[:limit]
    rows = [(int(ck[i]), int(rev[i])) for i in top]
    j2.destroy()
    for o in (oo, oc):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return rows


def q15(supp: Page, li: Page):
    """Q15 top supplier (q15.sql): per-supplier revenue over the 1996-Q1
    window by fused-agg probe; the scalar max subquery resolves on the
    group page (output stage).  Returns [(suppkey, revenue_1e4)] of the
    max-revenue supplier(s), suppkey ascending."""
    from .engine import lib
    bs = PlanHashBuild()
    bs.key_col = supp.channel("suppkey")
    bs.semijoin_table = -1
    bs.capacity_hint = supp.n_rows + 64
    bs.agg_table = 1
    os_ = Operator(OP_HASH_BUILD, bs)
    os_.add_input(supp)
    os_.finish()

    jp = PlanLookupJoin()
    jp.table = os_.table()
    jp.n_preds = 2
    jp.preds[0] = Pred(li.channel("shipdate"), CMP_GE, 9496, 0.0)
    jp.preds[1] = Pred(li.channel("shipdate"), CMP_LT, 9587, 0.0)
    jp.key_col = li.channel("suppkey")
    jp.mode = 1
    jp.proj = Proj(PROJ_DISC_PRICE, li.channel("extendedprice"),
                   li.channel("discount"), 0)
    jp.dec_scale = 4
    jp.dec_only = 1
    jo = Operator(OP_LOOKUP_JOIN, jp)
    jo.add_input(li)
    jo.finish()
    g = jo.get_output(["suppkey", "rev", "f64", "cnt"])
    import numpy as np
    rev = np.asarray(g["rev"])
    rows = []
    if len(rev):
        mx = int(rev.max())
        if mx > 0:
            sel = np.nonzero(rev == mx)[0]
            sk = np.asarray(g["suppkey"])[sel]
            order = np.argsort(sk)
            rows = [(int(sk[i]), mx) for i in order]
    jo.destroy()
    lib().c.pg_table_destroy(os_.table())
    os_.destroy()
    return rows


def q20(part: Page, ps: Page, supp: Page, li: Page):
    """Q20 potential part promotion (q20.sql): 'forest%' as a VARBIN
    PREFIX flag set; per-(part,supplier) 1994 quantities via a
    composite-key (KEYSHL) fused-agg probe; the correlated
    availqty > 0.5*sum compare runs as 2*availqty > sum using a KEYSHL
    doubling against the group page's zero column (missing sums drop at
    the inner join — SQL's NULL comparison); CANADA(3) semijoin;
    distinct suppliers via a final fused-agg probe.  Returns qualifying
    suppkeys ascending."""
    from .engine import lib

    bf = PlanHashBuild()
    bf.n_preds = 1
    pf = Pred(part.channel("name"), CMP_PREFIX, 0, 0.0)
    pf.sval = b"forest"
    pf.slen = 6
    bf.preds[0] = pf
    bf.key_col = part.channel("partkey")
    bf.semijoin_table = -1
    bf.capacity_hint = part.n_rows
    bf.key_set_only = 1
    bf.dense_array = 1
    of = Operator(OP_HASH_BUILD, bf)
    of.add_input(part)
    of.finish()

    # forest 1994 lineitems -> composite (pk<<32|sk, qty)
    fl = PlanFilterProject()
    fl.n_preds = 2
    fl.preds[0] = Pred(li.channel("shipdate"), CMP_GE, 8766, 0.0)
    fl.preds[1] = Pred(li.channel("shipdate"), CMP_LT, 9131, 0.0)
    fl.n_proj = 2
    fl.proj[0] = Proj(PROJ_KEYSHL, li.channel("partkey"),
                      li.channel("suppkey"), 32)
    fl.proj[1] = Proj(PROJ_IDENT, li.channel("quantity"), 0, 0)
    fl.semijoin_table = of.table()
    fl.semijoin_col = li.channel("partkey")
    f1 = Operator(OP_FILTER_PROJECT, fl)
    f1.add_input(li)
    lq = f1.get_output_raw()  # [key, qty]

    bk = PlanHashBuild()
    bk.key_col = 0
    bk.semijoin_table = -1
    bk.capacity_hint = max(lq.n_rows, 1024)
    ok_ = Operator(OP_HASH_BUILD, bk)
    ok_.add_input_raw(lq)
    ok_.finish()

    jq = PlanLookupJoin()
    jq.table = ok_.table()
    jq.key_col = 0
    jq.mode = 1
    jq.proj = Proj(PROJ_IDENT, 1, 0, 0)
    jq.dec_scale = 0
    jq.dec_only = 1
    jo = Operator(OP_LOOKUP_JOIN, jq)
    jo.add_input_raw(lq)
    jo.finish()
    sums = jo.get_output_raw()  # [key, sum_qty, f64(zeros), cnt]

    # forest partsupp rows -> [key, availqty, suppkey]
    fp = PlanFilterProject()
    fp.n_proj = 3
    fp.proj[0] = Proj(PROJ_KEYSHL, ps.channel("partkey"),
                      ps.channel("suppkey"), 32)
    fp.proj[1] = Proj(PROJ_IDENT, ps.channel("availqty"), 0, 0)
    fp.proj[2] = Proj(PROJ_IDENT, ps.channel("suppkey"), 0, 0)
    fp.semijoin_table = of.table()
    fp.semijoin_col = ps.channel("partkey")
    f2 = Operator(OP_FILTER_PROJECT, fp)
    f2.add_input(ps)
    psq = f2.get_output_raw()

    bs = PlanHashBuild()
    bs.key_col = 0
    bs.semijoin_table = -1
    bs.n_payload = 2
    bs.payload_col[0] = 1  # sum_qty
    bs.payload_col[1] = 2  # the zero column (for the KEYSHL doubling)
    bs.capacity_hint = max(sums.n_rows, 1024)
    osum = Operator(OP_HASH_BUILD, bs)
    osum.add_input_raw(sums)
    osum.finish()

    je = PlanLookupJoin()
    je.table = osum.table()
    je.key_col = 0
    je.mode = 0
    je.n_emit = 2
    je.emit_probe_cols[0] = 1  # availqty
    je.emit_probe_cols[1] = 2  # suppkey
    j3 = Operator(OP_LOOKUP_JOIN, je)
    j3.add_input_raw(psq)
    pj = j3.get_output_raw()  # [aq, sk, sum, zeros]

    bn = PlanHashBuild()
    bn.n_preds = 1
    bn.preds[0] = Pred(supp.channel("nationkey"), CMP_EQ, 3, 0.0)
    bn.key_col = supp.channel("suppkey")
    bn.semijoin_table = -1
    bn.capacity_hint = supp.n_rows
    bn.key_set_only = 1
    bn.dense_array = 1
    ocan = Operator(OP_HASH_BUILD, bn)
    ocan.add_input(supp)
    ocan.finish()

    f3p = PlanFilterProject()
    f3p.n_proj = 3
    f3p.proj[0] = Proj(PROJ_KEYSHL, 0, 3, 1)  # 2*availqty (zeros low bit)
    f3p.proj[1] = Proj(PROJ_IDENT, 1, 0, 0)   # suppkey
    f3p.proj[2] = Proj(PROJ_IDENT, 2, 0, 0)   # sum
    f3p.semijoin_table = ocan.table()
    f3p.semijoin_col = 1
    f3 = Operator(OP_FILTER_PROJECT, f3p)
    f3.add_input_raw(pj)
    pk2 = f3.get_output_raw()  # [2aq, sk, sum]

    f4p = PlanFilterProject()
    f4p.n_preds = 1
    p4 = Pred(0, CMP_GT, 0, 0.0)
    p4.rhs_col = 2 + 1
    f4p.preds[0] = p4
    f4p.n_proj = 1
    f4p.proj[0] = Proj(PROJ_IDENT, 1, 0, 0)
    f4 = Operator(OP_FILTER_PROJECT, f4p)
    f4.add_input_raw(pk2)
    excess = f4.get_output_raw()  # [sk] with dupes

    bq = PlanHashBuild()
    bq.key_col = supp.channel("suppkey")
    bq.semijoin_table = -1
    bq.capacity_hint = supp.n_rows + 64
    bq.agg_table = 1
    oq = Operator(OP_HASH_BUILD, bq)
    oq.add_input(supp)
    oq.finish()
    jd = PlanLookupJoin()
    jd.table = oq.table()
    jd.key_col = 0
    jd.mode = 1
    jd.proj = Proj(PROJ_IDENT, 0, 0, 0)
    jd.dec_scale = 0
    jd.dec_only = 1
    j5 = Operator(OP_LOOKUP_JOIN, jd)
    j5.add_input_raw(excess)
    j5.finish()
    out = j5.get_output(["suppkey", "s", "f", "c"])
    sks = sorted(int(v) for v in out["suppkey"])
    for o in (j5, f4, f3, j3, f2, jo, f1):
        o.destroy()
    for o in (of, ok_, osum, ocan, oq):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return sks


def q2(part: Page, ps: Page, supp: Page, s_abal, s_nat, limit=100):
    """Q2 minimum-cost supplier (q02.sql): size-15 '%BRASS' parts (the
    type%5 disjunction as 30 conjunctive flag-set fills), EUROPE
    suppliers, per-part MIN supplycost via a dec_min fused-agg probe,
    then an equality join back to emit the (supplier, part) pairs.  The
    final ORDER BY (acctbal desc, nation name, supplier, part) resolves
    on the host output stage from the pinned streams.  Returns
    [(suppkey, partkey)] in golden order."""
    import numpy as np
    from .engine import lib

    # qualifying parts: size == 15 AND type % 5 == 2 ('%BRASS')
    qual_pages = []
    fops = []
    for t in range(2, 150, 5):
        fp = PlanFilterProject()
        fp.n_preds = 2
        fp.preds[0] = Pred(part.channel("size"), CMP_EQ, 15, 0.0)
        fp.preds[1] = Pred(part.channel("type_id"), CMP_EQ, t, 0.0)
        fp.n_proj = 1
        fp.proj[0] = Proj(PROJ_IDENT, part.channel("partkey"), 0, 0)
        f = Operator(OP_FILTER_PROJECT, fp)
        f.add_input(part)
        qual_pages.append(f.get_output_raw())
        fops.append(f)
    bq = PlanHashBuild()
    bq.key_col = 0
    bq.semijoin_table = -1
    bq.capacity_hint = part.n_rows
    bq.key_set_only = 1
    bq.dense_array = 1
    oq = Operator(OP_HASH_BUILD, bq)
    for pg_ in qual_pages:
        oq.add_input_raw(pg_)
    oq.finish()
    for f in fops:
        f.destroy()

    # EUROPE supplier flag set: nation-key set semijoin (region 3
    # nations: FRANCE, GERMANY, ROMANIA, RUSSIA, UNITED KINGDOM)
    natp = Page({"nationkey": np.asarray((6, 7, 19, 22, 23),
                                         dtype=np.int64)})
    bn = PlanHashBuild()
    bn.key_col = 0
    bn.semijoin_table = -1
    bn.capacity_hint = 32
    bn.key_set_only = 1
    on = Operator(OP_HASH_BUILD, bn)
    on.add_input(natp)
    on.finish()
    feu = PlanFilterProject()
    feu.n_proj = 1
    feu.proj[0] = Proj(PROJ_IDENT, supp.channel("suppkey"), 0, 0)
    feu.semijoin_table = on.table()
    feu.semijoin_col = supp.channel("nationkey")
    fe = Operator(OP_FILTER_PROJECT, feu)
    fe.add_input(supp)
    eup = fe.get_output_raw()
    bf = PlanHashBuild()
    bf.key_col = 0
    bf.semijoin_table = -1
    bf.capacity_hint = supp.n_rows
    bf.key_set_only = 1
    bf.dense_array = 1
    oeu = Operator(OP_HASH_BUILD, bf)
    oeu.add_input_raw(eup)
    oeu.finish()
    fe.destroy()

    # partsupp restricted to qualifying parts AND european suppliers
    f1p = PlanFilterProject()
    f1p.n_proj = 3
    f1p.proj[0] = Proj(PROJ_IDENT, ps.channel("partkey"), 0, 0)
    f1p.proj[1] = Proj(PROJ_IDENT, ps.channel("suppkey"), 0, 0)
    f1p.proj[2] = Proj(PROJ_IDENT, ps.channel("supplycost"), 0, 0)
    f1p.semijoin_table = oq.table()
    f1p.semijoin_col = ps.channel("partkey")
    f1 = Operator(OP_FILTER_PROJECT, f1p)
    f1.add_input(ps)
    pa = f1.get_output_raw()
    f2p = PlanFilterProject()
    f2p.n_proj = 3
    for i in range(3):
        f2p.proj[i] = Proj(PROJ_IDENT, i, 0, 0)
    f2p.semijoin_table = oeu.table()
    f2p.semijoin_col = 1
    f2 = Operator(OP_FILTER_PROJECT, f2p)
    f2.add_input_raw(pa)
    pb = f2.get_output_raw()  # [pk, sk, cost_cents]

    # per-part MIN supplycost
    bm = PlanHashBuild()
    bm.key_col = part.channel("partkey")
    bm.semijoin_table = oq.table()
    bm.semijoin_col = part.channel("partkey")
    bm.capacity_hint = max(part.n_rows // 64, 4096)
    bm.agg_table = 1
    bm.bitmap_max_key = part.n_rows
    om = Operator(OP_HASH_BUILD, bm)
    om.add_input(part)
    om.finish()

    jm = PlanLookupJoin()
    jm.table = om.table()
    jm.key_col = 0
    jm.mode = 1
    jm.proj = Proj(PROJ_IDENT, 2, 0, 0)
    jm.dec_scale = 0
    jm.dec_min = 1
    j1 = Operator(OP_LOOKUP_JOIN, jm)
    j1.add_input_raw(pb)
    j1.finish()
    mins = j1.get_output_raw()  # [pk, min_cost, f64, cnt]

    bmm = PlanHashBuild()
    bmm.key_col = 0
    bmm.semijoin_table = -1
    bmm.n_payload = 1
    bmm.payload_col[0] = 1
    bmm.capacity_hint = max(mins.n_rows, 16)
    omm = Operator(OP_HASH_BUILD, bmm)
    omm.add_input_raw(mins)
    omm.finish()

    je = PlanLookupJoin()
    je.table = omm.table()
    je.key_col = 0
    je.mode = 0
    je.n_emit = 3
    for i in range(3):
        je.emit_probe_cols[i] = i
    j2 = Operator(OP_LOOKUP_JOIN, je)
    j2.add_input_raw(pb)
    pc = j2.get_output_raw()  # [pk, sk, cost, min]

    f3p = PlanFilterProject()
    f3p.n_preds = 1
    p3 = Pred(2, CMP_EQ, 0, 0.0)
    p3.rhs_col = 3 + 1
    f3p.preds[0] = p3
    f3p.n_proj = 2
    f3p.proj[0] = Proj(PROJ_IDENT, 1, 0, 0)
    f3p.proj[1] = Proj(PROJ_IDENT, 0, 0, 0)
    f3 = Operator(OP_FILTER_PROJECT, f3p)
    f3.add_input_raw(pc)
    out = f3.get_output(["suppkey", "partkey"])

    rows = sorted(
        ((int(out["suppkey"][i]), int(out["partkey"][i]))
         for i in range(len(out["suppkey"]))),
        key=lambda r: (-int(s_abal[r[0] - 1]),
                       NATION_NAMES[int(s_nat[r[0] - 1])], r[0], r[1]))
    for o in (f3, j2, j1, f2, f1):
        o.destroy()
    for o in (oq, on, oeu, om, omm):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    return rows[:limit]


NATION_NAMES = [
    "ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
    "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
    "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA", "ROMANIA",
    "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM", "UNITED STATES"]


Q22_CODE_NATIONS = (3, 7, 8, 13, 19, 20, 21)  # codes '13'..'31' ascending


def q22(cust: Page, orders: Page):
    """Q22 global sales opportunity (q22.sql): customers of the 7 phone
    country codes (code = nationkey+10) with above-average positive
    balance and no orders (anti-semijoin NOT-EXISTS pushdown).  The
    population average becomes a strict plan constant
    floor(sum/cnt).  Returns (counts[7], sums_cents[7]) in code order."""
    import numpy as np
    from .engine import lib

    natp = Page({"nationkey": np.asarray(Q22_CODE_NATIONS, dtype=np.int64)})
    bn = PlanHashBuild()
    bn.key_col = 0
    bn.semijoin_table = -1
    bn.capacity_hint = 32
    bn.key_set_only = 1
    on = Operator(OP_HASH_BUILD, bn)
    on.add_input(natp)
    on.finish()

    abc = cust.channel("acctbal")
    # positive-balance population aggregate (sum, count)
    fp = PlanFilterProject()
    fp.n_preds = 1
    fp.preds[0] = Pred(abc, CMP_GT, 0, 0.0)
    fp.n_proj = 1
    fp.proj[0] = Proj(PROJ_IDENT, abc, 0, 0)
    fp.semijoin_table = on.table()
    fp.semijoin_col = cust.channel("nationkey")
    f0 = Operator(OP_FILTER_PROJECT, fp)
    f0.add_input(cust)
    ppage = f0.get_output_raw()

    ap = PlanHashAggSmall()
    ap.n_keys = 0
    ap.n_aggs = 2
    ap.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_IDENT, 0, 0, 0), 0)
    ap.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    a0 = Operator(OP_HASH_AGG_SMALL, ap)
    a0.add_input_raw(ppage)
    a0.finish()
    r = a0.get_output(["hi", "lo", "cnt"])
    sum_pos = (int(r["hi"][0]) << 64) | int(np.uint64(r["lo"][0]))
    cnt_pos = int(r["cnt"][0])
    a0.destroy()
    f0.destroy()

    bo = PlanHashBuild()
    bo.key_col = orders.channel("custkey")
    bo.semijoin_table = -1
    bo.capacity_hint = orders.n_rows
    bo.key_set_only = 1
    oo = Operator(OP_HASH_BUILD, bo)
    oo.add_input(orders)
    oo.finish()

    f1p = PlanFilterProject()
    f1p.n_preds = 1
    f1p.preds[0] = Pred(abc, CMP_GT, sum_pos // cnt_pos, 0.0)
    f1p.n_proj = 3
    f1p.proj[0] = Proj(PROJ_IDENT, cust.channel("custkey"), 0, 0)
    f1p.proj[1] = Proj(PROJ_IDENT, cust.channel("nationkey"), 0, 0)
    f1p.proj[2] = Proj(PROJ_IDENT, abc, 0, 0)
    f1p.semijoin_table = on.table()
    f1p.semijoin_col = cust.channel("nationkey")
    f1 = Operator(OP_FILTER_PROJECT, f1p)
    f1.add_input(cust)
    page1 = f1.get_output_raw()

    f2p = PlanFilterProject()
    f2p.n_proj = 2
    f2p.proj[0] = Proj(PROJ_IDENT, 1, 0, 0)
    f2p.proj[1] = Proj(PROJ_IDENT, 2, 0, 0)
    f2p.semijoin_table = oo.table()
    f2p.semijoin_col = 0
    f2p.semijoin_anti = 1
    f2 = Operator(OP_FILTER_PROJECT, f2p)
    f2.add_input_raw(page1)
    page2 = f2.get_output_raw()  # [nationkey, acctbal]

    ag = PlanHashAggSmall()
    ag.n_keys = 1
    ag.key_col[0] = 0
    ag.n_vals[0] = len(Q22_CODE_NATIONS)
    for i, v in enumerate(Q22_CODE_NATIONS):
        ag.key_vals[0][i] = v
    ag.n_aggs = 2
    ag.aggs[0] = Agg(AGG_SUM_DEC, Proj(PROJ_IDENT, 1, 0, 0), 0)
    ag.aggs[1] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    a = Operator(OP_HASH_AGG_SMALL, ag)
    a.add_input_raw(page2)
    a.finish()
    out = a.get_output(["nationkey", "hi", "lo", "cnt"])
    a.destroy()
    f2.destroy()
    f1.destroy()
    for o in (on, oo):
        lib().c.pg_table_destroy(o.table())
        o.destroy()
    cnts = [0] * len(Q22_CODE_NATIONS)
    sums = [0] * len(Q22_CODE_NATIONS)
    for i in range(len(out["nationkey"])):
        j = Q22_CODE_NATIONS.index(int(out["nationkey"][i]))
        cnts[j] = int(out["cnt"][i])
        sums[j] = (int(out["hi"][i]) << 64) | int(np.uint64(out["lo"][i]))
    return cnts, sums


def q4(orders: Page, li_dates: Page):
    """Q4 order-priority checking (q04.sql): EXISTS(lineitem with
    commitdate < receiptdate) as a key-set build with a col-vs-col
    predicate, then orders filter (date range + semijoin) and a 5-value
    priority COUNT aggregation.  Returns counts per priority 0..4."""
    bs = PlanHashBuild()
    bs.n_preds = 1
    p = Pred(li_dates.channel("commitdate"), CMP_LT, 0, 0.0)
    p.rhs_col = li_dates.channel("receiptdate") + 1
    bs.preds[0] = p
    bs.key_col = li_dates.channel("orderkey")
    bs.semijoin_table = -1
    # dense flag set over the orderkey range (one byte per possible key,
    # ~600 MB at SF100): one predicated scan + flag stores replaces the
    # 380M-key chained set insert, and the semijoin probe is one byte
    # load (all lineitem orderkeys are in range by FK)
    bs.capacity_hint = okey_max(orders.n_rows)
    bs.key_set_only = 1
    bs.dense_array = 1
    b = Operator(OP_HASH_BUILD, bs)
    b.add_input(li_dates)
    b.finish()

    fp = PlanFilterProject()
    fp.n_preds = 2
    fp.preds[0] = Pred(orders.channel("orderdate"), CMP_GE, 8582, 0.0)
    fp.preds[1] = Pred(orders.channel("orderdate"), CMP_LT, 8674, 0.0)
    fp.n_proj = 1
    fp.proj[0] = Proj(PROJ_IDENT, orders.channel("priority"), 0, 0)
    fp.semijoin_table = b.table()
    fp.semijoin_col = orders.channel("orderkey")
    f = Operator(OP_FILTER_PROJECT, fp)
    f.add_input(orders)
    fpage = f.get_output_raw()

    ag = PlanHashAggSmall()
    ag.n_keys = 1
    ag.key_col[0] = 0
    ag.n_vals[0] = 5
    for i in range(5):
        ag.key_vals[0][i] = i
    ag.n_aggs = 1
    ag.aggs[0] = Agg(AGG_COUNT, Proj(PROJ_IDENT, 0, 0, 0), 0)
    a = Operator(OP_HASH_AGG_SMALL, ag)
    a.add_input_raw(fpage)
    a.finish()
    out = a.get_output(["priority", "count"])
    a.destroy()
    f.destroy()
    from .engine import lib
    lib().c.pg_table_destroy(b.table())
    b.destroy()
    counts = [0] * 5
    for i in range(len(out["priority"])):
        counts[int(out["priority"][i])] = int(out["count"][i])
    return counts
