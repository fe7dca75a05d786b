"""Multi-GPU exchange — the RCCL-over-xGMI analog of Presto's repartition
seam (PartitionedOutputOperator -> OutputBuffer -> HTTP pull ->
ExchangeOperator, SURVEY.md §3 stack 4): partition ids computed on-GPU with
the reference's partition math (fixed128.h pg_partition), bucketed columnar
split by the PARTITION operator, then torch.distributed all_to_all_single
(backend "nccl" == RCCL on ROCm) of the column buffers — no serialize/HTTP
hop.  Backend-agnostic (gloo on CPU for tests)."""
import torch
import torch.distributed as dist

# cumulative bytes this process moved through all_to_all exchanges — the
# xGMI-crossing volume (read by bench.py; compare against SURVEY.md §8d
# config 4's estimate, e.g. ~10.8 GB total for SF300 Q3)
EXCHANGE_BYTES = 0


def exchange_bytes_reset():
    global EXCHANGE_BYTES
    EXCHANGE_BYTES = 0


def exchange_split_counts(send_counts):
    """all_to_all of row counts: returns recv_counts (rows arriving from
    each rank)."""
    world = dist.get_world_size()
    send = torch.tensor(send_counts, dtype=torch.int64)
    allc = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
    dist.all_gather(allc, send)
    me = dist.get_rank()
    return [int(allc[r][me]) for r in range(world)]


def exchange_columns(cols, send_counts, recv_counts=None, device=None):
    """cols: dict name -> 1-D tensor laid out partition-major (rows for rank
    0 first, then rank 1, ...), with send_counts[r] rows per destination.
    Returns dict of received tensors (concatenated in source-rank order,
    stable within each source rank)."""
    if recv_counts is None:
        recv_counts = exchange_split_counts(send_counts)
    out = {}
    n_recv = sum(recv_counts)
    global EXCHANGE_BYTES
    me = dist.get_rank()
    for name, t in cols.items():
        r = torch.empty(n_recv, dtype=t.dtype,
                        device=device if device is not None else t.device)
        dist.all_to_all_single(r, t.contiguous(),
                               output_split_sizes=recv_counts,
                               input_split_sizes=list(send_counts))
        out[name] = r
        # bytes leaving this rank for OTHER ranks (xGMI-crossing)
        EXCHANGE_BYTES += (t.numel() - send_counts[me]) * t.element_size()
    return out


def _col_to_torch(col, n, device):
    """Copy a library-owned device column (pg_col) into a torch tensor."""
    import numpy as np  # noqa: F401
    from .engine import lib
    dtmap = {0: torch.uint8, 1: torch.int32, 2: torch.int64,
             3: torch.float64}
    t = torch.empty(n, dtype=dtmap[col.tag], device=device)
    nbytes = n * t.element_size()
    if nbytes:
        L = lib()
        if col.on_device:
            L.check(L.c.pg_memcpy_d2d(t.data_ptr(), col.data, nbytes), "d2d")
        else:
            import ctypes
            src = (ctypes.c_uint8 * nbytes).from_address(col.data)
            t.copy_(torch.frombuffer(bytearray(src), dtype=t.dtype))
    return t


def _gather_padded(t, cap, device, pad_value=0):
    """all_gather of a variable-length 1-D tensor, padded to cap."""
    world = dist.get_world_size()
    buf = torch.full((cap,), pad_value, dtype=t.dtype, device=device)
    buf[:t.numel()] = t
    outs = [torch.empty_like(buf) for _ in range(world)]
    dist.all_gather(outs, buf)
    return outs


def q3_distributed(cust_page, ord_page, li_page, world, rank, device,
                   limit=10, mode="dec", okey_bound=0):
    """Distributed Q3 (BASELINE config 4): the repartition seam over RCCL.

    customer: local filter (mktsegment='BUILDING') -> broadcast key set
      (all_gather of selected custkeys — the BroadcastOutputBuffer analog)
    orders:   local filter (orderdate<9204) -> PARTITION by orderkey hash
      -> all_to_all -> local HASH_BUILD with fused semijoin vs the set
    lineitem: local filter (shipdate>9204) -> PARTITION by orderkey hash
      -> all_to_all -> local LOOKUP_JOIN fused grouped sum -> local TopN
      -> gather candidates -> final TopN on rank 0.

    world==1 runs the same operator graph with the collectives as
    identities (used by the N=1 parity test).
    Returns dict of numpy arrays on rank 0, None elsewhere.
    """
    import numpy as np
    import presto_amd as P
    from .engine import lib

    use_dist = world > 1

    # ---- customer set ----
    fp = P.PlanFilterProject()
    fp.n_preds = 1
    fp.preds[0] = P.Pred(cust_page.channel("mktseg"), P.CMP_EQ, 1, 0.0)
    fp.n_proj = 1
    fp.proj[0] = P.Proj(P.PROJ_IDENT, cust_page.channel("custkey"), 0, 0)
    f = P.Operator(P.OP_FILTER_PROJECT, fp)
    f.add_input(cust_page)
    raw = f.get_output_raw()
    ck = _col_to_torch(raw.cols[0], raw.n_rows, device)
    f.destroy()
    if use_dist:
        # cap must agree across ranks (shards differ by +-1 row)
        capt = torch.tensor([int(cust_page.n_rows)], device=device)
        dist.all_reduce(capt, op=dist.ReduceOp.MAX)
        parts = _gather_padded(ck, int(capt.item()), device)
        ck = torch.cat(parts)
    bp = P.PlanHashBuild()
    bp.n_preds = 1
    bp.preds[0] = P.Pred(0, P.CMP_GE, 1, 0.0)  # drop the 0 padding
    bp.key_col = 0
    bp.semijoin_table = -1
    bp.n_payload = 0
    # dense membership flags sized to the global customer cardinality
    bp.capacity_hint = cust_page.n_rows * world + world
    bp.key_set_only = 1
    bp.dense_array = 1
    b1 = P.Operator(P.OP_HASH_BUILD, bp)
    b1.add_input(P.Page({"custkey": ck}))
    b1.finish()
    set_tbl = b1.table()

    def filter_partition_exchange(page, preds, emit_channels):
        """filter -> partition by col0 of the emitted page -> exchange."""
        fpl = P.PlanFilterProject()
        fpl.n_preds = len(preds)
        for i, pr in enumerate(preds):
            fpl.preds[i] = pr
        fpl.n_proj = len(emit_channels)
        for i, ch in enumerate(emit_channels):
            fpl.proj[i] = P.Proj(P.PROJ_IDENT, ch, 0, 0)
        fo = P.Operator(P.OP_FILTER_PROJECT, fpl)
        fo.add_input(page)
        fraw = fo.get_output_raw()
        pp = P.PlanPartition()
        pp.n_partitions = world
        pp.key_col = 0
        pp.n_emit = len(emit_channels)
        for i in range(len(emit_channels)):
            pp.emit_cols[i] = i
        po = P.Operator(P.OP_PARTITION, pp)
        po.add_input_raw(fraw)
        fo.destroy()
        counts = po.partition_counts(world)
        # concatenate partition slices into per-column send tensors
        pages = [po.get_output_raw() for _ in range(world)]
        total = sum(counts)
        cols = {}
        for c in range(len(emit_channels)):
            col0 = pages[0].cols[c]
            dtmap = {0: torch.uint8, 1: torch.int32, 2: torch.int64,
                     3: torch.float64}
            t = torch.empty(total, dtype=dtmap[col0.tag], device=device)
            off = 0
            for p in range(world):
                nrow = pages[p].n_rows
                if nrow:
                    nb = nrow * t.element_size()
                    L = lib()
                    L.check(L.c.pg_memcpy_d2d(
                        t.data_ptr() + off * t.element_size(),
                        pages[p].cols[c].data, nb), "d2d")
                off += nrow
            cols[f"c{c}"] = t
        po.destroy()
        if use_dist:
            return exchange_columns(cols, counts, device=device)
        return cols

    # ---- orders ----
    ocols = filter_partition_exchange(
        ord_page,
        [P.Pred(ord_page.channel("orderdate"), P.CMP_LT, 9204, 0.0)],
        [ord_page.channel("orderkey"), ord_page.channel("orderdate"),
         ord_page.channel("custkey")])
    b2p = P.PlanHashBuild()
    b2p.n_preds = 0
    b2p.key_col = 0
    b2p.semijoin_table = set_tbl
    b2p.semijoin_col = 2
    b2p.n_payload = 1
    b2p.payload_col[0] = 1
    b2p.capacity_hint = max(int(ocols["c0"].numel()) // 4, 16)
    b2p.agg_table = 1
    b2p.bitmap_max_key = okey_bound  # global orderkey range (keys are
    # hash-partitioned, so every rank sees global-range keys); 0 = off
    b2 = P.Operator(P.OP_HASH_BUILD, b2p)
    b2.add_input(P.Page({k: v for k, v in ocols.items()}))
    b2.finish()
    tbl = b2.table()

    # ---- lineitem probe ----
    lcols = filter_partition_exchange(
        li_page,
        [P.Pred(li_page.channel("shipdate"), P.CMP_GT, 9204, 0.0)],
        [li_page.channel("orderkey"), li_page.channel("extendedprice"),
         li_page.channel("discount")])
    jp = P.PlanLookupJoin()
    jp.table = tbl
    jp.n_preds = 0
    jp.key_col = 0
    jp.mode = 1
    jp.proj = P.Proj(P.PROJ_DISC_PRICE, 1, 2, 0)
    jp.dec_scale = 4
    j = P.Operator(P.OP_LOOKUP_JOIN, jp)
    j.add_input(P.Page({k: v for k, v in lcols.items()}))
    j.finish()
    groups = j.get_output_raw()
    tp = P.PlanTopN()
    tp.limit = limit
    tp.val_col = 2 if mode == "dec" else 3
    tp.date_col = 1
    tp.key_col = 0
    t = P.Operator(P.OP_TOPN, tp)
    t.add_input_raw(groups)
    t.finish()
    out = t.get_output(["orderkey", "rev", "orderdate"])
    t.destroy()
    j.destroy()
    lib().c.pg_table_destroy(set_tbl)
    lib().c.pg_table_destroy(tbl)
    b1.destroy()
    b2.destroy()

    if not use_dist:
        return out
    # gather per-rank candidates, final TopN on rank 0
    def pad(a, dt):
        x = torch.full((limit,), -1, dtype=dt, device=device)
        x[:len(a)] = torch.from_numpy(np.ascontiguousarray(a)).to(device)
        return x
    ks = _gather_padded(pad(out["orderkey"], torch.int64), limit, device, -1)
    if mode == "dec":
        vs = _gather_padded(pad(out["rev"], torch.int64), limit, device, -1)
    else:
        vs = _gather_padded(pad(out["rev"].view(np.int64), torch.int64),
                            limit, device, -1)
    ds = _gather_padded(pad(out["orderdate"], torch.int32), limit, device, -1)
    if rank != 0:
        return None
    rows = []
    for r in range(world):
        k = ks[r].cpu().numpy()
        v = vs[r].cpu().numpy()
        d = ds[r].cpu().numpy()
        for i in range(limit):
            if k[i] >= 0:
                rows.append((v[i], d[i], k[i]))
    if mode == "dec":
        rows.sort(key=lambda x: (-x[0], x[1], x[2]))
    else:
        rows.sort(key=lambda x: (-np.int64(x[0]).view(np.float64),
                                 x[1], x[2]))
    rows = rows[:limit]
    res = dict(orderkey=np.array([r[2] for r in rows], np.int64),
               rev=np.array([r[0] for r in rows], np.int64),
               orderdate=np.array([r[1] for r in rows], np.int32))
    if mode != "dec":
        res["rev"] = res["rev"].view(np.float64)
    return res


def q5_distributed(cust_page, ord_page, supp_page, li_page, world, rank,
                   device, sf_hint=None, okey_bound=0):
    """Distributed Q5 (BASELINE config 4): the nation dimensions are
    replicated (all_reduce of the dense custkey/suppkey -> nationkey
    arrays — the broadcast-join analog), orders and lineitem are
    repartitioned by orderkey hash over RCCL all_to_all, then each rank
    runs the fused single-pass probe on its partition; the five per-nation
    tick sums combine exactly with one small all_reduce.
    Returns dict {nationkey: revenue_1e4 ticks} on rank 0 (None elsewhere);
    world == 1 runs the same graph with collectives as identities."""
    import numpy as np
    import presto_amd as P
    from .engine import lib
    from .pipelines import Q5Pipeline

    use_dist = world > 1

    def global_dense(page, key_name, val_name, n_all):
        """u8 dense array key->val across all ranks (keys are dense 1..n,
        each rank holds a disjoint shard)."""
        full = torch.zeros(int(n_all), dtype=torch.uint8, device=device)
        k = page.cols[key_name]
        v = page.cols[val_name]
        full[k - 1] = v
        if use_dist:
            dist.all_reduce(full)  # disjoint shards -> sum == union
        return full

    # callers may set .n_total (global table cardinality) on the dimension
    # pages; otherwise derive it exactly as all_reduce(MAX) of the local
    # max key (dense 1..n keys), which is correct for uneven shards too
    def global_cardinality(page, key_name):
        n = getattr(page, "n_total", None)
        if n is not None:
            return int(n)
        local_max = int(page.cols[key_name].max().item()) \
            if page.n_rows else 0
        if use_dist:
            t = torch.tensor([local_max], dtype=torch.int64, device=device)
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            local_max = int(t.item())
        return local_max

    n_cust_all = global_cardinality(cust_page, "custkey")
    n_supp_all = global_cardinality(supp_page, "suppkey")
    cnat = global_dense(cust_page, "custkey", "nationkey", n_cust_all)
    snat = global_dense(supp_page, "suppkey", "nationkey", n_supp_all)

    # dense tables from the replicated arrays
    def dense_table(vals):
        n = vals.numel()
        keys = torch.arange(1, n + 1, dtype=torch.int64, device=device)
        bp = P.PlanHashBuild()
        bp.key_col = 0
        bp.semijoin_table = -1
        bp.n_payload = 1
        bp.payload_col[0] = 1
        bp.capacity_hint = n
        bp.dense_array = 1
        b = P.Operator(P.OP_HASH_BUILD, bp)
        b.add_input(P.Page({"k": keys, "v": vals}))
        b.finish()
        return b
    bc = dense_table(cnat)
    bs = dense_table(snat)

    # orders: filter (date) -> partition by orderkey -> exchange
    fp = P.PlanFilterProject()
    fp.n_preds = 2
    fp.preds[0] = P.Pred(ord_page.channel("orderdate"), P.CMP_GE,
                         Q5Pipeline.Q5_LO, 0.0)
    fp.preds[1] = P.Pred(ord_page.channel("orderdate"), P.CMP_LT,
                         Q5Pipeline.Q5_HI, 0.0)
    fp.n_proj = 2
    fp.proj[0] = P.Proj(P.PROJ_IDENT, ord_page.channel("orderkey"), 0, 0)
    fp.proj[1] = P.Proj(P.PROJ_IDENT, ord_page.channel("custkey"), 0, 0)
    fo = P.Operator(P.OP_FILTER_PROJECT, fp)
    fo.add_input(ord_page)
    fraw = fo.get_output_raw()

    def partition_exchange_raw(raw, ncols):
        pp = P.PlanPartition()
        pp.n_partitions = world
        pp.key_col = 0
        pp.n_emit = ncols
        for i in range(ncols):
            pp.emit_cols[i] = i
        po = P.Operator(P.OP_PARTITION, pp)
        po.add_input_raw(raw)
        counts = po.partition_counts(world)
        pages = [po.get_output_raw() for _ in range(world)]
        total = sum(counts)
        dtmap = {0: torch.uint8, 1: torch.int32, 2: torch.int64,
                 3: torch.float64}
        cols = {}
        for c in range(ncols):
            t = torch.empty(total, dtype=dtmap[pages[0].cols[c].tag],
                            device=device)
            off = 0
            for p in range(world):
                nrow = pages[p].n_rows
                if nrow:
                    lib().check(lib().c.pg_memcpy_d2d(
                        t.data_ptr() + off * t.element_size(),
                        pages[p].cols[c].data,
                        nrow * t.element_size()), "d2d")
                off += nrow
            cols[f"c{c}"] = t
        po.destroy()
        if use_dist:
            return exchange_columns(cols, counts, device=device)
        return cols

    ocols = partition_exchange_raw(fraw, 2)
    fo.destroy()

    b2p = P.PlanHashBuild()
    b2p.key_col = 0
    b2p.semijoin_table = -1
    b2p.n_payload = 1
    b2p.payload_col[0] = 0
    b2p.payload_lookup_table = bc.table()
    b2p.payload_lookup_key_col = 1
    b2p.capacity_hint = max(int(ocols["c0"].numel()) + 64, 64)
    b2p.agg_table = 1
    b2p.bitmap_max_key = okey_bound
    b2 = P.Operator(P.OP_HASH_BUILD, b2p)
    b2.add_input(P.Page(ocols))
    b2.finish()

    # lineitem: partition (no filter in Q5) -> exchange -> fused probe
    lcols_in = {"c0": li_page.cols["orderkey"],
                "c1": li_page.cols["suppkey"],
                "c2": li_page.cols["extendedprice"],
                "c3": li_page.cols["discount"]}
    lp = P.Page(lcols_in)
    lraw = lp.to_c()
    lcols = partition_exchange_raw(lraw, 4)

    jp = P.PlanLookupJoin()
    jp.table = b2.table()
    jp.key_col = 0
    jp.mode = 2
    jp.proj = P.Proj(P.PROJ_DISC_PRICE, 2, 3, 0)
    jp.dec_scale = 4
    jp.table2 = bs.table()
    jp.table2_key_col = 1
    jp.n_group_vals = len(Q5Pipeline.ASIA)
    for i, v in enumerate(Q5Pipeline.ASIA):
        jp.group_vals[i] = v
    j = P.Operator(P.OP_LOOKUP_JOIN, jp)
    j.add_input(P.Page(lcols))
    j.finish()
    out = j.get_output(["nationkey", "rev_lo", "rev_f64", "count"])
    j.destroy()
    for b in (bc, bs, b2):
        lib().c.pg_table_destroy(b.table())
        b.destroy()
    # exact final combine: per-nation tick sums.  Per-rank sums are
    # un-wrapped (the probe kernel raises on int64 tick overflow); bound
    # them so the cross-rank all_reduce cannot wrap either.
    ticks = torch.zeros(len(Q5Pipeline.ASIA), dtype=torch.int64,
                        device=device)
    for i in range(len(out["nationkey"])):
        t = int(out["rev_lo"][i])
        assert abs(t) < (1 << 62) // max(world, 1), \
            "per-rank nation revenue too large for exact int64 combine"
        ticks[Q5Pipeline.ASIA.index(int(out["nationkey"][i]))] = t
    if use_dist:
        dist.all_reduce(ticks)
    if rank != 0:
        return None
    t = ticks.cpu().numpy()
    return {n: int(t[i]) for i, n in enumerate(Q5Pipeline.ASIA) if t[i]}
