/* oracle.h — CPU restatement of the reference's per-Page operator hot path.
 *
 * TEST INFRASTRUCTURE ONLY.  Only tests/, __graft_entry__.smoke() and
 * bench.py's cpu_baseline leg may call this library; the product path
 * (presto_amd + libpresto_gpu.so) never links or loads it.
 *
 * Each function follows the cited Java implementation in /root/reference
 * (prestodb/presto 0.300-SNAPSHOT).  The oracle is pinned by the reference's
 * golden SF1 result vectors (see tpchgen.h header) via oracle_q1 / oracle_q3
 * in decimal mode, which reproduce
 *   presto-product-tests/.../hive_tpch/q01.result and q03.result
 * digit-for-digit (exact decimal arithmetic).
 */
#ifndef ORACLE_H
#define ORACLE_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------------- TPC-H Q1 ----------------
 * SQL: presto-benchto-benchmarks/.../tpch/q01.sql
 * Pipeline restated: ScanFilterAndProjectOperator (filter shipdate <= 10471,
 * PageProcessor.java:299-343 selection semantics) -> HashAggregationOperator
 * (HashAggregationOperator.java:413,451; grouping on (returnflag,linestatus),
 * MultiChannelGroupByHash.java:300-380 assigns dense group ids in first-seen
 * order) -> aggregates per DoubleSumAggregation.java (running f64 sum),
 * LongSumAggregation.java:33-37 (overflow-checked), CountAggregation.java:34,
 * AverageAggregations.java:34-63 ((count, sum) state, output sum/count).
 *
 * Groups are keyed by (returnflag<<8)|linestatus; results are emitted sorted
 * by (returnflag, linestatus) like the query's ORDER BY.
 *
 * Decimal outputs are exact integer ticks:
 *   sum_qty        : quantity summed in units (quantity is integral)
 *   sum_base_cents : extendedprice summed in cents            (scale 2)
 *   sum_disc_1e4   : sum extprice*(1-disc), ticks of 1e-4     (scale 4)
 *   sum_charge_1e6 : sum extprice*(1-disc)*(1+tax), 1e-6      (scale 6)
 *                    (returned as hi/lo int128 halves)
 *   sum_disc_cents : discount summed in hundredths            (scale 2)
 * f64 outputs follow the fixed-tree deterministic schedule the GPU kernels
 * use (documented in DESIGN.md): per virtual-lane sequential accumulation
 * over rows lane, lane+NLANES, ... then pairwise reduction — bit-exact
 * reproducible on CPU and GPU.
 */
typedef struct {
    uint8_t returnflag, linestatus;
    int64_t count_order;
    int64_t sum_qty_units;
    int64_t sum_base_cents;
    int64_t sum_disc_1e4;
    int64_t sum_charge_1e6_hi; /* int128 = hi*2^64 + lo (two's complement) */
    uint64_t sum_charge_1e6_lo;
    int64_t sum_disc_cents;
    /* f64 fixed-tree sums (same schedule as GPU kernels) */
    double f64_sum_qty, f64_sum_base, f64_sum_disc_price, f64_sum_charge,
        f64_sum_disc;
} q1_group_t;

/* returns number of groups (<=6), fills groups[] sorted by key */
int32_t oracle_q1(int64_t n_rows, const double* quantity,
                  const double* extendedprice, const double* discount,
                  const double* tax, const int32_t* shipdate,
                  const uint8_t* returnflag, const uint8_t* linestatus,
                  q1_group_t* groups);

/* ---------------- TPC-H Q3 ----------------
 * SQL: presto-benchto-benchmarks/.../tpch/q03.sql
 * Pipeline restated: build customer set (mktsegment='BUILDING'), hash join
 * orders (orderdate < 9204) x customer (HashBuilderOperator.java:333,
 * PagesHash.java:81-125 open-address linear probe with murmur3-finalizer
 * bucket, PagesHash.java:236-252), then lineitem (shipdate > 9204) probe
 * (LookupJoinOperator.java:481-604) with group-by on orderkey
 * (BigintGroupByHash.java:222-252) summing revenue; TopN 10 by
 * (revenue desc, orderdate asc, orderkey asc) (TopNOperator.java:90-111;
 * orderkey added as deterministic final tiebreak — revenue ties beyond the
 * SQL ORDER BY are unspecified in the reference).
 * revenue is exact decimal: ticks of 1e-4 = cents*(100-d). */
typedef struct {
    int64_t orderkey;
    int64_t revenue_1e4;
    int32_t orderdate;
    int32_t shippriority;
    double f64_revenue; /* fixed-tree f64 mode (same GPU schedule) */
} q3_row_t;

int32_t oracle_q3(int64_t n_cust, const int64_t* c_custkey,
                  const uint8_t* c_mktseg_id, int64_t n_ord,
                  const int64_t* o_orderkey, const int64_t* o_custkey,
                  const int32_t* o_orderdate, int64_t n_li,
                  const int64_t* l_orderkey, const double* l_extendedprice,
                  const double* l_discount, const int32_t* l_shipdate,
                  int32_t limit, q3_row_t* out);

/* ---------------- TPC-H Q5 ----------------
 * SQL: presto-benchto-benchmarks/.../tpch/q05.sql (region 'ASIA',
 * orderdate in [1994-01-01, 1995-01-01) = [8766, 9131) epoch days).
 * 6-way join: customer x orders x lineitem x supplier x nation x region
 * with the local-supplier condition c_nationkey = s_nationkey; grouped
 * revenue per nation, sorted desc.  Exact ticks of 1e-4. */
typedef struct {
    uint8_t nationkey;
    char name[32];
    int64_t revenue_1e4;
} q5_row_t;

int32_t oracle_q5(int64_t n_cust, const int64_t* c_custkey,
                  const uint8_t* c_nationkey, int64_t n_ord,
                  const int64_t* o_orderkey, const int64_t* o_custkey,
                  const int32_t* o_orderdate, int64_t n_li,
                  const int64_t* l_orderkey, const int64_t* l_suppkey,
                  const double* l_extendedprice, const double* l_discount,
                  int64_t n_supp, const uint8_t* s_nationkey,
                  q5_row_t* out /* capacity 25 */);

/* ---------------- TPC-H Q8 ----------------
 * SQL: q08.sql — national market share: volume of 'ECONOMY ANODIZED
 * STEEL' (type id 103) parts sold to AMERICA-region customers in
 * 1995/1996, share of BRAZIL(2)-nation suppliers.  Returns exact tick
 * sums: brazil[2] and total[2] for years 1995, 1996 (share =
 * brazil/total). */
void oracle_q8(int64_t n_cust, const int64_t* c_custkey,
               const uint8_t* c_nationkey, int64_t n_ord,
               const int64_t* o_orderkey, const int64_t* o_custkey,
               const int32_t* o_orderdate, int64_t n_li,
               const int64_t* l_orderkey, const int64_t* l_suppkey,
               const int64_t* l_partkey, const double* l_extendedprice,
               const double* l_discount, int64_t n_supp,
               const uint8_t* s_nationkey, int64_t n_part,
               const uint8_t* p_type, int64_t* brazil_1e4,
               int64_t* total_1e4);

/* ---------------- TPC-H Q4 ----------------
 * SQL: q04.sql — order priority checking: orders with orderdate in
 * [1993-07-01, 1993-10-01) = [8582, 8674) having EXISTS a lineitem with
 * commitdate < receiptdate; count per priority (0..4).  out_counts[5]. */
void oracle_q4(int64_t n_ord, const int64_t* o_orderkey,
               const int32_t* o_orderdate, const uint8_t* o_priority,
               int64_t n_li, const int64_t* l_orderkey,
               const int32_t* l_commitdate, const int32_t* l_receiptdate,
               int64_t* out_counts);

/* ---------------- TPC-H Q7 ----------------
 * SQL: q07.sql — volume shipping between FRANCE(6) and GERMANY(7):
 * s_nationkey/c_nationkey in {(6,7),(7,6)}, shipdate in
 * [1995-01-01, 1996-12-31] (epoch [9131, 9861]), volume =
 * extendedprice*(1-discount) grouped by (supp_nation, cust_nation,
 * year).  rows out in (supp_nation, cust_nation, year) order.
 * revenue in exact 1e-4 ticks. */
typedef struct {
    uint8_t supp_nation, cust_nation;
    int32_t year;
    int64_t revenue_1e4;
} q7_row_t;

int32_t oracle_q7(int64_t n_cust, const int64_t* c_custkey,
                  const uint8_t* c_nationkey, int64_t n_ord,
                  const int64_t* o_orderkey, const int64_t* o_custkey,
                  int64_t n_li, const int64_t* l_orderkey,
                  const int64_t* l_suppkey, const double* l_extendedprice,
                  const double* l_discount, const int32_t* l_shipdate,
                  int64_t n_supp, const uint8_t* s_nationkey,
                  q7_row_t* out /* capacity 4 */);

/* ---------------- TPC-H Q6 ----------------
 * SQL: q06.sql — scalar aggregate: sum(extendedprice*discount) over
 * shipdate in [1994-01-01, 1995-01-01), discount in [0.05, 0.07] (f64
 * compare; the generated hundredths are exactly representable), quantity
 * < 24.  Returns exact 1e-4 ticks (= cents * hundredths) and the count. */
void oracle_q6(int64_t n, const double* quantity,
               const double* extendedprice, const double* discount,
               const int32_t* shipdate, int64_t* revenue_1e4,
               int64_t* count);

/* ---------------- TPC-H Q14 ----------------
 * q14.sql — promo revenue: exact 1e-4 tick sums over September 1995;
 * promo = part type ids 125..149.  Result = 100.00*promo/total at
 * scale 6 HALF_UP (computed by the caller). */
void oracle_q14(int64_t n_li, const double* l_extendedprice,
                const double* l_discount, const int32_t* l_shipdate,
                const int64_t* l_partkey, int64_t n_part,
                const uint8_t* p_type, int64_t* promo_1e4,
                int64_t* total_1e4);

/* ---------------- TPC-H Q12 ----------------
 * q12.sql — shipmode priority: per shipmode id (0..6; MAIL=4, SHIP=6),
 * high (priority 0/1) and low line counts of late-commit lineitems
 * received in 1994. */
void oracle_q12(int64_t n_ord, const int64_t* o_orderkey,
                const uint8_t* o_priority, int64_t n_li,
                const int64_t* l_orderkey, const uint8_t* l_shipmode,
                const int32_t* l_shipdate, const int32_t* l_commitdate,
                const int32_t* l_receiptdate, int64_t* high_counts /*[7]*/,
                int64_t* low_counts /*[7]*/);

/* ---------------- TPC-H Q17 ----------------
 * q17.sql — small-quantity-order revenue for Brand#23 / 'MED BOX'
 * (container id 17): exact cents sum of extendedprice over rows with
 * quantity < 0.2*avg(part).  Result = sum/7.0 at scale 2 HALF_UP. */
void oracle_q17(int64_t n_li, const int64_t* l_partkey,
                const double* l_quantity, const double* l_extendedprice,
                int64_t n_part, const uint8_t* p_brand,
                const uint8_t* p_container, int64_t* out_cents);

/* ---------------- TPC-H Q11 ----------------
 * q11.sql — important stock in GERMANY(7): (partkey, value-cents) rows
 * sorted (value desc, partkey asc), value*10000 > total.  Returns rows
 * (out capacity n_part). */
int64_t oracle_q11(int64_t n_ps, const int64_t* ps_partkey,
                   const int64_t* ps_suppkey, const int32_t* ps_availqty,
                   const int64_t* ps_supplycost_cents, int64_t n_supp,
                   const uint8_t* s_nationkey, int64_t n_part,
                   int64_t* out_pk, int64_t* out_val);

/* ---------------- TPC-H Q18 ----------------
 * q18.sql — large-volume customers: orders with sum(l_quantity) > 300;
 * rows (custkey, orderkey, orderdate, totalprice_cents, sum_qty) sorted
 * (totalprice desc, orderdate asc, orderkey asc) LIMIT limit.  Returns
 * rows written. */
int64_t oracle_q18(int64_t n_ord, const int64_t* o_orderkey,
                   const int64_t* o_custkey, const int32_t* o_orderdate,
                   const int64_t* o_totalprice_cents, int64_t n_li,
                   const int64_t* l_orderkey, const double* l_quantity,
                   int32_t limit, int64_t* out_ck, int64_t* out_ok,
                   int32_t* out_od, int64_t* out_tp, int64_t* out_qty);

/* ---------------- TPC-H Q21 ----------------
 * q21.sql — suppliers who kept orders waiting (SAUDI ARABIA=20): late
 * lines in multi-supplier all-'F' orders whose only late supplier is
 * theirs.  Lineitem arrays must be grouped by orderkey.  Rows
 * (suppkey, numwait) sorted (numwait desc, suppkey asc) LIMIT limit. */
int64_t oracle_q21(int64_t n_supp, const uint8_t* s_nationkey, int64_t n_li,
                   const int64_t* l_orderkey, const int64_t* l_suppkey,
                   const uint8_t* l_linestatus, const int32_t* l_commitdate,
                   const int32_t* l_receiptdate, int32_t limit,
                   int64_t* out_sk, int64_t* out_cnt);

/* ---------------- TPC-H Q22 ----------------
 * q22.sql — global sales opportunity: per phone country code
 * (= nationkey+10; the 7 codes the query names), count + exact cents
 * sum of above-average-balance customers with no orders. */
void oracle_q22(int64_t n_cust, const int64_t* c_custkey,
                const uint8_t* c_nationkey, const int64_t* c_acctbal_cents,
                int64_t n_ord, const int64_t* o_custkey, int32_t n_codes,
                const uint8_t* code_nations, int64_t* out_cnt,
                int64_t* out_sum);

/* ---------------- TPC-H Q19 ----------------
 * q19.sql — discounted revenue over three brand/container/size/quantity
 * disjuncts ('AIR' shipmode id 1, 'DELIVER IN PERSON' instruct id 0).
 * Exact 1e-4 ticks. */
void oracle_q19(int64_t n_li, const int64_t* l_partkey,
                const double* l_quantity, const double* l_extendedprice,
                const double* l_discount, const uint8_t* l_shipmode,
                const uint8_t* l_shipinstruct, int64_t n_part,
                const uint8_t* p_brand, const uint8_t* p_container,
                const uint8_t* p_size, int64_t* revenue_1e4);

/* ---------------- TPC-H Q9 ----------------
 * q09.sql — product-type profit by (supplier nation, order year);
 * p_match flags mark parts whose name contains the query's word.
 * profit_1e4[nation*7 + (year-1992)] exact ticks. */
void oracle_q9(int64_t n_li, const int64_t* l_partkey,
               const int64_t* l_suppkey, const double* l_quantity,
               const double* l_extendedprice, const double* l_discount,
               const int64_t* l_orderkey, int64_t n_ord,
               const int64_t* o_orderkey, const int32_t* o_orderdate,
               int64_t n_supp, const uint8_t* s_nationkey, int64_t n_part,
               const uint8_t* p_match, const int64_t* ps_suppkey,
               const int64_t* ps_supplycost_cents, int64_t* profit_1e4);

/* ---------------- TPC-H Q13 ----------------
 * q13.sql — customer distribution: histogram of per-customer counts of
 * orders whose o_comment does not match '%special%requests%'
 * (LEFT OUTER: zero-order customers count).  Comment text is
 * pool[off:off+len].  Rows sorted (custdist desc, c_count desc). */
int64_t oracle_q13(int64_t n_cust, int64_t n_ord, const int64_t* o_custkey,
                   const int64_t* cmnt_off, const int32_t* cmnt_len,
                   const char* pool, int64_t* out_count, int64_t* out_dist,
                   int64_t cap);

/* ---------------- TPC-H Q16 ----------------
 * q16.sql — parts/supplier relationship: count(DISTINCT suppkey) per
 * (brand, type, size) over qualifying parts, excluding complaint
 * suppliers (bbb flag 1).  Rows sorted (cnt desc, brand, type NAME,
 * size).  Returns rows. */
int64_t oracle_q16(int64_t n_part, const uint8_t* p_brand,
                   const uint8_t* p_type, const uint8_t* p_size,
                   int64_t n_ps, const int64_t* ps_partkey,
                   const int64_t* ps_suppkey, int64_t n_supp,
                   const uint8_t* s_bbb, uint8_t* out_brand,
                   uint8_t* out_type, uint8_t* out_size, int32_t* out_cnt,
                   int64_t cap);

/* ---------------- TPC-H Q10 / Q15 / Q20 / Q2 ----------------
 * Display-light restatements: join/aggregate/order semantics over the
 * pinned streams; display columns (names, phones, balances, comments)
 * resolve from the generator streams in the callers.  Addresses
 * (v_string) remain the one unpinned generator column. */
int64_t oracle_q10(int64_t n_ord, const int64_t* o_orderkey,
                   const int64_t* o_custkey, const int32_t* o_orderdate,
                   int64_t n_li, const int64_t* l_orderkey,
                   const uint8_t* l_returnflag,
                   const double* l_extendedprice, const double* l_discount,
                   int64_t n_cust, int32_t limit, int64_t* out_ck,
                   int64_t* out_rev_1e4);
int64_t oracle_q15(int64_t n_li, const int64_t* l_suppkey,
                   const double* l_extendedprice, const double* l_discount,
                   const int32_t* l_shipdate, int64_t n_supp,
                   int64_t* out_sk, int64_t* out_rev_1e4, int64_t cap);
int64_t oracle_q20(int64_t n_part, const uint8_t* p_name_words,
                   int32_t forest_id, int64_t n_ps,
                   const int64_t* ps_partkey, const int64_t* ps_suppkey,
                   const int32_t* ps_availqty, int64_t n_li,
                   const int64_t* l_partkey, const int64_t* l_suppkey,
                   const double* l_quantity, const int32_t* l_shipdate,
                   int64_t n_supp, const uint8_t* s_nationkey,
                   int64_t* out_sk, int64_t cap);
int64_t oracle_q2(int64_t n_part, const uint8_t* p_type,
                  const uint8_t* p_size, int64_t n_ps,
                  const int64_t* ps_partkey, const int64_t* ps_suppkey,
                  const int64_t* ps_supplycost_cents, int64_t n_supp,
                  const uint8_t* s_nationkey,
                  const int64_t* s_acctbal_cents, int32_t limit,
                  int64_t* out_sk, int64_t* out_pk);

/* ---------------- operator-level primitives (parity targets) ---------- */

/* murmur3 finalizer bucket — PagesHash.java:236-252 /
 * BigintGroupByHash.getHashPosition */
uint64_t oracle_murmur3_finalize(uint64_t h);
/* bigint type hash — AbstractLongType.java:137-140 (xxhash64 mix) */
uint64_t oracle_bigint_hash(int64_t v);
/* partition id — HashGenerator.java:22-29:
 * (u32(Long.hashCode(rawHash)) * partitionCount) >> 32 */
int32_t oracle_partition(uint64_t raw_hash, int32_t partition_count);
/* varchar hash — AbstractVariableWidthBlock.java:102-105 (XxHash64 seed 0) */
uint64_t oracle_xxh64(const uint8_t* data, int64_t len);

/* Group-by over a single bigint key column, BigintGroupByHash.java:222-332:
 * open-address linear probe, fill 0.75, capacity = next pow2(ceil(hint/.75)),
 * group ids dense in first-seen order.  Writes group_ids[i] per row and
 * returns group count.  (No null handling: key column non-null.) */
int64_t oracle_bigint_group_by(int64_t n, const int64_t* keys,
                               int32_t* group_ids);

/* Join build+probe over bigint keys with duplicate chains,
 * PagesHash.java:81-181 + ArrayPositionLinks.java:25-56 (head-insert; probe
 * visits the LATEST inserted duplicate first, then walks to earlier ones).
 * For each probe row i, appends matched build-row indexes (in chain order)
 * to out_build_idx/out_probe_idx; returns number of emitted pairs
 * (capacity out_cap; excess dropped). */
int64_t oracle_join_bigint(int64_t n_build, const int64_t* build_keys,
                           int64_t n_probe, const int64_t* probe_keys,
                           int64_t* out_probe_idx, int64_t* out_build_idx,
                           int64_t out_cap);

#ifdef __cplusplus
}
#endif
#endif
