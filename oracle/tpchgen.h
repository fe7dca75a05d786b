/* tpchgen.h — columns-only TPC-H data generator (dbgen restatement).
 *
 * TEST / BENCH INPUT INFRASTRUCTURE + ORACLE DATA SOURCE.
 * This is NOT part of the product compute path: it synthesizes the input
 * columns that both the CPU oracle (oracle.c) and the GPU kernels consume.
 *
 * The reference (prestodb/presto) generates TPC-H rows through the external
 * dependency io.airlift.tpch:tpch:0.10 (pom.xml:1425-1429), called from
 * presto-tpch/src/main/java/com/facebook/presto/tpch/TpchRecordSet.java:46-51.
 * That dependency's source is NOT vendored in /root/reference, so this file
 * restates the published TPC-H dbgen algorithm (TPC-H specification §4.2.3
 * "Random number generation"; dbgen rnd.c / build.c / speed_seed.c semantics,
 * of which io.airlift.tpch is a faithful port).  Parity of the restatement is
 * pinned by the reference's own golden SF1 result vectors:
 *   presto-product-tests/src/main/resources/sql-tests/testcases/hive_tpch/q01.result
 *   presto-product-tests/src/main/resources/sql-tests/testcases/hive_tpch/q03.result
 * (restated as committed fixtures in tests/golden/), which are exact-decimal
 * aggregates over every generated row of lineitem/orders/customer at SF1.
 *
 * Dates are day indexes: idx 1 == 1992-01-01; epoch32 = 8035 + idx - 1
 * (8035 = days from 1970-01-01 to 1992-01-01).
 */
#ifndef TPCHGEN_H
#define TPCHGEN_H
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* row counts */
int64_t tpch_customer_count(double sf); /* 150,000 * sf */
int64_t tpch_orders_count(double sf);   /* 1,500,000 * sf */
/* scans the line-count RNG stream; 6,001,215 at SF1 */
int64_t tpch_lineitem_count(double sf);

/* customer columns for rows [start, start+count) (0-based row index).
 * custkey = row index + 1.  mktseg_id: 0=AUTOMOBILE 1=BUILDING 2=FURNITURE
 * 3=MACHINERY 4=HOUSEHOLD (distribution order of dbgen dists.dss msegmnt).
 * nationkey: uniform 0..24 (dbgen nation table order). */
void tpch_gen_customer(double sf, int64_t start, int64_t count,
                       int64_t* custkey /*nullable*/, uint8_t* mktseg_id);
void tpch_gen_customer2(double sf, int64_t start, int64_t count,
                        int64_t* custkey, uint8_t* mktseg_id,
                        uint8_t* nationkey);

/* supplier: 10,000 * sf rows; suppkey = index + 1 */
int64_t tpch_supplier_count(double sf);
void tpch_gen_supplier(double sf, int64_t start, int64_t count,
                       int64_t* suppkey, uint8_t* nationkey);

/* dbgen nation table (25 rows): returns region key of nation 0..24
 * (0 AFRICA, 1 AMERICA, 2 ASIA, 3 EUROPE, 4 MIDDLE EAST) */
int32_t tpch_nation_region(int32_t nationkey);
/* nation name (dbgen order) into buf (<=32B incl NUL); returns length */
int32_t tpch_nation_name(int32_t nationkey, char* buf);

/* lineitem with suppkey: same as tpch_gen_lineitem plus l_suppkey
 * (dbgen PART_SUPP bridge: selectPartSupplier(partkey, supplierNumber,
 * supplierCount), supplierNumber stream 0..3) */
int64_t tpch_gen_lineitem2(double sf, int64_t ord_start, int64_t ord_count,
                           int64_t* orderkey, double* quantity,
                           double* extendedprice, double* discount,
                           double* tax, int32_t* shipdate_epoch,
                           uint8_t* returnflag, uint8_t* linestatus,
                           int64_t* suppkey);

/* order priority (0..4 = 1-URGENT, 2-HIGH, 3-MEDIUM, 4-NOT SPECIFIED,
 * 5-LOW) and lineitem commit/receipt dates for Q4 */
void tpch_gen_orders_priority(double sf, int64_t start, int64_t count,
                              uint8_t* priority);
int64_t tpch_gen_lineitem_dates(double sf, int64_t ord_start,
                                int64_t ord_count, int64_t* orderkey,
                                int32_t* commitdate_epoch,
                                int32_t* receiptdate_epoch);

/* part type ids for part rows [start, start+count): id 0..149 indexes the
 * dbgen p_type 3-word combinations (Types1 x Types2 x Types3 nested);
 * 'ECONOMY ANODIZED STEEL' = 103. */
void tpch_gen_part_type(double sf, int64_t start, int64_t count,
                        uint8_t* type_id);
/* lineitem partkeys (the L_PKEY stream replayed standalone) */
int64_t tpch_gen_lineitem_partkey(double sf, int64_t ord_start,
                                  int64_t ord_count, int64_t* partkey);

/* lineitem shipmode ids 0..6 (pick 1..7 per line); id 4 = 'MAIL',
 * id 6 = 'SHIP' (pinned by the q12 golden counts) */
int64_t tpch_gen_lineitem_shipmode(double sf, int64_t ord_start,
                                   int64_t ord_count, uint8_t* shipmode);

/* part attribute columns: p_mfgr 1..5, p_brand = mfgr*10 + (1..5)
 * ('Brand#MB'), p_container id 0..39 ('MED BOX' = 17, q17 golden pin;
 * nested Cnt1 x Cnt2 order, 8 second words per first word). Any output
 * may be NULL. */
void tpch_gen_part2(double sf, int64_t start, int64_t count, uint8_t* mfgr,
                    uint8_t* brand, uint8_t* container);
/* as tpch_gen_part2 plus p_size 1..50 (q19 golden pin) */
void tpch_gen_part3(double sf, int64_t start, int64_t count, uint8_t* mfgr,
                    uint8_t* brand, uint8_t* container, uint8_t* size);

/* lineitem shipinstruct ids 0..3 ('DELIVER IN PERSON' = 0, q19 pin) */
int64_t tpch_gen_lineitem_shipinstruct(double sf, int64_t ord_start,
                                       int64_t ord_count,
                                       uint8_t* shipinstruct);

/* partsupp: 4 rows per part row [part_start, part_start+part_count);
 * suppkey via the PART_SUPP bridge; ps_availqty 1..9999;
 * ps_supplycost in cents 100..100000 (1.00..1000.00).  Outputs are
 * 4*part_count rows; any output may be NULL.  (q11 golden pin: all
 * 1048 result rows exact.) */
void tpch_gen_partsupp(double sf, int64_t part_start, int64_t part_count,
                       int64_t* partkey, int64_t* suppkey,
                       int32_t* availqty, int64_t* supplycost_cents);

/* p_name word ids: 5 words per part drawn from the 92-color list; per
 * part the identity permutation is re-permuted with a 92-draw swap pass
 * (the airlift port's agg_str semantics, q9 golden pin).  words has
 * 5*count entries, rows from part 1. */
void tpch_gen_part_name_words(double sf, int64_t count, uint8_t* words);
/* index of a color word in the dists order, -1 if absent */
int32_t tpch_color_id(const char* word);
/* color word of an id into buf (<=16B incl NUL); returns length */
int32_t tpch_color_name(int32_t id, char* buf);
/* part type name of id 0..149 into buf (<=32B); returns length */
int32_t tpch_part_type_name(int32_t id, char* buf);

/* ---- text pool + comment columns (validated against every airlift
 * comment fixture in the reference; see tpchgen.c) ---- */
/* lazily builds and returns the 300 MiB text pool (process lifetime) */
const char* tpch_text_pool(void);
int64_t tpch_text_pool_size(void);
/* comment (offset,length) streams; text = pool[off : off+len] */
void tpch_gen_orders_comment(double sf, int64_t start, int64_t count,
                             int64_t* off, int32_t* len);
void tpch_gen_supplier_comment(double sf, int64_t start, int64_t count,
                               int64_t* off, int32_t* len);
void tpch_gen_customer_comment(double sf, int64_t start, int64_t count,
                               int64_t* off, int32_t* len);
void tpch_gen_nation_comment(int64_t* off, int32_t* len);
/* supplier BBB comment splice flags: 0 plain, 1 'Customer Complaints',
 * 2 'Customer Recommends' (q16 golden pin) */
void tpch_gen_supplier_bbb(double sf, int64_t start, int64_t count,
                           uint8_t* bbb);

/* s_acctbal cents (q02 pin) + phone digit triples (CC = nationkey+10;
 * 'CC-AAA-BBB-CCCC'; q02/q10 pins) */
void tpch_gen_supplier_acctbal(double sf, int64_t start, int64_t count,
                               int64_t* acctbal_cents);
void tpch_gen_supplier_phone(double sf, int64_t start, int64_t count,
                             int32_t* a, int32_t* b, int32_t* c);
void tpch_gen_customer_phone(double sf, int64_t start, int64_t count,
                             int32_t* a, int32_t* b, int32_t* c);

/* c_acctbal in exact cents, -99999..999999 (q22 golden pin) */
void tpch_gen_customer_acctbal(double sf, int64_t start, int64_t count,
                               int64_t* acctbal_cents);

/* o_totalprice in exact cents (dbgen mk_order floor-div chain over the
 * order's lineitems; pinned by the q18 golden) */
void tpch_gen_orders_totalprice(double sf, int64_t ord_start,
                                int64_t ord_count,
                                int64_t* totalprice_cents);

/* orders columns for order rows [start, start+count) (0-based).
 * shippriority is the constant 0 (dbgen mk_order) and is not emitted.
 * lcnt: lineitems per order (1..7). Any output pointer may be NULL. */
void tpch_gen_orders(double sf, int64_t start, int64_t count,
                     int64_t* orderkey, int64_t* custkey,
                     int32_t* orderdate_epoch, int32_t* lcnt);

/* number of lineitem rows belonging to orders [0, ord_start) — prefix offset */
int64_t tpch_lineitem_offset(double sf, int64_t ord_start);

/* lineitem columns for all rows of orders [ord_start, ord_start+ord_count).
 * Writes rows densely from output index 0; the caller uses
 * tpch_lineitem_offset() to place chunks. Returns rows written.
 * Money columns are DOUBLE-typed like the reference's presto-tpch mapping
 * (TpchMetadata.java:537-553): quantity (integral 1..50), extendedprice
 * (cents/100), discount (0.00..0.10), tax (0.00..0.08).
 * returnflag in {'A','N','R'}, linestatus in {'F','O'} (ASCII u8). */
int64_t tpch_gen_lineitem(double sf, int64_t ord_start, int64_t ord_count,
                          int64_t* orderkey, double* quantity,
                          double* extendedprice, double* discount, double* tax,
                          int32_t* shipdate_epoch, uint8_t* returnflag,
                          uint8_t* linestatus);

#ifdef __cplusplus
}
#endif
#endif
