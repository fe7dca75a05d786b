/* tpchgen.c — columns-only TPC-H dbgen restatement. See tpchgen.h header for
 * provenance and the parity pin.  Plain C99 + OpenMP. */
#include "tpchgen.h"
#include <stdlib.h>

/* ---- RNG: TPC-H spec §4.2.3 / dbgen rnd.c ----
 * Lehmer LCG: seed' = seed * 16807 mod (2^31-1).
 * UnifInt(lo,hi): advance, then lo + floor(seed/2147483647.0 * (hi-lo+1)). */
#define RNG_M 2147483647LL
#define RNG_A 16807LL

static inline int64_t rng_next(int64_t s) { return (s * RNG_A) % RNG_M; }

/* seed * A^n mod M — dbgen NthElement / airlift advanceSeed32 */
static inline int64_t rng_skip(int64_t seed, uint64_t n)
{
    uint64_t a = RNG_A, r = 1;
    while (n) {
        if (n & 1) r = (r * a) % RNG_M;
        a = (a * a) % RNG_M;
        n >>= 1;
    }
    return (int64_t)(((uint64_t)seed * r) % RNG_M);
}

static inline int64_t unif(int64_t* s, int64_t lo, int64_t hi)
{
    *s = rng_next(*s);
    return lo + (int64_t)(((double)*s / 2147483647.0) * (double)(hi - lo + 1));
}

/* ---- per-column stream seeds (dbgen speed_seed.c, via io.airlift.tpch) ---- */
#define SEED_O_LCNT   1434868289LL /* line count per order, 1..7, usage 1/order */
#define SEED_O_ODATE  1066728069LL /* order date, usage 1/order */
#define SEED_O_CKEY    851767375LL /* customer key, usage 1/order */
#define SEED_L_QTY     209208115LL /* quantity 1..50, usage 7/order */
#define SEED_L_DISC    554590007LL /* discount 0..10 (%), usage 7/order */
#define SEED_L_TAX     721958466LL /* tax 0..8 (%), usage 7/order */
#define SEED_L_PKEY   1808217256LL /* part key, usage 7/order */
#define SEED_L_SDAYS  1769349045LL /* ship  = odate + 1..121, usage 7/order */
#define SEED_L_CDAYS   904914315LL /* commit= odate + 30..90, usage 7/order */
#define SEED_L_RDAYS   373135028LL /* receipt= ship + 1..30,  usage 7/order */
#define SEED_L_RFLAG   717419739LL /* returnflag pick, usage 7/order */
#define SEED_C_MSEG   1140279430LL /* mktsegment pick, usage 1/customer */
#define SEED_C_NATION 1489529863LL /* customer nationkey, usage 1/customer */
#define SEED_S_NATION  110356601LL /* supplier nationkey, usage 1/supplier */
#define SEED_L_SUPPN  2095021727LL /* lineitem supplier number 0..3, 7/order */
#define SEED_O_PRIO    591449447LL /* order priority pick 1..5, 1/order */
#define SEED_P_TYPE   1841581359LL /* part type pick 1..150, 1/part
                                    * (dbgen rnd.h P_TYPE_SD; pinned by the
                                    * q08 golden: id 103 'ECONOMY ANODIZED
                                    * STEEL' shares 0.0344/0.0415 match) */
#define SEED_L_SMODE   675466456LL /* shipmode pick 1..7, usage 7/order
                                    * (pinned by q12 golden: MAIL=4, SHIP=6) */
#define SEED_P_MFG             1LL /* p_mfgr 1..5, 1/part (q17 golden pin) */
#define SEED_P_BRND     46831694LL /* brand digit 1..5, 1/part (q17 pin) */
#define SEED_P_CNTR    727633698LL /* container pick 1..40, 1/part;
                                    * 'MED BOX' = id 17 (q17 golden pin) */
#define SEED_PS_QTY   1671059989LL /* ps_availqty 1..9999, 4/part (q11 pin) */
#define SEED_PS_SCST  1051288424LL /* ps_supplycost cents 100..100000,
                                    * 4/part (q11 golden pin, all 1048 rows) */
#define SEED_C_ABAL    298370230LL /* c_acctbal cents -99999..999999,
                                    * 1/customer (q22 golden pin) */
#define SEED_P_NAME    709314158LL /* p_name permutation, 92/part; fresh
                                    * identity re-permuted per part (the
                                    * airlift port's semantics — q9 pin) */
#define SEED_L_SINST  1371272478LL /* shipinstruct pick 1..4, 7/order
                                    * ('DELIVER IN PERSON' = id 0; q19 pin) */
#define SEED_P_SIZE   1193163244LL /* p_size 1..50, 1/part (q19 pin) */

/* ---- calendar ----
 * day index 1 = 1992-01-01; order-date index in [1, 2406]
 * (dbgen O_ODATE range STARTDATE .. STARTDATE+TOTDATE-151-1, TOTDATE=2557).
 * CURRENTDATE 1995-06-17 = index 1264. epoch32 = 8035 + idx - 1. */
#define ODATE_MIN 1
#define ODATE_MAX 2406
#define CURRENT_IDX 1264
#define EPOCH_1992 8035
#define CUSTOMER_MORTALITY 3

int64_t tpch_customer_count(double sf) { return (int64_t)(150000.0 * sf + 0.5); }
int64_t tpch_supplier_count(double sf) { return (int64_t)(10000.0 * sf + 0.5); }

/* dbgen nation table order; region keys per dists.dss */
static const int32_t NATION_REGION[25] = {
    0, 1, 1, 1, 4, 0, 3, 3, 2, 2, 4, 4, 2, 4, 0, 0, 0, 1, 2, 3, 4, 2, 3, 3,
    1};
static const char* NATION_NAME[25] = {
    "ALGERIA", "ARGENTINA", "BRAZIL", "CANADA", "EGYPT", "ETHIOPIA",
    "FRANCE", "GERMANY", "INDIA", "INDONESIA", "IRAN", "IRAQ", "JAPAN",
    "JORDAN", "KENYA", "MOROCCO", "MOZAMBIQUE", "PERU", "CHINA", "ROMANIA",
    "SAUDI ARABIA", "VIETNAM", "RUSSIA", "UNITED KINGDOM", "UNITED STATES"};
int32_t tpch_nation_region(int32_t nk)
{
    return nk >= 0 && nk < 25 ? NATION_REGION[nk] : -1;
}
int32_t tpch_nation_name(int32_t nk, char* buf)
{
    if (nk < 0 || nk >= 25) return -1;
    const char* s = NATION_NAME[nk];
    int32_t n = 0;
    while (s[n]) {
        buf[n] = s[n];
        n++;
    }
    buf[n] = 0;
    return n;
}

/* dbgen PART_SUPP bridge (build.c PART_SUPP_BRIDGE):
 * supplier of (partkey, i) = (partkey + i*(S/4 + (partkey-1)/S)) % S + 1 */
static inline int64_t part_supplier(int64_t partkey, int64_t i, int64_t S)
{
    return (partkey + i * (S / 4 + (partkey - 1) / S)) % S + 1;
}

void tpch_gen_supplier(double sf, int64_t start, int64_t count,
                       int64_t* suppkey, uint8_t* nationkey)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_S_NATION, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++) {
            int64_t nk = unif(&s, 0, 24);
            if (nationkey) nationkey[i] = (uint8_t)nk;
            if (suppkey) suppkey[i] = start + i + 1;
        }
    }
}
int64_t tpch_orders_count(double sf)   { return (int64_t)(1500000.0 * sf + 0.5); }

/* dbgen mk_sparse (build.c): keep low 3 bits, shift the rest up by 2.
 * order index is 1-based here. */
static inline int64_t make_orderkey(int64_t index1)
{
    return ((index1 >> 3) << 5) | (index1 & 7);
}

/* dbgen rpb_routine (build.c): part retail price in cents */
static inline int64_t part_price_cents(int64_t p)
{
    return 90000 + (p / 10) % 20001 + 100 * (p % 1000);
}

int64_t tpch_lineitem_count(double sf)
{
    int64_t n_ord = tpch_orders_count(sf);
    int64_t total = 0;
#pragma omp parallel reduction(+ : total)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = n_ord * tid / nt, hi = n_ord * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)lo);
        for (int64_t i = lo; i < hi; i++) total += unif(&s, 1, 7);
    }
    return total;
}

int64_t tpch_lineitem_offset(double sf, int64_t ord_start)
{
    int64_t total = 0;
#pragma omp parallel reduction(+ : total)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = ord_start * tid / nt, hi = ord_start * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)lo);
        for (int64_t i = lo; i < hi; i++) total += unif(&s, 1, 7);
    }
    return total;
}

void tpch_gen_customer2(double sf, int64_t start, int64_t count,
                        int64_t* custkey, uint8_t* mktseg_id,
                        uint8_t* nationkey)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_C_MSEG, (uint64_t)(start + lo));
        int64_t sn = rng_skip(SEED_C_NATION, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++) {
            int64_t j = unif(&s, 1, 5);
            int64_t nk = unif(&sn, 0, 24);
            if (mktseg_id) mktseg_id[i] = (uint8_t)(j - 1);
            if (nationkey) nationkey[i] = (uint8_t)nk;
            if (custkey) custkey[i] = start + i + 1;
        }
    }
}

void tpch_gen_customer(double sf, int64_t start, int64_t count,
                       int64_t* custkey, uint8_t* mktseg_id)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_C_MSEG, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++) {
            /* pick_str over 5 unit weights: UnifInt(1,5) -> id 0..4 */
            int64_t j = unif(&s, 1, 5);
            if (mktseg_id) mktseg_id[i] = (uint8_t)(j - 1);
            if (custkey) custkey[i] = start + i + 1;
        }
    }
}

/* generate one chunk of orders (single thread), rows [start, start+count) */
static void gen_orders_chunk(double sf, int64_t start, int64_t count,
                             int64_t* orderkey, int64_t* custkey,
                             int32_t* orderdate_epoch, int32_t* lcnt)
{
    int64_t max_ckey = tpch_customer_count(sf);
    int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)start);
    int64_t s_odate = rng_skip(SEED_O_ODATE, (uint64_t)start);
    int64_t s_ckey = rng_skip(SEED_O_CKEY, (uint64_t)start);
    for (int64_t i = 0; i < count; i++) {
        int64_t idx1 = start + i + 1;
        int64_t ck = unif(&s_ckey, 1, max_ckey);
        /* dbgen mk_order: customers with custkey % 3 == 0 never order */
        int64_t delta = 1;
        while (ck % CUSTOMER_MORTALITY == 0) {
            ck += delta;
            ck = ck > max_ckey ? max_ckey : ck;
            delta *= -1;
        }
        int64_t od = unif(&s_odate, ODATE_MIN, ODATE_MAX);
        int64_t lc = unif(&s_lcnt, 1, 7);
        if (orderkey) orderkey[i] = make_orderkey(idx1);
        if (custkey) custkey[i] = ck;
        if (orderdate_epoch) orderdate_epoch[i] = (int32_t)(EPOCH_1992 + od - 1);
        if (lcnt) lcnt[i] = (int32_t)lc;
    }
}

void tpch_gen_part_type(double sf, int64_t start, int64_t count,
                        uint8_t* type_id)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_P_TYPE, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++)
            type_id[i] = (uint8_t)(unif(&s, 1, 150) - 1);
    }
}

int64_t tpch_gen_lineitem_partkey(double sf, int64_t ord_start,
                                  int64_t ord_count, int64_t* partkey)
{
    int64_t max_pkey = (int64_t)(200000.0 * sf + 0.5);
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t tlo = ord_count * tid / nt, thi = ord_count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < tlo; i++) off += unif(&s, 1, 7);
        int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)(ord_start + tlo));
        int64_t s_pkey =
            rng_skip(SEED_L_PKEY, (uint64_t)(ord_start + tlo) * 7);
        int64_t out = off;
        for (int64_t o = tlo; o < thi; o++) {
            int64_t lc = unif(&s_lcnt, 1, 7);
            for (int64_t l = 0; l < lc; l++)
                partkey[out++] = unif(&s_pkey, 1, max_pkey);
            s_pkey = rng_skip(s_pkey, (uint64_t)(7 - lc));
            written += lc;
        }
    }
    return written;
}

int64_t tpch_gen_lineitem_shipinstruct(double sf, int64_t ord_start,
                                       int64_t ord_count,
                                       uint8_t* shipinstruct)
{
    (void)sf;
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t tlo = ord_count * tid / nt, thi = ord_count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < tlo; i++) off += unif(&s, 1, 7);
        int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)(ord_start + tlo));
        int64_t s_inst =
            rng_skip(SEED_L_SINST, (uint64_t)(ord_start + tlo) * 7);
        int64_t out = off;
        for (int64_t o = tlo; o < thi; o++) {
            int64_t lc = unif(&s_lcnt, 1, 7);
            for (int64_t l = 0; l < lc; l++)
                shipinstruct[out++] = (uint8_t)(unif(&s_inst, 1, 4) - 1);
            s_inst = rng_skip(s_inst, (uint64_t)(7 - lc));
            written += lc;
        }
    }
    return written;
}

int64_t tpch_gen_lineitem_shipmode(double sf, int64_t ord_start,
                                   int64_t ord_count, uint8_t* shipmode)
{
    (void)sf;
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t tlo = ord_count * tid / nt, thi = ord_count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < tlo; i++) off += unif(&s, 1, 7);
        int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)(ord_start + tlo));
        int64_t s_mode =
            rng_skip(SEED_L_SMODE, (uint64_t)(ord_start + tlo) * 7);
        int64_t out = off;
        for (int64_t o = tlo; o < thi; o++) {
            int64_t lc = unif(&s_lcnt, 1, 7);
            for (int64_t l = 0; l < lc; l++)
                shipmode[out++] = (uint8_t)(unif(&s_mode, 1, 7) - 1);
            s_mode = rng_skip(s_mode, (uint64_t)(7 - lc));
            written += lc;
        }
    }
    return written;
}

void tpch_gen_part2(double sf, int64_t start, int64_t count, uint8_t* mfgr,
                    uint8_t* brand, uint8_t* container)
{
    tpch_gen_part3(sf, start, count, mfgr, brand, container, 0);
}

void tpch_gen_part3(double sf, int64_t start, int64_t count, uint8_t* mfgr,
                    uint8_t* brand, uint8_t* container, uint8_t* size)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t sm = rng_skip(SEED_P_MFG, (uint64_t)(start + lo));
        int64_t sb = rng_skip(SEED_P_BRND, (uint64_t)(start + lo));
        int64_t sc = rng_skip(SEED_P_CNTR, (uint64_t)(start + lo));
        int64_t sz = rng_skip(SEED_P_SIZE, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++) {
            int64_t m = unif(&sm, 1, 5);
            int64_t b = unif(&sb, 1, 5);
            int64_t c = unif(&sc, 1, 40);
            int64_t z = unif(&sz, 1, 50);
            if (mfgr) mfgr[i] = (uint8_t)m;
            if (brand) brand[i] = (uint8_t)(m * 10 + b);
            if (container) container[i] = (uint8_t)(c - 1);
            if (size) size[i] = (uint8_t)z;
        }
    }
}

void tpch_gen_partsupp(double sf, int64_t part_start, int64_t part_count,
                       int64_t* partkey, int64_t* suppkey,
                       int32_t* availqty, int64_t* supplycost_cents)
{
    int64_t S = tpch_supplier_count(sf);
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = part_count * tid / nt, hi = part_count * (tid + 1) / nt;
        int64_t sq =
            rng_skip(SEED_PS_QTY, (uint64_t)(part_start + lo) * 4);
        int64_t sc =
            rng_skip(SEED_PS_SCST, (uint64_t)(part_start + lo) * 4);
        for (int64_t p = lo; p < hi; p++) {
            int64_t pk = part_start + p + 1;
            for (int64_t i = 0; i < 4; i++) {
                int64_t out = p * 4 + i;
                if (partkey) partkey[out] = pk;
                if (suppkey) suppkey[out] = part_supplier(pk, i, S);
                int32_t aq = (int32_t)unif(&sq, 1, 9999);
                int64_t cost = unif(&sc, 100, 100000);
                if (availqty) availqty[out] = aq;
                if (supplycost_cents) supplycost_cents[out] = cost;
            }
        }
    }
}

/* dbgen dists.dss "colors" (92 words, file order) — p_name draws 5 of
 * these via the persistent permutation stream (dbgen rnd.c permute /
 * build.c agg_str; airlift RandomStringSequence is the same algorithm) */
static const char* P_COLORS[92] = {
    "almond", "antique", "aquamarine", "azure", "beige", "bisque", "black",
    "blanched", "blue", "blush", "brown", "burlywood", "burnished",
    "chartreuse", "chiffon", "chocolate", "coral", "cornflower", "cornsilk",
    "cream", "cyan", "dark", "deep", "dim", "dodger", "drab", "firebrick",
    "floral", "forest", "frosted", "gainsboro", "ghost", "goldenrod",
    "green", "grey", "honeydew", "hot", "indian", "ivory", "khaki", "lace",
    "lavender", "lawn", "lemon", "light", "lime", "linen", "magenta",
    "maroon", "medium", "metallic", "midnight", "mint", "misty", "moccasin",
    "navajo", "navy", "olive", "orange", "orchid", "pale", "papaya",
    "peach", "peru", "pink", "plum", "powder", "puff", "purple", "red",
    "rose", "rosy", "royal", "saddle", "salmon", "sandy", "seashell",
    "sienna", "sky", "slate", "smoke", "snow", "spring", "steel", "tan",
    "thistle", "tomato", "turquoise", "violet", "wheat", "white", "yellow"};

int32_t tpch_color_name(int32_t id, char* buf)
{
    if (id < 0 || id >= 92) return -1;
    const char* s = P_COLORS[id];
    int32_t n = 0;
    while (s[n]) {
        buf[n] = s[n];
        n++;
    }
    buf[n] = 0;
    return n;
}

int32_t tpch_color_id(const char* word)
{
    for (int32_t i = 0; i < 92; i++) {
        const char* a = P_COLORS[i];
        const char* b = word;
        while (*a && *a == *b) {
            a++;
            b++;
        }
        if (!*a && !*b) return i;
    }
    return -1;
}

/* word ids (0..91 into the colors list) of the 5 p_name words per part,
 * rows [0, count): per part the identity permutation is re-permuted with
 * 92 draws (j in [i,91] swap pass) and the first 5 entries are the name.
 * words[p*5 + j] = j-th name word of part p+1. */
void tpch_gen_part_name_words(double sf, int64_t count, uint8_t* words)
{
    (void)sf;
    uint8_t perm[92];
    int64_t s = SEED_P_NAME;
    for (int64_t p = 0; p < count; p++) {
        for (int i = 0; i < 92; i++) perm[i] = (uint8_t)i;
        for (int i = 0; i < 92; i++) {
            int64_t j = unif(&s, i, 91);
            uint8_t t = perm[j];
            perm[j] = perm[i];
            perm[i] = t;
        }
        for (int j = 0; j < 5; j++) words[p * 5 + j] = perm[j];
    }
}

void tpch_gen_customer_acctbal(double sf, int64_t start, int64_t count,
                               int64_t* acctbal_cents)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_C_ABAL, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++)
            acctbal_cents[i] = unif(&s, -99999, 999999);
    }
}

void tpch_gen_orders_priority(double sf, int64_t start, int64_t count,
                              uint8_t* priority)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        (void)sf;
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_PRIO, (uint64_t)(start + lo));
        for (int64_t i = lo; i < hi; i++)
            priority[i] = (uint8_t)(unif(&s, 1, 5) - 1);
    }
}

int64_t tpch_gen_lineitem_dates(double sf, int64_t ord_start,
                                int64_t ord_count, int64_t* orderkey,
                                int32_t* commitd, int32_t* receiptd)
{
    (void)sf;
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t tlo = ord_count * tid / nt, thi = ord_count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < tlo; i++) off += unif(&s, 1, 7);
        int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)(ord_start + tlo));
        int64_t s_odate = rng_skip(SEED_O_ODATE, (uint64_t)(ord_start + tlo));
        uint64_t l7 = (uint64_t)(ord_start + tlo) * 7;
        int64_t s_sdays = rng_skip(SEED_L_SDAYS, l7);
        int64_t s_cdays = rng_skip(SEED_L_CDAYS, l7);
        int64_t s_rdays = rng_skip(SEED_L_RDAYS, l7);
        int64_t out = off;
        for (int64_t o = tlo; o < thi; o++) {
            int64_t idx1 = ord_start + o + 1;
            int64_t ok = make_orderkey(idx1);
            int64_t od = unif(&s_odate, ODATE_MIN, ODATE_MAX);
            int64_t lc = unif(&s_lcnt, 1, 7);
            for (int64_t l = 0; l < lc; l++) {
                int64_t sd = unif(&s_sdays, 1, 121);
                int64_t cd = unif(&s_cdays, 30, 90);
                int64_t rd = unif(&s_rdays, 1, 30);
                if (orderkey) orderkey[out] = ok;
                if (commitd)
                    commitd[out] = (int32_t)(EPOCH_1992 + od + cd - 1);
                if (receiptd)
                    receiptd[out] =
                        (int32_t)(EPOCH_1992 + od + sd + rd - 1);
                out++;
            }
            uint64_t rest = (uint64_t)(7 - lc);
            s_sdays = rng_skip(s_sdays, rest);
            s_cdays = rng_skip(s_cdays, rest);
            s_rdays = rng_skip(s_rdays, rest);
            written += lc;
        }
    }
    return written;
}

void tpch_gen_orders(double sf, int64_t start, int64_t count,
                     int64_t* orderkey, int64_t* custkey,
                     int32_t* orderdate_epoch, int32_t* lcnt)
{
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = count * tid / nt, hi = count * (tid + 1) / nt;
        gen_orders_chunk(sf, start + lo, hi - lo,
                         orderkey ? orderkey + lo : 0,
                         custkey ? custkey + lo : 0,
                         orderdate_epoch ? orderdate_epoch + lo : 0,
                         lcnt ? lcnt + lo : 0);
    }
}

/* one chunk of lineitem rows for orders [ord_start, ord_start+ord_count),
 * written densely at out index 0; returns rows written. Single thread. */
static int64_t gen_lineitem_chunk(double sf, int64_t ord_start,
                                  int64_t ord_count, int64_t* orderkey,
                                  double* quantity, double* extendedprice,
                                  double* discount, double* tax,
                                  int32_t* shipdate_epoch, uint8_t* returnflag,
                                  uint8_t* linestatus, int64_t* suppkey)
{
    int64_t max_pkey = (int64_t)(200000.0 * sf + 0.5);
    int64_t n_supp = tpch_supplier_count(sf);
    int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
    int64_t s_odate = rng_skip(SEED_O_ODATE, (uint64_t)ord_start);
    uint64_t l7 = (uint64_t)ord_start * 7;
    int64_t s_qty = rng_skip(SEED_L_QTY, l7);
    int64_t s_disc = rng_skip(SEED_L_DISC, l7);
    int64_t s_tax = rng_skip(SEED_L_TAX, l7);
    int64_t s_pkey = rng_skip(SEED_L_PKEY, l7);
    int64_t s_sdays = rng_skip(SEED_L_SDAYS, l7);
    int64_t s_cdays = rng_skip(SEED_L_CDAYS, l7);
    int64_t s_rdays = rng_skip(SEED_L_RDAYS, l7);
    int64_t s_rflag = rng_skip(SEED_L_RFLAG, l7);
    int64_t s_suppn = rng_skip(SEED_L_SUPPN, l7);
    int64_t out = 0;
    for (int64_t o = 0; o < ord_count; o++) {
        int64_t idx1 = ord_start + o + 1;
        int64_t ok = make_orderkey(idx1);
        int64_t od = unif(&s_odate, ODATE_MIN, ODATE_MAX);
        int64_t lc = unif(&s_lcnt, 1, 7);
        int64_t rflag_draws = 0;
        for (int64_t l = 0; l < lc; l++) {
            int64_t qty = unif(&s_qty, 1, 50);
            int64_t d = unif(&s_disc, 0, 10);
            int64_t t = unif(&s_tax, 0, 8);
            int64_t pk = unif(&s_pkey, 1, max_pkey);
            int64_t sn = unif(&s_suppn, 0, 3);
            int64_t sd = unif(&s_sdays, 1, 121);
            (void)unif(&s_cdays, 30, 90); /* commitdate: stream consumed */
            int64_t rd = unif(&s_rdays, 1, 30);
            int64_t ship = od + sd;    /* day index */
            int64_t receipt = ship + rd;
            uint8_t rf;
            if (receipt <= CURRENT_IDX) {
                /* dbgen: pick_str over rflag dist {R|1, A|1}, drawn ONLY
                 * when the receipt date is in the past */
                int64_t j = unif(&s_rflag, 1, 2);
                rflag_draws++;
                rf = (j == 1) ? 'R' : 'A';
            } else {
                rf = 'N';
            }
            int64_t cents = qty * part_price_cents(pk);
            if (orderkey) orderkey[out] = ok;
            if (quantity) quantity[out] = (double)qty;
            if (extendedprice) extendedprice[out] = (double)cents / 100.0;
            if (discount) discount[out] = (double)d / 100.0;
            if (tax) tax[out] = (double)t / 100.0;
            if (shipdate_epoch)
                shipdate_epoch[out] = (int32_t)(EPOCH_1992 + ship - 1);
            if (returnflag) returnflag[out] = rf;
            if (linestatus) linestatus[out] = (ship <= CURRENT_IDX) ? 'F' : 'O';
            if (suppkey) suppkey[out] = part_supplier(pk, sn, n_supp);
            out++;
        }
        /* rowFinished: advance every per-line stream to usage 7/order
         * (dbgen row_stop / airlift rowFinished) */
        uint64_t rest = (uint64_t)(7 - lc);
        s_qty = rng_skip(s_qty, rest);
        s_disc = rng_skip(s_disc, rest);
        s_tax = rng_skip(s_tax, rest);
        s_pkey = rng_skip(s_pkey, rest);
        s_sdays = rng_skip(s_sdays, rest);
        s_cdays = rng_skip(s_cdays, rest);
        s_rdays = rng_skip(s_rdays, rest);
        s_rflag = rng_skip(s_rflag, (uint64_t)(7 - rflag_draws));
        s_suppn = rng_skip(s_suppn, rest);
    }
    return out;
}

void tpch_gen_orders_totalprice(double sf, int64_t ord_start,
                                int64_t ord_count, int64_t* totalprice_cents)
{
    /* dbgen mk_order: totalprice = sum over lines of
     *   eprice*(100-disc)/100*(100+tax)/100   (integer floor divisions)
     * (pinned by the q18 golden o_totalprice column) */
    int64_t max_pkey = (int64_t)(200000.0 * sf + 0.5);
#pragma omp parallel
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = ord_count * tid / nt, hi = ord_count * (tid + 1) / nt;
        int64_t s_lcnt = rng_skip(SEED_O_LCNT, (uint64_t)(ord_start + lo));
        uint64_t l7 = (uint64_t)(ord_start + lo) * 7;
        int64_t s_qty = rng_skip(SEED_L_QTY, l7);
        int64_t s_disc = rng_skip(SEED_L_DISC, l7);
        int64_t s_tax = rng_skip(SEED_L_TAX, l7);
        int64_t s_pkey = rng_skip(SEED_L_PKEY, l7);
        for (int64_t o = lo; o < hi; o++) {
            int64_t lc = unif(&s_lcnt, 1, 7);
            int64_t tp = 0;
            for (int64_t l = 0; l < lc; l++) {
                int64_t qty = unif(&s_qty, 1, 50);
                int64_t d = unif(&s_disc, 0, 10);
                int64_t t = unif(&s_tax, 0, 8);
                int64_t pk = unif(&s_pkey, 1, max_pkey);
                int64_t cents = qty * part_price_cents(pk);
                tp += cents * (100 - d) / 100 * (100 + t) / 100;
            }
            totalprice_cents[o] = tp;
            uint64_t rest = (uint64_t)(7 - lc);
            s_qty = rng_skip(s_qty, rest);
            s_disc = rng_skip(s_disc, rest);
            s_tax = rng_skip(s_tax, rest);
            s_pkey = rng_skip(s_pkey, rest);
        }
    }
}

int64_t tpch_gen_lineitem(double sf, int64_t ord_start, int64_t ord_count,
                          int64_t* orderkey, double* quantity,
                          double* extendedprice, double* discount, double* tax,
                          int32_t* shipdate_epoch, uint8_t* returnflag,
                          uint8_t* linestatus)
{
    /* parallel: compute per-thread lineitem offsets first, then fill */
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = ord_count * tid / nt, hi = ord_count * (tid + 1) / nt;
        /* lineitem offset of this thread's first order relative to
         * ord_start's first lineitem: scan the lcnt stream */
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < lo; i++) off += unif(&s, 1, 7);
        written += gen_lineitem_chunk(
            sf, ord_start + lo, hi - lo, orderkey ? orderkey + off : 0,
            quantity ? quantity + off : 0,
            extendedprice ? extendedprice + off : 0,
            discount ? discount + off : 0, tax ? tax + off : 0,
            shipdate_epoch ? shipdate_epoch + off : 0,
            returnflag ? returnflag + off : 0,
            linestatus ? linestatus + off : 0, 0);
    }
    return written;
}

int64_t tpch_gen_lineitem2(double sf, int64_t ord_start, int64_t ord_count,
                           int64_t* orderkey, double* quantity,
                           double* extendedprice, double* discount,
                           double* tax, int32_t* shipdate_epoch,
                           uint8_t* returnflag, uint8_t* linestatus,
                           int64_t* suppkey)
{
    int64_t written = 0;
#pragma omp parallel reduction(+ : written)
    {
#ifdef _OPENMP
        extern int omp_get_num_threads(void), omp_get_thread_num(void);
        int nt = omp_get_num_threads(), tid = omp_get_thread_num();
#else
        int nt = 1, tid = 0;
#endif
        int64_t lo = ord_count * tid / nt, hi = ord_count * (tid + 1) / nt;
        int64_t s = rng_skip(SEED_O_LCNT, (uint64_t)ord_start);
        int64_t off = 0;
        for (int64_t i = 0; i < lo; i++) off += unif(&s, 1, 7);
        written += gen_lineitem_chunk(
            sf, ord_start + lo, hi - lo, orderkey ? orderkey + off : 0,
            quantity ? quantity + off : 0,
            extendedprice ? extendedprice + off : 0,
            discount ? discount + off : 0, tax ? tax + off : 0,
            shipdate_epoch ? shipdate_epoch + off : 0,
            returnflag ? returnflag + off : 0,
            linestatus ? linestatus + off : 0,
            suppkey ? suppkey + off : 0);
    }
    return written;
}
