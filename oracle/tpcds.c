/* tpcds.c — TPC-DS columns generator + Q17/Q72 CPU oracle.
 * See tpcds.h for the parity-pinning statement (oracle as the single
 * data source for config 5; spec-shaped scaling and domains).
 */
#include "tpcds.h"
#include <stdlib.h>
#include <string.h>

/* deterministic counter-based stream: splitmix64 over (table, col, row) */
static inline uint64_t mix64(uint64_t x)
{
    x += 0x9e3779b97f4a7c15ull;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
    return x ^ (x >> 31);
}
static inline uint64_t rnd(uint64_t table, uint64_t col, uint64_t row)
{
    return mix64(table * 0x100000001b3ull + col * 0x9e3779b1ull +
                 mix64(row + 0x5DEECE66Dull));
}

int64_t dsgen_store_sales_count(double sf) { return (int64_t)(2880404.0 * sf); }
int64_t dsgen_store_returns_count(double sf)
{
    return dsgen_store_sales_count(sf) / 10;
}
int64_t dsgen_catalog_sales_count(double sf)
{
    return (int64_t)(1441548.0 * sf);
}
int64_t dsgen_catalog_returns_count(double sf)
{
    return dsgen_catalog_sales_count(sf) / 10;
}
int64_t dsgen_item_count(double sf)
{
    /* spec: 18k @1, 102k @30, 204k @100, 300k @300 (stepped; linear
     * interpolation between the spec anchors keeps SF1/SF100 exact) */
    if (sf <= 1.0) return 18000;
    if (sf >= 100.0) return (int64_t)(204000 * (sf / 100.0));
    return 18000 + (int64_t)((204000 - 18000) * (sf - 1.0) / 99.0);
}
int64_t dsgen_store_count(double sf)
{
    if (sf <= 1.0) return 12;
    if (sf >= 100.0) return (int64_t)(402 * (sf / 100.0) < 402 ? 402
                                      : 402);
    return 12 + (int64_t)((402 - 12) * (sf - 1.0) / 99.0);
}
int64_t dsgen_warehouse_count(double sf)
{
    if (sf <= 1.0) return 5;
    if (sf >= 100.0) return 15;
    return 5 + (int64_t)(10 * (sf - 1.0) / 99.0);
}
int64_t dsgen_customer_count(double sf)
{
    int64_t n = (int64_t)(20000.0 * sf);
    return n < 100000 ? 100000 : n; /* 100k @1, 2M @100 (spec anchors) */
}
int64_t dsgen_promotion_count(double sf)
{
    if (sf <= 1.0) return 300;
    if (sf >= 100.0) return 1000;
    return 300 + (int64_t)(700 * (sf - 1.0) / 99.0);
}
int64_t dsgen_inventory_count(double sf)
{
    /* 261 weekly snapshots x catalog half of items x warehouses:
     * 11,745,000 @SF1, 399,330,000 @SF100 (spec row counts) */
    return 261ll * (dsgen_item_count(sf) / 2) * dsgen_warehouse_count(sf);
}

/* proleptic-Gregorian year/quarter from day index (day 0 = 1900-01-01) */
static void day_to_ymq(int32_t day, int32_t* year, int32_t* quarter)
{
    int32_t y = 1900;
    for (;;) {
        int leap = (y % 4 == 0 && y % 100 != 0) || y % 400 == 0;
        int32_t dy = 365 + leap;
        if (day < dy) break;
        day -= dy;
        y++;
    }
    static const int md[12] = {31, 28, 31, 30, 31, 30,
                               31, 31, 30, 31, 30, 31};
    int leap = (y % 4 == 0 && y % 100 != 0) || y % 400 == 0;
    int m = 0;
    for (; m < 12; m++) {
        int32_t dm = md[m] + (m == 1 && leap ? 1 : 0);
        if (day < dm) break;
        day -= dm;
    }
    *year = y;
    *quarter = m / 3 + 1;
}

void dsgen_date_dim(int32_t* d_year, int32_t* d_qname, int32_t* d_week_seq)
{
    for (int32_t i = 0; i < DSGEN_DATE_COUNT; i++) {
        int32_t y, q;
        day_to_ymq(i, &y, &q);
        if (d_year) d_year[i] = y;
        if (d_qname) d_qname[i] = y * 4 + (q - 1);
        if (d_week_seq) d_week_seq[i] = i / 7;
    }
}

void dsgen_item(double sf, int64_t* item_id_num, int64_t* price_cents)
{
    int64_t n = dsgen_item_count(sf);
    for (int64_t i = 0; i < n; i++) {
        if (item_id_num) item_id_num[i] = (i + 2) / 2; /* sk=i+1 pairs */
        if (price_cents) price_cents[i] = 100 + (int64_t)(rnd(7, 1, i) % 9900);
    }
}

void dsgen_store(double sf, uint8_t* state)
{
    int64_t n = dsgen_store_count(sf);
    for (int64_t i = 0; i < n; i++)
        state[i] = (uint8_t)(rnd(8, 1, i) % 9);
}

void dsgen_cdemo(uint8_t* marital)
{
    /* demographics are a cross product in the spec: codes cycle by sk */
    for (int64_t i = 0; i < DSGEN_CDEMO_COUNT; i++)
        marital[i] = (uint8_t)(i % 5);
}

void dsgen_hdemo(uint8_t* buy_potential)
{
    for (int64_t i = 0; i < DSGEN_HDEMO_COUNT; i++)
        buy_potential[i] = (uint8_t)(i % 6);
}

void dsgen_store_sales(double sf, int64_t start, int64_t count,
                       int32_t* sold_date, int64_t* item_sk,
                       int64_t* customer_sk, int64_t* store_sk,
                       int64_t* ticket, int32_t* quantity)
{
    int64_t n_item = dsgen_item_count(sf);
    int64_t n_store = dsgen_store_count(sf);
    int64_t n_cust = dsgen_customer_count(sf);
#pragma omp parallel for schedule(static)
    for (int64_t j = 0; j < count; j++) {
        int64_t i = start + j;
        int64_t t = i / 12; /* ~12 line items per ticket */
        if (sold_date)
            sold_date[j] = DSGEN_SALES_DATE0 +
                           (int32_t)(rnd(1, 1, t) % DSGEN_SALES_DAYS);
        if (item_sk) item_sk[j] = (int64_t)(rnd(1, 2, i) % n_item) + 1;
        if (customer_sk)
            customer_sk[j] = (int64_t)(rnd(1, 3, t) % n_cust) + 1;
        if (store_sk) store_sk[j] = (int64_t)(rnd(1, 4, t) % n_store) + 1;
        if (ticket) ticket[j] = t + 1;
        if (quantity) quantity[j] = 1 + (int32_t)(rnd(1, 5, i) % 100);
    }
}

void dsgen_store_returns(double sf, int64_t start, int64_t count,
                         int32_t* ret_date, int64_t* item_sk,
                         int64_t* customer_sk, int64_t* ticket,
                         int32_t* ret_quantity)
{
    int64_t n_ss = dsgen_store_sales_count(sf);
    int64_t n_item = dsgen_item_count(sf);
    int64_t n_cust = dsgen_customer_count(sf);
#pragma omp parallel for schedule(static)
    for (int64_t j = 0; j < count; j++) {
        int64_t i = start + j;
        /* each return references a sampled sale row (same customer,
         * item, ticket triple), returned 1..90 days later */
        int64_t r = (int64_t)(rnd(2, 1, i) % n_ss);
        int64_t t = r / 12;
        int32_t sold = DSGEN_SALES_DATE0 +
                       (int32_t)(rnd(1, 1, t) % DSGEN_SALES_DAYS);
        if (ret_date) ret_date[j] = sold + 1 + (int32_t)(rnd(2, 2, i) % 90);
        if (item_sk) item_sk[j] = (int64_t)(rnd(1, 2, r) % n_item) + 1;
        if (customer_sk)
            customer_sk[j] = (int64_t)(rnd(1, 3, t) % n_cust) + 1;
        if (ticket) ticket[j] = t + 1;
        if (ret_quantity)
            ret_quantity[j] = 1 + (int32_t)(rnd(2, 3, i) % 100);
    }
}

void dsgen_catalog_sales(double sf, int64_t start, int64_t count,
                         int32_t* sold_date, int32_t* ship_date,
                         int64_t* item_sk, int64_t* bill_customer_sk,
                         int64_t* order_number, int32_t* quantity,
                         int64_t* bill_cdemo_sk, int64_t* bill_hdemo_sk,
                         int64_t* promo_sk)
{
    int64_t n_item = dsgen_item_count(sf);
    int64_t n_cust = dsgen_customer_count(sf);
    int64_t n_promo = dsgen_promotion_count(sf);
#pragma omp parallel for schedule(static)
    for (int64_t j = 0; j < count; j++) {
        int64_t i = start + j;
        int64_t o = i / 8; /* ~8 lines per order */
        int32_t sold = DSGEN_SALES_DATE0 +
                       (int32_t)(rnd(3, 1, o) % DSGEN_SALES_DAYS);
        if (sold_date) sold_date[j] = sold;
        if (ship_date) ship_date[j] = sold + 2 + (int32_t)(rnd(3, 2, i) % 28);
        if (item_sk) item_sk[j] = (int64_t)(rnd(3, 3, i) % n_item) + 1;
        if (bill_customer_sk)
            bill_customer_sk[j] = (int64_t)(rnd(3, 4, o) % n_cust) + 1;
        if (order_number) order_number[j] = o + 1;
        if (quantity) quantity[j] = 1 + (int32_t)(rnd(3, 5, i) % 100);
        if (bill_cdemo_sk)
            bill_cdemo_sk[j] =
                (int64_t)(rnd(3, 6, o) % DSGEN_CDEMO_COUNT) + 1;
        if (bill_hdemo_sk)
            bill_hdemo_sk[j] =
                (int64_t)(rnd(3, 7, o) % DSGEN_HDEMO_COUNT) + 1;
        if (promo_sk) {
            uint64_t p = rnd(3, 8, i);
            promo_sk[j] = (p % 5) == 0 ? 0 /* NULL, ~20% */
                                       : (int64_t)(p % n_promo) + 1;
        }
    }
}

void dsgen_catalog_returns(double sf, int64_t start, int64_t count,
                           int64_t* item_sk, int64_t* order_number)
{
    int64_t n_cs = dsgen_catalog_sales_count(sf);
    int64_t n_item = dsgen_item_count(sf);
#pragma omp parallel for schedule(static)
    for (int64_t j = 0; j < count; j++) {
        int64_t i = start + j;
        int64_t r = (int64_t)(rnd(4, 1, i) % n_cs);
        if (item_sk) item_sk[j] = (int64_t)(rnd(3, 3, r) % n_item) + 1;
        if (order_number) order_number[j] = r / 8 + 1;
    }
}

void dsgen_inventory(double sf, int64_t start, int64_t count,
                     int32_t* inv_date, int64_t* item_sk,
                     int64_t* warehouse_sk, int32_t* qty_on_hand)
{
    int64_t n_wh = dsgen_warehouse_count(sf);
    int64_t n_ci = dsgen_item_count(sf) / 2; /* catalog half of items */
#pragma omp parallel for schedule(static)
    for (int64_t j = 0; j < count; j++) {
        int64_t i = start + j;
        int64_t wh = i % n_wh;
        int64_t it = (i / n_wh) % n_ci;
        int64_t wk = i / (n_wh * n_ci);
        if (inv_date)
            inv_date[j] = DSGEN_SALES_DATE0 + (int32_t)(wk * 7);
        if (item_sk) item_sk[j] = it * 2 + 1; /* odd skus = catalog */
        if (warehouse_sk) warehouse_sk[j] = wh + 1;
        if (qty_on_hand) qty_on_hand[j] = (int32_t)(rnd(5, 1, i) % 1000);
    }
}

/* ------------------------------------------------------------------ */
/* oracle                                                              */
/* ------------------------------------------------------------------ */
static inline uint64_t h3(int64_t a, int64_t b, int64_t c)
{
    return mix64(mix64((uint64_t)a * 0x9e3779b1ull + (uint64_t)b) ^
                 (uint64_t)c * 0x85ebca6bull);
}
static int64_t cap_for(int64_t n)
{
    int64_t c = 64;
    while (c < n * 2) c <<= 1;
    return c;
}

typedef struct {
    int64_t item, state;
    int64_t c1, s1, q1, c2, s2, q2, c3, s3, q3;
} q17_group;

static int cmp_q17(const void* a, const void* b)
{
    const q17_group *x = a, *y = b;
    if (x->item != y->item) return x->item < y->item ? -1 : 1;
    return x->state < y->state ? -1 : (x->state > y->state ? 1 : 0);
}

int64_t oracle_ds_q17(double sf, int32_t q0, int64_t max_out,
                      int64_t* g_item, int32_t* g_state, int64_t* cnt_ss,
                      int64_t* sum_ss, int64_t* sq_ss, int64_t* cnt_sr,
                      int64_t* sum_sr, int64_t* sq_sr, int64_t* cnt_cs,
                      int64_t* sum_cs, int64_t* sq_cs)
{
    int64_t n_ss = dsgen_store_sales_count(sf);
    int64_t n_sr = dsgen_store_returns_count(sf);
    int64_t n_cs = dsgen_catalog_sales_count(sf);
    int64_t n_item = dsgen_item_count(sf);
    int32_t* dy = malloc(DSGEN_DATE_COUNT * 4);
    int32_t* dq = malloc(DSGEN_DATE_COUNT * 4);
    dsgen_date_dim(dy, dq, NULL);
    int64_t* iid = malloc(n_item * 8);
    dsgen_item(sf, iid, NULL);
    int64_t n_store = dsgen_store_count(sf);
    uint8_t* sstate = malloc(n_store);
    dsgen_store(sf, sstate);

    /* store_sales in quarter q0, keyed by (customer, item, ticket) */
    int32_t* ss_d = malloc(n_ss * 4);
    int64_t* ss_i = malloc(n_ss * 8);
    int64_t* ss_c = malloc(n_ss * 8);
    int64_t* ss_s = malloc(n_ss * 8);
    int64_t* ss_t = malloc(n_ss * 8);
    int32_t* ss_q = malloc(n_ss * 4);
    dsgen_store_sales(sf, 0, n_ss, ss_d, ss_i, ss_c, ss_s, ss_t, ss_q);
    int64_t n_sel = 0;
    for (int64_t i = 0; i < n_ss; i++)
        if (dq[ss_d[i]] == q0) n_sel++;
    int64_t cap = cap_for(n_sel < 16 ? 16 : n_sel);
    int64_t* hk1 = malloc(cap * 8); /* customer */
    int64_t* hk2 = malloc(cap * 8); /* item */
    int64_t* hk3 = malloc(cap * 8); /* ticket */
    int64_t* hrow = malloc(cap * 8);
    int32_t* hnext = malloc((n_sel < 16 ? 16 : n_sel) * 4);
    memset(hrow, 0xff, cap * 8);
    int64_t* sel_rows = malloc((n_sel < 16 ? 16 : n_sel) * 8);
    int64_t ns = 0;
    for (int64_t i = 0; i < n_ss; i++) {
        if (dq[ss_d[i]] != q0) continue;
        int64_t row = ns++;
        sel_rows[row] = i;
        uint64_t h = h3(ss_c[i], ss_i[i], ss_t[i]);
        int64_t p = (int64_t)(h & (uint64_t)(cap - 1));
        for (;;) {
            if (hrow[p] < 0) {
                hk1[p] = ss_c[i];
                hk2[p] = ss_i[i];
                hk3[p] = ss_t[i];
                hnext[row] = -1;
                hrow[p] = row;
                break;
            }
            if (hk1[p] == ss_c[i] && hk2[p] == ss_i[i] &&
                hk3[p] == ss_t[i]) {
                hnext[row] = (int32_t)hrow[p];
                hrow[p] = row;
                break;
            }
            p = (p + 1) & (cap - 1);
        }
    }

    /* store_returns joined on (cust, item, ticket), return quarter in
     * {q0..q0+2}: collect matched (ss_row, sr_row) pairs keyed by
     * (cust, item) for the catalog leg */
    int32_t* sr_d = malloc(n_sr * 4);
    int64_t* sr_i = malloc(n_sr * 8);
    int64_t* sr_c = malloc(n_sr * 8);
    int64_t* sr_t = malloc(n_sr * 8);
    int32_t* sr_q = malloc(n_sr * 4);
    dsgen_store_returns(sf, 0, n_sr, sr_d, sr_i, sr_c, sr_t, sr_q);
    /* second map keyed (cust,item) -> chain of (ss_row, sr_row) pairs.
     * One return row can match several sale rows of its ticket (the
     * same item may appear twice), so size with headroom and fail
     * loudly on overflow rather than dropping pairs. */
    int64_t pair_cap_n = n_sr * 4 + 16;
    int64_t* p_ss = malloc(pair_cap_n * 8 * 4);
    int64_t* p_sr = p_ss + pair_cap_n;
    int64_t* p_cust = p_sr + pair_cap_n;
    int64_t* p_item = p_cust + pair_cap_n;
    int64_t np = 0;
    for (int64_t i = 0; i < n_sr; i++) {
        int32_t rq = dq[sr_d[i]];
        if (rq < q0 || rq > q0 + 2) continue;
        uint64_t h = h3(sr_c[i], sr_i[i], sr_t[i]);
        int64_t p = (int64_t)(h & (uint64_t)(cap - 1));
        for (;;) {
            if (hrow[p] < 0) break;
            if (hk1[p] == sr_c[i] && hk2[p] == sr_i[i] &&
                hk3[p] == sr_t[i]) {
                for (int64_t r = hrow[p]; r >= 0; r = hnext[r]) {
                    if (np >= pair_cap_n) abort(); /* never drop pairs */
                    p_ss[np] = sel_rows[r];
                    p_sr[np] = i;
                    p_cust[np] = sr_c[i];
                    p_item[np] = sr_i[i];
                    np++;
                }
                break;
            }
            p = (p + 1) & (cap - 1);
        }
    }

    /* pair map keyed by (cust, item) */
    int64_t cap2 = cap_for(np < 16 ? 16 : np);
    int64_t* m_c = malloc(cap2 * 8);
    int64_t* m_i = malloc(cap2 * 8);
    int64_t* m_head = malloc(cap2 * 8);
    int32_t* m_next = malloc((np < 16 ? 16 : np) * 4);
    memset(m_head, 0xff, cap2 * 8);
    for (int64_t j = 0; j < np; j++) {
        uint64_t h = h3(p_cust[j], p_item[j], 0x1234);
        int64_t p = (int64_t)(h & (uint64_t)(cap2 - 1));
        for (;;) {
            if (m_head[p] < 0) {
                m_c[p] = p_cust[j];
                m_i[p] = p_item[j];
                m_next[j] = -1;
                m_head[p] = j;
                break;
            }
            if (m_c[p] == p_cust[j] && m_i[p] == p_item[j]) {
                m_next[j] = (int32_t)m_head[p];
                m_head[p] = j;
                break;
            }
            p = (p + 1) & (cap2 - 1);
        }
    }

    /* catalog_sales leg + grouping by (item_id_num, state) */
    int64_t gcap = cap_for(4096);
    q17_group* groups = calloc(gcap, sizeof(q17_group));
    int64_t* g_used = malloc(gcap * 8);
    memset(g_used, 0xff, gcap * 8);
    int64_t n_groups = 0;
    int32_t* cs_d = malloc(n_cs * 4);
    int64_t* cs_i = malloc(n_cs * 8);
    int64_t* cs_c = malloc(n_cs * 8);
    int32_t* cs_q = malloc(n_cs * 4);
    dsgen_catalog_sales(sf, 0, n_cs, cs_d, NULL, cs_i, cs_c, NULL, cs_q,
                        NULL, NULL, NULL);
    for (int64_t k = 0; k < n_cs; k++) {
        int32_t cq = dq[cs_d[k]];
        if (cq < q0 || cq > q0 + 2) continue;
        uint64_t h = h3(cs_c[k], cs_i[k], 0x1234);
        int64_t p = (int64_t)(h & (uint64_t)(cap2 - 1));
        for (;;) {
            if (m_head[p] < 0) break;
            if (m_c[p] == cs_c[k] && m_i[p] == cs_i[k]) {
                for (int64_t j = m_head[p]; j >= 0; j = m_next[j]) {
                    int64_t ssr = p_ss[j], srr = p_sr[j];
                    int64_t item = iid[cs_i[k] - 1];
                    int64_t state = sstate[ss_s[ssr] - 1];
                    uint64_t gh = h3(item, state, 0x77);
                    int64_t gp = (int64_t)(gh & (uint64_t)(gcap - 1));
                    for (;;) {
                        if (g_used[gp] < 0) {
                            g_used[gp] = 1;
                            groups[gp].item = item;
                            groups[gp].state = state;
                            n_groups++;
                            break;
                        }
                        if (groups[gp].item == item &&
                            groups[gp].state == state)
                            break;
                        gp = (gp + 1) & (gcap - 1);
                    }
                    q17_group* g = &groups[gp];
                    int64_t q_ss = ss_q[ssr], q_sr = sr_q[srr],
                            q_cs = cs_q[k];
                    g->c1++; g->s1 += q_ss; g->q1 += q_ss * q_ss;
                    g->c2++; g->s2 += q_sr; g->q2 += q_sr * q_sr;
                    g->c3++; g->s3 += q_cs; g->q3 += q_cs * q_cs;
                }
                break;
            }
            p = (p + 1) & (cap2 - 1);
        }
    }

    q17_group* out = malloc((n_groups ? n_groups : 1) *
                            sizeof(q17_group));
    int64_t m = 0;
    for (int64_t p = 0; p < gcap; p++)
        if (g_used[p] >= 0) out[m++] = groups[p];
    qsort(out, m, sizeof(q17_group), cmp_q17);
    int64_t emit = m < max_out ? m : max_out;
    for (int64_t j = 0; j < emit; j++) {
        g_item[j] = out[j].item;
        g_state[j] = (int32_t)out[j].state;
        cnt_ss[j] = out[j].c1; sum_ss[j] = out[j].s1; sq_ss[j] = out[j].q1;
        cnt_sr[j] = out[j].c2; sum_sr[j] = out[j].s2; sq_sr[j] = out[j].q2;
        cnt_cs[j] = out[j].c3; sum_cs[j] = out[j].s3; sq_cs[j] = out[j].q3;
    }
    free(dy); free(dq); free(iid); free(sstate);
    free(ss_d); free(ss_i); free(ss_c); free(ss_s); free(ss_t); free(ss_q);
    free(hk1); free(hk2); free(hk3); free(hrow); free(hnext);
    free(sel_rows); free(sr_d); free(sr_i); free(sr_c); free(sr_t);
    free(sr_q); free(p_ss); free(m_c); free(m_i); free(m_head);
    free(m_next); free(groups); free(g_used); free(cs_d); free(cs_i);
    free(cs_c); free(cs_q); free(out);
    return m;
}

typedef struct {
    int64_t item, wh;
    int32_t week;
    int64_t no_promo, promo, total;
} q72_group;

static int cmp_q72(const void* a, const void* b)
{
    const q72_group *x = a, *y = b;
    if (x->item != y->item) return x->item < y->item ? -1 : 1;
    if (x->wh != y->wh) return x->wh < y->wh ? -1 : 1;
    return x->week - y->week;
}

int64_t oracle_ds_q72(double sf, int32_t year, int32_t marital,
                      int32_t buypot, int64_t max_out, int64_t* g_item,
                      int64_t* g_wh, int32_t* g_week, int64_t* no_promo,
                      int64_t* promo, int64_t* total)
{
    int64_t n_cs = dsgen_catalog_sales_count(sf);
    int64_t n_cr = dsgen_catalog_returns_count(sf);
    int64_t n_inv = dsgen_inventory_count(sf);
    int64_t n_item = dsgen_item_count(sf);
    int32_t* dy = malloc(DSGEN_DATE_COUNT * 4);
    int32_t* dw = malloc(DSGEN_DATE_COUNT * 4);
    dsgen_date_dim(dy, NULL, dw);
    int64_t* iid = malloc(n_item * 8);
    dsgen_item(sf, iid, NULL);

    /* catalog_returns multiplicity per (item, order): LEFT JOIN
     * multiplies rows by max(1, matches) */
    int64_t crcap = cap_for(n_cr < 16 ? 16 : n_cr);
    int64_t* cr_k = malloc(crcap * 8);
    int32_t* cr_n = malloc(crcap * 4);
    memset(cr_k, 0xff, crcap * 8);
    {
        int64_t* ci = malloc(n_cr * 8);
        int64_t* co = malloc(n_cr * 8);
        dsgen_catalog_returns(sf, 0, n_cr, ci, co);
        for (int64_t i = 0; i < n_cr; i++) {
            int64_t key = ci[i] * 0x100000000ll + co[i];
            uint64_t h = mix64((uint64_t)key);
            int64_t p = (int64_t)(h & (uint64_t)(crcap - 1));
            for (;;) {
                if (cr_k[p] == -1) {
                    cr_k[p] = key;
                    cr_n[p] = 1;
                    break;
                }
                if (cr_k[p] == key) {
                    cr_n[p]++;
                    break;
                }
                p = (p + 1) & (crcap - 1);
            }
        }
        free(ci); free(co);
    }

    /* inventory qty by (item, warehouse, week-date) — the join is
     * inv_date in the sold week (d1.d_week_seq == d2.d_week_seq);
     * inventory snapshots are weekly so at most one row matches per
     * warehouse */
    int64_t n_wh = dsgen_warehouse_count(sf);
    int64_t n_ci = n_item / 2;
    (void)n_inv;

    int64_t gcap = cap_for(n_cs / 32 + 65536);
    q72_group* groups = calloc(gcap, sizeof(q72_group));
    int64_t* g_used = malloc(gcap * 8);
    memset(g_used, 0xff, gcap * 8);
    int64_t n_groups = 0;

    int32_t* cs_sold = malloc(n_cs * 4);
    int32_t* cs_ship = malloc(n_cs * 4);
    int64_t* cs_i = malloc(n_cs * 8);
    int64_t* cs_on = malloc(n_cs * 8);
    int32_t* cs_q = malloc(n_cs * 4);
    int64_t* cs_cd = malloc(n_cs * 8);
    int64_t* cs_hd = malloc(n_cs * 8);
    int64_t* cs_p = malloc(n_cs * 8);
    dsgen_catalog_sales(sf, 0, n_cs, cs_sold, cs_ship, cs_i, NULL, cs_on,
                        cs_q, cs_cd, cs_hd, cs_p);
    for (int64_t k = 0; k < n_cs; k++) {
        if (dy[cs_sold[k]] != year) continue;
        /* demographics filters: the demographics dimensions are exact
         * cross products (spec), so cd_marital_status of sk is
         * (sk-1)%5 — the filter is evaluated through that column
         * definition instead of materializing the dimension */
        if ((cs_cd[k] - 1) % 5 != marital) continue;
        if ((cs_hd[k] - 1) % 6 != buypot) continue;
        if (cs_ship[k] <= cs_sold[k] + 5) continue;
        if ((cs_i[k] & 1) == 0) continue; /* inventory covers odd skus */
        int64_t it_idx = (cs_i[k] - 1) / 2;
        if (it_idx >= n_ci) continue;
        /* inventory snapshot of the sold week: snapshots carry week
         * numbers D0/7 .. D0/7+260, so the matching inventory week
         * index is wk - D0/7 */
        int32_t wk = dw[cs_sold[k]];
        int64_t wk_idx = wk - DSGEN_SALES_DATE0 / 7;
        if (wk_idx < 0 || wk_idx >= 261) continue;
        /* catalog-returns multiplicity for this (item, order) */
        int64_t key = cs_i[k] * 0x100000000ll + cs_on[k];
        uint64_t h = mix64((uint64_t)key);
        int64_t p = (int64_t)(h & (uint64_t)(crcap - 1));
        int32_t mult = 1;
        for (;;) {
            if (cr_k[p] == -1) break;
            if (cr_k[p] == key) {
                mult = cr_n[p];
                break;
            }
            p = (p + 1) & (crcap - 1);
        }
        for (int64_t wh = 0; wh < n_wh; wh++) {
            /* the weekly inventory snapshot for (item, warehouse, week)
             * is exactly row inv_row of the inventory table; its
             * quantity is read through the column definition instead of
             * materializing the 400M-row table on the host */
            int64_t inv_row =
                wk_idx * (n_wh * n_ci) + it_idx * n_wh + wh;
            int32_t qoh = (int32_t)(rnd(5, 1, inv_row) % 1000);
            if (qoh >= cs_q[k]) continue;
            int64_t item = iid[cs_i[k] - 1];
            uint64_t gh = h3(item, wh + 1, wk);
            int64_t gp = (int64_t)(gh & (uint64_t)(gcap - 1));
            for (;;) {
                if (g_used[gp] < 0) {
                    g_used[gp] = 1;
                    groups[gp].item = item;
                    groups[gp].wh = wh + 1;
                    groups[gp].week = wk;
                    n_groups++;
                    break;
                }
                if (groups[gp].item == item && groups[gp].wh == wh + 1 &&
                    groups[gp].week == wk)
                    break;
                gp = (gp + 1) & (gcap - 1);
            }
            q72_group* g = &groups[gp];
            if (cs_p[k] == 0)
                g->no_promo += mult;
            else
                g->promo += mult;
            g->total += mult;
        }
    }
    q72_group* out = malloc((n_groups ? n_groups : 1) *
                            sizeof(q72_group));
    int64_t m = 0;
    for (int64_t p = 0; p < gcap; p++)
        if (g_used[p] >= 0) out[m++] = groups[p];
    qsort(out, m, sizeof(q72_group), cmp_q72);
    int64_t emit = m < max_out ? m : max_out;
    for (int64_t j = 0; j < emit; j++) {
        g_item[j] = out[j].item;
        g_wh[j] = out[j].wh;
        g_week[j] = out[j].week;
        no_promo[j] = out[j].no_promo;
        promo[j] = out[j].promo;
        total[j] = out[j].total;
    }
    free(dy); free(dw); free(iid); free(cr_k); free(cr_n);
    free(groups); free(g_used); free(cs_sold); free(cs_ship); free(cs_i);
    free(cs_on); free(cs_q); free(cs_cd); free(cs_hd); free(cs_p);
    free(out);
    return m;
}
