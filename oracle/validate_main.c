/* validate_main.c — generate TPC-H at a given SF, run the oracle's Q1/Q3,
 * print results in the golden-file format of
 * presto-product-tests/.../hive_tpch/q01.result / q03.result
 * (exact decimal columns; avg columns rounded HALF_UP to scale 2).
 * Usage: validate [sf]           (default 1) */
#include "oracle.h"
#include "tpchgen.h"
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

static void print_dec(FILE* f, __int128 v, int scale)
{
    /* print v / 10^scale with 'scale' decimals */
    char buf[64];
    int pos = 63;
    buf[pos] = 0;
    __int128 x = v < 0 ? -v : v;
    for (int i = 0; i < scale; i++) {
        buf[--pos] = '0' + (int)(x % 10);
        x /= 10;
    }
    buf[--pos] = '.';
    if (x == 0) buf[--pos] = '0';
    while (x > 0) {
        buf[--pos] = '0' + (int)(x % 10);
        x /= 10;
    }
    if (v < 0) buf[--pos] = '-';
    fputs(buf + pos, f);
}

/* round_half_up(num/den) at scale 2: returns hundredths */
static int64_t avg_half_up(__int128 num_scaled, int64_t den, int add_scale)
{
    /* num is in 10^-k ticks; we want hundredths: num*10^(2-k)/den rounded */
    __int128 n = num_scaled;
    for (int i = 0; i < add_scale; i++) n *= 10;
    __int128 q = (2 * n + den) / (2 * (__int128)den); /* half-up, n>=0 */
    return (int64_t)q;
}

static void epoch_to_ymd(int32_t e, int* y, int* m, int* d)
{
    /* civil from days since 1970-01-01 (Howard Hinnant's algorithm) */
    long z = e + 719468;
    long era = (z >= 0 ? z : z - 146096) / 146097;
    unsigned doe = (unsigned)(z - era * 146097);
    unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
    long yy = (long)yoe + era * 400;
    unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
    unsigned mp = (5 * doy + 2) / 153;
    unsigned dd = doy - (153 * mp + 2) / 5 + 1;
    unsigned mm = mp + (mp < 10 ? 3 : -9);
    *y = (int)(yy + (mm <= 2));
    *m = (int)mm;
    *d = (int)dd;
}

int main(int argc, char** argv)
{
    double sf = argc > 1 ? atof(argv[1]) : 1.0;
    int64_t n_li = tpch_lineitem_count(sf);
    int64_t n_ord = tpch_orders_count(sf);
    int64_t n_cust = tpch_customer_count(sf);
    fprintf(stderr, "sf=%g lineitem=%lld orders=%lld customer=%lld\n", sf,
            (long long)n_li, (long long)n_ord, (long long)n_cust);

    double *qty = malloc(n_li * 8), *ep = malloc(n_li * 8),
           *disc = malloc(n_li * 8), *tax = malloc(n_li * 8);
    int32_t* sd = malloc(n_li * 4);
    uint8_t *rf = malloc(n_li), *ls = malloc(n_li);
    int64_t* lok = malloc(n_li * 8);
    tpch_gen_lineitem(sf, 0, n_ord, lok, qty, ep, disc, tax, sd, rf, ls);

    q1_group_t g[6];
    int32_t ng = oracle_q1(n_li, qty, ep, disc, tax, sd, rf, ls, g);
    for (int i = 0; i < ng; i++) {
        printf("%c|%c|", g[i].returnflag, g[i].linestatus);
        print_dec(stdout, (__int128)g[i].sum_qty_units * 100, 2);
        printf("|");
        print_dec(stdout, g[i].sum_base_cents, 2);
        printf("|");
        print_dec(stdout, g[i].sum_disc_1e4, 4);
        printf("|");
        print_dec(stdout,
                  ((__int128)g[i].sum_charge_1e6_hi << 64) |
                      (__int128)(unsigned __int128)g[i].sum_charge_1e6_lo,
                  6);
        printf("|");
        print_dec(stdout, avg_half_up((__int128)g[i].sum_qty_units * 100,
                                      g[i].count_order, 0), 2);
        printf("|");
        print_dec(stdout,
                  avg_half_up(g[i].sum_base_cents, g[i].count_order, 0), 2);
        printf("|");
        print_dec(stdout,
                  avg_half_up(g[i].sum_disc_cents, g[i].count_order, 0), 2);
        printf("|%lld|\n", (long long)g[i].count_order);
    }

    int64_t *ook = malloc(n_ord * 8), *ock = malloc(n_ord * 8);
    int32_t* od = malloc(n_ord * 4);
    tpch_gen_orders(sf, 0, n_ord, ook, ock, od, 0);
    int64_t* ckey = malloc(n_cust * 8);
    uint8_t* cseg = malloc(n_cust);
    tpch_gen_customer(sf, 0, n_cust, ckey, cseg);

    q3_row_t rows[10];
    int32_t nr = oracle_q3(n_cust, ckey, cseg, n_ord, ook, ock, od, n_li, lok,
                           ep, disc, sd, 10, rows);
    for (int i = 0; i < nr; i++) {
        int y, m, d;
        epoch_to_ymd(rows[i].orderdate, &y, &m, &d);
        printf("%lld|", (long long)rows[i].orderkey);
        /* golden q03.result prints revenue with trailing zeros trimmed
         * (DECIMAL rendering, e.g. 390324.061) */
        char rev[64];
        snprintf(rev, 64, "%lld.%04lld", (long long)(rows[i].revenue_1e4 / 10000),
                 (long long)(rows[i].revenue_1e4 % 10000));
        int len = (int)strlen(rev);
        while (rev[len - 1] == '0' && rev[len - 2] != '.') rev[--len] = 0;
        printf("%s|%04d-%02d-%02d|%d|\n", rev, y, m, d, rows[i].shippriority);
    }
    return 0;
}
