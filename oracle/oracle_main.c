/*
 * oracle_main.c — CLI driver for the CPU oracle. Used by bench.py's
 * cpu_baseline leg (kind "port", scalar, 1 thread) and for debugging.
 *
 * Usage:
 *   oracle_cli q1   --rows N [--rank R --nranks W] [--cutoff D]
 *   oracle_cli q3   --rows N [--rank R --nranks W] [--date D] [--segment S]
 *   oracle_cli scan --rows N [--rank R --nranks W] [--cutoff D]
 *   oracle_cli q9   --rows N [--rank R --nranks W]
 * --rows is the GLOBAL lineitem row count (orders = rows/4, customer = rows/40).
 * Prints one JSON line: timing covers the executor only, not generation.
 */
#include "oracle.h"
#include "otbx_gen.h"
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <time.h>

static double now_s(void)
{
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return ts.tv_sec + 1e-9 * ts.tv_nsec;
}

int main(int argc, char **argv)
{
    if (argc < 2) { fprintf(stderr, "usage: oracle_cli q1|q3|scan --rows N ...\n"); return 2; }
    const char *query = argv[1];
    int64_t rows = 6000000;
    uint32_t rank = 0, nranks = 1;
    int32_t cutoff = OTBX_Q1_CUTOFF, q3date = OTBX_Q3_DATE;
    int segment = 0;
    uint64_t seed = OTBX_DEFAULT_SEED;
    for (int i = 2; i + 1 < argc; i += 2) {
        if (!strcmp(argv[i], "--rows")) rows = atoll(argv[i + 1]);
        else if (!strcmp(argv[i], "--rank")) rank = (uint32_t)atoi(argv[i + 1]);
        else if (!strcmp(argv[i], "--nranks")) nranks = (uint32_t)atoi(argv[i + 1]);
        else if (!strcmp(argv[i], "--cutoff")) cutoff = atoi(argv[i + 1]);
        else if (!strcmp(argv[i], "--date")) q3date = atoi(argv[i + 1]);
        else if (!strcmp(argv[i], "--segment")) segment = atoi(argv[i + 1]);
        else if (!strcmp(argv[i], "--seed")) seed = strtoull(argv[i + 1], NULL, 10);
        else { fprintf(stderr, "unknown arg %s\n", argv[i]); return 2; }
    }

    ora_lineitem li;
    if (ora_gen_lineitem(&li, seed, rows, rank, nranks)) { fprintf(stderr, "gen failed\n"); return 1; }

    if (!strcmp(query, "scan")) {
        double t0 = now_s();
        int64_t cnt = ora_scan_count_shipdate_le(li.l_shipdate, li.n, cutoff);
        double dt = now_s() - t0;
        printf("{\"query\":\"scan\",\"rows\":%lld,\"count\":%lld,\"seconds\":%.6f}\n",
               (long long)li.n, (long long)cnt, dt);
    } else if (!strcmp(query, "q1")) {
        ora_q1_group g[8];
        int ng = 0;
        double t0 = now_s();
        if (ora_q1_partial(&li, cutoff, g, &ng)) { fprintf(stderr, "q1 failed\n"); return 1; }
        ora_q1_finalize(g, ng);
        double dt = now_s() - t0;
        printf("{\"query\":\"q1\",\"rows\":%lld,\"seconds\":%.6f,\"groups\":[",
               (long long)li.n, dt);
        for (int i = 0; i < ng; i++)
            printf("%s{\"rf\":\"%c\",\"ls\":\"%c\",\"sum_qty\":%.17g,\"sum_base_price\":%.17g,"
                   "\"sum_disc_price\":%.17g,\"sum_charge\":%.17g,\"avg_qty\":%.17g,"
                   "\"avg_price\":%.17g,\"avg_disc\":%.17g,\"count\":%lld}",
                   i ? "," : "", g[i].returnflag, g[i].linestatus, g[i].sum_qty,
                   g[i].sum_base_price, g[i].sum_disc_price, g[i].sum_charge,
                   g[i].avg_qty, g[i].avg_price, g[i].avg_disc,
                   (long long)g[i].count_order);
        printf("]}\n");
    } else if (!strcmp(query, "q3")) {
        ora_orders od;
        ora_customer cu;
        if (ora_gen_orders(&od, seed, rows / 4, rows / 40, rank, nranks, 0) ||
            ora_gen_customer(&cu, seed, rows / 40, rank, nranks)) {
            fprintf(stderr, "gen failed\n"); return 1;
        }
        ora_q3_row *out = NULL;
        int64_t ng = 0;
        double t0 = now_s();
        if (ora_q3_partial(&cu, &od, &li, (uint8_t)segment, q3date, &out, &ng)) {
            fprintf(stderr, "q3 failed\n"); return 1;
        }
        int64_t k = ora_q3_topk(out, ng, 10);
        double dt = now_s() - t0;
        printf("{\"query\":\"q3\",\"rows\":%lld,\"ngroups\":%lld,\"seconds\":%.6f,\"top\":[",
               (long long)li.n, (long long)ng, dt);
        for (int64_t i = 0; i < k; i++)
            printf("%s{\"orderkey\":%lld,\"revenue\":%.17g,\"orderdate\":%d,\"prio\":%d}",
                   i ? "," : "", (long long)out[i].l_orderkey, out[i].revenue,
                   out[i].o_orderdate, out[i].o_shippriority);
        printf("]}\n");
        free(out);
        ora_free_orders(&od);
        ora_free_customer(&cu);
    } else if (!strcmp(query, "q9")) {
        ora_orders od;
        ora_part pt;
        if (ora_gen_orders(&od, seed, rows / 4, rows / 40, rank, nranks, 0) ||
            ora_gen_part(&pt, seed, rows / 30 > 0 ? rows / 30 : 1)) {
            fprintf(stderr, "gen failed\n"); return 1;
        }
        ora_q9_group g[8];
        int ng = 0;
        double t0 = now_s();
        if (ora_q9_partial(&pt, &od, &li, 17, 0, g, &ng)) {
            fprintf(stderr, "q9 failed\n"); return 1;
        }
        double dt = now_s() - t0;
        printf("{\"query\":\"q9\",\"rows\":%lld,\"seconds\":%.6f,\"groups\":[",
               (long long)li.n, dt);
        for (int i = 0; i < ng; i++)
            printf("%s{\"year\":%d,\"revenue\":%.17g,\"count\":%lld}",
                   i ? "," : "", 1992 + g[i].year, g[i].revenue,
                   (long long)g[i].count_rows);
        printf("]}\n");
        ora_free_orders(&od);
        ora_free_part(&pt);
    } else {
        fprintf(stderr, "unknown query %s\n", query);
        return 2;
    }
    ora_free_lineitem(&li);
    return 0;
}
