/*
 * otbx_gen.h — deterministic, counter-based synthetic TPC-H-shaped data
 * generator, shared between the CPU oracle (gcc) and the GPU generator
 * kernels (hipcc). Pure functions of (seed, table, column, global row index)
 * so CPU and GPU produce bit-identical tables with no staging transfer.
 *
 * Shapes per SURVEY.md §8(d) / BASELINE.md: lineitem SF×6M, orders SF×1.5M,
 * customer SF×150k; dates are int32 day numbers relative to 1992-01-01.
 *
 * Floating-point determinism: value construction uses only int→double
 * conversion, a division, an addition not fed by a multiply, and a single
 * multiply — no mul+add chains a compiler could contract into FMA. Compile
 * with -ffp-contract=off anyway (both gcc and hipcc) for belt and braces.
 *
 * This header is test/bench INFRASTRUCTURE (the dbgen analog), not part of
 * the offloaded executor path.
 */
#ifndef OTBX_GEN_H
#define OTBX_GEN_H

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define OTBX_FN __host__ __device__ static inline
#else
#define OTBX_FN static inline
#endif

/* --- calendar constants (day 0 = 1992-01-01) --- */
#define OTBX_DATE_MIN 1       /* 1992-01-02: earliest l_shipdate            */
#define OTBX_DATE_MAX 2526    /* 1998-12-01: latest  l_shipdate             */
#define OTBX_ODATE_MAX 2405   /* 1998-08-02: latest  o_orderdate            */
#define OTBX_Q1_CUTOFF 2436   /* 1998-09-02 = date '1998-12-01' - 90 days   */
#define OTBX_Q3_DATE 1169     /* 1995-03-15 (Q3 order/ship date constant)   */
#define OTBX_LS_CUTOFF 1263   /* 1995-06-17: linestatus O/F boundary        */

#define OTBX_DEFAULT_SEED 42ull

/* rows per SF */
#define OTBX_LINEITEM_PER_SF 6000000ll
#define OTBX_ORDERS_PER_SF   1500000ll
#define OTBX_CUSTOMER_PER_SF 150000ll
#define OTBX_LINES_PER_ORDER 4ll

/* table ids */
#define OTBX_T_LINEITEM 1u
#define OTBX_T_ORDERS   2u
#define OTBX_T_CUSTOMER 3u
#define OTBX_T_PART     4u

#define OTBX_PART_PER_SF 200000ll
#define OTBX_PART_NTYPES 150    /* p_type domain; Q9's p_name LIKE filter is
                                 * restated as p_type % 17 == 0 (~6%) */

/* column ids (per table) */
#define OTBX_C_SHIPDATE  1u
#define OTBX_C_QTY       2u
#define OTBX_C_PRICE     3u
#define OTBX_C_DISCOUNT  4u
#define OTBX_C_TAX       5u
#define OTBX_C_RFLAG     6u
#define OTBX_C_PARTKEY   7u
#define OTBX_C_ODATE     1u
#define OTBX_C_CUSTKEY   2u
#define OTBX_C_MKTSEG    1u

OTBX_FN uint64_t otbx_splitmix64(uint64_t x)
{
    x += 0x9e3779b97f4a7c15ull;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
    return x ^ (x >> 31);
}

OTBX_FN uint64_t otbx_rnd(uint64_t seed, uint32_t table, uint32_t col, uint64_t row)
{
    return otbx_splitmix64(seed ^ ((uint64_t)table << 56) ^ ((uint64_t)col << 48) ^ row);
}

/* ---- lineitem: global row i (0-based) ---- */

OTBX_FN int64_t otbx_li_orderkey(uint64_t i)
{
    return (int64_t)(i / OTBX_LINES_PER_ORDER) + 1; /* 4 lines/order, clustered */
}

OTBX_FN int32_t otbx_li_shipdate(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_SHIPDATE, i);
    return (int32_t)(OTBX_DATE_MIN + (int32_t)(r % (OTBX_DATE_MAX - OTBX_DATE_MIN + 1)));
}

OTBX_FN double otbx_li_quantity(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_QTY, i);
    return (double)(1 + (int32_t)(r % 50)); /* l_quantity ∈ [1,50] */
}

OTBX_FN double otbx_li_extendedprice(uint64_t seed, uint64_t i)
{
    /* qty × unit price in [900.00, 2100.00] with cent granularity.
     * price = 900 + cents/100 : div then add (not contractible);
     * qty*price : lone multiply. */
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_PRICE, i);
    double cents = (double)(int32_t)(r % 120001);
    double price = 900.0 + cents / 100.0;
    return otbx_li_quantity(seed, i) * price;
}

OTBX_FN double otbx_li_discount(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_DISCOUNT, i);
    return (double)(int32_t)(r % 11) / 100.0; /* [0.00, 0.10] */
}

OTBX_FN double otbx_li_tax(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_TAX, i);
    return (double)(int32_t)(r % 9) / 100.0; /* [0.00, 0.08] */
}

/* returnflag/linestatus tied to shipdate as in dbgen (flag depends on
 * receipt vs current date) so exactly the 4 real TPC-H Q1 groups occur:
 * A/F, R/F (old), N/F (boundary band), N/O (recent). */
OTBX_FN uint8_t otbx_li_linestatus(uint64_t seed, uint64_t i)
{
    (void)seed;
    return otbx_li_shipdate(seed, i) > OTBX_LS_CUTOFF ? (uint8_t)'O' : (uint8_t)'F';
}

OTBX_FN uint8_t otbx_li_returnflag(uint64_t seed, uint64_t i)
{
    int32_t d = otbx_li_shipdate(seed, i);
    if (d > OTBX_LS_CUTOFF - 180)
        return (uint8_t)'N';
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_RFLAG, i);
    return (r & 1) ? (uint8_t)'A' : (uint8_t)'R';
}

OTBX_FN int64_t otbx_li_partkey(uint64_t seed, uint64_t i, int64_t nparts)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_LINEITEM, OTBX_C_PARTKEY, i);
    return 1 + (int64_t)(r % (uint64_t)nparts);
}

/* ---- part: global row i (0-based) ---- */

OTBX_FN int64_t otbx_part_partkey(uint64_t i) { return (int64_t)i + 1; }

OTBX_FN uint8_t otbx_part_type(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_PART, 1u, i);
    return (uint8_t)(r % OTBX_PART_NTYPES);
}

/* calendar year index of a day number (day 0 = 1992-01-01); boundaries use
 * the real 1992..1998 lengths (leap 1992/1996) */
OTBX_FN int32_t otbx_year_of_day(int32_t d)
{
    const int32_t b[8] = {0, 366, 731, 1096, 1461, 1827, 2192, 2558};
    int32_t y = 0;
    for (int i = 1; i < 8; i++)
        if (d >= b[i]) y = i;
    return y; /* 0 = 1992 … 6 = 1998 */
}

/* ---- orders: global row i (0-based); o_orderkey dense 1..norders ---- */

OTBX_FN int64_t otbx_ord_orderkey(uint64_t i) { return (int64_t)i + 1; }

OTBX_FN int64_t otbx_ord_custkey(uint64_t seed, uint64_t i, int64_t ncust)
{
    /* ncust = n_lineitem/40 is 0 for tiny tables; clamp like li_partkey's
     * call sites do (a bare % 0 is SIGFPE on the CPU side, UB on GPU) */
    if (ncust < 1) ncust = 1;
    uint64_t r = otbx_rnd(seed, OTBX_T_ORDERS, OTBX_C_CUSTKEY, i);
    return 1 + (int64_t)(r % (uint64_t)ncust);
}

/* BASELINE config 5: skewed distribution keys — 20% of custkeys ("hot",
 * keys 1..ncust/5) receive 80% of the orders. Pure integer arithmetic:
 * bit-identical on CPU and GPU. */
OTBX_FN int64_t otbx_ord_custkey_skewed(uint64_t seed, uint64_t i, int64_t ncust)
{
    if (ncust < 1) ncust = 1;
    uint64_t r = otbx_rnd(seed, OTBX_T_ORDERS, OTBX_C_CUSTKEY, i);
    uint64_t r2 = otbx_splitmix64(r);
    int64_t nhot = ncust / 5 > 0 ? ncust / 5 : 1;
    if ((int32_t)(r % 10) < 8)
        return 1 + (int64_t)(r2 % (uint64_t)nhot);
    int64_t ncold = ncust - nhot > 0 ? ncust - nhot : 1;
    return 1 + nhot + (int64_t)(r2 % (uint64_t)ncold);
}

OTBX_FN int32_t otbx_ord_orderdate(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_ORDERS, OTBX_C_ODATE, i);
    return (int32_t)(OTBX_DATE_MIN + (int32_t)(r % (OTBX_ODATE_MAX - OTBX_DATE_MIN + 1)));
}

OTBX_FN int32_t otbx_ord_shippriority(uint64_t i) { (void)i; return 0; }

/* ---- customer: global row i (0-based) ---- */

OTBX_FN int64_t otbx_cust_custkey(uint64_t i) { return (int64_t)i + 1; }

/* 5 segments, 0 = 'BUILDING' (the Q3 predicate segment, 20% selectivity) */
OTBX_FN uint8_t otbx_cust_mktsegment(uint64_t seed, uint64_t i)
{
    uint64_t r = otbx_rnd(seed, OTBX_T_CUSTOMER, OTBX_C_MKTSEG, i);
    return (uint8_t)(r % 5);
}

/* ---- sharding: 1 DataNode shard ↔ 1 GPU.
 * Distribution keys: lineitem+orders by orderkey, customer by custkey
 * (locator 'H' semantics, shardid = hash mod nShards — pgxc/shard/shardmap.c:2231;
 * restated as key % nranks with dense keys). Row counts must divide nranks. */

/* global row index of local row l of the lineitem shard on rank r of n */
OTBX_FN uint64_t otbx_li_global_row(uint64_t l, uint32_t rank, uint32_t nranks)
{
    uint64_t order_local = l / OTBX_LINES_PER_ORDER;
    uint64_t line = l % OTBX_LINES_PER_ORDER;
    /* orderkey = global/4 + 1 ≡ rank (mod n)  ⇒ global/4 ≡ rank-1 (mod n) */
    uint64_t res = (rank + nranks - 1) % nranks;
    return OTBX_LINES_PER_ORDER * (order_local * nranks + res) + line;
}

OTBX_FN uint64_t otbx_ord_global_row(uint64_t l, uint32_t rank, uint32_t nranks)
{
    /* orderkey = global + 1 ≡ rank (mod n) */
    uint64_t res = (rank + nranks - 1) % nranks;
    return l * nranks + res;
}

OTBX_FN uint64_t otbx_cust_global_row(uint64_t l, uint32_t rank, uint32_t nranks)
{
    uint64_t res = (rank + nranks - 1) % nranks;
    return l * nranks + res;
}

#endif /* OTBX_GEN_H */
