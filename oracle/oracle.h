/*
 * oracle.h — CPU oracle for the otbx offload: a standalone C restatement of
 * the OpenTenBase/PostgreSQL DataNode executor hot path
 * (SeqScan → HashJoin → HashAggregate) plus the two-phase combine the
 * Coordinator applies.
 *
 * TEST INFRASTRUCTURE ONLY. Only tests/, __graft_entry__.smoke() and
 * bench.py's cpu_baseline leg may call this library — it is the parity
 * checker and the reported CPU baseline, never the shipped compute path.
 *
 * Reference citations are to /root/reference (OpenTenBase @2025-08-08);
 * this is a restatement of behaviour, not a copy.
 */
#ifndef OTBX_ORACLE_H
#define OTBX_ORACLE_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum {
    ORA_OK = 0,
    ORA_ERR_OVERFLOW = 1,  /* analog of ereport(ERROR, value out of range) */
    ORA_ERR_OOM = 2,
    ORA_ERR_INVALID = 3
} ora_status;

/* ---- columnar tables (SoA; the staged form of heap pages) ---- */

typedef struct {
    int64_t n;
    int64_t *l_orderkey;
    double  *l_quantity, *l_extendedprice, *l_discount, *l_tax;
    uint8_t *l_returnflag, *l_linestatus;
    int32_t *l_shipdate;
    int64_t *l_partkey;   /* appended: Q9-mix FK into part */
} ora_lineitem;

typedef struct {
    int64_t n;
    int64_t *o_orderkey, *o_custkey;
    int32_t *o_orderdate, *o_shippriority;
} ora_orders;

typedef struct {
    int64_t n;
    int64_t *c_custkey;
    uint8_t *c_mktsegment;
} ora_customer;

typedef struct {
    int64_t n;
    int64_t *p_partkey;
    uint8_t *p_type;
} ora_part;

/* generation (otbx_gen.h): n_global must be divisible by nranks */
ora_status ora_gen_lineitem(ora_lineitem *t, uint64_t seed, int64_t n_global,
                            uint32_t rank, uint32_t nranks);
ora_status ora_gen_orders(ora_orders *t, uint64_t seed, int64_t n_global,
                          int64_t ncust_global, uint32_t rank, uint32_t nranks,
                          int skew /* config-5 hot-custkey skew */);
ora_status ora_gen_customer(ora_customer *t, uint64_t seed, int64_t n_global,
                            uint32_t rank, uint32_t nranks);
/* part is replicated (small dimension table): every rank generates all rows */
ora_status ora_gen_part(ora_part *t, uint64_t seed, int64_t n_global);
void ora_free_lineitem(ora_lineitem *t);
void ora_free_orders(ora_orders *t);
void ora_free_customer(ora_customer *t);
void ora_free_part(ora_part *t);

/* ---- config 2: SeqScan + qual + COUNT(*) ----
 * restates ExecScan/ExecQual per-tuple loop (execScan.c:140-363) +
 * int8inc (int8.c:714). */
int64_t ora_scan_count_shipdate_le(const int32_t *shipdate, int64_t n, int32_t cutoff);

/* ---- TPC-H Q1: scan + filter + hash aggregate, two-phase ---- */

typedef struct {
    uint8_t returnflag, linestatus;
    /* partial (AGGSPLIT_INITIAL_SERIAL) transition states */
    double sum_qty, sum_base_price, sum_disc_price, sum_charge; /* float8pl chains */
    double qty_acc[3], price_acc[3], disc_acc[3]; /* float8_accum [N,Sx,Sxx] (float.c:2823) */
    int64_t count_order;                          /* int8inc (int8.c:714) */
    /* finalized */
    double avg_qty, avg_price, avg_disc;          /* float8_avg: Sx/N */
} ora_q1_group;

/* DataNode side: Partial Aggregate below RemoteSubplan
 * (agg_fill_hash_table nodeAgg.c:2609; groups sorted by (rf,ls) on emit). */
ora_status ora_q1_partial(const ora_lineitem *t, int32_t cutoff_day,
                          ora_q1_group out[8], int *ngroups);
/* Coordinator side: Finalize Aggregate combine (float8pl for plain sums,
 * float8_combine float.c:2725 for accum states, int8pl for counts;
 * nodeAgg.c:3912,4260). */
ora_status ora_q1_combine(ora_q1_group acc[8], int *nacc,
                          const ora_q1_group *part, int npart);
void ora_q1_finalize(ora_q1_group *g, int n);

/* ---- TPC-H Q3: customer ⋈ orders ⋈ lineitem + hash agg + top-k ---- */

typedef struct {
    int64_t l_orderkey;
    double revenue;        /* sum(l_extendedprice * (1 - l_discount)) */
    int32_t o_orderdate, o_shippriority;
} ora_q3_row;

/* DataNode side: returns ALL groups (caller frees *out). Group key is
 * (l_orderkey, o_orderdate, o_shippriority) ≡ l_orderkey (o_* functionally
 * determined). */
ora_status ora_q3_partial(const ora_customer *c, const ora_orders *o,
                          const ora_lineitem *l, uint8_t segment,
                          int32_t q3date, ora_q3_row **out, int64_t *ngroups);
/* ORDER BY revenue DESC, o_orderdate ASC LIMIT k — sorts in place, returns
 * min(k, n) (the Coordinator merge-sort analog, execFragment.c:4035-4059). */
int64_t ora_q3_topk(ora_q3_row *rows, int64_t n, int64_t k);

/* ---- Q9-mix (BASELINE config 5): revenue by order year over parts of a
 * filtered type class — lineitem ⋈ part (p_type % typemod == typeval) ⋈
 * orders, GROUP BY year(o_orderdate). The Q9 shape: two joins below an
 * aggregate over a COMPUTED key. ---- */

typedef struct {
    int32_t year;          /* 0 = 1992 … 6 = 1998 */
    double revenue;        /* sum(l_extendedprice * (1 - l_discount)) */
    int64_t count_rows;
} ora_q9_group;

ora_status ora_q9_partial(const ora_part *p, const ora_orders *o,
                          const ora_lineitem *l, uint8_t typemod,
                          uint8_t typeval, ora_q9_group out[8], int *ngroups);

/* ---- generic entry points for NULL/edge-case parity tests ---- */

typedef struct {
    int64_t key;
    int key_isnull;        /* NULL keys form one group (execGrouping.c match) */
    int64_t count_star;    /* count(*)  — counts every row                    */
    int64_t count_v;       /* count(v)  — strict, skips NULL inputs           */
    double sum_v;          /* sum(v)    — strict float8pl chain               */
    int sum_isnull;        /* SUM over empty/all-NULL group = NULL            */
    double acc[3];         /* avg(v) float8_accum state                       */
} ora_agg_group;

/* hash aggregate: group by (nullable) i64 key, aggregate (nullable) f64 v.
 * val_null/key_null may be NULL pointers (= no NULLs). Caller frees *out.
 * Groups emitted sorted by (key_isnull, key). */
ora_status ora_agg_i64(const int64_t *keys, const uint8_t *key_null,
                       const double *vals, const uint8_t *val_null,
                       int64_t n, ora_agg_group **out, int64_t *ngroups);

/* inner hash join on i64 keys: build over b, probe with p; emits (bidx,pidx)
 * pairs. NULL keys never match (strict equality op). Caller frees outputs. */
ora_status ora_join_i64(const int64_t *bkeys, const uint8_t *bnull, int64_t nb,
                        const int64_t *pkeys, const uint8_t *pnull, int64_t np,
                        int64_t **out_bidx, int64_t **out_pidx, int64_t *nout);



/* ---- extended join types (nodeHashjoin.c FSM fill states) ----
 * join_type: 0 inner, 1 left, 2 semi, 3 anti, 4 right, 5 full — restating
 * the HJ_* state machine (nodeHashjoin.c:139-144):
 *   inner match                       -> (bidx, pidx)
 *   left/full unmatched probe row     -> (-1, pidx)   HJ_FILL_OUTER_TUPLE
 *     (incl. NULL-key probe rows)        (nodeHashjoin.c:142,668; NULL-key
 *                                         outer path :543-552)
 *   right/full unmatched build row    -> (bidx, -1)   HJ_FILL_INNER_TUPLES
 *     (incl. NULL-key build rows)        (nodeHashjoin.c:143,693;
 *                                         ExecScanHashTableForUnmatched
 *                                         nodeHash.c:2322)
 *   semi: (-1, pidx) once per probe row with >=1 match (JOIN_SEMI advances
 *     to the next outer after the first match, nodeHashjoin.c:572)
 *   anti: (-1, pidx) per probe row with no match (nodeHashjoin.c:631)
 * bk2/pk2 NULL => single-key join. With two keys, the join key is the ROW
 * (k1,k2): a row with EITHER key NULL never matches (strict equality,
 * ExecHashGetHashValue nodeHash.c:2026 keep_nulls semantics; multi-key
 * hash combine = rotate-left-1 then xor, nodeHash.c:2059 — restated at
 * 64 bit; parity is on result sets, not hash values). Caller frees. */
ora_status ora_join_ext(const int64_t *bk1, const uint8_t *bn1,
                        const int64_t *bk2, const uint8_t *bn2, int64_t nb,
                        const int64_t *pk1, const uint8_t *pn1,
                        const int64_t *pk2, const uint8_t *pn2, int64_t np,
                        int join_type,
                        int64_t **out_bidx, int64_t **out_pidx, int64_t *nout);

/* ---- two-key hash aggregate (multi-key GROUP BY) ----
 * Group identity is (k1_isnull, k1, k2_isnull, k2) with NULL==NULL for
 * grouping (execGrouping.c:295 + null-match semantics :525); aggregates as
 * ora_agg_i64. Emitted sorted by (k1_isnull, k1, k2_isnull, k2). */
typedef struct {
    int64_t key1, key2;
    int key1_isnull, key2_isnull;
    int64_t count_star;
    int64_t count_v;
    double sum_v;
    int sum_isnull;
    double acc[3];
} ora_agg_group2;

ora_status ora_agg_i64x2(const int64_t *k1, const uint8_t *k1null,
                         const int64_t *k2, const uint8_t *k2null,
                         const double *vals, const uint8_t *val_null,
                         int64_t n, ora_agg_group2 **out, int64_t *ngroups);


/* ---- exact decimal (scaled-int64) aggregate with int128 sum ----
 * Restates the reference's HAVE_INT128 numeric aggregation: group state =
 * Int128AggState {N, sumX} (numeric.c:5072, do_int128_accum :4998 region),
 * transition int8_avg_accum (numeric.c:5365); sum(bigint) promotes to
 * numeric and cannot overflow (int8_sum numeric.c:6206). Values are
 * scaled-decimal int64 (e.g. NUMERIC(15,2) money in cents); the emitted
 * 128-bit sum (two's-complement hi/lo) is exact, so parity is bit-exact.
 * AVG finalization (numeric division, sum/N) is the caller's / combine
 * phase's job — the partial state is what a DataNode ships. */
typedef struct {
    int64_t key;
    int key_isnull;
    int64_t count_star;
    int64_t count_v;
    int64_t sum_hi;          /* int128 two's-complement high word */
    uint64_t sum_lo;
    int sum_isnull;
} ora_dec_group;

ora_status ora_agg_i64_dec(const int64_t *keys, const uint8_t *key_null,
                           const int64_t *vals, const uint8_t *val_null,
                           int64_t n, ora_dec_group **out, int64_t *ngroups);


/* ---- N-key (1..8 columns) group-by and join ----
 * Group/join identity is the ROW of key columns with NULL==NULL for
 * grouping (execGrouping.c:295,:525) and any-NULL-never-matches for joins
 * (nodeHash.c:2026); hash = iterated rotate-left-1 xor over per-column
 * hashes (ExecHashGetHashValue multi-key combine, nodeHash.c:2059).
 * Groups carry the DEFINING ROW INDEX instead of N key values — the
 * reference's hash table likewise stores the representative tuple
 * (execGrouping.c firstTuple); the caller reads the keys back by index. */
#define ORA_MAX_KEYS 8
typedef struct {
    int nkeys;
    const int64_t *keys[ORA_MAX_KEYS];
    const uint8_t *nulls[ORA_MAX_KEYS];   /* per column; entries may be NULL */
} ora_keyset;

typedef struct {
    int64_t row_idx;          /* defining row (representative tuple) */
    int64_t count_star;
    int64_t count_v;
    double sum_v;
    int sum_isnull;
} ora_aggn_group;

/* emitted sorted by row_idx (deterministic; group identity is the row) */
ora_status ora_agg_i64n(const ora_keyset *ks, const double *vals,
                        const uint8_t *val_null, int64_t n,
                        ora_aggn_group **out, int64_t *ngroups);

/* join_type as ora_join_ext (0..5) */
ora_status ora_join_i64n(const ora_keyset *bks, int64_t nb,
                         const ora_keyset *pks, int64_t np, int join_type,
                         int64_t **out_bidx, int64_t **out_pidx,
                         int64_t *nout);

#ifdef __cplusplus
}
#endif
#endif
