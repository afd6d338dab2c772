/*
 * oracle.c — CPU restatement of the OpenTenBase DataNode executor hot path.
 * See oracle.h header comment: TEST INFRASTRUCTURE, never the product path.
 *
 * Structure mirrors the reference executor tuple-at-a-time loops:
 *   - scan+qual:       ExecScan/ExecScanFetch loop, execScan.c:140-363
 *   - hash join:       chained-bucket table; insert-at-head
 *                      ExecHashTableInsert nodeHash.c:1828, bucket split
 *                      ExecHashGetBucketAndBatch nodeHash.c:2141 (single
 *                      batch: our build sides fit memory; grace batching is
 *                      SURVEY §8f.4), probe walk ExecScanHashBucket
 *                      nodeHash.c:2174, join FSM ExecHashJoinImpl
 *                      nodeHashjoin.c:186 (inner join states only)
 *   - hash aggregate:  simplehash-style open addressing, linear probe,
 *                      fillfactor 0.9 (simplehash.h:187), lookup-or-insert
 *                      LookupTupleHashEntry execGrouping.c:295,
 *                      agg_fill_hash_table nodeAgg.c:2609
 *   - transition fns:  int8inc int8.c:714 (overflow → error), float8pl
 *                      float.c:970 (CHECKFLOATVAL float.c:58), float8_accum
 *                      float.c:2823 (Youngs-Cramer [N,Sx,Sxx])
 *   - NULL semantics:  strict transfns skip NULL input (nodeAgg.c:743 /
 *                      EEOP_AGG_STRICT_TRANS checks), sum state starts NULL
 *                      (nodeAgg.c:614: strict transfn with no initcond),
 *                      NULL group keys compare equal for grouping
 *                      (TupleHashTableMatch/execGrouping.c:520 via
 *                      ExecQualAndReset with IS NOT DISTINCT semantics)
 *   - two-phase:       float8_combine float.c:2725, int8pl; the Finalize
 *                      Aggregate / Remote Subquery Scan / Partial Aggregate
 *                      plan shape of opentenbase_c_aggregation.out:504-511
 *
 * Hashing: parity is defined on RESULTS, not hash values,
 * so this oracle hashes with splitmix64 rather than restating hash_any
 * (access/hash/hashfunc.c:110) — bucket population differs, result sets do not.
 */
#include "oracle.h"
#include "otbx_gen.h"

#include <stdlib.h>
#include <string.h>
#include <math.h>

/* ================= generation ================= */

ora_status ora_gen_lineitem(ora_lineitem *t, uint64_t seed, int64_t n_global,
                            uint32_t rank, uint32_t nranks)
{
    if (nranks == 0 || n_global % nranks) return ORA_ERR_INVALID;
    int64_t n = n_global / nranks;
    t->n = n;
    t->l_orderkey = malloc(n * 8);
    t->l_quantity = malloc(n * 8);
    t->l_extendedprice = malloc(n * 8);
    t->l_discount = malloc(n * 8);
    t->l_tax = malloc(n * 8);
    t->l_returnflag = malloc(n);
    t->l_linestatus = malloc(n);
    t->l_shipdate = malloc(n * 4);
    t->l_partkey = malloc(n * 8);
    if (!t->l_orderkey || !t->l_quantity || !t->l_extendedprice || !t->l_discount ||
        !t->l_tax || !t->l_returnflag || !t->l_linestatus || !t->l_shipdate ||
        !t->l_partkey)
        return ORA_ERR_OOM;
    for (int64_t l = 0; l < n; l++) {
        uint64_t i = otbx_li_global_row((uint64_t)l, rank, nranks);
        t->l_orderkey[l] = otbx_li_orderkey(i);
        t->l_quantity[l] = otbx_li_quantity(seed, i);
        t->l_extendedprice[l] = otbx_li_extendedprice(seed, i);
        t->l_discount[l] = otbx_li_discount(seed, i);
        t->l_tax[l] = otbx_li_tax(seed, i);
        t->l_returnflag[l] = otbx_li_returnflag(seed, i);
        t->l_linestatus[l] = otbx_li_linestatus(seed, i);
        t->l_shipdate[l] = otbx_li_shipdate(seed, i);
        /* nparts = SF×200k = n_global/30 (same formula as the GPU side) */
        t->l_partkey[l] = otbx_li_partkey(
            seed, i, n_global / 30 > 0 ? n_global / 30 : 1);
    }
    return ORA_OK;
}

ora_status ora_gen_orders(ora_orders *t, uint64_t seed, int64_t n_global,
                          int64_t ncust_global, uint32_t rank, uint32_t nranks,
                          int skew)
{
    if (nranks == 0 || n_global % nranks) return ORA_ERR_INVALID;
    int64_t n = n_global / nranks;
    t->n = n;
    t->o_orderkey = malloc(n * 8);
    t->o_custkey = malloc(n * 8);
    t->o_orderdate = malloc(n * 4);
    t->o_shippriority = malloc(n * 4);
    if (!t->o_orderkey || !t->o_custkey || !t->o_orderdate || !t->o_shippriority)
        return ORA_ERR_OOM;
    for (int64_t l = 0; l < n; l++) {
        uint64_t i = otbx_ord_global_row((uint64_t)l, rank, nranks);
        t->o_orderkey[l] = otbx_ord_orderkey(i);
        t->o_custkey[l] = skew ? otbx_ord_custkey_skewed(seed, i, ncust_global)
                               : otbx_ord_custkey(seed, i, ncust_global);
        t->o_orderdate[l] = otbx_ord_orderdate(seed, i);
        t->o_shippriority[l] = otbx_ord_shippriority(i);
    }
    return ORA_OK;
}

ora_status ora_gen_customer(ora_customer *t, uint64_t seed, int64_t n_global,
                            uint32_t rank, uint32_t nranks)
{
    if (nranks == 0 || n_global % nranks) return ORA_ERR_INVALID;
    int64_t n = n_global / nranks;
    t->n = n;
    t->c_custkey = malloc(n * 8);
    t->c_mktsegment = malloc(n);
    if (!t->c_custkey || !t->c_mktsegment) return ORA_ERR_OOM;
    for (int64_t l = 0; l < n; l++) {
        uint64_t i = otbx_cust_global_row((uint64_t)l, rank, nranks);
        t->c_custkey[l] = otbx_cust_custkey(i);
        t->c_mktsegment[l] = otbx_cust_mktsegment(seed, i);
    }
    return ORA_OK;
}

void ora_free_lineitem(ora_lineitem *t)
{
    free(t->l_orderkey); free(t->l_quantity); free(t->l_extendedprice);
    free(t->l_discount); free(t->l_tax); free(t->l_returnflag);
    free(t->l_linestatus); free(t->l_shipdate); free(t->l_partkey);
    memset(t, 0, sizeof(*t));
}

ora_status ora_gen_part(ora_part *t, uint64_t seed, int64_t n_global)
{
    t->n = n_global;
    t->p_partkey = malloc((n_global > 0 ? n_global : 1) * 8);
    t->p_type = malloc(n_global > 0 ? n_global : 1);
    if (!t->p_partkey || !t->p_type) return ORA_ERR_OOM;
    for (int64_t i = 0; i < n_global; i++) {
        t->p_partkey[i] = otbx_part_partkey((uint64_t)i);
        t->p_type[i] = otbx_part_type(seed, (uint64_t)i);
    }
    return ORA_OK;
}

void ora_free_part(ora_part *t)
{
    free(t->p_partkey); free(t->p_type);
    memset(t, 0, sizeof(*t));
}
void ora_free_orders(ora_orders *t)
{
    free(t->o_orderkey); free(t->o_custkey); free(t->o_orderdate);
    free(t->o_shippriority);
    memset(t, 0, sizeof(*t));
}
void ora_free_customer(ora_customer *t)
{
    free(t->c_custkey); free(t->c_mktsegment);
    memset(t, 0, sizeof(*t));
}

/* ================= transition functions ================= */

/* float8pl, float.c:970: result = a+b; error when finite inputs overflow
 * to inf (CHECKFLOATVAL(result, isinf(arg1)||isinf(arg2), true)) */
static inline ora_status float8pl(double a, double b, double *out)
{
    double r = a + b;
    if (isinf(r) && !(isinf(a) || isinf(b))) return ORA_ERR_OVERFLOW;
    *out = r;
    return ORA_OK;
}

/* int8inc, int8.c:714: +1 with signed-overflow error */
static inline ora_status int8inc(int64_t *v)
{
    if (*v == INT64_MAX) return ORA_ERR_OVERFLOW;
    (*v)++;
    return ORA_OK;
}

/* float8_accum, float.c:2823 (Youngs-Cramer) on state [N,Sx,Sxx] */
static ora_status float8_accum(double st[3], double newval)
{
    double N = st[0], Sx = st[1], Sxx = st[2], tmp;
    N += 1.0;
    Sx += newval;
    if (st[0] > 0.0) {
        tmp = newval * N - Sx;
        Sxx += tmp * tmp / (N * st[0]);
        if (isinf(Sx) || isinf(Sxx)) {
            if (!isinf(st[1]) && !isinf(newval)) return ORA_ERR_OVERFLOW;
            Sxx = NAN;
        }
    } else {
        if (isinf(newval) || isnan(newval)) Sxx = NAN;
    }
    st[0] = N; st[1] = Sx; st[2] = Sxx;
    return ORA_OK;
}

/* float8_combine, float.c:2725 */
static ora_status float8_combine(double a[3], const double b[3])
{
    double N1 = a[0], Sx1 = a[1], Sxx1 = a[2];
    double N2 = b[0], Sx2 = b[1], Sxx2 = b[2];
    if (N1 == 0.0) { a[0] = N2; a[1] = Sx2; a[2] = Sxx2; return ORA_OK; }
    if (N2 == 0.0) return ORA_OK;
    double N = N1 + N2;
    double Sx = Sx1 + Sx2;
    if (isinf(Sx) && !isinf(Sx1) && !isinf(Sx2)) return ORA_ERR_OVERFLOW;
    double tmp = Sx1 / N1 - Sx2 / N2;
    double Sxx = Sxx1 + Sxx2 + N1 * N2 * tmp * tmp / N;
    if (isinf(Sxx) && !isinf(Sxx1) && !isinf(Sxx2)) return ORA_ERR_OVERFLOW;
    a[0] = N; a[1] = Sx; a[2] = Sxx;
    return ORA_OK;
}

/* ================= config 2: scan + qual + count ================= */

int64_t ora_scan_count_shipdate_le(const int32_t *shipdate, int64_t n, int32_t cutoff)
{
    /* ExecScan loop (execScan.c:140): fetch, ExecQual, advance aggregate.
     * count(*) = int8inc per qualifying tuple. */
    int64_t count = 0;
    for (int64_t i = 0; i < n; i++)
        if (shipdate[i] <= cutoff)
            count++;
    return count;
}

/* ================= Q1 partial aggregate ================= */

ora_status ora_q1_partial(const ora_lineitem *t, int32_t cutoff_day,
                          ora_q1_group out[8], int *ngroups)
{
    /* ≤6 possible (returnflag,linestatus) combos; the reference uses the
     * simplehash group table (execGrouping.c:295) — with a key domain this
     * small the oracle keeps a direct-mapped array, same lookup-or-create
     * + advance_aggregates (nodeAgg.c:856) structure, same emit order as the
     * plan's ORDER BY (rf,ls). */
    ora_q1_group g[6];
    int used[6] = {0};
    memset(g, 0, sizeof(g));
    ora_status st;

    for (int64_t i = 0; i < t->n; i++) {
        if (!(t->l_shipdate[i] <= cutoff_day)) /* ExecQual */
            continue;
        uint8_t rf = t->l_returnflag[i], ls = t->l_linestatus[i];
        int ri = rf == 'A' ? 0 : rf == 'N' ? 1 : 2;
        int li = ls == 'F' ? 0 : 1;
        int k = ri * 2 + li;
        ora_q1_group *e = &g[k];
        if (!used[k]) {
            used[k] = 1;
            e->returnflag = rf;
            e->linestatus = ls;
            /* sum states start NULL (strict transfn, no initcond,
             * nodeAgg.c:614); represent as has-rows-implied: count==0 */
        }
        double qty = t->l_quantity[i];
        double price = t->l_extendedprice[i];
        double disc = t->l_discount[i];
        double disc_price = price * (1.0 - disc);           /* ExecProject */
        double charge = disc_price * (1.0 + t->l_tax[i]);
        if (e->count_order == 0) {
            /* first row: strict-transfn state initialization */
            e->sum_qty = qty;
            e->sum_base_price = price;
            e->sum_disc_price = disc_price;
            e->sum_charge = charge;
        } else {
            if ((st = float8pl(e->sum_qty, qty, &e->sum_qty))) return st;
            if ((st = float8pl(e->sum_base_price, price, &e->sum_base_price))) return st;
            if ((st = float8pl(e->sum_disc_price, disc_price, &e->sum_disc_price))) return st;
            if ((st = float8pl(e->sum_charge, charge, &e->sum_charge))) return st;
        }
        if ((st = float8_accum(e->qty_acc, qty))) return st;
        if ((st = float8_accum(e->price_acc, price))) return st;
        if ((st = float8_accum(e->disc_acc, disc))) return st;
        if ((st = int8inc(&e->count_order))) return st;
    }

    /* emit in (rf,ls) order: A < N < R, F < O — matches Q1's ORDER BY and
     * ASCII order of the key bytes */
    int ng = 0;
    for (int k = 0; k < 6; k++)
        if (used[k])
            out[ng++] = g[k];
    *ngroups = ng;
    return ORA_OK;
}

ora_status ora_q1_combine(ora_q1_group acc[8], int *nacc,
                          const ora_q1_group *part, int npart)
{
    ora_status st;
    for (int p = 0; p < npart; p++) {
        const ora_q1_group *src = &part[p];
        ora_q1_group *dst = NULL;
        for (int i = 0; i < *nacc; i++)
            if (acc[i].returnflag == src->returnflag &&
                acc[i].linestatus == src->linestatus) { dst = &acc[i]; break; }
        if (!dst) {
            dst = &acc[(*nacc)++];
            *dst = *src;
            continue;
        }
        /* combinefn wiring per AGGSPLIT_FINAL_DESERIAL (nodeAgg.c:3912,4260):
         * sum(float8) combines with float8pl, count with int8pl,
         * avg states with float8_combine (float.c:2725). */
        if (src->count_order > 0) {
            if (dst->count_order == 0) {
                dst->sum_qty = src->sum_qty;
                dst->sum_base_price = src->sum_base_price;
                dst->sum_disc_price = src->sum_disc_price;
                dst->sum_charge = src->sum_charge;
            } else {
                if ((st = float8pl(dst->sum_qty, src->sum_qty, &dst->sum_qty))) return st;
                if ((st = float8pl(dst->sum_base_price, src->sum_base_price, &dst->sum_base_price))) return st;
                if ((st = float8pl(dst->sum_disc_price, src->sum_disc_price, &dst->sum_disc_price))) return st;
                if ((st = float8pl(dst->sum_charge, src->sum_charge, &dst->sum_charge))) return st;
            }
        }
        if ((st = float8_combine(dst->qty_acc, src->qty_acc))) return st;
        if ((st = float8_combine(dst->price_acc, src->price_acc))) return st;
        if ((st = float8_combine(dst->disc_acc, src->disc_acc))) return st;
        if (dst->count_order > INT64_MAX - src->count_order) return ORA_ERR_OVERFLOW;
        dst->count_order += src->count_order;
    }
    /* keep (rf,ls) sort order */
    for (int i = 1; i < *nacc; i++)
        for (int j = i; j > 0; j--) {
            ora_q1_group *a = &acc[j - 1], *b = &acc[j];
            if (a->returnflag > b->returnflag ||
                (a->returnflag == b->returnflag && a->linestatus > b->linestatus)) {
                ora_q1_group tmpv = *a; *a = *b; *b = tmpv;
            } else break;
        }
    return ORA_OK;
}

void ora_q1_finalize(ora_q1_group *g, int n)
{
    /* float8_avg: Sx/N (NULL when N==0 — cannot occur for emitted groups) */
    for (int i = 0; i < n; i++) {
        g[i].avg_qty = g[i].qty_acc[0] > 0 ? g[i].qty_acc[1] / g[i].qty_acc[0] : 0;
        g[i].avg_price = g[i].price_acc[0] > 0 ? g[i].price_acc[1] / g[i].price_acc[0] : 0;
        g[i].avg_disc = g[i].disc_acc[0] > 0 ? g[i].disc_acc[1] / g[i].disc_acc[0] : 0;
    }
}

/* ================= chained-bucket hash join machinery ================= */

/* nodeHash.c semantics: nbuckets = power of 2 with NTUP_PER_BUCKET=1
 * (ExecChooseHashTableSize nodeHash.c:677 region), bucketno = hash &
 * (nbuckets-1) (ExecHashGetBucketAndBatch nodeHash.c:2141), insert at bucket
 * head (ExecHashTableInsert nodeHash.c:1828), probe walks the chain
 * comparing hash then key (ExecScanHashBucket nodeHash.c:2174). */
typedef struct {
    int64_t nbuckets;      /* power of 2 */
    int64_t *bucket_head;  /* index into entries, -1 empty */
    int64_t *next;         /* chain links */
    const int64_t *keys;   /* build keys (borrowed) */
} ora_hashtab;

static inline uint64_t ora_hash_i64(int64_t k)
{
    return otbx_splitmix64((uint64_t)k);
}

static int64_t next_pow2(int64_t v)
{
    int64_t p = 1;
    while (p < v) p <<= 1;
    return p;
}

static ora_status ht_build(ora_hashtab *ht, const int64_t *keys,
                           const uint8_t *knull, int64_t n)
{
    ht->nbuckets = next_pow2(n < 16 ? 16 : n);
    ht->bucket_head = malloc(ht->nbuckets * 8);
    ht->next = malloc((n > 0 ? n : 1) * 8);
    ht->keys = keys;
    if (!ht->bucket_head || !ht->next) return ORA_ERR_OOM;
    for (int64_t i = 0; i < ht->nbuckets; i++) ht->bucket_head[i] = -1;
    for (int64_t i = 0; i < n; i++) {
        if (knull && knull[i])
            continue; /* NULL join key never matches; hashable path drops it
                       * (ExecHashGetHashValue nodeHash.c:2026 keep_nulls=false) */
        int64_t b = (int64_t)(ora_hash_i64(keys[i]) & (uint64_t)(ht->nbuckets - 1));
        ht->next[i] = ht->bucket_head[b];
        ht->bucket_head[b] = i;
    }
    return ORA_OK;
}

static void ht_free(ora_hashtab *ht)
{
    free(ht->bucket_head);
    free(ht->next);
}

/* probe: returns first matching build index for key, then use ht_next_match */
static inline int64_t ht_first_match(const ora_hashtab *ht, int64_t key)
{
    int64_t b = (int64_t)(ora_hash_i64(key) & (uint64_t)(ht->nbuckets - 1));
    int64_t i = ht->bucket_head[b];
    while (i >= 0 && ht->keys[i] != key)
        i = ht->next[i];
    return i;
}
static inline int64_t ht_next_match(const ora_hashtab *ht, int64_t i, int64_t key)
{
    i = ht->next[i];
    while (i >= 0 && ht->keys[i] != key)
        i = ht->next[i];
    return i;
}

ora_status ora_join_i64(const int64_t *bkeys, const uint8_t *bnull, int64_t nb,
                        const int64_t *pkeys, const uint8_t *pnull, int64_t np,
                        int64_t **out_bidx, int64_t **out_pidx, int64_t *nout)
{
    ora_hashtab ht;
    ora_status st = ht_build(&ht, bkeys, bnull, nb);
    if (st) { ht_free(&ht); return st; }
    int64_t cap = 64, n = 0;
    int64_t *bi = malloc(cap * 8), *pi = malloc(cap * 8);
    if (!bi || !pi) { free(bi); free(pi); ht_free(&ht); return ORA_ERR_OOM; }
    for (int64_t p = 0; p < np; p++) {
        if (pnull && pnull[p])
            continue; /* HJ_FILL_OUTER etc. not modeled: inner join only */
        for (int64_t m = ht_first_match(&ht, pkeys[p]); m >= 0;
             m = ht_next_match(&ht, m, pkeys[p])) {
            if (n == cap) {
                cap *= 2;
                int64_t *nb2 = realloc(bi, cap * 8), *np2 = realloc(pi, cap * 8);
                if (!nb2 || !np2) { free(nb2 ? nb2 : bi); free(np2 ? np2 : pi); ht_free(&ht); return ORA_ERR_OOM; }
                bi = nb2; pi = np2;
            }
            bi[n] = m; pi[n] = p; n++;
        }
    }
    ht_free(&ht);
    *out_bidx = bi; *out_pidx = pi; *nout = n;
    return ORA_OK;
}

/* ================= simplehash-style open-addressing agg table ============ */

/* simplehash.h: linear probing, fillfactor 0.9 (SH_FILLFACTOR
 * simplehash.h:187), size = power of 2. Used for the generic agg and Q3. */

typedef struct {
    int64_t key;
    int used;
    int key_isnull;
    void *payload;
} sh_slot;

/* ---- generic agg (ora_agg_i64) ---- */

/* emit order (key_isnull, key) — deterministic comparison only, not a
 * semantic of the reference (hash iteration order there is arbitrary) */
static int agg_group_cmp(const void *pa, const void *pb)
{
    const ora_agg_group *a = pa, *b = pb;
    if (a->key_isnull != b->key_isnull) return a->key_isnull ? 1 : -1;
    if (a->key_isnull) return 0;
    return a->key < b->key ? -1 : a->key > b->key ? 1 : 0;
}

ora_status ora_agg_i64(const int64_t *keys, const uint8_t *key_null,
                       const double *vals, const uint8_t *val_null,
                       int64_t n, ora_agg_group **out, int64_t *ngroups)
{
    int64_t cap = next_pow2(n < 16 ? 16 : (int64_t)((double)n / 0.85) + 1);
    ora_agg_group *slots = calloc(cap, sizeof(ora_agg_group));
    uint8_t *used = calloc(cap, 1);
    if (!slots || !used) { free(slots); free(used); return ORA_ERR_OOM; }
    int64_t ng = 0;
    ora_status st;

    for (int64_t i = 0; i < n; i++) {
        int isnull = key_null && key_null[i];
        int64_t k = isnull ? 0 : keys[i];
        /* LookupTupleHashEntry (execGrouping.c:295): hash, linear probe,
         * NULL==NULL for grouping purposes */
        uint64_t h = isnull ? 0x9e3779b97f4a7c15ull : ora_hash_i64(k);
        int64_t s = (int64_t)(h & (uint64_t)(cap - 1));
        for (;;) {
            if (!used[s]) {
                used[s] = 1;
                slots[s].key = k;
                slots[s].key_isnull = isnull;
                slots[s].sum_isnull = 1; /* strict sum starts NULL */
                ng++;
                break;
            }
            if (slots[s].key_isnull == isnull && (isnull || slots[s].key == k))
                break;
            s = (s + 1) & (cap - 1); /* linear probe, simplehash SH_NEXT */
        }
        ora_agg_group *e = &slots[s];
        if ((st = int8inc(&e->count_star))) goto fail;         /* count(*) */
        int vnull = val_null && val_null[i];
        if (!vnull) {                                          /* strict aggs */
            if ((st = int8inc(&e->count_v))) goto fail;        /* count(v) */
            if (e->sum_isnull) {
                e->sum_v = vals[i];                            /* init state */
                e->sum_isnull = 0;
            } else if ((st = float8pl(e->sum_v, vals[i], &e->sum_v))) goto fail;
            if ((st = float8_accum(e->acc, vals[i]))) goto fail; /* avg(v) */
        }
    }

    /* emit sorted by (key_isnull, key) for deterministic comparison */
    ora_agg_group *res = malloc((ng > 0 ? ng : 1) * sizeof(ora_agg_group));
    if (!res) { st = ORA_ERR_OOM; goto fail; }
    int64_t j = 0;
    for (int64_t s = 0; s < cap; s++)
        if (used[s]) res[j++] = slots[s];
    qsort(res, (size_t)ng, sizeof(ora_agg_group), agg_group_cmp);
    free(slots); free(used);
    *out = res; *ngroups = ng;
    return ORA_OK;
fail:
    free(slots); free(used);
    return st;
}

/* ================= Q9-mix partial aggregate ================= */

ora_status ora_q9_partial(const ora_part *p, const ora_orders *o,
                          const ora_lineitem *l, uint8_t typemod,
                          uint8_t typeval, ora_q9_group out[8], int *ngroups)
{
    /* same executor structure as Q3 (two chained-bucket hash joins feeding
     * a hash aggregate, nodeHash.c/nodeHashjoin.c/nodeAgg.c) but the group
     * key is COMPUTED (year(o_orderdate)) — the Q9 plan shape. The tiny
     * year domain makes the group table a direct-mapped array (the
     * simplehash instance degenerates). */
    ora_status st = ORA_OK;

    /* build 1: filtered part key set */
    int64_t *pkeys = malloc((p->n > 0 ? p->n : 1) * 8);
    if (!pkeys) return ORA_ERR_OOM;
    int64_t npf = 0;
    for (int64_t i = 0; i < p->n; i++)
        if (p->p_type[i] % typemod == typeval)
            pkeys[npf++] = p->p_partkey[i];
    ora_hashtab pht;
    if ((st = ht_build(&pht, pkeys, NULL, npf))) goto done1;

    /* build 2: orders keyed on o_orderkey (no filter in Q9) */
    {
        ora_hashtab oht;
        if ((st = ht_build(&oht, o->o_orderkey, NULL, o->n))) {
            ht_free(&oht);
            goto done2;
        }
        ora_q9_group g[8];
        memset(g, 0, sizeof(g));
        for (int64_t i = 0; i < l->n; i++) {
            if (ht_first_match(&pht, l->l_partkey[i]) < 0)
                continue; /* ⋈ part (inner) */
            int64_t m = ht_first_match(&oht, l->l_orderkey[i]);
            if (m < 0)
                continue; /* ⋈ orders */
            int32_t y = otbx_year_of_day(o->o_orderdate[m]);
            double rev = l->l_extendedprice[i] * (1.0 - l->l_discount[i]);
            if (g[y].count_rows == 0)
                g[y].revenue = rev;        /* strict sum init */
            else if ((st = float8pl(g[y].revenue, rev, &g[y].revenue))) {
                ht_free(&oht);
                goto done2;
            }
            if ((st = int8inc(&g[y].count_rows))) {
                ht_free(&oht);
                goto done2;
            }
        }
        ht_free(&oht);
        int ng = 0;
        for (int y = 0; y < 8; y++)
            if (g[y].count_rows > 0) {
                out[ng] = g[y];
                out[ng].year = y;
                ng++;
            }
        *ngroups = ng;
    }
done2:
    ht_free(&pht);
done1:
    free(pkeys);
    return st;
}

/* ================= Q3 ================= */

typedef struct {
    int64_t orderkey;
    int32_t orderdate;
    int32_t shippriority;
    double revenue;
    int used;
} q3_slot;

ora_status ora_q3_partial(const ora_customer *c, const ora_orders *o,
                          const ora_lineitem *l, uint8_t segment,
                          int32_t q3date, ora_q3_row **out, int64_t *ngroups)
{
    ora_status st = ORA_OK;

    /* --- build 1: filtered customer (c_mktsegment = seg) key set --- */
    int64_t *ckeys = malloc((c->n > 0 ? c->n : 1) * 8);
    if (!ckeys) return ORA_ERR_OOM;
    int64_t ncf = 0;
    for (int64_t i = 0; i < c->n; i++)
        if (c->c_mktsegment[i] == segment)
            ckeys[ncf++] = c->c_custkey[i];
    ora_hashtab cht;
    if ((st = ht_build(&cht, ckeys, NULL, ncf))) goto done1;

    /* --- probe with orders (o_orderdate < date), collect matched orders,
     *     build 2 keyed on o_orderkey --- */
    int64_t *okeys = malloc((o->n > 0 ? o->n : 1) * 8);
    int32_t *odates = malloc((o->n > 0 ? o->n : 1) * 4);
    int32_t *oprios = malloc((o->n > 0 ? o->n : 1) * 4);
    if (!okeys || !odates || !oprios) { st = ORA_ERR_OOM; goto done2; }
    int64_t nof = 0;
    for (int64_t i = 0; i < o->n; i++) {
        if (!(o->o_orderdate[i] < q3date))
            continue;
        if (ht_first_match(&cht, o->o_custkey[i]) < 0)
            continue; /* inner join, custkey unique in customer */
        okeys[nof] = o->o_orderkey[i];
        odates[nof] = o->o_orderdate[i];
        oprios[nof] = o->o_shippriority[i];
        nof++;
    }
    ora_hashtab oht;
    if ((st = ht_build(&oht, okeys, NULL, nof))) { ht_free(&oht); goto done2; }

    /* --- probe with lineitem (l_shipdate > date), agg revenue by group
     *     (l_orderkey, o_orderdate, o_shippriority) ≡ l_orderkey --- */
    int64_t cap = next_pow2(nof < 16 ? 16 : (int64_t)((double)nof / 0.85) + 1);
    q3_slot *slots = calloc(cap, sizeof(q3_slot));
    if (!slots) { st = ORA_ERR_OOM; ht_free(&oht); goto done2; }
    int64_t ng = 0;
    for (int64_t i = 0; i < l->n; i++) {
        if (!(l->l_shipdate[i] > q3date))
            continue;
        int64_t k = l->l_orderkey[i];
        int64_t m = ht_first_match(&oht, k);
        if (m < 0)
            continue;
        double rev = l->l_extendedprice[i] * (1.0 - l->l_discount[i]);
        /* group lookup-or-insert (simplehash linear probe) */
        int64_t s = (int64_t)(ora_hash_i64(k) & (uint64_t)(cap - 1));
        for (;;) {
            if (!slots[s].used) {
                slots[s].used = 1;
                slots[s].orderkey = k;
                slots[s].orderdate = odates[m];
                slots[s].shippriority = oprios[m];
                slots[s].revenue = rev; /* strict sum init */
                ng++;
                break;
            }
            if (slots[s].orderkey == k) {
                if ((st = float8pl(slots[s].revenue, rev, &slots[s].revenue))) {
                    free(slots); ht_free(&oht); goto done2;
                }
                break;
            }
            s = (s + 1) & (cap - 1);
        }
    }
    ht_free(&oht);

    ora_q3_row *res = malloc((ng > 0 ? ng : 1) * sizeof(ora_q3_row));
    if (!res) { st = ORA_ERR_OOM; free(slots); goto done2; }
    int64_t j = 0;
    for (int64_t s = 0; s < cap; s++)
        if (slots[s].used) {
            res[j].l_orderkey = slots[s].orderkey;
            res[j].revenue = slots[s].revenue;
            res[j].o_orderdate = slots[s].orderdate;
            res[j].o_shippriority = slots[s].shippriority;
            j++;
        }
    free(slots);
    *out = res;
    *ngroups = ng;
done2:
    free(okeys); free(odates); free(oprios);
    ht_free(&cht);
done1:
    free(ckeys);
    return st;
}

/* ORDER BY revenue DESC, o_orderdate ASC LIMIT k (tuplesort merge analog) */
static int q3_cmp(const void *pa, const void *pb)
{
    const ora_q3_row *a = pa, *b = pb;
    if (a->revenue > b->revenue) return -1;
    if (a->revenue < b->revenue) return 1;
    if (a->o_orderdate < b->o_orderdate) return -1;
    if (a->o_orderdate > b->o_orderdate) return 1;
    /* tiebreak on orderkey for determinism (not part of SQL contract) */
    if (a->l_orderkey < b->l_orderkey) return -1;
    if (a->l_orderkey > b->l_orderkey) return 1;
    return 0;
}

int64_t ora_q3_topk(ora_q3_row *rows, int64_t n, int64_t k)
{
    qsort(rows, (size_t)n, sizeof(ora_q3_row), q3_cmp);
    return n < k ? n : k;
}

/* ================= extended joins (ora_join_ext) ================= */
/* Restates the HJ_* FSM (nodeHashjoin.c:139-144) incl. the fill states —
 * see oracle.h for the per-state citations. Chained-bucket table as
 * ora_join_i64, generalized to an optional second key (multi-key combine
 * semantics nodeHash.c:2059 restated at 64 bit). */

static inline int ext_rownull(const uint8_t *n1, const uint8_t *n2,
                              int has_k2, int64_t i)
{
    return (n1 && n1[i]) || (has_k2 && n2 && n2[i]);
}

static inline uint64_t ext_hash(int64_t k1, int64_t k2, int has_k2)
{
    uint64_t h = ora_hash_i64(k1);
    if (has_k2) {
        h = (h << 1) | (h >> 63);   /* rotate-left-1 (nodeHash.c:2059) */
        h ^= ora_hash_i64(k2);
    }
    return h;
}

static ora_status ext_push(int64_t **bi, int64_t **pi, int64_t *n,
                           int64_t *cap, int64_t b, int64_t p)
{
    if (*n == *cap) {
        *cap *= 2;
        int64_t *nb = realloc(*bi, (size_t)*cap * 8);
        int64_t *np = realloc(*pi, (size_t)*cap * 8);
        if (nb) *bi = nb;
        if (np) *pi = np;
        if (!nb || !np) return ORA_ERR_OOM;
    }
    (*bi)[*n] = b;
    (*pi)[*n] = p;
    (*n)++;
    return ORA_OK;
}

ora_status ora_join_ext(const int64_t *bk1, const uint8_t *bn1,
                        const int64_t *bk2, const uint8_t *bn2, int64_t nb,
                        const int64_t *pk1, const uint8_t *pn1,
                        const int64_t *pk2, const uint8_t *pn2, int64_t np,
                        int join_type,
                        int64_t **out_bidx, int64_t **out_pidx, int64_t *nout)
{
    const int has_k2 = bk2 != NULL;
    const int jt = join_type;
    if (jt < 0 || jt > 5 || (has_k2 != (pk2 != NULL)))
        return ORA_ERR_INVALID;
    const int fill_probe = (jt == 1 || jt == 5);      /* left, full  */
    const int fill_build = (jt == 4 || jt == 5);      /* right, full */
    const int emit_match = (jt == 0 || jt == 1 || jt == 4 || jt == 5);

    /* chained-bucket build over non-null rows (ExecHashTableInsert
     * nodeHash.c:1828; NULL keys dropped, nodeHash.c:2026) */
    int64_t nbuckets = next_pow2(nb < 16 ? 16 : nb);
    int64_t *head = malloc((size_t)nbuckets * 8);
    int64_t *next = malloc((size_t)(nb > 0 ? nb : 1) * 8);
    uint8_t *matched = NULL;
    if (fill_build) matched = calloc((size_t)(nb > 0 ? nb : 1), 1);
    if (!head || !next || (fill_build && !matched)) {
        free(head); free(next); free(matched);
        return ORA_ERR_OOM;
    }
    for (int64_t i = 0; i < nbuckets; i++) head[i] = -1;
    for (int64_t i = 0; i < nb; i++) {
        if (ext_rownull(bn1, bn2, has_k2, i)) continue;
        int64_t b = (int64_t)(ext_hash(bk1[i], has_k2 ? bk2[i] : 0, has_k2) &
                              (uint64_t)(nbuckets - 1));
        next[i] = head[b];
        head[b] = i;
    }

    int64_t cap = 64, n = 0;
    int64_t *bi = malloc(cap * 8), *pi = malloc(cap * 8);
    ora_status st = ORA_OK;
    if (!bi || !pi) { st = ORA_ERR_OOM; goto done; }

    for (int64_t p = 0; p < np; p++) {
        if (ext_rownull(pn1, pn2, has_k2, p)) {
            /* NULL-key outer tuple: cannot match; left/full fill
             * (nodeHashjoin.c:543-552), anti emits (no match exists) */
            if (fill_probe || jt == 3)
                if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
            continue;
        }
        int64_t k1 = pk1[p], k2 = has_k2 ? pk2[p] : 0;
        int64_t bkt = (int64_t)(ext_hash(k1, k2, has_k2) &
                                (uint64_t)(nbuckets - 1));
        int64_t nmatch = 0;
        for (int64_t m = head[bkt]; m >= 0; m = next[m]) {
            if (bk1[m] != k1 || (has_k2 && bk2[m] != k2)) continue;
            nmatch++;
            if (matched) matched[m] = 1;
            if (emit_match) {
                if ((st = ext_push(&bi, &pi, &n, &cap, m, p))) goto done;
            }
            if (jt == 2) break;     /* JOIN_SEMI: first match suffices
                                     * (nodeHashjoin.c:572) */
        }
        if (jt == 2 && nmatch > 0) {
            if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
        } else if ((jt == 3 || fill_probe) && nmatch == 0) {
            if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
        }
    }

    /* HJ_FILL_INNER_TUPLES (nodeHashjoin.c:693) / unmatched-build scan
     * (ExecScanHashTableForUnmatched nodeHash.c:2322): NULL-key build rows
     * never entered the table, so they are unmatched by construction */
    if (fill_build)
        for (int64_t i = 0; i < nb; i++)
            if (ext_rownull(bn1, bn2, has_k2, i) || !matched[i])
                if ((st = ext_push(&bi, &pi, &n, &cap, i, -1))) goto done;

done:
    free(head); free(next); free(matched);
    if (st) { free(bi); free(pi); return st; }
    *out_bidx = bi; *out_pidx = pi; *nout = n;
    return ORA_OK;
}

/* ================= two-key hash aggregate (ora_agg_i64x2) ================= */

static int agg_group2_cmp(const void *pa, const void *pb)
{
    const ora_agg_group2 *a = pa, *b = pb;
    if (a->key1_isnull != b->key1_isnull) return a->key1_isnull ? 1 : -1;
    if (!a->key1_isnull) {
        if (a->key1 != b->key1) return a->key1 < b->key1 ? -1 : 1;
    }
    if (a->key2_isnull != b->key2_isnull) return a->key2_isnull ? 1 : -1;
    if (!a->key2_isnull) {
        if (a->key2 != b->key2) return a->key2 < b->key2 ? -1 : 1;
    }
    return 0;
}

ora_status ora_agg_i64x2(const int64_t *k1, const uint8_t *k1null,
                         const int64_t *k2, const uint8_t *k2null,
                         const double *vals, const uint8_t *val_null,
                         int64_t n, ora_agg_group2 **out, int64_t *ngroups)
{
    int64_t cap = next_pow2(n < 16 ? 16 : (int64_t)((double)n / 0.85) + 1);
    ora_agg_group2 *slots = calloc(cap, sizeof(ora_agg_group2));
    uint8_t *used = calloc(cap, 1);
    if (!slots || !used) { free(slots); free(used); return ORA_ERR_OOM; }
    int64_t ng = 0;
    ora_status st;

    for (int64_t i = 0; i < n; i++) {
        int n1 = k1null && k1null[i], n2 = k2null && k2null[i];
        int64_t a = n1 ? 0 : k1[i], b = n2 ? 0 : k2[i];
        /* LookupTupleHashEntry (execGrouping.c:295): NULL==NULL for
         * grouping (:525); per-key hash with NULL sentinel, combined by
         * rotate-left-1 + xor (nodeHash.c:2059 restated) */
        uint64_t h1 = n1 ? 0x9e3779b97f4a7c15ull : ora_hash_i64(a);
        uint64_t h2 = n2 ? 0xc2b2ae3d27d4eb4full : ora_hash_i64(b);
        uint64_t h = ((h1 << 1) | (h1 >> 63)) ^ h2;
        int64_t s = (int64_t)(h & (uint64_t)(cap - 1));
        for (;;) {
            if (!used[s]) {
                used[s] = 1;
                slots[s].key1 = a; slots[s].key1_isnull = n1;
                slots[s].key2 = b; slots[s].key2_isnull = n2;
                slots[s].sum_isnull = 1;
                ng++;
                break;
            }
            if (slots[s].key1_isnull == n1 && slots[s].key2_isnull == n2 &&
                (n1 || slots[s].key1 == a) && (n2 || slots[s].key2 == b))
                break;
            s = (s + 1) & (cap - 1);
        }
        ora_agg_group2 *e = &slots[s];
        if ((st = int8inc(&e->count_star))) goto fail;
        int vnull = val_null && val_null[i];
        if (!vnull) {
            if ((st = int8inc(&e->count_v))) goto fail;
            if (e->sum_isnull) {
                e->sum_v = vals[i];
                e->sum_isnull = 0;
            } else if ((st = float8pl(e->sum_v, vals[i], &e->sum_v))) goto fail;
            if ((st = float8_accum(e->acc, vals[i]))) goto fail;
        }
    }

    ora_agg_group2 *res = malloc((size_t)(ng > 0 ? ng : 1) *
                                 sizeof(ora_agg_group2));
    if (!res) { st = ORA_ERR_OOM; goto fail; }
    int64_t j = 0;
    for (int64_t s = 0; s < cap; s++)
        if (used[s]) res[j++] = slots[s];
    qsort(res, (size_t)ng, sizeof(ora_agg_group2), agg_group2_cmp);
    free(slots); free(used);
    *out = res; *ngroups = ng;
    return ORA_OK;
fail:
    free(slots); free(used);
    return st;
}


/* ============== exact decimal aggregate (ora_agg_i64_dec) ==============
 * Int128AggState semantics (numeric.c:5072 / do_int128_accum :4998):
 * strict transition skips NULL inputs; the int128 sum of int64 values
 * cannot overflow (|sum| <= 2^63 * 2^63 = 2^126). */

static int dec_group_cmp(const void *pa, const void *pb)
{
    const ora_dec_group *a = pa, *b = pb;
    if (a->key_isnull != b->key_isnull) return a->key_isnull ? 1 : -1;
    if (a->key_isnull) return 0;
    return a->key < b->key ? -1 : a->key > b->key ? 1 : 0;
}

ora_status ora_agg_i64_dec(const int64_t *keys, const uint8_t *key_null,
                           const int64_t *vals, const uint8_t *val_null,
                           int64_t n, ora_dec_group **out, int64_t *ngroups)
{
    typedef struct {
        int64_t key;
        int key_isnull;
        int64_t count_star, count_v;
        __int128 sum;
        int sum_isnull;
    } slot_t;
    int64_t cap = next_pow2(n < 16 ? 16 : (int64_t)((double)n / 0.85) + 1);
    slot_t *slots = calloc(cap, sizeof(slot_t));
    uint8_t *used = calloc(cap, 1);
    if (!slots || !used) { free(slots); free(used); return ORA_ERR_OOM; }
    int64_t ng = 0;
    ora_status st;

    for (int64_t i = 0; i < n; i++) {
        int isnull = key_null && key_null[i];
        int64_t k = isnull ? 0 : keys[i];
        uint64_t h = isnull ? 0x9e3779b97f4a7c15ull : ora_hash_i64(k);
        int64_t s = (int64_t)(h & (uint64_t)(cap - 1));
        for (;;) {
            if (!used[s]) {
                used[s] = 1;
                slots[s].key = k;
                slots[s].key_isnull = isnull;
                slots[s].sum_isnull = 1;
                ng++;
                break;
            }
            if (slots[s].key_isnull == isnull && (isnull || slots[s].key == k))
                break;
            s = (s + 1) & (cap - 1);
        }
        slot_t *e = &slots[s];
        if ((st = int8inc(&e->count_star))) goto fail;
        if (!(val_null && val_null[i])) {           /* strict transition */
            if ((st = int8inc(&e->count_v))) goto fail;
            e->sum += (__int128)vals[i];            /* do_int128_accum */
            e->sum_isnull = 0;
        }
    }

    ora_dec_group *res = malloc((size_t)(ng > 0 ? ng : 1) *
                                sizeof(ora_dec_group));
    if (!res) { st = ORA_ERR_OOM; goto fail; }
    int64_t j = 0;
    for (int64_t s = 0; s < cap; s++)
        if (used[s]) {
            res[j].key = slots[s].key;
            res[j].key_isnull = slots[s].key_isnull;
            res[j].count_star = slots[s].count_star;
            res[j].count_v = slots[s].count_v;
            res[j].sum_hi = (int64_t)(slots[s].sum >> 64);
            res[j].sum_lo = (uint64_t)slots[s].sum;
            res[j].sum_isnull = slots[s].sum_isnull;
            j++;
        }
    qsort(res, (size_t)ng, sizeof(ora_dec_group), dec_group_cmp);
    free(slots); free(used);
    *out = res; *ngroups = ng;
    return ORA_OK;
fail:
    free(slots); free(used);
    return st;
}

/* ============== N-key group-by / join (ora_agg_i64n, ora_join_i64n) ======
 * See oracle.h for semantics + citations. The defining-row-index output is
 * the representative-tuple pattern (execGrouping.c firstTuple). */

static inline int nk_rownull(const ora_keyset *ks, int64_t i)
{
    for (int c = 0; c < ks->nkeys; c++)
        if (ks->nulls[c] && ks->nulls[c][i])
            return 1;
    return 0;
}

static inline uint64_t nk_hash(const ora_keyset *ks, int64_t i)
{
    uint64_t h = 0;
    for (int c = 0; c < ks->nkeys; c++) {
        h = (h << 1) | (h >> 63);       /* rotate-left-1, nodeHash.c:2059 */
        int isnull = ks->nulls[c] && ks->nulls[c][i];
        h ^= isnull ? (0x9e3779b97f4a7c15ull + (uint64_t)c)
                    : ora_hash_i64(ks->keys[c][i]);
    }
    return h;
}

/* group identity: rows a and b agree on every column's (isnull, value) */
static inline int nk_row_eq(const ora_keyset *ks, int64_t a, int64_t b)
{
    for (int c = 0; c < ks->nkeys; c++) {
        int na = ks->nulls[c] && ks->nulls[c][a];
        int nb = ks->nulls[c] && ks->nulls[c][b];
        if (na != nb) return 0;
        if (!na && ks->keys[c][a] != ks->keys[c][b]) return 0;
    }
    return 1;
}

/* join match: no column NULL on either side, values equal */
static inline int nk_match(const ora_keyset *bks, int64_t b,
                           const ora_keyset *pks, int64_t p)
{
    for (int c = 0; c < bks->nkeys; c++)
        if (bks->keys[c][b] != pks->keys[c][p])
            return 0;
    return 1;
}

static int aggn_cmp(const void *pa, const void *pb)
{
    const ora_aggn_group *a = pa, *b = pb;
    return a->row_idx < b->row_idx ? -1 : a->row_idx > b->row_idx ? 1 : 0;
}

ora_status ora_agg_i64n(const ora_keyset *ks, const double *vals,
                        const uint8_t *val_null, int64_t n,
                        ora_aggn_group **out, int64_t *ngroups)
{
    if (!ks || ks->nkeys < 1 || ks->nkeys > ORA_MAX_KEYS)
        return ORA_ERR_INVALID;
    int64_t cap = next_pow2(n < 16 ? 16 : (int64_t)((double)n / 0.85) + 1);
    int64_t *slot_row = malloc((size_t)cap * 8);       /* -1 empty */
    ora_aggn_group *slots = calloc(cap, sizeof(ora_aggn_group));
    if (!slot_row || !slots) { free(slot_row); free(slots); return ORA_ERR_OOM; }
    for (int64_t s = 0; s < cap; s++) slot_row[s] = -1;
    int64_t ng = 0;
    ora_status st;

    for (int64_t i = 0; i < n; i++) {
        int64_t s = (int64_t)(nk_hash(ks, i) & (uint64_t)(cap - 1));
        for (;;) {
            if (slot_row[s] < 0) {
                slot_row[s] = i;
                slots[s].row_idx = i;
                slots[s].sum_isnull = 1;
                ng++;
                break;
            }
            if (nk_row_eq(ks, slot_row[s], i))
                break;
            s = (s + 1) & (cap - 1);
        }
        ora_aggn_group *e = &slots[s];
        if ((st = int8inc(&e->count_star))) goto fail;
        if (!(val_null && val_null[i])) {
            if ((st = int8inc(&e->count_v))) goto fail;
            if (e->sum_isnull) {
                e->sum_v = vals[i];
                e->sum_isnull = 0;
            } else if ((st = float8pl(e->sum_v, vals[i], &e->sum_v)))
                goto fail;
        }
    }

    ora_aggn_group *res = malloc((size_t)(ng > 0 ? ng : 1) *
                                 sizeof(ora_aggn_group));
    if (!res) { st = ORA_ERR_OOM; goto fail; }
    int64_t j = 0;
    for (int64_t s = 0; s < cap; s++)
        if (slot_row[s] >= 0) res[j++] = slots[s];
    qsort(res, (size_t)ng, sizeof(ora_aggn_group), aggn_cmp);
    free(slot_row); free(slots);
    *out = res; *ngroups = ng;
    return ORA_OK;
fail:
    free(slot_row); free(slots);
    return st;
}

ora_status ora_join_i64n(const ora_keyset *bks, int64_t nb,
                         const ora_keyset *pks, int64_t np, int join_type,
                         int64_t **out_bidx, int64_t **out_pidx,
                         int64_t *nout)
{
    if (!bks || !pks || bks->nkeys != pks->nkeys || bks->nkeys < 1 ||
        bks->nkeys > ORA_MAX_KEYS || join_type < 0 || join_type > 5)
        return ORA_ERR_INVALID;
    const int jt = join_type;
    const int fill_probe = (jt == 1 || jt == 5);
    const int fill_build = (jt == 4 || jt == 5);
    const int emit_match = (jt == 0 || jt == 1 || jt == 4 || jt == 5);

    int64_t nbuckets = next_pow2(nb < 16 ? 16 : nb);
    int64_t *head = malloc((size_t)nbuckets * 8);
    int64_t *next = malloc((size_t)(nb > 0 ? nb : 1) * 8);
    uint8_t *matched = fill_build ? calloc((size_t)(nb > 0 ? nb : 1), 1)
                                  : NULL;
    if (!head || !next || (fill_build && !matched)) {
        free(head); free(next); free(matched);
        return ORA_ERR_OOM;
    }
    for (int64_t i = 0; i < nbuckets; i++) head[i] = -1;
    for (int64_t i = 0; i < nb; i++) {
        if (nk_rownull(bks, i)) continue;
        int64_t b = (int64_t)(nk_hash(bks, i) & (uint64_t)(nbuckets - 1));
        next[i] = head[b];
        head[b] = i;
    }

    int64_t cap = 64, n = 0;
    int64_t *bi = malloc(cap * 8), *pi = malloc(cap * 8);
    ora_status st = ORA_OK;
    if (!bi || !pi) { st = ORA_ERR_OOM; goto done; }

    for (int64_t p = 0; p < np; p++) {
        if (nk_rownull(pks, p)) {
            if (fill_probe || jt == 3)
                if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
            continue;
        }
        int64_t bkt = (int64_t)(nk_hash(pks, p) & (uint64_t)(nbuckets - 1));
        int64_t nmatch = 0;
        for (int64_t m = head[bkt]; m >= 0; m = next[m]) {
            if (nk_rownull(bks, m) || !nk_match(bks, m, pks, p)) continue;
            nmatch++;
            if (matched) matched[m] = 1;
            if (emit_match)
                if ((st = ext_push(&bi, &pi, &n, &cap, m, p))) goto done;
            if (jt == 2) break;
        }
        if (jt == 2 && nmatch > 0) {
            if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
        } else if ((jt == 3 || fill_probe) && nmatch == 0) {
            if ((st = ext_push(&bi, &pi, &n, &cap, -1, p))) goto done;
        }
    }
    if (fill_build)
        for (int64_t i = 0; i < nb; i++)
            if (nk_rownull(bks, i) || !matched[i])
                if ((st = ext_push(&bi, &pi, &n, &cap, i, -1))) goto done;

done:
    free(head); free(next); free(matched);
    if (st) { free(bi); free(pi); return st; }
    *out_bidx = bi; *out_pidx = pi; *nout = n;
    return ORA_OK;
}
