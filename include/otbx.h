/*
 * otbx.h — C-ABI of the MI355X-native OpenTenBase executor offload.
 *
 * This is the drop-in boundary: the entry points a
 * CustomScan provider (include/nodes/extensible.h:117-152, driven by
 * executor/nodeCustom.c:31-125 in the reference) calls from
 * BeginCustomScan / ExecCustomScan / EndCustomScan, callable equally from a
 * standalone C harness. Plain pointers + sizes only; all device work is
 * stream-ordered on the caller's HIP stream (passed as void*); errors are
 * status codes the provider shim converts to ereport(ERROR) (the reference's
 * error convention, utils/error/elog.c).
 *
 * Replaced reference interfaces, per entry point:
 *   otbx_init/_finish      — _PG_init library load + per-GPU arbiter
 *                            (utils/fmgr/dfmgr.c:98 load path; one backend
 *                            process per DN ↔ one HIP device context)
 *   otbx_stage_*           — heap → device-resident columnar staging; stands
 *                            where heapgetpage/heapgettup_pagemode
 *                            (access/heap/heapam.c:388,920) feed the scan
 *   otbx_scan_count        — SeqScan + ExecQual + COUNT (execScan.c:140,
 *                            execExprInterp.c:324, int8inc int8.c:714)
 *   otbx_q1_partial        — the whole DN fragment of TPC-H Q1:
 *                            SeqScan → qual → project → Partial HashAgg
 *                            (nodeAgg.c:2609/856; AGGSPLIT_INITIAL_SERIAL,
 *                            include/nodes/nodes.h:964) as one CustomScan
 *                            covering the fragment subtree
 *   otbx_q3_partial        — DN fragment of TPC-H Q3: two hash joins
 *                            (nodeHash.c:1828/2174, nodeHashjoin.c:186) +
 *                            Partial HashAgg keyed on l_orderkey
 *   otbx_agg_i64 /
 *   otbx_join_i64          — the composable HashAggregate / inner HashJoin
 *                            operators for generic plan shapes (+ NULL
 *                            semantics parity: execGrouping.c:295 NULL==NULL
 *                            grouping; nodeHash.c:2026 NULL keys dropped)
 *   merge (Coordinator)    — NOT here: the shard merge (execFragment.c:3877)
 *                            is a collective over ranks, done by the host
 *                            layer with RCCL (torch.distributed) on the
 *                            partial-state buffers these calls return.
 *
 * THREADING / STREAM CONTRACT: the library is single-threaded per process,
 * matching the reference's execution model (each DataNode backend is a
 * single-threaded process). Entry points may be called from one
 * thread at a time with one stream in flight; concurrent calls from
 * multiple threads or interleaved streams are NOT supported (several entry
 * points keep per-process cached scratch). Multiple backends on one host
 * each load their own copy (process isolation). otbx_finish releases all
 * cached scratch, so init → work → finish → init(other_device) is clean.
 *
 * Tuning/test environment variables (read at call time; all optional — the
 * GUC analog of the provider shim, guc.c):
 *   OTBX_PART_TILE=0       — select the legacy cursor-scatter partitioner
 *                            in otbx_agg_i64/otbx_join_i64 (default is the
 *                            tile-staged counting sort)
 *   OTBX_JOINP_FORCE=1     — force the partitioned join path below its
 *                            8 M-row build threshold (tests)
 *   OTBX_Q9_BITMAP_BITS=N  — cap the Q9 part-bitmap slice width (bits);
 *                            affects otbx_q9_workspace_bytes AND
 *                            otbx_q9_partial identically (default 2^30 =
 *                            single-pass for any realistic part table)
 *   OTBX_DIRECT_CAP / OTBX_Q3_FORCE_HASH / OTBX_Q3_HASH_BUDGET
 *                          — Q3 dense-direct vs hash+bloom path selection
 *                            overrides (tests force the fallback)
 *   OTBX_Q9_FILTER_WAVE=1  — legacy per-wave appender in the Q9 part filter
 *                            (default is the tile-staged compaction; A/B)
 *   OTBX_Q3_COMPACT_LEGACY=1 — legacy block-chunk Q3 group compaction
 *                            (default is the word-granular bitmap walk)
 *   OTBX_Q3_COMPACT_TILE=1 — quad-granular tile-staged compaction (A/B)
 *   OTBX_NK_FORCE_CAP=N    — force a tiny first-attempt table in the
 *                            generality-tier aggregates (tests the
 *                            estimator-overflow abort + full-cap rerun)
 *   OTBX_JOINX_VIA_INNER=1 — route 1-key left/right/full via the inner
 *                            join + mark-and-fill (measured slower than
 *                            the FSM table at all tested shapes; kept
 *                            parity-tested for future shapes). Affects
 *                            otbx_join_ext_workspace_bytes AND the route
 *                            identically (query sizing under the same env)
 */
#ifndef OTBX_H
#define OTBX_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum {
    OTBX_OK = 0,
    OTBX_ERR_HIP = 1,       /* HIP runtime failure (provider → ereport)   */
    OTBX_ERR_OOM = 2,
    OTBX_ERR_INVALID = 3,
    OTBX_ERR_OVERFLOW = 4,  /* numeric value out of range                 */
    OTBX_ERR_NO_GPU = 5
} otbx_status;

const char *otbx_version(void);
const char *otbx_status_str(otbx_status s);

/* Device lifecycle. otbx_init selects the HIP device (one DN backend ↔ one
 * GPU; the per-GPU arbiter of SURVEY §8b). */
otbx_status otbx_init(int device);
otbx_status otbx_finish(void);

/* Stream-ordered device allocation for standalone (non-torch) callers. */
otbx_status otbx_device_malloc(void **ptr, size_t bytes);
otbx_status otbx_device_free(void *ptr);
otbx_status otbx_memcpy_h2d(void *dst_dev, const void *src_host, size_t bytes, void *stream);

/* ---- heap-page → SoA staging shim (host-side; the provider's
 * BeginCustomScan staging step — INTEGRATION.md §4). Walks PostgreSQL-format
 * heap pages (PageHeaderData storage/bufpage.h:157; ItemIdData
 * storage/itemid.h:25; HeapTupleHeaderData access/htup_details.h:118) and
 * deforms each LP_NORMAL tuple (heap_deform_tuple,
 * access/common/heaptuple.c:936) into caller-provided per-column host
 * arrays, ready for otbx_memcpy_h2d. Fixed-width attributes only; MVCC
 * visibility stays server-side (LP_NORMAL = post-heapgetpage state).
 * out_nulls[a] (byte-per-row, may be NULL per column) receives the NULL
 * bitmap; a NULL attribute with no out_nulls[a] is an error. */
typedef struct {
    uint16_t attlen;   /* attribute width: 1, 2, 4 or 8 bytes */
    uint16_t attalign; /* typalign in bytes: 1, 2, 4 or 8 */
} otbx_attdesc;
otbx_status otbx_stage_pages(const void *pages_host, int64_t npages,
                             size_t page_size, const otbx_attdesc *atts,
                             int32_t natts, void **out_cols_host,
                             uint8_t **out_nulls_host, int64_t cap_rows,
                             int64_t *nrows_out);
otbx_status otbx_memcpy_d2h(void *dst_host, const void *src_dev, size_t bytes, void *stream);
otbx_status otbx_stream_sync(void *stream);

/* ---- staged columnar tables (device pointers; SoA — DESIGN.md §2) ---- */

typedef struct {
    int64_t n;
    int64_t *l_orderkey;      /* may be NULL if not staged (Q1 needs none) */
    double  *l_quantity, *l_extendedprice, *l_discount, *l_tax;
    uint8_t *l_returnflag, *l_linestatus;
    int32_t *l_shipdate;
    int64_t *l_partkey;       /* appended; may be NULL (Q9-mix only) */
    /* optional Q9 probe-side record cache {l_orderkey i64, l_extendedprice
     * f64, l_discount f64, pad} (32 B/row), built ONCE at staging
     * (otbx_build_q9recs — a derived layout like the zone-map metadata):
     * the Q9 probe's per-survivor gather then touches one cache line
     * instead of three column lines. NULL = probe gathers the columns. */
    void *q9rec;
    /* optional compact-key cache (otbx_build_key32, built once at staging):
     * int32 copy of l_orderkey, valid iff every key fits [0, 2^31) — at
     * TPC-H scales orderkeys do (SF100: max 6e8). Halves the probe's key
     * stream. NULL = kernels read the i64 column. */
    int32_t *l_orderkey32;
    int32_t *l_partkey32;     /* same compact-key cache for l_partkey
                               * (halves the Q9 filter's key stream) */
} otbx_lineitem_dev;

typedef struct {
    int64_t n;
    int64_t *o_orderkey, *o_custkey;
    int32_t *o_orderdate, *o_shippriority;
    /* staged-table metadata (zone-map style, computed once at staging —
     * otbx_gen_orders_dev fills it; a provider computes it while walking
     * pages). has_minmax = 0 → executors run their own minmax kernel. */
    int64_t okey_min, okey_max;
    int32_t has_minmax;
    /* optional compact-key caches (otbx_build_key32; see lineitem note) */
    int32_t *o_orderkey32, *o_custkey32;
} otbx_orders_dev;

typedef struct {
    int64_t n;
    int64_t *c_custkey;
    uint8_t *c_mktsegment;
} otbx_customer_dev;

typedef struct {
    int64_t n;
    int64_t *p_partkey;
    uint8_t *p_type;
} otbx_part_dev;

/* On-device synthetic generation (the dbgen analog; same counter-based
 * functions as the CPU oracle — oracle/otbx_gen.h — so tables are
 * bit-identical on both sides). Caller provides the device buffers
 * (column pointers in the struct, each sized for n_global/nranks rows). */
/* build an int32 compact-key cache from an i64 key column: writes
 * saturated casts into dst32_dev and *ok_host = 1 if every value fit
 * [0, 2^31), else 0 (caller must then leave the cache pointer NULL).
 * Synchronous (staging-time only). */
otbx_status otbx_build_key32(const int64_t *src_dev, int64_t n,
                             int32_t *dst32_dev, int32_t *ok_host,
                             void *stream);
/* build the q9rec cache from the staged columns (32 B/row into recs_dev) */
otbx_status otbx_build_q9recs(const otbx_lineitem_dev *l, void *recs_dev,
                              void *stream);
otbx_status otbx_gen_lineitem_dev(const otbx_lineitem_dev *t, uint64_t seed,
                                  int64_t n_global, uint32_t rank,
                                  uint32_t nranks, void *stream);
otbx_status otbx_gen_orders_dev(otbx_orders_dev *t, uint64_t seed,
                                int64_t n_global, int64_t ncust_global,
                                uint32_t rank, uint32_t nranks, int skew,
                                void *stream);
otbx_status otbx_gen_customer_dev(const otbx_customer_dev *t, uint64_t seed,
                                  int64_t n_global, uint32_t rank,
                                  uint32_t nranks, void *stream);
/* part is a replicated dimension table: every rank holds all n_global rows */
otbx_status otbx_gen_part_dev(const otbx_part_dev *t, uint64_t seed,
                              int64_t n_global, void *stream);

/* ---- config 2: SeqScan + qual + COUNT(*) (scan-bandwidth kernel) ----
 * count_dev: one int64 device slot (zeroed by the call). */
otbx_status otbx_scan_count(const int32_t *shipdate_dev, int64_t n,
                            int32_t cutoff, int64_t *count_dev, void *stream);

/* ---- TPC-H Q1 DN fragment ----
 * Partial-aggregate state per (l_returnflag,l_linestatus) group, dense over
 * the 6 possible combos, fixed slot order (A,F)(A,O)(N,F)(N,O)(R,F)(R,O):
 *   sums_dev:   double[6][5] = {sum_qty, sum_base_price, sum_disc_price,
 *                               sum_charge, sum_disc}
 *   counts_dev: int64[6]     = count(*)
 * Both zeroed by the call; the Coordinator-side merge all-gathers exactly
 * these buffers. kernel_ms (host, may be NULL): HIP-event time of the fused
 * kernel on `stream` (the bench's roofline numerator — DESIGN.md §4). */
otbx_status otbx_q1_partial(const otbx_lineitem_dev *t, int32_t cutoff_day,
                            double *sums_dev, int64_t *counts_dev,
                            void *stream, float *kernel_ms);
/* A/B harness entry: variant 0 = 2 rows/lane, 1 = 4 rows/lane,
 * 2 = 4 rows/lane + non-temporal loads. otbx_q1_partial dispatches the
 * measured-best variant. */
otbx_status otbx_q1_partial_variant(const otbx_lineitem_dev *t,
                                    int32_t cutoff_day, double *sums_dev,
                                    int64_t *counts_dev, void *stream,
                                    float *kernel_ms, int variant);

/* ---- TPC-H Q3 DN fragment ----
 * customer/orders/lineitem staged on-device; custkeys of the replicated
 * (broadcast) customer build side are read from cust_keys_dev when non-NULL
 * (ncust_keys rows; the post-all-gather buffer), else from c->... filtered
 * by segment locally.
 * Outputs (device, caller-allocated):
 *   groups_dev:  capacity cap_groups entries of otbx_q3_group
 *   ngroups_dev: int64[1] — compacted group count
 * ws_dev: workspace (hash tables), size from otbx_q3_workspace_bytes. */
typedef struct {
    int64_t l_orderkey;
    double revenue;
    int32_t o_orderdate, o_shippriority;
} otbx_q3_group;

otbx_status otbx_q3_workspace_bytes(int64_t ncust, int64_t norders,
                                    int64_t nlineitem, size_t *bytes);
/* kernel_ms (host, may be NULL): float[4] = {customer-keyset build,
 * orders build+probe, lineitem probe+partial-agg, compact} HIP-event times.
 * stats_dev (may be NULL): int64[1] = probe hits (lineitem rows passing the
 * date qual AND matching an order — the N_probe_hits of the §8d roofline
 * formula); zeroed by the call. */
otbx_status otbx_q3_partial(const otbx_customer_dev *c,
                            const otbx_orders_dev *o,
                            const otbx_lineitem_dev *l,
                            const int64_t *cust_keys_dev, int64_t ncust_keys,
                            uint8_t segment, int32_t q3date,
                            void *ws_dev, size_t ws_bytes,
                            otbx_q3_group *groups_dev, int64_t cap_groups,
                            int64_t *ngroups_dev, int64_t *stats_dev,
                            void *stream, float *kernel_ms);

/* top-k selection over compacted Q3 groups (ORDER BY revenue DESC …
 * LIMIT k pre-selection; the final k-way ordering happens host-side on the
 * ≤ cap_cand candidates). hist_dev: uint32[16384] workspace (zeroed by the
 * call). ncand_dev ≥ k unless n < k; candidates are all groups with revenue
 * ≥ the selection threshold (a bit-pattern bin boundary). */
otbx_status otbx_topk_by_revenue(const otbx_q3_group *groups_dev, int64_t n,
                                 int64_t k, otbx_q3_group *cand_dev,
                                 int64_t cap_cand, int64_t *ncand_dev,
                                 uint32_t *hist_dev, void *stream);
/* helper for the broadcast build side: compact custkeys where
 * c_mktsegment == segment into keys_out_dev, count into nkeys_dev (zeroed). */
otbx_status otbx_filter_customer(const otbx_customer_dev *c, uint8_t segment,
                                 int64_t *keys_out_dev, int64_t *nkeys_dev,
                                 void *stream);

/* ---- Q9-mix DN fragment (BASELINE config 5's second query shape) ----
 * lineitem ⋈ part (p_type % typemod == typeval) ⋈ orders, partial aggregate
 * GROUP BY year(o_orderdate) — two joins under an aggregate on a COMPUTED
 * key. Dense outputs over the 7 order years (0 = 1992):
 *   sums_dev: double[7] revenue, counts_dev: int64[7]; both zeroed by the
 * call; the Coordinator merge all-gathers them like Q1's states.
 * ws: otbx_q9_workspace_bytes(nparts, norders_local, nranks). */
otbx_status otbx_q9_workspace_bytes(int64_t nparts, int64_t norders,
                                    int64_t nlineitem, uint32_t nranks,
                                    size_t *bytes);
otbx_status otbx_q9_partial(const otbx_part_dev *p, const otbx_orders_dev *o,
                            const otbx_lineitem_dev *l, uint8_t typemod,
                            uint8_t typeval, void *ws, size_t ws_bytes,
                            double *sums_dev, int64_t *counts_dev,
                            void *stream, float *kernel_ms);

/* ---- composable operators (generic plan shapes + NULL-semantics parity) --

 * Hash aggregate: group by nullable i64 key, aggregate nullable f64 value;
 * count(*), count(v), sum(v), avg-N/Sx. Output: open-addressing table
 * compacted to groups_dev (cap = n). Null bitmaps are byte-per-row (1 =
 * NULL), may be NULL pointers. */
typedef struct {
    int64_t key;
    int64_t count_star;
    int64_t count_v;
    double sum_v;
    int32_t key_isnull;
    int32_t sum_isnull;
} otbx_agg_group;

otbx_status otbx_agg_i64_workspace_bytes(int64_t n, size_t *bytes);
otbx_status otbx_agg_i64(const int64_t *keys_dev, const uint8_t *key_null_dev,
                         const double *vals_dev, const uint8_t *val_null_dev,
                         int64_t n, void *ws_dev, size_t ws_bytes,
                         otbx_agg_group *groups_dev, int64_t *ngroups_dev,
                         void *stream);

/* ---- extended join types + two-key variants ----
 * The HJ_* fill-state FSM (executor/nodeHashjoin.c:139-144) on the generic
 * join, and multi-key (2 x i64) variants of join and group-by.
 * join_type: 0 inner, 1 left, 2 semi, 3 anti, 4 right, 5 full.
 * Pair encoding in (out_bidx, out_pidx):
 *   match                          -> (bidx, pidx)
 *   left/full unmatched probe row  -> (-1, pidx)  [HJ_FILL_OUTER_TUPLE,
 *     incl. NULL-key probe rows]      nodeHashjoin.c:142,668]
 *   right/full unmatched build row -> (bidx, -1)  [HJ_FILL_INNER_TUPLES,
 *     incl. NULL-key build rows]      nodeHashjoin.c:143,693;
 *                                     ExecScanHashTableForUnmatched
 *                                     nodeHash.c:2322]
 *   semi: (-1, pidx) once per probe row with >= 1 match (JOIN_SEMI
 *     advances after the first match, nodeHashjoin.c:572)
 *   anti: (-1, pidx) per probe row with no match (nodeHashjoin.c:631)
 * Two-key variants join/group on the ROW (k1,k2): a row with EITHER key
 * NULL never matches (strict equality; multi-key hash combine =
 * rotate-left-1 xor, nodeHash.c:2059, restated at 64 bit — parity is on
 * result sets). Overflow contract identical to otbx_join_i64. */
otbx_status otbx_join_ext_workspace_bytes(int64_t nb, int64_t np,
                                          size_t *bytes);
otbx_status otbx_join_i64_ext(const int64_t *bkeys_dev,
                              const uint8_t *bnull_dev, int64_t nb,
                              const int64_t *pkeys_dev,
                              const uint8_t *pnull_dev, int64_t np,
                              int32_t join_type, void *ws_dev,
                              size_t ws_bytes, int64_t *out_bidx_dev,
                              int64_t *out_pidx_dev, int64_t cap_pairs,
                              int64_t *npairs_dev, void *stream);
otbx_status otbx_join_i64x2(const int64_t *bk1_dev, const uint8_t *bn1_dev,
                            const int64_t *bk2_dev, const uint8_t *bn2_dev,
                            int64_t nb, const int64_t *pk1_dev,
                            const uint8_t *pn1_dev, const int64_t *pk2_dev,
                            const uint8_t *pn2_dev, int64_t np,
                            int32_t join_type, void *ws_dev, size_t ws_bytes,
                            int64_t *out_bidx_dev, int64_t *out_pidx_dev,
                            int64_t cap_pairs, int64_t *npairs_dev,
                            void *stream);

/* two-key hash aggregate: group by (nullable k1, nullable k2) with
 * NULL==NULL grouping (execGrouping.c:295,:525); aggregates as
 * otbx_agg_i64. groups_dev capacity = n rows; *ngroups_dev receives the
 * group count; emission order is arbitrary (hash-table order, as the
 * reference's simplehash iteration). */
typedef struct {
    int64_t key1;
    int64_t key2;
    int64_t count_star;
    int64_t count_v;
    double sum_v;
    int32_t key1_isnull;
    int32_t key2_isnull;
    int32_t sum_isnull;
    int32_t _pad;
} otbx_agg2_group; /* 56 B */

otbx_status otbx_agg_i64x2_workspace_bytes(int64_t n, size_t *bytes);
otbx_status otbx_agg_i64x2(const int64_t *k1_dev, const uint8_t *k1null_dev,
                           const int64_t *k2_dev, const uint8_t *k2null_dev,
                           const double *vals_dev,
                           const uint8_t *val_null_dev, int64_t n,
                           void *ws_dev, size_t ws_bytes,
                           otbx_agg2_group *groups_dev, int64_t *ngroups_dev,
                           void *stream);

/* ---- N-key (1..8 columns) group-by and join ----
 * Group/join identity is the ROW of key columns: NULL==NULL for grouping
 * (execGrouping.c:295,:525), any-NULL-never-matches for joins
 * (nodeHash.c:2026); hash = iterated rotate-left-1 xor over per-column
 * hashes (nodeHash.c:2059). Groups carry the DEFINING ROW INDEX instead of
 * N key values — the reference's hash table stores the representative
 * tuple the same way (execGrouping.c firstTuple); the caller reads the key
 * values back through the index. Joins emit (bidx, pidx) pairs exactly as
 * otbx_join_i64_ext (same join_type codes, fills and overflow contract). */
#define OTBX_MAX_KEYS 8
typedef struct {
    int32_t nkeys;                           /* 1..OTBX_MAX_KEYS */
    const int64_t *keys[OTBX_MAX_KEYS];      /* device pointers */
    const uint8_t *nulls[OTBX_MAX_KEYS];     /* per column; may be NULL */
} otbx_keyset;

typedef struct {
    int64_t row_idx;          /* defining row (representative tuple) */
    int64_t count_star;
    int64_t count_v;
    double sum_v;
    int32_t sum_isnull;
    int32_t _pad;
} otbx_aggn_group; /* 40 B */

otbx_status otbx_agg_i64n_workspace_bytes(int64_t n, size_t *bytes);
otbx_status otbx_agg_i64n(const otbx_keyset *ks, const double *vals_dev,
                          const uint8_t *val_null_dev, int64_t n,
                          void *ws_dev, size_t ws_bytes,
                          otbx_aggn_group *groups_dev, int64_t *ngroups_dev,
                          void *stream);

otbx_status otbx_join_i64n_workspace_bytes(int64_t nb, int64_t np,
                                           size_t *bytes);
otbx_status otbx_join_i64n(const otbx_keyset *bks, int64_t nb,
                           const otbx_keyset *pks, int64_t np,
                           int32_t join_type, void *ws_dev, size_t ws_bytes,
                           int64_t *out_bidx_dev, int64_t *out_pidx_dev,
                           int64_t cap_pairs, int64_t *npairs_dev,
                           void *stream);

/* ---- exact decimal (scaled-int64) aggregate with int128 sum ----
 * The reference's HAVE_INT128 numeric aggregation: group state =
 * Int128AggState {N, sumX} (utils/adt/numeric.c:5072; do_int128_accum
 * :4998; transition int8_avg_accum :5365; sum(bigint) promotes to numeric
 * and cannot overflow, int8_sum :6206). Values are scaled-decimal int64
 * (e.g. NUMERIC(15,2) money in cents); the emitted two's-complement
 * 128-bit sum is EXACT, so parity with the reference is bit-exact —
 * no float tolerance. groups_dev capacity = n. */
typedef struct {
    int64_t key;
    int64_t count_star;
    int64_t count_v;
    int64_t sum_hi;          /* int128 two's-complement high word */
    uint64_t sum_lo;
    int32_t key_isnull;
    int32_t sum_isnull;
} otbx_dec_group; /* 48 B */

otbx_status otbx_agg_i64_dec_workspace_bytes(int64_t n, size_t *bytes);
otbx_status otbx_agg_i64_dec(const int64_t *keys_dev,
                             const uint8_t *key_null_dev,
                             const int64_t *vals_dev,
                             const uint8_t *val_null_dev, int64_t n,
                             void *ws_dev, size_t ws_bytes,
                             otbx_dec_group *groups_dev,
                             int64_t *ngroups_dev, void *stream);

/* ---- GPU ORDER BY ----
 * Full sort of Q3 group rows by (revenue DESC, o_orderdate ASC) — the
 * tuplesort.c analog for the no-LIMIT ORDER BY case (LIMIT queries use
 * otbx_topk_by_revenue). Stable LSD radix sort; out_dev must not alias
 * groups_dev. */
otbx_status otbx_order_groups_workspace_bytes(int64_t n, size_t *bytes);
otbx_status otbx_order_groups(const otbx_q3_group *groups_dev, int64_t n,
                              otbx_q3_group *out_dev, void *ws,
                              size_t ws_bytes, void *stream);

/* ---- repartition exchange ----
 * The GPU half of the reference's "Distribute results by H: col" exchange
 * (make_remotesubplan, optimizer/plan/createplan.c:8671; locator semantics
 * shardid → node, pgxc/shard/shardmap.c:2231 restated as key % nranks for
 * the dense-key locator of DESIGN.md §2 — owner = (uint64_t)key % nranks,
 * i.e. the UNSIGNED-cast modulo: negative keys map to
 * (2^64 + key) % nranks, NOT the C signed remainder and NOT Python's
 * floored modulo; server-side locator code must use the same cast): groups
 * rows by owning rank into a
 * permutation with contiguous per-rank segments; the host layer then
 * all-to-alls the gathered segments over RCCL (fragment.py). nranks ≤ 64.
 * counts_host: int64[nranks], written synchronously (the call syncs). */
otbx_status otbx_partition_by_key(const int64_t *keys_dev, int64_t n,
                                  uint32_t nranks, int64_t *perm_dev,
                                  int64_t *counts_host, void *stream);

/* permutation gathers (dst[i] = src[perm[i]]) for applying the partition to
 * payload columns without leaving the native path */
otbx_status otbx_gather_i64(const int64_t *src, const int64_t *perm, int64_t n,
                            int64_t *dst, void *stream);
otbx_status otbx_gather_f64(const double *src, const int64_t *perm, int64_t n,
                            double *dst, void *stream);
otbx_status otbx_gather_i32(const int32_t *src, const int64_t *perm, int64_t n,
                            int32_t *dst, void *stream);
otbx_status otbx_gather_u8(const uint8_t *src, const int64_t *perm, int64_t n,
                           uint8_t *dst, void *stream);

/* Inner hash join on i64 keys: emits (build_idx, probe_idx) pairs in
 * arbitrary order (result-set parity; SQL imposes no order).
 * OVERFLOW CONTRACT: *npairs_dev always receives the TRUE match count; if
 * it exceeds cap_pairs the output arrays hold only a cap_pairs-bounded
 * subset and the caller MUST treat the result as overflowed (re-run with
 * cap_pairs >= *npairs_dev). The call itself still returns OTBX_OK — the
 * count lives on the device and is not visible to the host entry point;
 * this mirrors how the reference sizes hash tables from observed counts
 * (ExecHashTableInsert growth, nodeHash.c:1876) rather than failing
 * mid-scan. Pinned by tests/test_gpu_parity.py::test_join_overflow_contract. */
otbx_status otbx_join_i64_workspace_bytes(int64_t nb, int64_t np,
                                           size_t *bytes);
otbx_status otbx_join_i64(const int64_t *bkeys_dev, const uint8_t *bnull_dev,
                          int64_t nb,
                          const int64_t *pkeys_dev, const uint8_t *pnull_dev,
                          int64_t np, void *ws_dev, size_t ws_bytes,
                          int64_t *out_bidx_dev, int64_t *out_pidx_dev,
                          int64_t cap_pairs, int64_t *npairs_dev,
                          void *stream);

#ifdef __cplusplus
}
#endif
#endif
