"""Property-based pinning of the CPU oracle against brute-force Python
restatements (hypothesis). The oracle is the parity anchor for the GPU path
(SURVEY §8c); these tests guard the anchor itself on adversarial small
inputs: duplicate / negative / INT64_MIN keys, NULL masks on keys and
values, empty inputs, and float accumulation in scan order (reference
semantics: nodeAgg.c:856 advance_aggregates via int8inc int8.c:714 and
float8pl float.c:970 — strict transfns skip NULL inputs, NULL keys form one
group, inner-join NULL keys never match)."""
import numpy as np
from hypothesis import given, settings, strategies as st

from oracle import oracle_py as ora

I64_MIN = -(2**63)

small_key = st.integers(min_value=-4, max_value=4)
wild_key = st.one_of(small_key,
                     st.sampled_from([I64_MIN, I64_MIN + 1, 2**63 - 1]))
val = st.floats(min_value=-1e12, max_value=1e12,
                allow_nan=False, width=64)


@st.composite
def agg_input(draw):
    n = draw(st.integers(min_value=0, max_value=80))
    keys = np.array(draw(st.lists(wild_key, min_size=n, max_size=n)),
                    dtype=np.int64)
    vals = np.array(draw(st.lists(val, min_size=n, max_size=n)),
                    dtype=np.float64)
    kn = np.array(draw(st.lists(st.booleans(), min_size=n, max_size=n)),
                  dtype=np.uint8)
    vn = np.array(draw(st.lists(st.booleans(), min_size=n, max_size=n)),
                  dtype=np.uint8)
    return keys, vals, kn, vn


@settings(max_examples=80, deadline=None, derandomize=True)
@given(agg_input())
def test_agg_matches_bruteforce(inp):
    keys, vals, kn, vn = inp
    got = ora.agg_i64(keys, vals, key_null=kn, val_null=vn)
    # brute force in SCAN ORDER (float sums must match bit-exactly: the
    # oracle restates the reference's tuple-at-a-time accumulation)
    bf = {}  # (key, isnull) -> [count_star, count_v, sum_v]
    for i in range(len(keys)):
        gk = (0, True) if kn[i] else (int(keys[i]), False)
        g = bf.setdefault(gk, [0, 0, 0.0])
        g[0] += 1
        if not vn[i]:
            g[1] += 1
            g[2] += float(vals[i])
    assert len(got) == len(bf)
    for g in got:
        e = bf[(int(g.key) if not g.key_isnull else 0, bool(g.key_isnull))]
        assert g.count_star == e[0]
        assert g.count_v == e[1]
        if e[1]:
            assert g.sum_v == e[2]  # bit-exact: same accumulation order
        else:
            assert g.sum_isnull  # SUM over all-NULL = NULL


@st.composite
def join_input(draw):
    nb = draw(st.integers(min_value=0, max_value=40))
    npr = draw(st.integers(min_value=0, max_value=60))
    bk = np.array(draw(st.lists(wild_key, min_size=nb, max_size=nb)),
                  dtype=np.int64)
    pk = np.array(draw(st.lists(wild_key, min_size=npr, max_size=npr)),
                  dtype=np.int64)
    bn = np.array(draw(st.lists(st.booleans(), min_size=nb, max_size=nb)),
                  dtype=np.uint8)
    pn = np.array(draw(st.lists(st.booleans(), min_size=npr, max_size=npr)),
                  dtype=np.uint8)
    return bk, pk, bn, pn


@settings(max_examples=80, deadline=None, derandomize=True)
@given(join_input())
def test_join_matches_bruteforce(inp):
    bk, pk, bn, pn = inp
    bi, pi = ora.join_i64(bk, pk, bnull=bn, pnull=pn)
    exp = sorted((i, j)
                 for i in range(len(bk)) if not bn[i]
                 for j in range(len(pk)) if not pn[j]
                 if bk[i] == pk[j])
    assert sorted(zip(bi.tolist(), pi.tolist())) == exp


@settings(max_examples=25, deadline=None, derandomize=True)
@given(st.integers(min_value=0, max_value=5000),
       st.integers(min_value=-1, max_value=4000),
       st.integers(min_value=1, max_value=2**31 - 1))
def test_q1_partial_matches_numpy_at_any_size(n, cutoff, seed_shift):
    """q1_partial (the fused Q1 shape the GPU mirrors) vs a numpy
    restatement at arbitrary sizes incl. 0 and cutoffs that pass nothing /
    everything; shard parameters vary via rank to move the generator's
    stream (gen_tables is deterministic per (n, rank))."""
    if n == 0:
        return  # gen_tables requires n>=1; empty covered in semantics tests
    # sharded generation requires n divisible by nranks (generator
    # contract, oracle.c ora_gen_lineitem)
    rank, nranks = (seed_shift % 7, 8) if n >= 8 and n % 8 == 0 else (0, 1)
    t = ora.gen_tables(n, rank=rank, nranks=nranks)
    li = t["lineitem"]
    groups = ora.q1_partial(t, cutoff=cutoff)
    mask = li["l_shipdate"] <= cutoff
    assert sum(g.count_order for g in groups) == int(mask.sum())
    for g in groups:
        m = mask & (li["l_returnflag"] == g.returnflag) & \
            (li["l_linestatus"] == g.linestatus)
        assert g.count_order == int(m.sum())
        if m.any():
            assert abs(g.sum_qty - float(li["l_quantity"][m].sum())) <= \
                1e-9 * max(abs(g.sum_qty), 1.0)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(st.integers(min_value=1, max_value=3000),
       st.integers(min_value=1, max_value=50),
       st.integers(min_value=0, max_value=49))
def test_q9_partial_matches_numpy_any_size_and_filter(n, typemod, typeval):
    """q9_partial across sizes and part-filter parameters (p_type %
    typemod == typeval), incl. typeval >= typemod (matches nothing) and
    typemod=1 (matches everything) — the selectivity extremes the fixed
    reference test does not reach."""
    t = ora.gen_tables(n, need=("lineitem", "orders", "part"))
    li, od, pt = t["lineitem"], t["orders"], t["part"]
    sel = pt["p_type"] % typemod == typeval
    part_ok = np.zeros(int(pt["p_partkey"].max(initial=0)) + 1, dtype=bool)
    part_ok[pt["p_partkey"][sel]] = True
    # orderkey -> orderdate, with -1 for keys absent from orders (tiny
    # tables: norders = n//4 may not cover every l_orderkey; the inner
    # join drops those rows)
    kmax = max(int(od["o_orderkey"].max(initial=0)),
               int(li["l_orderkey"].max(initial=0)))
    odate = np.full(kmax + 1, -1, dtype=np.int32)
    odate[od["o_orderkey"]] = od["o_orderdate"]
    mask = part_ok[li["l_partkey"]] & (odate[li["l_orderkey"]] >= 0)
    got = ora.q9_partial(t, typemod=typemod, typeval=typeval)
    if not mask.any():
        assert got == []
        return
    d = odate[li["l_orderkey"][mask]]
    rev = li["l_extendedprice"][mask] * (1.0 - li["l_discount"][mask])
    bounds = np.array([0, 366, 731, 1096, 1461, 1827, 2192, 2558])
    yr = np.searchsorted(bounds, d, side="right") - 1
    exp = {}
    for y in range(7):
        m = yr == y
        if m.sum():
            exp[y] = (float(rev[m].sum()), int(m.sum()))
    assert {g.year for g in got} == set(exp)
    for g in got:
        assert g.count_rows == exp[g.year][1]
        assert abs(g.revenue - exp[g.year][0]) <= \
            1e-9 * max(abs(exp[g.year][0]), 1e-9)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(st.integers(min_value=1, max_value=3000),
       st.integers(min_value=0, max_value=5),
       st.integers(min_value=-1, max_value=3000))
def test_q3_partial_matches_numpy_any_size(n, segment, date):
    """q3_partial across sizes (incl. customer/orders tables of 0 rows at
    n<40), market segments (incl. 5 = matches nothing; domain is 0..4) and
    date cutoffs at both extremes."""
    t = ora.gen_tables(n, need=("lineitem", "orders", "customer"))
    li, od, cu = t["lineitem"], t["orders"], t["customer"]
    rows = ora.q3_partial(t, segment=segment, date=date)
    seg_keys = cu["c_custkey"][cu["c_mktsegment"] == segment]
    omask = (od["o_orderdate"] < date) & np.isin(od["o_custkey"], seg_keys)
    okeys = od["o_orderkey"][omask]
    lmask = (li["l_shipdate"] > date) & np.isin(li["l_orderkey"], okeys)
    exp_groups = set(li["l_orderkey"][lmask].tolist())
    assert set(rows["l_orderkey"].tolist()) == exp_groups
    if lmask.any():
        rev = li["l_extendedprice"][lmask] * (1.0 - li["l_discount"][lmask])
        byk = {}
        for k, r in zip(li["l_orderkey"][lmask].tolist(), rev.tolist()):
            byk[k] = byk.get(k, 0.0) + r
        got = dict(zip(rows["l_orderkey"].tolist(),
                       rows["revenue"].tolist()))
        for k, v in byk.items():
            assert abs(got[k] - v) <= 1e-9 * max(abs(v), 1e-9)


# ---------------- round-2 operators: hypothesis pinning ----------------

@st.composite
def dec_input(draw):
    n = draw(st.integers(min_value=0, max_value=60))
    wild_val = st.one_of(st.integers(min_value=-100, max_value=100),
                         st.sampled_from([I64_MIN, I64_MIN + 1, 2**63 - 1]))
    keys = np.array(draw(st.lists(small_key, min_size=n, max_size=n)),
                    dtype=np.int64)
    vals = np.array(draw(st.lists(wild_val, min_size=n, max_size=n)),
                    dtype=np.int64)
    kn = np.array(draw(st.lists(st.booleans(), min_size=n, max_size=n)),
                  dtype=np.uint8)
    vn = np.array(draw(st.lists(st.booleans(), min_size=n, max_size=n)),
                  dtype=np.uint8)
    return keys, vals, kn, vn


@settings(max_examples=80, deadline=None, derandomize=True)
@given(dec_input())
def test_dec_agg_matches_bigint_bruteforce(inp):
    """Exact int128 decimal agg vs Python's unbounded ints — including
    INT64_MIN/MAX values whose sums leave the int64 range."""
    keys, vals, kn, vn = inp
    got = ora.agg_i64_dec(keys, vals, key_null=kn, val_null=vn)
    groups = {}
    for i in range(len(keys)):
        a = (bool(kn[i]), 0 if kn[i] else int(keys[i]))
        g = groups.setdefault(a, [0, 0, None])
        g[0] += 1
        if not vn[i]:
            g[1] += 1
            g[2] = (g[2] or 0) + int(vals[i])
    assert len(got) == len(groups)
    for o in got:
        key = (bool(o.key_isnull), 0 if o.key_isnull else int(o.key))
        cs, cv, sm = groups[key]
        assert o.count_star == cs and o.count_v == cv
        if sm is None:
            assert o.sum_isnull
        else:
            assert o.sum128 == sm


@st.composite
def joinext_input(draw):
    nb = draw(st.integers(min_value=0, max_value=40))
    npr = draw(st.integers(min_value=0, max_value=60))
    bk = np.array(draw(st.lists(wild_key, min_size=nb, max_size=nb)),
                  dtype=np.int64)
    pk = np.array(draw(st.lists(wild_key, min_size=npr, max_size=npr)),
                  dtype=np.int64)
    bn = np.array(draw(st.lists(st.booleans(), min_size=nb, max_size=nb)),
                  dtype=np.uint8)
    pn = np.array(draw(st.lists(st.booleans(), min_size=npr, max_size=npr)),
                  dtype=np.uint8)
    jt = draw(st.integers(min_value=0, max_value=5))
    return bk, pk, bn, pn, jt


@settings(max_examples=80, deadline=None, derandomize=True)
@given(joinext_input())
def test_join_ext_matches_bruteforce(inp):
    """All six join types vs brute force under hypothesis — INT64_MIN
    keys, NULL masks, empty sides, duplicates."""
    bk, pk, bn, pn, jt = inp
    bi, pi = ora.join_ext(bk, pk, jt, bnull=bn, pnull=pn)
    got = sorted(zip(bi.tolist(), pi.tolist()))
    pairs, matched = [], set()
    for p in range(len(pk)):
        if pn[p]:
            if jt in (1, 3, 5):
                pairs.append((-1, p))
            continue
        ms = [b for b in range(len(bk)) if not bn[b] and bk[b] == pk[p]]
        matched.update(ms)
        if jt in (0, 1, 4, 5):
            pairs.extend((b, p) for b in ms)
        if jt == 2 and ms:
            pairs.append((-1, p))
        if jt in (1, 3, 5) and not ms:
            pairs.append((-1, p))
    if jt in (4, 5):
        for b in range(len(bk)):
            if bn[b] or b not in matched:
                pairs.append((b, -1))
    assert got == sorted(pairs)
