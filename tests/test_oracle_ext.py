"""Oracle extended-join (inner/left/semi/anti/right/full) and two-key
aggregate vs brute force. The brute force is
an independent Python restatement of the SQL semantics; the oracle restates
the reference's executor (nodeHashjoin.c FSM / execGrouping.c) — agreeing
on adversarial inputs pins both."""
import numpy as np
import pytest

from oracle import oracle_py as ora

JT = {"inner": 0, "left": 1, "semi": 2, "anti": 3, "right": 4, "full": 5}


def brute_join(bk, pk, jt, bn=None, pn=None, bk2=None, pk2=None,
               bn2=None, pn2=None):
    has2 = bk2 is not None

    def bnull(i):
        return (bn is not None and bn[i]) or \
               (has2 and bn2 is not None and bn2[i])

    def pnull(i):
        return (pn is not None and pn[i]) or \
               (has2 and pn2 is not None and pn2[i])

    pairs = []
    matched = set()
    for p in range(len(pk)):
        if pnull(p):
            if jt in (1, 3, 5):          # left, anti, full
                pairs.append((-1, p))
            continue
        ms = [b for b in range(len(bk))
              if not bnull(b) and bk[b] == pk[p] and
              (not has2 or bk2[b] == pk2[p])]
        matched.update(ms)
        if jt in (0, 1, 4, 5):
            pairs.extend((b, p) for b in ms)
        if jt == 2 and ms:
            pairs.append((-1, p))
        if jt in (1, 3, 5) and not ms:
            pairs.append((-1, p))
    if jt in (4, 5):
        for b in range(len(bk)):
            if bnull(b) or b not in matched:
                pairs.append((b, -1))
    return sorted(pairs)


def check(bk, pk, jt, **kw):
    bi, pi = ora.join_ext(bk, pk, jt, **{
        {"bn": "bnull", "pn": "pnull", "bk2": "bkeys2", "pk2": "pkeys2",
         "bn2": "bnull2", "pn2": "pnull2"}[k]: v for k, v in kw.items()})
    got = sorted(zip(bi.tolist(), pi.tolist()))
    exp = brute_join(bk, pk, jt, **kw)
    assert got == exp, (jt, got[:10], exp[:10], len(got), len(exp))


@pytest.mark.parametrize("jt", list(JT.values()))
def test_join_ext_random(jt):
    rng = np.random.default_rng(jt + 1)
    for trial in range(8):
        nb, npr = rng.integers(0, 200, 2)
        bk = rng.integers(-5, 15, nb)
        pk = rng.integers(-5, 15, npr)
        bn = (rng.random(nb) < 0.15).astype(np.uint8)
        pn = (rng.random(npr) < 0.15).astype(np.uint8)
        check(bk, pk, jt, bn=bn, pn=pn)


@pytest.mark.parametrize("jt", list(JT.values()))
def test_join_ext_two_key(jt):
    rng = np.random.default_rng(100 + jt)
    for trial in range(8):
        nb, npr = rng.integers(0, 150, 2)
        bk = rng.integers(0, 6, nb)
        bk2 = rng.integers(0, 6, nb)
        pk = rng.integers(0, 6, npr)
        pk2 = rng.integers(0, 6, npr)
        bn2 = (rng.random(nb) < 0.1).astype(np.uint8)
        pn2 = (rng.random(npr) < 0.1).astype(np.uint8)
        check(bk, pk, jt, bk2=bk2, pk2=pk2, bn2=bn2, pn2=pn2)


@pytest.mark.parametrize("jt", list(JT.values()))
def test_join_ext_edges(jt):
    e = np.empty(0, dtype=np.int64)
    k = np.array([1, 2, 2, 3], dtype=np.int64)
    # empty build side: left/anti emit every probe row, right/full emit none
    check(e, k, jt)
    # empty probe side: right/full emit every build row
    check(k, e, jt)
    # both empty
    check(e, e, jt)
    # all-NULL sides
    nn = np.ones(4, dtype=np.uint8)
    check(k, k, jt, bn=nn)
    check(k, k, jt, pn=nn)
    check(k, k, jt, bn=nn, pn=nn)
    # INT64_MIN / INT64_MAX keys join fine
    ext = np.array([np.iinfo(np.int64).min, np.iinfo(np.int64).max, 0],
                   dtype=np.int64)
    check(ext, ext, jt)


def test_join_ext_semi_dup_build():
    """Semi emits ONCE per probe row regardless of build duplicates
    (JOIN_SEMI advances after the first match, nodeHashjoin.c:572)."""
    bk = np.array([7, 7, 7, 7], dtype=np.int64)
    pk = np.array([7, 7, 8], dtype=np.int64)
    bi, pi = ora.join_ext(bk, pk, JT["semi"])
    assert sorted(pi.tolist()) == [0, 1]
    assert all(b == -1 for b in bi.tolist())


def test_join_ext_invalid_type():
    k = np.array([1], dtype=np.int64)
    with pytest.raises(AssertionError):
        ora.join_ext(k, k, 9)


# ---------------- two-key aggregate ----------------

def brute_agg2(k1, k2, vals, n1=None, n2=None, vn=None):
    groups = {}
    for i in range(len(k1)):
        a = (bool(n1[i]) if n1 is not None else False,
             0 if (n1 is not None and n1[i]) else int(k1[i]),
             bool(n2[i]) if n2 is not None else False,
             0 if (n2 is not None and n2[i]) else int(k2[i]))
        g = groups.setdefault(a, {"cs": 0, "cv": 0, "sum": None})
        g["cs"] += 1
        if vn is None or not vn[i]:
            g["cv"] += 1
            g["sum"] = vals[i] if g["sum"] is None else g["sum"] + vals[i]
    return groups


def test_agg2_random():
    rng = np.random.default_rng(7)
    for trial in range(10):
        n = int(rng.integers(0, 400))
        k1 = rng.integers(-3, 4, n)
        k2 = rng.integers(-3, 4, n)
        v = rng.standard_normal(n)
        n1 = (rng.random(n) < 0.2).astype(np.uint8)
        n2 = (rng.random(n) < 0.2).astype(np.uint8)
        vn = (rng.random(n) < 0.2).astype(np.uint8)
        got = ora.agg_i64x2(k1, k2, v, k1null=n1, k2null=n2, val_null=vn)
        exp = brute_agg2(k1, k2, v, n1, n2, vn)
        assert len(got) == len(exp)
        for g in got:
            key = (bool(g.key1_isnull), int(g.key1) if not g.key1_isnull else 0,
                   bool(g.key2_isnull), int(g.key2) if not g.key2_isnull else 0)
            e = exp[key]
            assert g.count_star == e["cs"]
            assert g.count_v == e["cv"]
            if e["sum"] is None:
                assert g.sum_isnull
            else:
                assert not g.sum_isnull
                assert abs(g.sum_v - e["sum"]) <= 1e-9 * max(abs(e["sum"]), 1)


def test_agg2_empty_and_sorted():
    e = np.empty(0, dtype=np.int64)
    assert ora.agg_i64x2(e, e, np.empty(0)) == []
    k1 = np.array([2, 1, 2, 1], dtype=np.int64)
    k2 = np.array([0, 1, 1, 0], dtype=np.int64)
    v = np.ones(4)
    got = ora.agg_i64x2(k1, k2, v)
    assert [(g.key1, g.key2) for g in got] == [(1, 0), (1, 1), (2, 0), (2, 1)]


def test_agg2_grid_golden():
    """The xc_FQS_join.out:89-105 grid self-join shape on two REAL key
    columns — the multi-key case the round-1 golden test handled via
    host-side key packing: count over a 2-key group-by of a
    5x2 grid crossed with itself via join_ext inner on both keys."""
    # grid: (a, b) for a in 0..4, b in 0..1
    a = np.repeat(np.arange(5), 2).astype(np.int64)
    b = np.tile(np.arange(2), 5).astype(np.int64)
    bi, pi = ora.join_ext(a, a, JT["inner"], bkeys2=b, pkeys2=b)
    assert len(bi) == 10            # self-join on both keys: one pair each
    assert np.array_equal(np.sort(bi), np.arange(10))
    assert np.array_equal(bi, pi)   # identical rows pair with themselves
    g = ora.agg_i64x2(a, b, np.ones(10))
    assert len(g) == 10
    assert all(x.count_star == 1 for x in g)


# ---------------- exact decimal (int128) aggregate ----------------

def brute_dec(keys, vals, kn=None, vn=None):
    groups = {}
    for i in range(len(keys)):
        a = (bool(kn[i]) if kn is not None else False,
             0 if (kn is not None and kn[i]) else int(keys[i]))
        g = groups.setdefault(a, {"cs": 0, "cv": 0, "sum": None})
        g["cs"] += 1
        if vn is None or not vn[i]:
            g["cv"] += 1
            g["sum"] = (g["sum"] or 0) + int(vals[i])   # exact Python int
    return groups


def test_agg_dec_exact_random():
    rng = np.random.default_rng(31)
    for trial in range(6):
        n = int(rng.integers(1, 500))
        k = rng.integers(-3, 4, n)
        v = rng.integers(np.iinfo(np.int64).min // 2,
                         np.iinfo(np.int64).max // 2, n)
        kn = (rng.random(n) < 0.2).astype(np.uint8)
        vn = (rng.random(n) < 0.2).astype(np.uint8)
        got = ora.agg_i64_dec(k, v, key_null=kn, val_null=vn)
        exp = brute_dec(k, v, kn, vn)
        assert len(got) == len(exp)
        for g in got:
            key = (bool(g.key_isnull), 0 if g.key_isnull else int(g.key))
            e = exp[key]
            assert g.count_star == e["cs"] and g.count_v == e["cv"]
            if e["sum"] is None:
                assert g.sum_isnull
            else:
                assert g.sum128 == e["sum"]            # EXACT int128


def test_agg_dec_beyond_int64():
    """Sums that overflow int64 stay exact in the int128 state — the
    numeric-promotion property of int8_sum (numeric.c:6206)."""
    n = 10000
    k = np.zeros(n, dtype=np.int64)
    v = np.full(n, np.iinfo(np.int64).max, dtype=np.int64)
    g = ora.agg_i64_dec(k, v)
    assert len(g) == 1
    assert g[0].sum128 == n * (2**63 - 1)              # > int64 range
    vneg = np.full(n, np.iinfo(np.int64).min, dtype=np.int64)
    g = ora.agg_i64_dec(k, vneg)
    assert g[0].sum128 == n * (-2**63)


def test_agg_dec_empty_allnull():
    e = np.empty(0, dtype=np.int64)
    assert ora.agg_i64_dec(e, e) == []
    k = np.zeros(4, dtype=np.int64)
    v = np.arange(4, dtype=np.int64)
    g = ora.agg_i64_dec(k, v, val_null=np.ones(4, dtype=np.uint8))
    assert len(g) == 1 and g[0].sum_isnull and g[0].count_star == 4 \
        and g[0].count_v == 0


# ---------------- N-key (1..8) group-by / join ----------------

def brute_aggn(key_cols, vals, null_cols=None, vn=None):
    groups = {}
    nk = len(key_cols)
    for i in range(len(vals)):
        ident = []
        for c in range(nk):
            isnull = null_cols is not None and null_cols[c] is not None \
                and null_cols[c][i]
            ident.append((bool(isnull), 0 if isnull else int(key_cols[c][i])))
        a = tuple(ident)
        g = groups.setdefault(a, {"cs": 0, "cv": 0, "sum": None, "first": i})
        g["cs"] += 1
        if vn is None or not vn[i]:
            g["cv"] += 1
            g["sum"] = (g["sum"] or 0.0) + float(vals[i])
    return groups


def _ident_of(key_cols, null_cols, i):
    out = []
    for c in range(len(key_cols)):
        isnull = null_cols is not None and null_cols[c] is not None \
            and null_cols[c][i]
        out.append((bool(isnull), 0 if isnull else int(key_cols[c][i])))
    return tuple(out)


@pytest.mark.parametrize("nk", [1, 3, 4, 8])
def test_aggn_random(nk):
    rng = np.random.default_rng(50 + nk)
    n = 300
    keys = [rng.integers(-2, 3, n) for _ in range(nk)]
    nulls = [(rng.random(n) < 0.15).astype(np.uint8) if c % 2 == 0 else None
             for c in range(nk)]
    v = rng.standard_normal(n)
    vn = (rng.random(n) < 0.2).astype(np.uint8)
    got = ora.agg_i64n(keys, v, null_cols=nulls, val_null=vn)
    exp = brute_aggn(keys, v, nulls, vn)
    assert len(got) == len(exp)
    for g in got:
        ident = _ident_of(keys, nulls, g.row_idx)
        e = exp[ident]
        assert g.count_star == e["cs"] and g.count_v == e["cv"]
        # the defining row must be a member of its own group
        assert _ident_of(keys, nulls, e["first"]) == ident
        if e["sum"] is None:
            assert g.sum_isnull
        else:
            assert abs(g.sum_v - e["sum"]) <= 1e-9 * max(abs(e["sum"]), 1)


@pytest.mark.parametrize("jt", list(JT.values()))
def test_joinn_random_3key(jt):
    rng = np.random.default_rng(70 + jt)
    nb, npr = 120, 260
    bkeys = [rng.integers(0, 4, nb) for _ in range(3)]
    pkeys = [rng.integers(0, 4, npr) for _ in range(3)]
    bnulls = [None, (rng.random(nb) < 0.1).astype(np.uint8), None]
    pnulls = [(rng.random(npr) < 0.1).astype(np.uint8), None, None]
    bi, pi = ora.join_i64n(bkeys, pkeys, jt, bnull_cols=bnulls,
                           pnull_cols=pnulls)
    got = sorted(zip(bi.tolist(), pi.tolist()))

    def bnull(i):
        return any(nc is not None and nc[i] for nc in bnulls)

    def pnull(i):
        return any(nc is not None and nc[i] for nc in pnulls)

    pairs, matched = [], set()
    for p in range(npr):
        if pnull(p):
            if jt in (1, 3, 5):
                pairs.append((-1, p))
            continue
        ms = [b for b in range(nb) if not bnull(b) and
              all(bkeys[c][b] == pkeys[c][p] for c in range(3))]
        matched.update(ms)
        if jt in (0, 1, 4, 5):
            pairs.extend((b, p) for b in ms)
        if jt == 2 and ms:
            pairs.append((-1, p))
        if jt in (1, 3, 5) and not ms:
            pairs.append((-1, p))
    if jt in (4, 5):
        for b in range(nb):
            if bnull(b) or b not in matched:
                pairs.append((b, -1))
    assert got == sorted(pairs)


def test_aggn_matches_single_key_path():
    """nkeys=1 N-key agg ≡ ora_agg_i64 on the same data."""
    rng = np.random.default_rng(90)
    n = 500
    k = rng.integers(-5, 5, n)
    v = rng.standard_normal(n)
    kn = (rng.random(n) < 0.1).astype(np.uint8)
    gn = ora.agg_i64n([k], v, null_cols=[kn])
    g1 = ora.agg_i64(k, v, key_null=kn)
    assert len(gn) == len(g1)
    m = {}
    for g in gn:
        isnull = bool(kn[g.row_idx])
        m[(isnull, 0 if isnull else int(k[g.row_idx]))] = g
    for o in g1:
        g = m[(bool(o.key_isnull), 0 if o.key_isnull else int(o.key))]
        assert g.count_star == o.count_star and g.count_v == o.count_v


# ---------------- cross-type join invariants ----------------

def _pairs(bk, pk, jt, **kw):
    bi, pi = ora.join_ext(bk, pk, jt, **kw)
    return sorted(zip(bi.tolist(), pi.tolist()))


def test_join_type_algebra():
    """Relational identities across the six types (random inputs):
    left = inner + probe-fills; right = inner + build-fills;
    full = inner + probe-fills + build-fills;
    semi and anti partition the non-NULL probe rows;
    anti fills + semi fills = left fills' NULL-extended probe set."""
    rng = np.random.default_rng(99)
    for trial in range(6):
        nb, npr = rng.integers(0, 120, 2)
        bk = rng.integers(-4, 10, nb)
        pk = rng.integers(-4, 10, npr)
        bn = (rng.random(nb) < 0.15).astype(np.uint8)
        pn = (rng.random(npr) < 0.15).astype(np.uint8)
        kw = dict(bnull=bn, pnull=pn)
        inner = _pairs(bk, pk, JT["inner"], **kw)
        left = _pairs(bk, pk, JT["left"], **kw)
        right = _pairs(bk, pk, JT["right"], **kw)
        full = _pairs(bk, pk, JT["full"], **kw)
        semi = _pairs(bk, pk, JT["semi"], **kw)
        anti = _pairs(bk, pk, JT["anti"], **kw)

        probe_fills = [p for p in left if p[0] == -1]
        build_fills = [p for p in right if p[1] == -1]
        assert sorted(inner + probe_fills) == left
        assert sorted(inner + build_fills) == right
        assert sorted(inner + probe_fills + build_fills) == full
        # semi ∪ anti = every probe row exactly once... except NULL-key
        # probe rows, which appear only on the anti side
        semi_rows = {p for _, p in semi}
        anti_rows = {p for _, p in anti}
        assert not (semi_rows & anti_rows)
        assert semi_rows | anti_rows == set(range(npr))
        # matched probe rows in semi == distinct probe ids in inner
        assert semi_rows == {p for _, p in inner}


# ---------------- decimal AVG finalizer ----------------

def test_dec_avg_half_up():
    """Exact round-half-up-away-from-zero division (numeric.c round_var
    HALF_ADJUST_ROUND) vs Fraction, incl. negatives and exact halves."""
    from fractions import Fraction
    from opentenbase_amd.executor import dec_avg
    import pytest as _pt

    cases = [(7, 2), (-7, 2), (5, 2), (-5, 2), (1, 3), (-1, 3),
             (10**30 + 5, 10), (-(10**30 + 5), 10), (0, 5),
             (2**100 + 1, 7), (-(2**100) - 1, 7)]
    for s, c in cases:
        got = dec_avg(s, c)
        f = Fraction(s, c)
        # round half away from zero
        import math
        exp = int(math.floor(f + Fraction(1, 2))) if f >= 0 \
            else -int(math.floor(-f + Fraction(1, 2)))
        assert got == exp, (s, c, got, exp)
    # extra scale: cents -> hundredth-cents
    assert dec_avg(1, 3, extra_scale=2) == 33
    assert dec_avg(2, 3, extra_scale=2) == 67      # .666.. -> 67
    assert dec_avg(1, 2, extra_scale=2) == 50
    assert dec_avg(-1, 2, extra_scale=2) == -50
    with _pt.raises(Exception):
        dec_avg(1, 0)


def test_dec_avg_roundtrip_with_agg():
    """AVG through the oracle dec state: exact vs Fraction on the real
    aggregate output."""
    from fractions import Fraction
    from opentenbase_amd.executor import dec_avg
    rng = np.random.default_rng(123)
    k = rng.integers(0, 5, 300)
    v = rng.integers(-10**15, 10**15, 300)
    for g in ora.agg_i64_dec(k, v):
        if g.sum_isnull:
            continue
        got = dec_avg(g.sum128, g.count_v)
        f = Fraction(g.sum128, g.count_v)
        half = Fraction(1, 2)
        import math
        exp = int(math.floor(f + half)) if f >= 0 \
            else -int(math.floor(-f + half))
        assert got == exp
