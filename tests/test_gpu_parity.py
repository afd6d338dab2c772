"""GPU parity tests: the HIP path vs the CPU oracle on identical inputs
(bit-identical tables via the shared counter-based generator), through the
C-ABI. The parity bar: COUNT and group keys bit-exact;
SUM/AVG(float8) within 1e-6 relative."""
import json
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")

REL = 1e-6
GOLDEN = os.path.join(os.path.dirname(__file__), "golden")


@pytest.fixture(scope="module")
def ex():
    from opentenbase_amd import executor
    executor.init_device(0)
    return executor


@pytest.fixture(scope="module")
def ora():
    from oracle import oracle_py
    return oracle_py


def drain(node):
    node.BeginCustomScan()
    rows = []
    while True:
        r = node.ExecCustomScan()
        if r is None:
            break
        rows.append(r)
    node.EndCustomScan()
    return rows


def approx(a, b, rel=REL):
    return abs(a - b) <= rel * max(abs(a), abs(b), 1e-300)


# ---------------- datagen bit-identity ----------------

def test_datagen_bit_identical(ex, ora):
    n = 80000
    li = ex.GpuLineitem.generate(n)
    t = ora.gen_tables(n)["lineitem"]
    for col in ["l_orderkey", "l_quantity", "l_extendedprice", "l_discount",
                "l_tax", "l_returnflag", "l_linestatus", "l_shipdate"]:
        gpu = li.t[col].cpu().numpy()
        cpu = t[col]
        assert gpu.dtype.itemsize == cpu.dtype.itemsize
        assert np.array_equal(gpu.view(np.uint8), cpu.view(np.uint8)), col


def test_datagen_sharded_bit_identical(ex, ora):
    n = 80000
    for rank in range(4):
        li = ex.GpuLineitem.generate(n, rank=rank, nranks=4)
        t = ora.gen_tables(n, rank=rank, nranks=4)["lineitem"]
        assert np.array_equal(li.t["l_shipdate"].cpu().numpy(), t["l_shipdate"])
        assert np.array_equal(li.t["l_extendedprice"].cpu().numpy().view(np.uint8),
                              t["l_extendedprice"].view(np.uint8))
        od = ex.GpuOrders.generate(n // 4, n // 40, rank=rank, nranks=4)
        ot = ora.gen_tables(n, rank=rank, nranks=4, need=("orders",))["orders"]
        assert np.array_equal(od.t["o_custkey"].cpu().numpy(), ot["o_custkey"])


# ---------------- config 2: scan + count ----------------

def test_scan_count_parity(ex, ora):
    n = 600000
    li = ex.GpuLineitem.generate(n, with_orderkey=False)
    t = ora.gen_tables(n)
    for cutoff in (2436, 1, 0, 2526, 3000, -5):
        rows = drain(ex.GpuSeqScanCount(li, cutoff=cutoff))
        expect = ora.scan_count(t["lineitem"]["l_shipdate"], cutoff)
        assert rows[0][0] == expect, cutoff


def test_scan_count_odd_n(ex, ora):
    # exercise the vector-tail path (n not divisible by 4)
    n = 600000
    li = ex.GpuLineitem.generate(n, with_orderkey=False)
    t = ora.gen_tables(n)
    sd = li.t["l_shipdate"][:123457]
    import ctypes as C
    out = torch.zeros(1, dtype=torch.int64, device="cuda")
    from opentenbase_amd._lib import call
    call("otbx_scan_count", C.c_void_p(sd.data_ptr()), C.c_int64(123457),
         C.c_int32(2436),
         C.c_void_p(out.data_ptr()),
         C.c_void_p(torch.cuda.current_stream().cuda_stream))
    exp = int((t["lineitem"]["l_shipdate"][:123457] <= 2436).sum())
    assert int(out.cpu().item()) == exp


# ---------------- Q1 fragment ----------------

@pytest.mark.parametrize("n,cutoff", [(400000, 2436), (400000, 1200),
                                      (400002, 2436), (399999, 2436),
                                      (399998, -1), (400000, 9999)])
def test_q1_parity(ex, ora, n, cutoff):
    li = ex.GpuLineitem.generate(n, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li, cutoff=cutoff)
    rows = ex.q1_finalize(drain(node))
    t = ora.gen_tables(n)
    og = ora.q1_finalize(ora.q1_partial(t, cutoff=cutoff))
    assert len(rows) == len(og)
    for g, o in zip(rows, og):
        assert g["l_returnflag"] == chr(o.returnflag)
        assert g["l_linestatus"] == chr(o.linestatus)
        assert g["count_order"] == o.count_order
        for fld, ov in [("sum_qty", o.sum_qty), ("sum_base_price", o.sum_base_price),
                        ("sum_disc_price", o.sum_disc_price),
                        ("sum_charge", o.sum_charge), ("avg_qty", o.avg_qty),
                        ("avg_price", o.avg_price), ("avg_disc", o.avg_disc)]:
            assert approx(g[fld], ov), (fld, g[fld], ov)


def test_q1_two_phase_merge_parity(ex, ora):
    """4 shards on one GPU, partial states combined as fragment.py does
    (elementwise dense combine) ≡ oracle single-node run."""
    n = 400000
    tot_s = torch.zeros((6, 5), dtype=torch.float64, device="cuda")
    tot_c = torch.zeros(6, dtype=torch.int64, device="cuda")
    for rank in range(4):
        li = ex.GpuLineitem.generate(n, rank=rank, nranks=4, with_orderkey=False)
        node = ex.GpuQ1PartialAgg(li)
        drain(node)
        s, c = node.partial_state_tensors()
        tot_s += s
        tot_c += c
    rows = ex.q1_finalize(ex.q1_rows_from_state(tot_s, tot_c))
    og = ora.q1_finalize(ora.q1_partial(ora.gen_tables(n)))
    assert [r["count_order"] for r in rows] == [o.count_order for o in og]
    for g, o in zip(rows, og):
        assert approx(g["sum_charge"], o.sum_charge)
        assert approx(g["avg_disc"], o.avg_disc)


# ---------------- Q3 fragment ----------------

def q3_oracle(ora, n, rank=0, nranks=1, replicate_customer=False):
    t = ora.gen_tables(n, rank=rank, nranks=nranks,
                       need=("lineitem", "orders", "customer"))
    if replicate_customer:
        t["customer"] = ora.gen_tables(n, need=("customer",))["customer"]
    return ora.q3_partial(t)


@pytest.mark.parametrize("n", [400000, 400004, 399998])
def test_q3_parity(ex, ora, n):
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    node = ex.GpuQ3Fragment(cu, od, li)
    top = drain(node)
    og = q3_oracle(ora, n)
    # full group-set parity
    got = {int(r["l_orderkey"]): float(r["revenue"]) for r in node.fetch_groups()}
    exp = {int(k): float(v) for k, v in zip(og["l_orderkey"], og["revenue"])}
    assert got.keys() == exp.keys()
    for k in exp:
        assert approx(got[k], exp[k]), k
    # dates/prios bit-exact
    gd = {int(r["l_orderkey"]): (int(r["o_orderdate"]), int(r["o_shippriority"]))
          for r in node.fetch_groups()}
    ed = {int(k): (int(d), int(p)) for k, d, p in
          zip(og["l_orderkey"], og["o_orderdate"], og["o_shippriority"])}
    assert gd == ed
    # top-k parity (revenue ties broken identically by orderkey)
    ot = ora.q3_topk(og, 10)
    assert [t_[0] for t_ in top] == [int(x) for x in ot["l_orderkey"]]


def test_q3_broadcast_path_parity(ex, ora):
    """Replicated customer build side (the multi-GPU path) on 2 shards ==
    oracle with replicated customer; union of shard groups == full run."""
    from opentenbase_amd import fragment
    n = 400000
    all_groups = []
    for rank in range(2):
        li = ex.GpuLineitem.generate(n, rank=rank, nranks=2)
        od = ex.GpuOrders.generate(n // 4, n // 40, rank=rank, nranks=2)
        cu = ex.GpuCustomer.generate(n // 40, rank=rank, nranks=2)
        # local filter + (degenerate world=1) broadcast
        import ctypes as C
        from opentenbase_amd._lib import call
        keys = torch.empty(cu.n, dtype=torch.int64, device="cuda")
        nk = torch.zeros(1, dtype=torch.int64, device="cuda")
        call("otbx_filter_customer", C.byref(cu.cstruct), C.c_uint8(0),
             C.c_void_p(keys.data_ptr()), C.c_void_p(nk.data_ptr()),
             C.c_void_p(torch.cuda.current_stream().cuda_stream))
        local_keys = keys[: int(nk.cpu().item())]
        all_groups.append((rank, local_keys))
    # emulate the all-gather: concatenate both ranks' filtered keys
    bcast = torch.cat([k for _, k in all_groups])
    union = []
    for rank in range(2):
        li = ex.GpuLineitem.generate(n, rank=rank, nranks=2)
        od = ex.GpuOrders.generate(n // 4, n // 40, rank=rank, nranks=2)
        cu = ex.GpuCustomer.generate(n // 40, rank=rank, nranks=2)
        node = ex.GpuQ3Fragment(cu, od, li, cust_keys=bcast)
        drain(node)
        union.append(node.fetch_groups())
    union = np.concatenate(union)
    og = q3_oracle(ora, n)
    assert set(union["l_orderkey"].tolist()) == set(og["l_orderkey"].tolist())
    exp = dict(zip(og["l_orderkey"], og["revenue"]))
    for k, v in zip(union["l_orderkey"], union["revenue"]):
        assert approx(float(v), float(exp[k]))


# ---------------- Q9-mix fragment (BASELINE config 5) ----------------

def test_q9_datagen_bit_identical(ex, ora):
    n = 120000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    pt = ex.GpuPart.generate(n // 30)
    t = ora.gen_tables(n, need=("lineitem", "part"))
    assert np.array_equal(li.t["l_partkey"].cpu().numpy(),
                          t["lineitem"]["l_partkey"])
    assert np.array_equal(pt.t["p_partkey"].cpu().numpy(),
                          t["part"]["p_partkey"])
    assert np.array_equal(pt.t["p_type"].cpu().numpy(), t["part"]["p_type"])


@pytest.mark.parametrize("n", [400000, 400004, 399998])
def test_q9_parity(ex, ora, n):
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(max(n // 30, 1))
    node = ex.GpuQ9Fragment(pt, od, li)
    rows = drain(node)
    t = ora.gen_tables(n, need=("lineitem", "orders", "part"))
    exp = ora.q9_partial(t)
    assert [r["o_year"] - 1992 for r in rows] == [g.year for g in exp]
    for r, g in zip(rows, exp):
        assert r["count_rows"] == g.count_rows          # bit-exact
        assert approx(r["sum_revenue"], g.revenue)      # float8 ≤ 1e-6 rel
    assert node.kernel_ms is not None and node.kernel_ms > 0


def test_q9_typemod_variants(ex, ora):
    """Different part-filter selectivities (the p_type % m == v predicate
    family) agree with the oracle, including empty-result cases."""
    n = 200000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)
    t = ora.gen_tables(n, need=("lineitem", "orders", "part"))
    for m, v in [(2, 1), (150, 149), (151, 150), (1, 0)]:
        rows = drain(ex.GpuQ9Fragment(pt, od, li, typemod=m, typeval=v))
        exp = ora.q9_partial(t, typemod=m, typeval=v)
        assert len(rows) == len(exp), (m, v)
        for r, g in zip(rows, exp):
            assert r["o_year"] - 1992 == g.year
            assert r["count_rows"] == g.count_rows
            assert approx(r["sum_revenue"], g.revenue)


def test_q9_bitmap_slice_multipass(ex, ora, monkeypatch):
    """Grace-style part-bitmap slice passes (production: bitmap > the
    ~3 MB L2 budget at SF300+; forced tiny here) must match the oracle."""
    monkeypatch.setenv("OTBX_Q9_BITMAP_BITS", "4096")  # 400k parts -> 4 passes
    n = 400000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)
    rows = drain(ex.GpuQ9Fragment(pt, od, li))
    exp = ora.q9_partial(ora.gen_tables(n, need=("lineitem", "orders",
                                                 "part")))
    assert [r["o_year"] - 1992 for r in rows] == [g.year for g in exp]
    for r, g in zip(rows, exp):
        assert r["count_rows"] == g.count_rows
        assert approx(r["sum_revenue"], g.revenue)


def test_q9_sharded_merge_parity(ex, ora):
    """2-shard dense-state combine == full run (counts bit-exact, sums
    ≤1e-9 relative): the RemoteSubplan merge payload for Q9."""
    n = 400000
    full = ex.GpuQ9Fragment(ex.GpuPart.generate(n // 30),
                            ex.GpuOrders.generate(n // 4, n // 40),
                            ex.GpuLineitem.generate(n, with_partkey=True))
    drain(full)
    fs, fc = full.partial_state_tensors()
    tot_s = torch.zeros_like(fs)
    tot_c = torch.zeros_like(fc)
    for r in range(2):
        node = ex.GpuQ9Fragment(
            ex.GpuPart.generate(n // 30),
            ex.GpuOrders.generate(n // 4, n // 40, rank=r, nranks=2),
            ex.GpuLineitem.generate(n, rank=r, nranks=2, with_partkey=True),
            nranks=2)
        drain(node)
        s, c = node.partial_state_tensors()
        tot_s += s
        tot_c += c
    assert torch.equal(tot_c, fc)
    rel = ((tot_s - fs).abs() / fs.abs().clamp(min=1e-300)).max().item()
    assert rel < 1e-9


# ---------------- composable operators: NULL semantics on GPU -------------

def _agg(ex, keys, vals, kn=None, vn=None):
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashAgg(
        dev(keys, torch.int64), dev(vals, torch.float64),
        None if kn is None else dev(kn, torch.uint8),
        None if vn is None else dev(vn, torch.uint8))
    return drain(node)


def test_agg_null_semantics_gpu(ex, ora):
    rng = np.random.default_rng(7)
    n = 100000
    keys = rng.integers(0, 500, n)
    vals = rng.random(n) * 100
    kn = (rng.random(n) < 0.1).astype(np.uint8)
    vn = (rng.random(n) < 0.2).astype(np.uint8)
    got = _agg(ex, keys, vals, kn, vn)
    exp = ora.agg_i64(keys, vals, key_null=kn, val_null=vn)
    assert len(got) == len(exp)
    for g, e in zip(got, exp):
        assert bool(g["key_isnull"]) == bool(e.key_isnull)
        if not e.key_isnull:
            assert g["key"] == e.key
        assert g["count_star"] == e.count_star
        assert g["count_v"] == e.count_v
        assert bool(g["sum_isnull"]) == bool(e.sum_isnull)
        if not e.sum_isnull:
            assert approx(float(g["sum_v"]), e.sum_v)


def test_agg_empty_and_all_null(ex):
    assert _agg(ex, np.empty(0, np.int64), np.empty(0, np.float64)) == []
    got = _agg(ex, np.zeros(5, np.int64), np.arange(5, dtype=np.float64),
               vn=np.ones(5, np.uint8))
    assert len(got) == 1
    assert got[0]["sum_isnull"] == 1 and got[0]["count_v"] == 0 \
        and got[0]["count_star"] == 5


def test_join_parity_gpu(ex, ora):
    rng = np.random.default_rng(11)
    nb, npr = 40000, 120000
    bk = rng.integers(0, 30000, nb)
    pk = rng.integers(0, 30000, npr)
    bn = (rng.random(nb) < 0.05).astype(np.uint8)
    pn = (rng.random(npr) < 0.05).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                          dev(bn, torch.uint8), dev(pn, torch.uint8),
                          cap_pairs=4 * npr)
    pairs = drain(node)
    obi, opi = ora.join_i64(bk, pk, bnull=bn, pnull=pn)
    assert len(pairs) == len(obi)
    got = sorted(pairs)
    exp = sorted(zip(obi.tolist(), opi.tolist()))
    assert got == exp


def test_agg_small_table_abort_path(ex):
    """Low-cardinality estimate with a rare-key tail the sample misses: the
    estimator-sized table aborts (bounded probes) and the call redoes at
    full capacity — group count and count conservation must hold."""
    import ctypes as C
    from opentenbase_amd._lib import call, lib
    n = 8_400_000  # ≥ AGGP_THRESHOLD so the estimator runs; stride = n/2^20
    stride = n // (1 << 20)
    keys_np = np.arange(n, dtype=np.int64) + 1_000_000
    keys_np[::stride] = 7  # every sampled position is the hot key
    keys = torch.as_tensor(keys_np, device="cuda")
    vals = torch.ones(n, dtype=torch.float64, device="cuda")
    L = lib()
    wsb = C.c_size_t(0)
    L.otbx_agg_i64_workspace_bytes(C.c_int64(n), C.byref(wsb))
    ws = torch.empty(wsb.value, dtype=torch.uint8, device="cuda")
    out = torch.empty(n * 40, dtype=torch.uint8, device="cuda")
    ng = torch.zeros(1, dtype=torch.int64, device="cuda")
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
    call("otbx_agg_i64", C.c_void_p(keys.data_ptr()), None,
         C.c_void_p(vals.data_ptr()), None, C.c_int64(n),
         C.c_void_p(ws.data_ptr()), C.c_size_t(wsb.value),
         C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream)
    g = int(ng.cpu().item())
    assert g == len(np.unique(keys_np))
    dt = np.dtype([("key", "i8"), ("count_star", "i8"), ("count_v", "i8"),
                   ("sum_v", "f8"), ("key_isnull", "i4"),
                   ("sum_isnull", "i4")])
    groups = out[: g * 40].cpu().numpy().view(dt)
    assert int(groups["count_star"].sum()) == n
    hot = groups[groups["key"] == 7]
    assert len(hot) == 1
    assert int(hot["count_star"][0]) == len(np.arange(0, n, stride))


def test_join_partitioned_parity(ex, ora, monkeypatch):
    """The partitioned join path (build ≥ 8M rows in production; forced via
    OTBX_JOINP_FORCE here) against the oracle — including buckets flagged
    to the global fallback by INT64_MIN build keys (the LDS table's
    reserved sentinel) and NULL keys on both sides."""
    monkeypatch.setenv("OTBX_JOINP_FORCE", "1")
    rng = np.random.default_rng(17)
    nb, npr = 300000, 900000
    bk = rng.integers(0, 200000, nb)
    pk = rng.integers(0, 200000, npr)
    bk[::5000] = np.iinfo(np.int64).min   # sentinel keys -> flagged buckets
    pk[::7000] = np.iinfo(np.int64).min
    bn = (rng.random(nb) < 0.03).astype(np.uint8)
    pn = (rng.random(npr) < 0.03).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                          dev(bn, torch.uint8), dev(pn, torch.uint8),
                          cap_pairs=8 * npr)
    pairs = drain(node)
    obi, opi = ora.join_i64(bk, pk, bnull=bn, pnull=pn)
    assert len(pairs) == len(obi)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))


def test_join_partitioned_large_count(ex):
    """Pair-count parity at a size that takes the partitioned path without
    forcing (build 10M > the 8M threshold): expected count computed from
    the key histograms (sum over keys of cb[k]*cp[k])."""
    g = torch.Generator(device="cuda").manual_seed(5)
    nb, npr = 10_000_000, 20_000_000
    dom = 4_000_000
    bk = torch.randint(0, dom, (nb,), dtype=torch.int64, device="cuda",
                       generator=g)
    pk = torch.randint(0, dom, (npr,), dtype=torch.int64, device="cuda",
                       generator=g)
    cb = torch.bincount(bk, minlength=dom)
    cp = torch.bincount(pk, minlength=dom)
    expected = int((cb * cp).sum().item())
    node = ex.GpuHashJoin(bk, pk, cap_pairs=expected + 64)
    node.BeginCustomScan()
    node._rows = node._run()
    assert len(node._rows) == expected
    # spot-check: every emitted pair joins equal keys
    import numpy as np_
    pairs = np_.array(node._rows[:100000], dtype=np_.int64)
    bkh = bk.cpu().numpy()
    pkh = pk.cpu().numpy()
    assert (bkh[pairs[:, 0]] == pkh[pairs[:, 1]]).all()


# ---------------- golden vectors on the GPU path ----------------

def test_golden_onek_gpu(ex):
    tabs = np.load(os.path.join(GOLDEN, "regress_tables.npz"))
    with open(os.path.join(GOLDEN, "expected.json")) as f:
        expjs = json.load(f)
    four = tabs["onek_four"].astype(np.float64)
    got = _agg(ex, np.zeros(len(four), np.int64), four)
    g = got[0]
    assert g["count_star"] == expjs["onek"]["nrows"]
    assert g["count_v"] == expjs["onek"]["count_four"]
    assert float(g["sum_v"]) == expjs["onek"]["sum_four"]  # ints: exact in f64
    assert float(g["sum_v"]) / g["count_v"] == expjs["onek"]["avg_four"]


def test_golden_tenk_join_gpu(ex):
    tabs = np.load(os.path.join(GOLDEN, "regress_tables.npz"))
    with open(os.path.join(GOLDEN, "expected.json")) as f:
        expjs = json.load(f)
    mask = (tabs["tenk_fivethous"] % 10) < 10
    bk = tabs["tenk_thousand"][mask]
    pk = tabs["tenk_hundred"]
    dev = lambda a: torch.as_tensor(np.ascontiguousarray(a),  # noqa: E731
                                    dtype=torch.int64, device="cuda")
    node = ex.GpuHashJoin(dev(bk), dev(pk), cap_pairs=200000)
    pairs = drain(node)
    assert len(pairs) == expjs["tenk_selfjoin"]["count"]


# ---------------- config-5 skewed distribution keys ----------------

def test_skewed_custkey_datagen_and_q3_parity(ex, ora):
    """Skewed orders (20% hot custkeys get 80% of orders, config 5):
    bit-identical GPU/CPU generation and full Q3 result parity."""
    n = 400000
    od = ex.GpuOrders.generate(n // 4, n // 40, skew=True)
    ot = ora.gen_tables(n, need=("orders",), skew=True)["orders"]
    ck = od.t["o_custkey"].cpu().numpy()
    assert np.array_equal(ck, ot["o_custkey"])
    # skew property: hot 20% of custkeys hold ~80% of orders
    ncust = n // 40
    hot = (ck <= ncust // 5).mean()
    assert 0.75 < hot < 0.85
    li = ex.GpuLineitem.generate(n)
    cu = ex.GpuCustomer.generate(n // 40)
    node = ex.GpuQ3Fragment(cu, od, li)
    drain(node)
    t = ora.gen_tables(n, need=("lineitem", "orders", "customer"), skew=True)
    og = ora.q3_partial(t)
    got = {int(r["l_orderkey"]): float(r["revenue"]) for r in node.fetch_groups()}
    exp = dict(zip(og["l_orderkey"].tolist(), og["revenue"].tolist()))
    assert got.keys() == exp.keys()
    for k in list(exp)[:200]:
        assert approx(got[k], exp[k])


def test_q3_hash_fallback_path(ex, ora):
    """The wide-range hash+bloom fallback (forced via env) produces identical
    results to the dense-direct default path and the oracle."""
    import os
    n = 400000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    os.environ["OTBX_Q3_FORCE_HASH"] = "1"
    try:
        node = ex.GpuQ3Fragment(cu, od, li)
        drain(node)
        hash_groups = {int(k): float(v) for k, v in
                       zip(node.fetch_groups()["l_orderkey"],
                           node.fetch_groups()["revenue"])}
    finally:
        del os.environ["OTBX_Q3_FORCE_HASH"]
    node2 = ex.GpuQ3Fragment(cu, od, li)
    drain(node2)
    direct_groups = {int(k): float(v) for k, v in
                     zip(node2.fetch_groups()["l_orderkey"],
                         node2.fetch_groups()["revenue"])}
    assert hash_groups.keys() == direct_groups.keys()
    for k in hash_groups:
        assert approx(hash_groups[k], direct_groups[k])
    og = q3_oracle(ora, n)
    exp = dict(zip(og["l_orderkey"].tolist(), og["revenue"].tolist()))
    assert direct_groups.keys() == exp.keys()


# ---------------- repartition exchange ----------------

def test_partition_by_key_and_gather(ex):
    """GPU partition groups rows into contiguous per-rank segments
    (owner = key % nranks) with exact counts; native gather applies the
    permutation. Checked against numpy."""
    import ctypes as Ct
    from opentenbase_amd._lib import call
    rng = np.random.default_rng(3)
    n = 1_000_000
    keys_h = rng.integers(1, 10_000_000, n)
    keys = torch.as_tensor(keys_h, dtype=torch.int64, device="cuda")
    vals = torch.as_tensor(keys_h * 2 + 1, dtype=torch.int64, device="cuda")
    nranks = 4
    perm = torch.empty(n, dtype=torch.int64, device="cuda")
    counts = (Ct.c_int64 * nranks)()
    call("otbx_partition_by_key", Ct.c_void_p(keys.data_ptr()), Ct.c_int64(n),
         Ct.c_uint32(nranks), Ct.c_void_p(perm.data_ptr()), counts,
         Ct.c_void_p(torch.cuda.current_stream().cuda_stream))
    counts = list(counts)
    exp_counts = [int((keys_h % nranks == r).sum()) for r in range(nranks)]
    assert counts == exp_counts
    pk = ex.gather(keys, perm).cpu().numpy()
    pv = ex.gather(vals, perm).cpu().numpy()
    off = 0
    for r in range(nranks):
        seg = pk[off:off + counts[r]]
        assert (seg % nranks == r).all()
        assert sorted(seg.tolist()) == sorted(
            keys_h[keys_h % nranks == r].tolist())
        assert (pv[off:off + counts[r]] == seg * 2 + 1).all()
        off += counts[r]
    # the permutation is a bijection over [0, n)
    assert np.array_equal(np.sort(perm.cpu().numpy()), np.arange(n))


def test_q3_grace_multipass_parity(ex, ora):
    """Grace-style multi-pass (SURVEY §8f.4 analog): shrink the direct-table
    capacity so the key range needs ~7 passes; results must equal the
    single-pass run and the oracle."""
    import os
    n = 400000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    os.environ["OTBX_DIRECT_CAP"] = "16384"   # norders=100k → ~7 passes
    try:
        node = ex.GpuQ3Fragment(cu, od, li)
        drain(node)
        got = {int(k): float(v) for k, v in
               zip(node.fetch_groups()["l_orderkey"],
                   node.fetch_groups()["revenue"])}
    finally:
        del os.environ["OTBX_DIRECT_CAP"]
    og = q3_oracle(ora, n)
    exp = dict(zip(og["l_orderkey"].tolist(), og["revenue"].tolist()))
    assert got.keys() == exp.keys()
    for k in exp:
        assert approx(got[k], exp[k]), k


# ---------------- full GPU ORDER BY ----------------

def test_order_groups_full_sort(ex, ora):
    """Full sort of Q3 groups by (revenue DESC, o_orderdate ASC) matches the
    oracle's ordering key-for-key (ties in both keys order-insensitive)."""
    n = 400000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    node = ex.GpuQ3Fragment(cu, od, li)
    drain(node)
    got = ex.order_groups(node._groups_dev, node.ngroups)
    assert len(got) == node.ngroups
    # sortedness by the composite key
    key = list(zip((-got["revenue"]).tolist(), got["o_orderdate"].tolist()))
    assert key == sorted(key)
    # same multiset as the oracle's groups, and identical (rev, date) seq
    og = q3_oracle(ora, n)
    order = np.lexsort((og["o_orderdate"], -og["revenue"]))
    exp = og[order]
    assert len(exp) == len(got)
    assert np.allclose(np.sort(got["revenue"]), np.sort(exp["revenue"]),
                       rtol=1e-9)
    # spot-check strict prefix against oracle top-100 (no revenue ties there)
    for i in range(100):
        assert abs(got["revenue"][i] - exp["revenue"][i]) <= \
            1e-9 * abs(exp["revenue"][i])


def test_order_groups_edges(ex):
    import ctypes as Ct
    import numpy as np
    dt = np.dtype(ex.GpuQ3Fragment.NP_DTYPE)
    # n = 0 and n = 1
    g0 = torch.empty(24, dtype=torch.uint8, device="cuda")
    out = ex.order_groups(g0, 0)
    assert len(out) == 0
    one = np.zeros(1, dtype=dt)
    one["l_orderkey"] = 7
    one["revenue"] = 3.5
    g1 = torch.from_numpy(one.view(np.uint8).reshape(-1).copy()).cuda()
    out = ex.order_groups(g1, 1)
    assert out["l_orderkey"][0] == 7
    # equal revenues → date ascending
    eq = np.zeros(1000, dtype=dt)
    eq["l_orderkey"] = np.arange(1000)
    eq["revenue"] = 42.0
    eq["o_orderdate"] = np.random.default_rng(5).integers(0, 3000, 1000)
    geq = torch.from_numpy(eq.view(np.uint8).reshape(-1).copy()).cuda()
    out = ex.order_groups(geq, 1000)
    assert (np.diff(out["o_orderdate"]) >= 0).all()


def test_agg_int64min_key(ex, ora):
    """ANY i64 key is groupable, including the open-addressing sentinel
    value INT64_MIN (routed to a dedicated accumulator)."""
    imin = -(2**63)
    keys = np.array([imin, 5, imin, 7, imin], dtype=np.int64)
    vals = np.array([1.0, 2.0, 3.0, 4.0, 5.0])
    got = _agg(ex, keys, vals)
    exp = ora.agg_i64(keys, vals)
    assert len(got) == len(exp) == 3
    gm = {int(g["key"]): (int(g["count_star"]), float(g["sum_v"])) for g in got}
    assert gm[imin] == (3, 9.0)
    assert gm[5] == (1, 2.0)
    assert gm[7] == (1, 4.0)


def test_q3_hash_grace_multipass_parity(ex, ora):
    """Grace batching on the HASH fallback: force both the hash path and a
    tiny per-pass table budget (→ ~13 passes); results must equal the
    oracle."""
    import os
    n = 400000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    os.environ["OTBX_Q3_FORCE_HASH"] = "1"
    os.environ["OTBX_Q3_HASH_BUDGET"] = "1024"
    try:
        node = ex.GpuQ3Fragment(cu, od, li)
        drain(node)
        got = {int(k): float(v) for k, v in
               zip(node.fetch_groups()["l_orderkey"],
                   node.fetch_groups()["revenue"])}
        hits = node.probe_hits
    finally:
        del os.environ["OTBX_Q3_FORCE_HASH"]
        del os.environ["OTBX_Q3_HASH_BUDGET"]
    og = q3_oracle(ora, n)
    exp = dict(zip(og["l_orderkey"].tolist(), og["revenue"].tolist()))
    assert got.keys() == exp.keys()
    for k in exp:
        assert approx(got[k], exp[k]), k
    assert hits >= len(exp)


def test_fetch_groups_ordered(ex):
    n = 400000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    node = ex.GpuQ3Fragment(cu, od, li)
    drain(node)
    g = node.fetch_groups(ordered=True)
    key = list(zip((-g["revenue"]).tolist(), g["o_orderdate"].tolist()))
    assert key == sorted(key)
    assert len(g) == node.ngroups


def test_agg_partitioned_path_parity(ex, ora):
    """The partitioned mid-cardinality agg path (triggers at ≥8M rows with a
    1536 < distinct-estimate ≤ 32M) must match the oracle — both the
    (key,val)-record fast path and the NULL-carrying gather path, including
    sentinel-valued keys."""
    rng = np.random.default_rng(31)
    n = 12_000_000
    keys = rng.integers(0, 100_000, n)
    keys[::1_000_000] = -(2**63)      # sprinkle sentinel-valued keys
    vals = rng.random(n) * 1e3
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa

    # fast path: no null bitmaps
    got = _agg(ex, keys, vals)
    exp = ora.agg_i64(keys, vals)
    assert len(got) == len(exp)
    gm = {(int(g["key"]), bool(g["key_isnull"])): g for g in got}
    for e in exp:
        g = gm[(e.key if not e.key_isnull else 0, bool(e.key_isnull))]
        assert g["count_star"] == e.count_star
        assert g["count_v"] == e.count_v
        assert abs(float(g["sum_v"]) - e.sum_v) <= 1e-9 * abs(e.sum_v)

    # gather path: NULL keys + NULL values present
    kn = (rng.random(n) < 0.02).astype(np.uint8)
    vn = (rng.random(n) < 0.05).astype(np.uint8)
    got = _agg(ex, keys, vals, kn, vn)
    exp = ora.agg_i64(keys, vals, key_null=kn, val_null=vn)
    assert len(got) == len(exp)
    gm = {(int(g["key"]) if not g["key_isnull"] else 0, bool(g["key_isnull"])): g
          for g in got}
    for e in exp:
        g = gm[(e.key if not e.key_isnull else 0, bool(e.key_isnull))]
        assert g["count_star"] == e.count_star
        assert g["count_v"] == e.count_v
        if not e.sum_isnull:
            assert abs(float(g["sum_v"]) - e.sum_v) <= \
                1e-9 * max(abs(e.sum_v), 1e-6)


def test_agg_partitioned_tile_level2_parity(ex, ora):
    """The tile-staged counting-sort partitioner (default path, §8b.0,
    A/B in profiles/r01_scatter_ab.txt) through BOTH partition levels
    (600k distinct -> nb=256, nb2=8), incl. sentinel keys. Asserts sample
    the group map (the full 600k-iteration Python loop was the slow part,
    not the GPU) plus count-conservation invariants."""
    rng = np.random.default_rng(41)
    n = 12_000_000
    keys = rng.integers(0, 600_000, n)
    keys[::1_000_000] = -(2**63)
    vals = rng.random(n) * 1e3
    got = _agg(ex, keys, vals)
    exp = ora.agg_i64(keys, vals)
    assert len(got) == len(exp)
    assert sum(g["count_star"] for g in got) == n
    gm = {(int(g["key"]), bool(g["key_isnull"])): g for g in got}
    for e in exp[::29]:  # ~21k sampled groups
        g = gm[(e.key if not e.key_isnull else 0, bool(e.key_isnull))]
        assert g["count_star"] == e.count_star
        assert g["count_v"] == e.count_v
        assert abs(float(g["sum_v"]) - e.sum_v) <= 1e-9 * abs(e.sum_v)


def test_agg_partitioned_legacy_parity(ex, ora, monkeypatch):
    """OTBX_PART_TILE=0 pins the legacy cursor-scatter partitioner (the
    fallback for NULL-carrying inputs and n >= 2^32) on the same two-level
    shape."""
    monkeypatch.setenv("OTBX_PART_TILE", "0")
    rng = np.random.default_rng(43)
    n = 10_000_000
    keys = rng.integers(0, 600_000, n)
    vals = rng.random(n) * 1e3
    got = _agg(ex, keys, vals)
    exp = ora.agg_i64(keys, vals)
    assert len(got) == len(exp)
    assert sum(g["count_star"] for g in got) == n
    gm = {int(g["key"]): g for g in got}
    for e in exp[::29]:
        g = gm[e.key]
        assert g["count_star"] == e.count_star
        assert abs(float(g["sum_v"]) - e.sum_v) <= 1e-9 * abs(e.sum_v)


def test_join_partitioned_legacy_parity(ex, ora, monkeypatch):
    """OTBX_PART_TILE=0 pins the legacy k_joinp_scatter/kv2 partitioner;
    the default-path equivalent (tile scatter with row-index payload and
    NULL skip list) is covered by test_join_partitioned_parity and
    test_join_partitioned_large_count above. Sentinel keys still flag
    buckets to the global fallback."""
    monkeypatch.setenv("OTBX_JOINP_FORCE", "1")
    monkeypatch.setenv("OTBX_PART_TILE", "0")
    rng = np.random.default_rng(19)
    nb, npr = 300000, 900000
    bk = rng.integers(0, 200000, nb)
    pk = rng.integers(0, 200000, npr)
    bk[::5000] = np.iinfo(np.int64).min
    pk[::7000] = np.iinfo(np.int64).min
    bn = (rng.random(nb) < 0.03).astype(np.uint8)
    pn = (rng.random(npr) < 0.03).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                          dev(bn, torch.uint8), dev(pn, torch.uint8),
                          cap_pairs=8 * npr)
    pairs = drain(node)
    obi, opi = ora.join_i64(bk, pk, bnull=bn, pnull=pn)
    assert len(pairs) == len(obi)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))


def test_host_staging_path(ex, ora):
    """otbx_stage_table flow: oracle-generated HOST columns staged through
    otbx_memcpy_h2d, then the Q1 fragment — exercises the real provider
    staging boundary end to end."""
    import time
    n = 2_000_000
    t = ora.gen_tables(n)["lineitem"]
    ex.GpuLineitem.from_host(t, with_orderkey=False)  # warm the copy engine
    t0 = time.time()
    li = ex.GpuLineitem.from_host(t, with_orderkey=False)
    stage_s = time.time() - t0
    rows = ex.q1_finalize(drain(ex.GpuQ1PartialAgg(li)))
    og = ora.q1_finalize(ora.q1_partial(ora.gen_tables(n)))
    assert [r["count_order"] for r in rows] == [o.count_order for o in og]
    for g, o in zip(rows, og):
        assert approx(g["sum_charge"], o.sum_charge)
    gb = 38 * n / 1e9
    print(f"\nhost staging: {gb / stage_s:.1f} GB/s PCIe-inclusive "
          f"({stage_s * 1e3:.1f} ms for {gb:.2f} GB)")


def test_q9_multipass_selfconsistent(ex, monkeypatch):
    """Single-pass vs forced 4-pass bitmap slicing on the SAME staged
    tables: counts bit-exact, sums within 1e-12 relative (the passes only
    change float accumulation order). Both paths are independently
    oracle-verified above; this pins them against each other directly."""
    n = 400000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)
    a = drain(ex.GpuQ9Fragment(pt, od, li))
    monkeypatch.setenv("OTBX_Q9_BITMAP_BITS", "4096")
    b = drain(ex.GpuQ9Fragment(pt, od, li))
    assert [(r["o_year"], r["count_rows"]) for r in a] == \
        [(r["o_year"], r["count_rows"]) for r in b]
    for ra, rb in zip(a, b):
        # order-dependent rounding only; 1e-9 bounds the worst-case linear
        # error growth over the per-group value count with wide margin
        assert abs(ra["sum_revenue"] - rb["sum_revenue"]) <= \
            1e-9 * abs(ra["sum_revenue"])


def test_partitioner_tile_vs_legacy_selfconsistent(ex, monkeypatch):
    """Tile vs legacy partitioner on the SAME device arrays: group sets
    and counts bit-exact, sums within 1e-12 (bucket record order differs).
    Each path is oracle-verified above; this compares them directly at a
    two-level shape (600k distinct)."""
    g = torch.Generator(device="cuda").manual_seed(77)
    n = 10_000_000
    keys = torch.randint(0, 600_000, (n,), dtype=torch.int64, device="cuda",
                         generator=g)
    vals = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
    import numpy as np  # noqa: F811
    ka, va = keys.cpu().numpy(), vals.cpu().numpy()
    a = _agg(ex, ka, va)
    monkeypatch.setenv("OTBX_PART_TILE", "0")
    b = _agg(ex, ka, va)
    assert len(a) == len(b)
    ma = {int(r["key"]): r for r in a}
    for rb in b[:: 17]:
        ra = ma[int(rb["key"])]
        assert ra["count_star"] == rb["count_star"]
        assert abs(float(ra["sum_v"]) - float(rb["sum_v"])) <= \
            1e-12 * max(abs(float(rb["sum_v"])), 1e-12)


def test_join_overflow_contract(ex):
    """Pins the otbx.h overflow contract (ADVICE r1 / VERDICT r1 #6):
    *npairs_dev receives the TRUE match count even when it exceeds
    cap_pairs; the call returns OTBX_OK; the output arrays hold only a
    bounded subset of VALID pairs; the host layer raises."""
    import ctypes as C
    from opentenbase_amd._lib import lib
    from opentenbase_amd.executor import _stream

    L = lib()
    nb, npr = 1000, 50000
    # every probe key matches exactly one build key -> 50000 true pairs
    bk = torch.arange(nb, dtype=torch.int64, device="cuda")
    pk = torch.arange(npr, dtype=torch.int64, device="cuda") % nb
    ws_bytes = C.c_size_t(0)
    assert L.otbx_join_i64_workspace_bytes(C.c_int64(nb), C.c_int64(npr),
                                           C.byref(ws_bytes)) == 0
    ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8, device="cuda")
    cap = 128          # far below the 50000 true pairs
    ob = torch.full((cap,), -1, dtype=torch.int64, device="cuda")
    op = torch.full((cap,), -1, dtype=torch.int64, device="cuda")
    npairs = torch.zeros(1, dtype=torch.int64, device="cuda")
    st = L.otbx_join_i64(C.c_void_p(bk.data_ptr()), None, C.c_int64(nb),
                         C.c_void_p(pk.data_ptr()), None, C.c_int64(npr),
                         C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
                         C.c_void_p(ob.data_ptr()), C.c_void_p(op.data_ptr()),
                         C.c_int64(cap), C.c_void_p(npairs.data_ptr()),
                         _stream())
    assert st == 0                      # OTBX_OK per the documented contract
    torch.cuda.synchronize()
    n_true = int(npairs.cpu().item())
    assert n_true == npr                # TRUE count, not clamped to cap
    # whatever subset was written must be valid join pairs
    obh, oph = ob.cpu().numpy(), op.cpu().numpy()
    written = obh >= 0
    assert written.any()
    assert np.array_equal(obh[written], oph[written] % nb)
    # the host layer surfaces overflow as an error (never truncates silently)
    node = ex.GpuHashJoin(bk, pk, cap_pairs=cap)
    node.BeginCustomScan()
    with pytest.raises(ex.OtbxError):
        node._run()


def test_q9_filter_tile_vs_wave(ex, monkeypatch):
    """Tile-staged Q9 part filter (append_ab v6 pattern, default) vs the
    legacy per-wave appender on the same tables: counts exact, sums within
    tree-rounding tolerance (survivor order differs)."""
    n = 2_000_000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)

    def run():
        node = ex.GpuQ9Fragment(pt, od, li)
        node.BeginCustomScan()
        node._run()
        s, c = node.partial_state_tensors()
        return s.cpu().numpy().copy(), c.cpu().numpy().copy()

    s_tile, c_tile = run()
    monkeypatch.setenv("OTBX_Q9_FILTER_WAVE", "1")
    s_wave, c_wave = run()
    assert np.array_equal(c_tile, c_wave)
    for a, b in zip(s_tile, s_wave):
        assert abs(a - b) <= 1e-9 * max(abs(b), 1.0)


def test_q3_compact_tile_vs_legacy(ex, monkeypatch):
    """Tile-staged Q3 group compaction (default) vs the legacy block-chunk
    kernel: identical group sets (revenues bit-exact — compaction reads the
    same finished rtab either way)."""
    n = 1_000_000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)

    def run():
        node = ex.GpuQ3Fragment(cu, od, li)
        node.BeginCustomScan()
        node._run()
        g = node.fetch_groups()
        return sorted((int(r["l_orderkey"]), float(r["revenue"]),
                       int(r["o_orderdate"]), int(r["o_shippriority"]))
                      for r in g)

    word = run()                      # default: word-granular
    monkeypatch.setenv("OTBX_Q3_COMPACT_TILE", "1")
    tile = run()
    monkeypatch.delenv("OTBX_Q3_COMPACT_TILE")
    monkeypatch.setenv("OTBX_Q3_COMPACT_LEGACY", "1")
    legacy = run()
    assert word == tile == legacy


# ---------------- extended joins + two-key operators (r2 widening) --------

JOIN_TYPES6 = ["inner", "left", "semi", "anti", "right", "full"]


@pytest.mark.parametrize("jt", JOIN_TYPES6)
def test_join_ext_parity_gpu(ex, ora, jt):
    """All six join types vs the oracle's FSM restatement, with NULLs and
    duplicate keys (result-set parity; -1 = NULL-fill side)."""
    rng = np.random.default_rng(hash(jt) % 2**31)
    nb, npr = 30000, 90000
    bk = rng.integers(-50, 20000, nb)
    pk = rng.integers(-50, 20000, npr)
    bn = (rng.random(nb) < 0.07).astype(np.uint8)
    pn = (rng.random(npr) < 0.07).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                          dev(bn, torch.uint8), dev(pn, torch.uint8),
                          join_type=jt)
    pairs = drain(node)
    obi, opi = ora.join_ext(bk, pk, ex.JOIN_TYPES[jt], bnull=bn, pnull=pn)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))


@pytest.mark.parametrize("jt", ["inner", "left", "full"])
def test_join_ext_two_key_parity_gpu(ex, ora, jt):
    rng = np.random.default_rng(17)
    nb, npr = 20000, 60000
    bk = rng.integers(0, 300, nb)
    bk2 = rng.integers(0, 50, nb)
    pk = rng.integers(0, 300, npr)
    pk2 = rng.integers(0, 50, npr)
    pn2 = (rng.random(npr) < 0.05).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                          join_type=jt, build_keys2=dev(bk2, torch.int64),
                          probe_keys2=dev(pk2, torch.int64),
                          probe_null2=dev(pn2, torch.uint8))
    pairs = drain(node)
    obi, opi = ora.join_ext(bk, pk, ex.JOIN_TYPES[jt], bkeys2=bk2,
                            pkeys2=pk2, pnull2=pn2)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))


@pytest.mark.parametrize("jt", JOIN_TYPES6)
def test_join_ext_edges_gpu(ex, ora, jt):
    """Empty sides, all-NULL sides, INT64_MIN/MAX keys."""
    dev = lambda a: torch.as_tensor(a, dtype=torch.int64, device="cuda")  # noqa: E731
    k = np.array([1, 2, 2, 3], dtype=np.int64)
    e = np.empty(0, dtype=np.int64)
    nn = np.ones(4, dtype=np.uint8)
    ext = np.array([np.iinfo(np.int64).min, np.iinfo(np.int64).max, 0],
                   dtype=np.int64)
    cases = [
        (e, k, None, None), (k, e, None, None), (e, e, None, None),
        (k, k, nn, None), (k, k, None, nn), (ext, ext, None, None),
    ]
    for bk, pk, bn, pn in cases:
        devn = lambda a: (torch.as_tensor(a, device="cuda")  # noqa: E731
                          if a is not None else None)
        node = ex.GpuHashJoin(dev(bk), dev(pk), devn(bn), devn(pn),
                              join_type=jt)
        pairs = drain(node)
        obi, opi = ora.join_ext(bk, pk, ex.JOIN_TYPES[jt], bnull=bn, pnull=pn)
        assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist())), \
            (jt, len(bk), len(pk))


def test_agg2_parity_gpu(ex, ora):
    rng = np.random.default_rng(23)
    n = 500000
    k1 = rng.integers(-10, 1000, n)
    k2 = rng.integers(0, 40, n)
    v = rng.standard_normal(n)
    n1 = (rng.random(n) < 0.1).astype(np.uint8)
    vn = (rng.random(n) < 0.1).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashAgg2(dev(k1, torch.int64), dev(k2, torch.int64),
                          dev(v, torch.float64),
                          key1_null=dev(n1, torch.uint8),
                          val_null=dev(vn, torch.uint8))
    got = drain(node)
    exp = ora.agg_i64x2(k1, k2, v, k1null=n1, val_null=vn)
    assert len(got) == len(exp)
    for g, o in zip(got, exp):
        assert bool(g["key1_isnull"]) == bool(o.key1_isnull)
        assert bool(g["key2_isnull"]) == bool(o.key2_isnull)
        if not o.key1_isnull:
            assert int(g["key1"]) == o.key1
        if not o.key2_isnull:
            assert int(g["key2"]) == o.key2
        assert int(g["count_star"]) == o.count_star    # bit-exact
        assert int(g["count_v"]) == o.count_v
        assert bool(g["sum_isnull"]) == bool(o.sum_isnull)
        if not o.sum_isnull:
            assert approx(float(g["sum_v"]), o.sum_v, rel=1e-9)


def test_agg2_grid_golden_gpu(ex, ora):
    """xc_FQS_join.out:89-105 grid-join shape on two REAL key columns — no
    host-side key packing."""
    a = np.repeat(np.arange(5), 2).astype(np.int64)
    b = np.tile(np.arange(2), 5).astype(np.int64)
    dev = lambda x: torch.as_tensor(x, dtype=torch.int64, device="cuda")  # noqa: E731
    node = ex.GpuHashJoin(dev(a), dev(a), join_type="inner",
                          build_keys2=dev(b), probe_keys2=dev(b))
    pairs = drain(node)
    assert len(pairs) == 10
    assert sorted(p for _, p in pairs) == list(range(10))
    assert all(bi == pi for bi, pi in pairs)
    agg = ex.GpuHashAgg2(dev(a), dev(b),
                         torch.ones(10, dtype=torch.float64, device="cuda"))
    rows = drain(agg)
    assert len(rows) == 10
    assert all(int(r["count_star"]) == 1 for r in rows)


def test_q9_rec_vs_columnar_selfconsistent(ex):
    """The staged q9rec probe path (default) vs the columnar-gather
    fallback (q9rec = NULL) on the same tables: counts exact, sums within
    rounding tolerance."""
    import ctypes as CT
    n = 2_000_000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)

    def run():
        node = ex.GpuQ9Fragment(pt, od, li)
        node.BeginCustomScan()
        node._run()
        s, c = node.partial_state_tensors()
        return s.cpu().numpy().copy(), c.cpu().numpy().copy()

    s_rec, c_rec = run()
    li.cstruct.q9rec = CT.c_void_p(0)      # force the columnar fallback
    li.cstruct.l_partkey32 = CT.c_void_p(0)  # ... and the i64 key stream
    s_col, c_col = run()
    assert np.array_equal(c_rec, c_col)
    for a, b in zip(s_rec, s_col):
        assert abs(a - b) <= 1e-9 * max(abs(b), 1.0)


def test_agg_dec_parity_gpu(ex, ora):
    """Exact int128 decimal aggregate: GPU vs oracle BIT-EXACT (integer
    math — no float tolerance), incl. sums beyond the int64 range and
    negative-carry paths."""
    rng = np.random.default_rng(41)
    n = 400000
    k = rng.integers(-5, 100, n)
    v = rng.integers(np.iinfo(np.int64).min // 2,
                     np.iinfo(np.int64).max // 2, n)
    kn = (rng.random(n) < 0.1).astype(np.uint8)
    vn = (rng.random(n) < 0.1).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashAggDec(dev(k, torch.int64), dev(v, torch.int64),
                            key_null=dev(kn, torch.uint8),
                            val_null=dev(vn, torch.uint8))
    got = drain(node)
    exp = ora.agg_i64_dec(k, v, key_null=kn, val_null=vn)
    assert len(got) == len(exp)
    for g, o in zip(got, exp):
        assert bool(g["key_isnull"]) == bool(o.key_isnull)
        if not o.key_isnull:
            assert g["key"] == o.key
        assert g["count_star"] == o.count_star
        assert g["count_v"] == o.count_v
        assert bool(g["sum_isnull"]) == bool(o.sum_isnull)
        if not o.sum_isnull:
            assert g["sum128"] == o.sum128             # bit-exact int128
    # carry stress: one group, all INT64_MAX (sum far beyond int64)
    m = 200000
    k1 = torch.zeros(m, dtype=torch.int64, device="cuda")
    v1 = torch.full((m,), 2**63 - 1, dtype=torch.int64, device="cuda")
    rows = drain(ex.GpuHashAggDec(k1, v1))
    assert len(rows) == 1 and rows[0]["sum128"] == m * (2**63 - 1)


def test_q3_key32_vs_i64_selfconsistent(ex):
    """The staged compact-key (int32) path vs the i64 fallback (cache
    pointers nulled) on the same tables: identical group sets and top-k."""
    import ctypes as CT
    n = 1_000_000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    assert li.cstruct.l_orderkey32      # caches built at this scale
    assert od.cstruct.o_orderkey32 and od.cstruct.o_custkey32

    def run():
        node = ex.GpuQ3Fragment(cu, od, li)
        node.BeginCustomScan()
        node._run()
        g = node.fetch_groups()
        return sorted((int(r["l_orderkey"]), float(r["revenue"]),
                       int(r["o_orderdate"]), int(r["o_shippriority"]))
                      for r in g)

    k32 = run()
    li.cstruct.l_orderkey32 = CT.c_void_p(0)
    od.cstruct.o_orderkey32 = CT.c_void_p(0)
    od.cstruct.o_custkey32 = CT.c_void_p(0)
    i64 = run()
    assert k32 == i64


def test_build_key32_overflow_detect(ex):
    """Keys outside [0, 2^31) must void the cache (staging-time check)."""
    from opentenbase_amd.executor import _build_key32
    ok = torch.tensor([1, 2, 3], dtype=torch.int64, device="cuda")
    assert _build_key32(ok) is not None
    bad = torch.tensor([1, 2**31, 3], dtype=torch.int64, device="cuda")
    assert _build_key32(bad) is None
    neg = torch.tensor([1, -2, 3], dtype=torch.int64, device="cuda")
    assert _build_key32(neg) is None


# ---------------- N-key operators ----------------

@pytest.mark.parametrize("nk", [1, 3, 8])
def test_aggn_parity_gpu(ex, ora, nk):
    """N-key group-by GPU vs oracle: same group count and, keyed by each
    group's identity tuple, identical counts and 1e-9-close sums; the
    defining row must belong to its own group."""
    rng = np.random.default_rng(60 + nk)
    n = 200000
    keys = [rng.integers(-3, 6, n) for _ in range(nk)]
    nulls = [(rng.random(n) < 0.1).astype(np.uint8) if c == 0 else None
             for c in range(nk)]
    v = rng.standard_normal(n)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731
    node = ex.GpuHashAggN([dev(k, torch.int64) for k in keys],
                          dev(v, torch.float64),
                          null_tensors=[dev(nc, torch.uint8)
                                        if nc is not None else None
                                        for nc in nulls])
    got = drain(node)
    exp = ora.agg_i64n(keys, v, null_cols=nulls)

    def ident(i):
        out = []
        for c in range(nk):
            isnull = nulls[c] is not None and nulls[c][i]
            out.append((bool(isnull), 0 if isnull else int(keys[c][i])))
        return tuple(out)

    assert len(got) == len(exp)
    em = {ident(o.row_idx): o for o in exp}
    for g in got:
        o = em[ident(int(g["row_idx"]))]
        assert int(g["count_star"]) == o.count_star
        assert int(g["count_v"]) == o.count_v
        if o.sum_isnull:
            assert g["sum_isnull"]
        else:
            assert abs(float(g["sum_v"]) - o.sum_v) <= \
                1e-9 * max(abs(o.sum_v), 1.0)


@pytest.mark.parametrize("jt", ["inner", "left", "semi", "anti", "full"])
def test_joinn_parity_gpu(ex, ora, jt):
    rng = np.random.default_rng(80)
    nb, npr = 20000, 50000
    bkeys = [rng.integers(0, 40, nb) for _ in range(3)]
    pkeys = [rng.integers(0, 40, npr) for _ in range(3)]
    bnulls = [(rng.random(nb) < 0.05).astype(np.uint8), None, None]
    dev = lambda a: torch.as_tensor(a, dtype=torch.int64, device="cuda")  # noqa: E731
    devn = lambda a: (torch.as_tensor(a, device="cuda")  # noqa: E731
                      if a is not None else None)
    node = ex.GpuHashJoinN([dev(k) for k in bkeys], [dev(k) for k in pkeys],
                           join_type=jt,
                           bnulls=[devn(x) for x in bnulls])
    pairs = drain(node)
    obi, opi = ora.join_i64n(bkeys, pkeys, ex.JOIN_TYPES[jt],
                             bnull_cols=bnulls)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))


def test_finish_releases_scratch_and_reinit(ex):
    """otbx_finish releases the per-process cached scratch and nulls the
    owning statics: init -> ops that allocate lazy scratch
    (partition, Q3 with its pinned readbacks) -> finish -> re-init ->
    the same ops run correctly again on fresh scratch."""
    from opentenbase_amd._lib import call

    def work():
        keys = torch.arange(100000, dtype=torch.int64, device="cuda") % 7
        perm, counts = ex.partition_by_key(keys)
        assert sum(counts) == 100000
        n = 200000
        li = ex.GpuLineitem.generate(n)
        od = ex.GpuOrders.generate(n // 4, n // 40)
        cu = ex.GpuCustomer.generate(n // 40)
        node = ex.GpuQ3Fragment(cu, od, li)
        node.BeginCustomScan()
        node._run()
        return node.ngroups, counts

    a = work()
    call("otbx_finish")
    ex.init_device(0)
    b = work()
    assert a == b


def test_nk_agg_estimator_fallback(ex, monkeypatch):
    """The estimator-sized first attempt vs a forced-overflow run (tiny
    first table -> abort -> full-cap rerun): identical results. Covers the
    bounded-probe abort path of agg2/dec/aggn."""
    g = torch.Generator(device="cuda").manual_seed(13)
    n = 9_000_000                       # above AGGP_THRESHOLD
    k1 = torch.randint(0, 100_000, (n,), dtype=torch.int64, device="cuda",
                       generator=g)
    k2 = torch.randint(0, 10, (n,), dtype=torch.int64, device="cuda",
                       generator=g)
    v = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
    vi = torch.randint(-10**9, 10**9, (n,), dtype=torch.int64,
                       device="cuda", generator=g)

    def run_all():
        a2 = ex.GpuHashAgg2(k1, k2, v)
        a2.BeginCustomScan()
        r2 = [(int(r["key1"]), int(r["key2"]), int(r["count_star"]))
              for r in a2._run()]
        dc = ex.GpuHashAggDec(k1, vi)
        dc.BeginCustomScan()
        rd = [(int(r["key"]), int(r["count_star"]), r["sum128"])
              for r in dc._run()]
        an = ex.GpuHashAggN([k1, k2], v)
        an.BeginCustomScan()
        rn = sorted((int(r["count_star"]), int(r["count_v"]))
                    for r in an._run())
        return r2, rd, rn

    normal = run_all()
    monkeypatch.setenv("OTBX_NK_FORCE_CAP", "4096")
    forced = run_all()
    assert normal[0] == forced[0]
    assert normal[1] == forced[1]
    assert normal[2] == forced[2]


@pytest.mark.parametrize("jt", ["left", "right", "full"])
def test_join_ext_via_inner_parity(ex, ora, jt, monkeypatch):
    """The via-inner outer-join route (fast inner join + mark-and-fill)
    vs the oracle AND vs the FSM-table route on the same inputs."""
    rng = np.random.default_rng(55)
    nb, npr = 40000, 120000
    bk = rng.integers(-10, 25000, nb)
    pk = rng.integers(-10, 25000, npr)
    bn = (rng.random(nb) < 0.08).astype(np.uint8)
    pn = (rng.random(npr) < 0.08).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa: E731

    def run():
        node = ex.GpuHashJoin(dev(bk, torch.int64), dev(pk, torch.int64),
                              dev(bn, torch.uint8), dev(pn, torch.uint8),
                              join_type=jt)
        return sorted(drain(node))

    monkeypatch.setenv("OTBX_JOINX_VIA_INNER", "1")
    via = run()
    monkeypatch.setenv("OTBX_JOINX_VIA_INNER", "0")
    fsm = run()
    obi, opi = ora.join_ext(bk, pk, ex.JOIN_TYPES[jt], bnull=bn, pnull=pn)
    exp = sorted(zip(obi.tolist(), opi.tolist()))
    assert via == exp
    assert fsm == exp
