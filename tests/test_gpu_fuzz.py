"""Randomized parity fuzz of the generic operators vs the oracle (seeded,
fast — runs in the round-end GPU suite)."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")


@pytest.fixture(scope="module")
def ex():
    from opentenbase_amd import executor
    executor.init_device(0)
    return executor


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


@pytest.fixture(scope="module")
def ora():
    from oracle import oracle_py
    return oracle_py


@pytest.mark.parametrize("seed,n,kspace,nullp", [
    (11, 1000, 3, 0.0),       # hot keys, LDS level only
    (12, 50000, 50000, 0.3),  # mostly-unique keys + NULLs
    (13, 200000, 700, 0.05),  # LDS-resident cardinality
    (14, 5000, 2, 0.5),       # two hot keys, heavy NULLs
    (15, 100000, 1 << 62, 0.1),  # huge sparse key domain
])
def test_fuzz_agg(ex, ora, seed, n, kspace, nullp):
    rng = np.random.default_rng(seed)
    keys = rng.integers(-(kspace // 2 + 1), kspace // 2 + 1, n)
    vals = rng.normal(0, 1e6, n)
    kn = (rng.random(n) < nullp).astype(np.uint8)
    vn = (rng.random(n) < nullp).astype(np.uint8)
    dev = lambda a, dt: torch.as_tensor(a, dtype=dt, device="cuda")  # noqa
    node = ex.GpuHashAgg(dev(keys, torch.int64), dev(vals, torch.float64),
                         dev(kn, torch.uint8), dev(vn, torch.uint8))
    got = drain(node)
    exp = ora.agg_i64(keys, vals, key_null=kn, val_null=vn)
    assert len(got) == len(exp)
    for g, e in zip(got, exp):
        assert bool(g["key_isnull"]) == bool(e.key_isnull)
        if not e.key_isnull:
            assert g["key"] == e.key
        assert g["count_star"] == e.count_star
        assert g["count_v"] == e.count_v
        if not e.sum_isnull:
            assert abs(float(g["sum_v"]) - e.sum_v) <= \
                1e-9 * max(abs(e.sum_v), 1e-6)


@pytest.mark.parametrize("seed,nb,np_,kspace", [
    (21, 1000, 5000, 100),     # heavy duplicates
    (22, 30000, 90000, 10**12),  # sparse
    (23, 1, 50000, 5),         # single build row
    (24, 0, 1000, 5),          # empty build side
])
def test_fuzz_join(ex, ora, seed, nb, np_, kspace):
    rng = np.random.default_rng(seed)
    bk = rng.integers(0, kspace, max(nb, 1))[:nb]
    pk = rng.integers(0, kspace, np_)
    dev = lambda a: torch.as_tensor(np.ascontiguousarray(a),  # noqa
                                    dtype=torch.int64, device="cuda")
    node = ex.GpuHashJoin(dev(bk), dev(pk), cap_pairs=max(20 * np_, 64))
    pairs = drain(node)
    obi, opi = ora.join_i64(bk, pk)
    assert sorted(pairs) == sorted(zip(obi.tolist(), opi.tolist()))
