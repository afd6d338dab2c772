"""Full-scale DIRECT parity (VERDICT r1 missing #4): GPU vs the oracle at
the BASELINE contract sizes, comparing actual results — not just invariants.
The oracle side runs through the C CLI (scalar restatement, ~5 s at SF100
Q1), on the SAME bit-identical counter-based tables (seed 42).

Bar: COUNT and group keys bit-exact; SUM/AVG(float8) within
1e-6 relative.
"""
import json
import os
import subprocess

import pytest

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")

REL = 1e-6
ORACLE_DIR = os.path.join(os.path.dirname(__file__), "..", "oracle")


@pytest.fixture(scope="module")
def ex():
    from opentenbase_amd import executor
    executor.init_device(0)
    return executor


def oracle_cli(*args):
    subprocess.run(["make", "-C", ORACLE_DIR], check=True,
                   capture_output=True)
    out = subprocess.run(
        [os.path.join(ORACLE_DIR, "oracle_cli"), *args],
        check=True, capture_output=True, text=True).stdout
    return json.loads(out)


def approx(a, b, rel=REL):
    return abs(a - b) <= rel * max(abs(a), abs(b), 1e-300)


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


def test_q1_full_scale_sf100(ex):
    """BASELINE config 3 (the bench headline): Q1 over SF100 = 600 M rows,
    GPU vs oracle, all 4 group states compared directly."""
    n = 600_000_000
    li = ex.GpuLineitem.generate(n, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    rows = ex.q1_finalize(drain(node))
    del li
    torch.cuda.empty_cache()
    og = oracle_cli("q1", "--rows", str(n))["groups"]
    assert len(rows) == len(og) == 4
    for g, o in zip(rows, og):
        assert g["l_returnflag"] == o["rf"]
        assert g["l_linestatus"] == o["ls"]
        assert g["count_order"] == o["count"]          # bit-exact
        for fld in ("sum_qty", "sum_base_price", "sum_disc_price",
                    "sum_charge", "avg_qty", "avg_price", "avg_disc"):
            assert approx(g[fld], o[fld]), (fld, g[fld], o[fld])


def test_q3_full_scale_sf10(ex):
    """Q3 over SF10 (60 M lineitem / 15 M orders / 1.5 M customer): group
    count and the top-10 rows (revenue DESC, date ASC; ties by orderkey)
    GPU vs oracle."""
    n = 60_000_000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    node = ex.GpuQ3Fragment(cu, od, li)
    top = drain(node)
    ngroups = node.ngroups
    del li, od, cu
    torch.cuda.empty_cache()
    j = oracle_cli("q3", "--rows", str(n))
    assert ngroups == j["ngroups"]                      # group keys exact
    ot = j["top"]
    assert len(top) == len(ot) == 10
    for g, o in zip(top, ot):
        assert int(g[0]) == o["orderkey"]
        assert int(g[2]) == o["orderdate"]
        assert int(g[3]) == o["prio"]
        assert approx(float(g[1]), o["revenue"])


def test_q9_full_scale_sf10(ex):
    """Q9-mix over SF10: the 7 year-group states GPU vs oracle."""
    from opentenbase_amd import fragment
    n = 60_000_000
    li = ex.GpuLineitem.generate(n, with_partkey=True)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    pt = ex.GpuPart.generate(n // 30)
    node = ex.GpuQ9Fragment(pt, od, li)
    node.BeginCustomScan()
    node._rows = node._run()
    s, c = node.partial_state_tensors()
    rows = fragment.merge_q9_partials(s, c)
    del li, od, pt
    torch.cuda.empty_cache()
    og = oracle_cli("q9", "--rows", str(n))["groups"]
    assert len(rows) == len(og)
    for g, o in zip(rows, og):
        assert int(g["o_year"]) == o["year"]
        assert int(g["count_rows"]) == o["count"]       # bit-exact
        assert approx(float(g["sum_revenue"]), o["revenue"])
