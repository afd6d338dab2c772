"""CustomScan lifecycle conformance (the nodeCustom.c driver contract):
ReScan produces identical results, Exec-before-Begin errors, End resets."""
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
    rows = []
    while True:
        r = node.ExecCustomScan()
        if r is None:
            break
        rows.append(r)
    return rows


def test_rescan_identical(ex):
    """ReScan re-runs the fragment: group keys/counts bit-identical; float
    sums within the 1e-6 contract (atomic f64 accumulation order varies
    between runs — BASELINE.md parity gate)."""
    li = ex.GpuLineitem.generate(200000, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    node.BeginCustomScan()
    first = drain(node)
    node.ReScanCustomScan()          # ReScanCustomScan (extensible.h:131)
    second = drain(node)
    assert len(first) == len(second)
    for a, b in zip(first, second):
        assert a["l_returnflag"] == b["l_returnflag"]
        assert a["count_order"] == b["count_order"]
        for f in ("sum_qty", "sum_base_price", "sum_disc_price", "sum_charge"):
            assert abs(a[f] - b[f]) <= 1e-6 * abs(a[f])
    node.EndCustomScan()


def test_exec_before_begin_errors(ex):
    from opentenbase_amd import OtbxError
    li = ex.GpuLineitem.generate(4000, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    with pytest.raises(OtbxError):
        node.ExecCustomScan()
    # after End, Exec errors again (ereport analog)
    node.BeginCustomScan()
    drain(node)
    node.EndCustomScan()
    with pytest.raises(OtbxError):
        node.ExecCustomScan()


def test_volcano_one_row_per_call(ex):
    li = ex.GpuLineitem.generate(200000, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    node.BeginCustomScan()
    r1 = node.ExecCustomScan()
    r2 = node.ExecCustomScan()
    assert r1 is not None and r2 is not None and r1 != r2
    node.EndCustomScan()


def test_explain_instrumentation(ex):
    li = ex.GpuLineitem.generate(200000, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    node.BeginCustomScan()
    assert node.explain() is None  # nothing run yet
    drain(node)
    exp = node.explain()
    assert exp and all(v >= 0 for v in exp.values())
    node.EndCustomScan()


def test_invalid_args_status(ex):
    """C-ABI returns OTBX_ERR_INVALID (→ OtbxError, the ereport analog) for
    bad arguments rather than crashing."""
    import ctypes as C
    from opentenbase_amd import OtbxError
    from opentenbase_amd._lib import call
    t = ex.GpuLineitem.generate(4000, with_orderkey=False)
    with pytest.raises(OtbxError) as ei:
        # n_global not divisible by nranks
        call("otbx_gen_lineitem_dev", C.byref(t.cstruct), C.c_uint64(42),
             C.c_int64(4001), C.c_uint32(0), C.c_uint32(2),
             C.c_void_p(torch.cuda.current_stream().cuda_stream))
    assert ei.value.status == 3
