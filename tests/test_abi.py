"""CPU-side checks of the C-ABI boundary: the library builds, loads, and
exports exactly the entry points include/otbx.h declares (no compute without
a GPU)."""
import os
import re

import pytest

from opentenbase_amd import _lib

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def so():
    if not os.path.exists(_lib._SO):
        _lib.build()
    return _lib.lib()


def header_symbols():
    hdr = open(os.path.join(REPO, "include", "otbx.h")).read()
    # function declarations: "otbx_status otbx_foo(" / "const char *otbx_foo("
    names = re.findall(r"^\s*(?:otbx_status|const char \*)\s*\*?(otbx_\w+)\s*\(",
                       hdr, re.M)
    return sorted(set(names))


def test_so_loads_and_version(so):
    assert b"otbx" in so.otbx_version()


def test_every_header_symbol_exported(so):
    syms = header_symbols()
    assert len(syms) >= 15
    for name in syms:
        assert getattr(so, name, None) is not None, f"missing export {name}"
    # and the python-side list matches the header exactly
    assert sorted(_lib.EXPORTED_SYMBOLS) == syms


def test_status_strings(so):
    for s in range(6):
        assert so.otbx_status_str(s)


def test_error_class():
    e = _lib.OtbxError(4, "x")
    assert "value out of range" in str(e)


def test_oracle_is_not_imported_by_product_code():
    """The product package must never route through the oracle
    (the oracle is the checker, not the shipped path)."""
    import subprocess
    import sys
    code = ("import sys; import opentenbase_amd, opentenbase_amd.executor, "
            "opentenbase_amd.fragment; "
            "bad=[m for m in sys.modules if 'oracle' in m]; "
            "assert not bad, bad")
    subprocess.run([sys.executable, "-c", code], check=True, cwd=REPO)


def _ws(so, fn, *args):
    import ctypes as C
    out = C.c_size_t(0)
    st = getattr(so, fn)(*[C.c_int64(a) if not isinstance(a, C.c_uint32)
                           else a for a in args], C.byref(out))
    assert st == 0
    return out.value


def test_workspace_sizing_monotone_and_sane(so):
    """The *_workspace_bytes functions are pure host code in libotbx.so —
    callable without a GPU. Sizing must be monotone in the row counts
    (a caller growing its input must never get a SMALLER requirement —
    the kernels check ws_bytes >= need and return OTBX_ERR_INVALID on a
    mismatch) and nonzero for nonzero inputs."""
    import ctypes as C
    prev = 0
    for n in (0, 1, 1000, 10**6, 8 << 20, 100 * 10**6):
        b = _ws(so, "otbx_agg_i64_workspace_bytes", n)
        assert b >= prev
        prev = b
    prev = 0
    for nb in (0, 1, 1000, 8 << 20, 50 * 10**6):
        b = _ws(so, "otbx_join_i64_workspace_bytes", nb, 4 * nb)
        assert b >= prev
        prev = b
    prev = 0
    for n in (1, 1000, 10**6, 10**8):
        b = _ws(so, "otbx_order_groups_workspace_bytes", n)
        assert b >= prev
        prev = b
    b3 = _ws(so, "otbx_q3_workspace_bytes", 150_000, 1_500_000, 6_000_000)
    assert b3 > 0
    assert _ws(so, "otbx_q3_workspace_bytes", 1_500_000, 15_000_000,
               60_000_000) > b3


def test_q9_workspace_respects_bitmap_slice_cap(so, monkeypatch):
    """otbx_q9_workspace_bytes and otbx_q9_partial share q9_slice_bits():
    the workspace for a huge part table must stop growing once the slice
    cap is hit (multipass), and the OTBX_Q9_BITMAP_BITS hook must shrink
    it further — if the two sides ever diverged, every q9 call at SF300+
    would fail with OTBX_ERR_INVALID on the GPU."""
    import ctypes as C

    def q9(nparts):
        out = C.c_size_t(0)
        st = so.otbx_q9_workspace_bytes(C.c_int64(nparts),
                                        C.c_int64(1_000_000),
                                        C.c_int64(4_000_000),
                                        C.c_uint32(1), C.byref(out))
        assert st == 0
        return out.value

    cap_bits = 1 << 30
    small = q9(cap_bits // 2)
    at_cap = q9(cap_bits)
    beyond = q9(4 * cap_bits)
    assert small < at_cap  # still single-pass: bitmap grows with nparts
    assert beyond == at_cap  # sliced: bitmap capped at the slice width
    monkeypatch.setenv("OTBX_Q9_BITMAP_BITS", "4096")
    assert q9(cap_bits) < small  # the test hook shrinks the bitmap
    monkeypatch.delenv("OTBX_Q9_BITMAP_BITS")
    assert q9(cap_bits) == at_cap
