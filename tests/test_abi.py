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
