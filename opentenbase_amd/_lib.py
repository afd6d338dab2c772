"""ctypes binding for libotbx.so (the C-ABI in include/otbx.h).

The .so is built in-tree (opentenbase_amd/csrc/Makefile, hipcc gfx950) and
travels with the repo snapshot. On a machine WITH a GPU, a missing or
unloadable extension is a hard error — the HIP path is the only compute path
(no CPU fallback; the oracle is test infrastructure only).
"""
import ctypes as C
import os
import subprocess

_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
_SO = os.path.join(_DIR, "libotbx.so")

OTBX_OK = 0
_STATUS = {0: "ok", 1: "hip runtime error", 2: "out of memory",
           3: "invalid argument", 4: "value out of range", 5: "no gpu"}


class OtbxError(RuntimeError):
    """Analog of ereport(ERROR): any non-OK status from the C-ABI."""

    def __init__(self, status, what=""):
        self.status = status
        super().__init__(f"otbx error {status} ({_STATUS.get(status, '?')}) {what}")


class LineitemDev(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("l_orderkey", C.c_void_p),
        ("l_quantity", C.c_void_p), ("l_extendedprice", C.c_void_p),
        ("l_discount", C.c_void_p), ("l_tax", C.c_void_p),
        ("l_returnflag", C.c_void_p), ("l_linestatus", C.c_void_p),
        ("l_shipdate", C.c_void_p),
        ("l_partkey", C.c_void_p),
        ("q9rec", C.c_void_p),
        ("l_orderkey32", C.c_void_p),
        ("l_partkey32", C.c_void_p),
    ]


class OrdersDev(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("o_orderkey", C.c_void_p), ("o_custkey", C.c_void_p),
        ("o_orderdate", C.c_void_p), ("o_shippriority", C.c_void_p),
        ("okey_min", C.c_int64), ("okey_max", C.c_int64),
        ("has_minmax", C.c_int32),
        ("o_orderkey32", C.c_void_p), ("o_custkey32", C.c_void_p),
    ]


class CustomerDev(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("c_custkey", C.c_void_p), ("c_mktsegment", C.c_void_p),
    ]


class KeysetDev(C.Structure):
    """otbx_keyset (include/otbx.h): up to 8 key columns (device ptrs)."""
    _fields_ = [("nkeys", C.c_int32),
                ("keys", C.c_void_p * 8),
                ("nulls", C.c_void_p * 8)]


class PartDev(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("p_partkey", C.c_void_p), ("p_type", C.c_void_p),
    ]


def build():
    """Compile the extension (hipcc cross-compiles without a GPU)."""
    subprocess.run(["make", "-C", _DIR], check=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            raise OtbxError(5, f"HIP extension not built: {_SO} missing — "
                               "run __graft_entry__.build()")
        _lib = C.CDLL(_SO)
        _lib.otbx_version.restype = C.c_char_p
        _lib.otbx_status_str.restype = C.c_char_p
        # argtypes for the pure-host sizing functions: a wrong arity from
        # a caller becomes a ctypes error instead of a segfault (the
        # compute entry points keep the explicit-ctypes call style)
        i64, u32, szp = C.c_int64, C.c_uint32, C.POINTER(C.c_size_t)
        _lib.otbx_agg_i64_workspace_bytes.argtypes = [i64, szp]
        _lib.otbx_join_i64_workspace_bytes.argtypes = [i64, i64, szp]
        _lib.otbx_order_groups_workspace_bytes.argtypes = [i64, szp]
        _lib.otbx_q3_workspace_bytes.argtypes = [i64, i64, i64, szp]
        _lib.otbx_q9_workspace_bytes.argtypes = [i64, i64, i64, u32, szp]
    return _lib


def check(status, what=""):
    if status != OTBX_OK:
        raise OtbxError(status, what)


def call(name, *args):
    check(getattr(lib(), name)(*args), what=name)


EXPORTED_SYMBOLS = [
    "otbx_version", "otbx_status_str", "otbx_init", "otbx_finish",
    "otbx_device_malloc", "otbx_device_free", "otbx_memcpy_h2d",
    "otbx_memcpy_d2h", "otbx_stream_sync",
    "otbx_gen_lineitem_dev", "otbx_gen_orders_dev", "otbx_gen_customer_dev",
    "otbx_gen_part_dev", "otbx_q9_workspace_bytes", "otbx_q9_partial",
    "otbx_stage_pages", "otbx_build_q9recs", "otbx_build_key32",
    "otbx_scan_count", "otbx_q1_partial", "otbx_q1_partial_variant",
    "otbx_q3_workspace_bytes", "otbx_q3_partial", "otbx_filter_customer",
    "otbx_topk_by_revenue",
    "otbx_order_groups_workspace_bytes", "otbx_order_groups",
    "otbx_agg_i64_workspace_bytes", "otbx_agg_i64",
    "otbx_partition_by_key", "otbx_gather_i64", "otbx_gather_f64",
    "otbx_gather_i32", "otbx_gather_u8",
    "otbx_join_i64_workspace_bytes", "otbx_join_i64",
    "otbx_join_ext_workspace_bytes", "otbx_join_i64_ext", "otbx_join_i64x2",
    "otbx_agg_i64x2_workspace_bytes", "otbx_agg_i64x2",
    "otbx_agg_i64_dec_workspace_bytes", "otbx_agg_i64_dec",
    "otbx_agg_i64n_workspace_bytes", "otbx_agg_i64n",
    "otbx_join_i64n_workspace_bytes", "otbx_join_i64n",
]
