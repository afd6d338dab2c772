"""ctypes binding for the CPU oracle (oracle/liboracle.so).

TEST INFRASTRUCTURE: only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this module. The product path (opentenbase_amd)
must never route through it.
"""
import ctypes as C
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


class OraQ1Group(C.Structure):
    _fields_ = [
        ("returnflag", C.c_uint8), ("linestatus", C.c_uint8),
        ("sum_qty", C.c_double), ("sum_base_price", C.c_double),
        ("sum_disc_price", C.c_double), ("sum_charge", C.c_double),
        ("qty_acc", C.c_double * 3), ("price_acc", C.c_double * 3),
        ("disc_acc", C.c_double * 3),
        ("count_order", C.c_int64),
        ("avg_qty", C.c_double), ("avg_price", C.c_double),
        ("avg_disc", C.c_double),
    ]


class OraQ3Row(C.Structure):
    _fields_ = [
        ("l_orderkey", C.c_int64), ("revenue", C.c_double),
        ("o_orderdate", C.c_int32), ("o_shippriority", C.c_int32),
    ]


class OraAggGroup(C.Structure):
    _fields_ = [
        ("key", C.c_int64), ("key_isnull", C.c_int),
        ("count_star", C.c_int64), ("count_v", C.c_int64),
        ("sum_v", C.c_double), ("sum_isnull", C.c_int),
        ("acc", C.c_double * 3),
    ]


class _Lineitem(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("l_orderkey", C.POINTER(C.c_int64)),
        ("l_quantity", C.POINTER(C.c_double)),
        ("l_extendedprice", C.POINTER(C.c_double)),
        ("l_discount", C.POINTER(C.c_double)),
        ("l_tax", C.POINTER(C.c_double)),
        ("l_returnflag", C.POINTER(C.c_uint8)),
        ("l_linestatus", C.POINTER(C.c_uint8)),
        ("l_shipdate", C.POINTER(C.c_int32)),
        ("l_partkey", C.POINTER(C.c_int64)),
    ]


class _Orders(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("o_orderkey", C.POINTER(C.c_int64)),
        ("o_custkey", C.POINTER(C.c_int64)),
        ("o_orderdate", C.POINTER(C.c_int32)),
        ("o_shippriority", C.POINTER(C.c_int32)),
    ]


class _Customer(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("c_custkey", C.POINTER(C.c_int64)),
        ("c_mktsegment", C.POINTER(C.c_uint8)),
    ]


class _Part(C.Structure):
    _fields_ = [
        ("n", C.c_int64),
        ("p_partkey", C.POINTER(C.c_int64)),
        ("p_type", C.POINTER(C.c_uint8)),
    ]


class OraQ9Group(C.Structure):
    _fields_ = [
        ("year", C.c_int32), ("revenue", C.c_double),
        ("count_rows", C.c_int64),
    ]


def _build():
    subprocess.run(["make", "-C", _DIR, "liboracle.so"], check=True,
                   capture_output=True)


def load():
    if not os.path.exists(_SO):
        _build()
    lib = C.CDLL(_SO)
    lib.ora_scan_count_shipdate_le.restype = C.c_int64
    lib.ora_scan_count_shipdate_le.argtypes = [C.POINTER(C.c_int32), C.c_int64, C.c_int32]
    lib.ora_q1_partial.restype = C.c_int
    lib.ora_q3_partial.restype = C.c_int
    lib.ora_q3_topk.restype = C.c_int64
    lib.ora_agg_i64.restype = C.c_int
    lib.ora_join_i64.restype = C.c_int
    lib.ora_join_ext.restype = C.c_int
    lib.ora_agg_i64x2.restype = C.c_int
    lib.ora_agg_i64_dec.restype = C.c_int
    lib.ora_agg_i64n.restype = C.c_int
    lib.ora_join_i64n.restype = C.c_int
    lib.ora_q1_combine.restype = C.c_int
    return lib


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = load()
    return _lib


def _p(arr, ctype):
    if arr is None:
        return None
    return arr.ctypes.data_as(C.POINTER(ctype))


def gen_tables(n_lineitem, rank=0, nranks=1, seed=42, need=("lineitem",),
               skew=False):
    """Generate tables via the oracle's generator; returns dict of numpy views
    plus the C structs (kept alive). n_lineitem is the GLOBAL row count."""
    L = lib()
    out = {}
    keep = []
    if "lineitem" in need:
        t = _Lineitem()
        st = L.ora_gen_lineitem(C.byref(t), C.c_uint64(seed), C.c_int64(n_lineitem),
                                rank, nranks)
        assert st == 0, st
        n = t.n
        out["lineitem"] = {
            "struct": t,
            "l_orderkey": np.ctypeslib.as_array(t.l_orderkey, (n,)),
            "l_quantity": np.ctypeslib.as_array(t.l_quantity, (n,)),
            "l_extendedprice": np.ctypeslib.as_array(t.l_extendedprice, (n,)),
            "l_discount": np.ctypeslib.as_array(t.l_discount, (n,)),
            "l_tax": np.ctypeslib.as_array(t.l_tax, (n,)),
            "l_returnflag": np.ctypeslib.as_array(t.l_returnflag, (n,)),
            "l_linestatus": np.ctypeslib.as_array(t.l_linestatus, (n,)),
            "l_shipdate": np.ctypeslib.as_array(t.l_shipdate, (n,)),
            "l_partkey": np.ctypeslib.as_array(t.l_partkey, (n,)),
        }
        keep.append(t)
    if "orders" in need:
        t = _Orders()
        st = L.ora_gen_orders(C.byref(t), C.c_uint64(seed), C.c_int64(n_lineitem // 4),
                              C.c_int64(n_lineitem // 40), rank, nranks,
                              1 if skew else 0)
        assert st == 0, st
        n = t.n
        out["orders"] = {
            "struct": t,
            "o_orderkey": np.ctypeslib.as_array(t.o_orderkey, (n,)),
            "o_custkey": np.ctypeslib.as_array(t.o_custkey, (n,)),
            "o_orderdate": np.ctypeslib.as_array(t.o_orderdate, (n,)),
            "o_shippriority": np.ctypeslib.as_array(t.o_shippriority, (n,)),
        }
    if "part" in need:
        t = _Part()
        st = L.ora_gen_part(C.byref(t), C.c_uint64(seed),
                            C.c_int64(max(n_lineitem // 30, 1)))
        assert st == 0, st
        n = t.n
        out["part"] = {
            "struct": t,
            "p_partkey": np.ctypeslib.as_array(t.p_partkey, (n,)),
            "p_type": np.ctypeslib.as_array(t.p_type, (n,)),
        }
    if "customer" in need:
        t = _Customer()
        st = L.ora_gen_customer(C.byref(t), C.c_uint64(seed),
                                C.c_int64(n_lineitem // 40), rank, nranks)
        assert st == 0, st
        n = t.n
        out["customer"] = {
            "struct": t,
            "c_custkey": np.ctypeslib.as_array(t.c_custkey, (n,)),
            "c_mktsegment": np.ctypeslib.as_array(t.c_mktsegment, (n,)),
        }
    return out


def q1_partial(tables, cutoff=2436):
    L = lib()
    g = (OraQ1Group * 8)()
    ng = C.c_int(0)
    st = L.ora_q1_partial(C.byref(tables["lineitem"]["struct"]), C.c_int32(cutoff),
                          g, C.byref(ng))
    assert st == 0, st
    return [g[i] for i in range(ng.value)]


def q1_combine(parts_lists):
    """parts_lists: list of lists of OraQ1Group (one per shard)."""
    L = lib()
    acc = (OraQ1Group * 8)()
    nacc = C.c_int(0)
    for parts in parts_lists:
        arr = (OraQ1Group * len(parts))(*parts)
        st = L.ora_q1_combine(acc, C.byref(nacc), arr, len(parts))
        assert st == 0, st
    return [acc[i] for i in range(nacc.value)]


def q1_finalize(groups):
    L = lib()
    arr = (OraQ1Group * len(groups))(*groups)
    L.ora_q1_finalize(arr, len(groups))
    return list(arr)


def q9_partial(tables, typemod=17, typeval=0):
    L = lib()
    g = (OraQ9Group * 8)()
    ng = C.c_int(0)
    st = L.ora_q9_partial(C.byref(tables["part"]["struct"]),
                          C.byref(tables["orders"]["struct"]),
                          C.byref(tables["lineitem"]["struct"]),
                          C.c_uint8(typemod), C.c_uint8(typeval),
                          g, C.byref(ng))
    assert st == 0, st
    return [g[i] for i in range(ng.value)]


def q3_partial(tables, segment=0, date=1169):
    L = lib()
    out = C.POINTER(OraQ3Row)()
    ng = C.c_int64(0)
    st = L.ora_q3_partial(C.byref(tables["customer"]["struct"]),
                          C.byref(tables["orders"]["struct"]),
                          C.byref(tables["lineitem"]["struct"]),
                          C.c_uint8(segment), C.c_int32(date),
                          C.byref(out), C.byref(ng))
    assert st == 0, st
    n = ng.value
    rows = np.empty(n, dtype=[("l_orderkey", "i8"), ("revenue", "f8"),
                              ("o_orderdate", "i4"), ("o_shippriority", "i4")])
    if n:
        C.memmove(rows.ctypes.data, out, n * C.sizeof(OraQ3Row))
    return rows


def q3_topk(rows, k=10):
    order = np.lexsort((rows["l_orderkey"], rows["o_orderdate"], -rows["revenue"]))
    return rows[order][:k]


def scan_count(shipdate, cutoff=2436):
    shipdate = np.ascontiguousarray(shipdate, dtype=np.int32)
    return lib().ora_scan_count_shipdate_le(_p(shipdate, C.c_int32),
                                            len(shipdate), cutoff)


def agg_i64(keys, vals, key_null=None, val_null=None):
    L = lib()
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    kn = None if key_null is None else np.ascontiguousarray(key_null, dtype=np.uint8)
    vn = None if val_null is None else np.ascontiguousarray(val_null, dtype=np.uint8)
    out = C.POINTER(OraAggGroup)()
    ng = C.c_int64(0)
    st = L.ora_agg_i64(_p(keys, C.c_int64), _p(kn, C.c_uint8),
                       _p(vals, C.c_double), _p(vn, C.c_uint8),
                       len(keys), C.byref(out), C.byref(ng))
    assert st == 0, st
    return [out[i] for i in range(ng.value)]


def join_i64(bkeys, pkeys, bnull=None, pnull=None):
    L = lib()
    bkeys = np.ascontiguousarray(bkeys, dtype=np.int64)
    pkeys = np.ascontiguousarray(pkeys, dtype=np.int64)
    bn = None if bnull is None else np.ascontiguousarray(bnull, dtype=np.uint8)
    pn = None if pnull is None else np.ascontiguousarray(pnull, dtype=np.uint8)
    ob = C.POINTER(C.c_int64)()
    op = C.POINTER(C.c_int64)()
    n = C.c_int64(0)
    st = L.ora_join_i64(_p(bkeys, C.c_int64), _p(bn, C.c_uint8), len(bkeys),
                        _p(pkeys, C.c_int64), _p(pn, C.c_uint8), len(pkeys),
                        C.byref(ob), C.byref(op), C.byref(n))
    assert st == 0, st
    nn = n.value
    bi = np.ctypeslib.as_array(ob, (nn,)).copy() if nn else np.empty(0, np.int64)
    pi = np.ctypeslib.as_array(op, (nn,)).copy() if nn else np.empty(0, np.int64)
    return bi, pi


class OraAggGroup2(C.Structure):
    _fields_ = [
        ("key1", C.c_int64), ("key2", C.c_int64),
        ("key1_isnull", C.c_int), ("key2_isnull", C.c_int),
        ("count_star", C.c_int64), ("count_v", C.c_int64),
        ("sum_v", C.c_double), ("sum_isnull", C.c_int),
        ("acc", C.c_double * 3),
    ]


def join_ext(bkeys, pkeys, join_type, bnull=None, pnull=None,
             bkeys2=None, pkeys2=None, bnull2=None, pnull2=None):
    """Extended join (oracle.h ora_join_ext): join_type 0 inner / 1 left /
    2 semi / 3 anti / 4 right / 5 full; optional second key column."""
    L = lib()
    bkeys = np.ascontiguousarray(bkeys, dtype=np.int64)
    pkeys = np.ascontiguousarray(pkeys, dtype=np.int64)
    arrs = {}
    for nm, a, dt in [("bn", bnull, np.uint8), ("pn", pnull, np.uint8),
                      ("bk2", bkeys2, np.int64), ("pk2", pkeys2, np.int64),
                      ("bn2", bnull2, np.uint8), ("pn2", pnull2, np.uint8)]:
        arrs[nm] = None if a is None else np.ascontiguousarray(a, dtype=dt)
    ob = C.POINTER(C.c_int64)()
    op = C.POINTER(C.c_int64)()
    n = C.c_int64(0)
    st = L.ora_join_ext(
        _p(bkeys, C.c_int64), _p(arrs["bn"], C.c_uint8),
        _p(arrs["bk2"], C.c_int64), _p(arrs["bn2"], C.c_uint8),
        C.c_int64(len(bkeys)),
        _p(pkeys, C.c_int64), _p(arrs["pn"], C.c_uint8),
        _p(arrs["pk2"], C.c_int64), _p(arrs["pn2"], C.c_uint8),
        C.c_int64(len(pkeys)), C.c_int(join_type),
        C.byref(ob), C.byref(op), C.byref(n))
    assert st == 0, st
    nn = n.value
    bi = np.ctypeslib.as_array(ob, (nn,)).copy() if nn else np.empty(0, np.int64)
    pi = np.ctypeslib.as_array(op, (nn,)).copy() if nn else np.empty(0, np.int64)
    return bi, pi


def agg_i64x2(k1, k2, vals, k1null=None, k2null=None, val_null=None):
    """Two-key hash aggregate (oracle.h ora_agg_i64x2); returns a list of
    OraAggGroup2 sorted by (k1_isnull, k1, k2_isnull, k2)."""
    L = lib()
    k1 = np.ascontiguousarray(k1, dtype=np.int64)
    k2 = np.ascontiguousarray(k2, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    n1 = None if k1null is None else np.ascontiguousarray(k1null, np.uint8)
    n2 = None if k2null is None else np.ascontiguousarray(k2null, np.uint8)
    vn = None if val_null is None else np.ascontiguousarray(val_null, np.uint8)
    out = C.POINTER(OraAggGroup2)()
    ng = C.c_int64(0)
    st = L.ora_agg_i64x2(_p(k1, C.c_int64), _p(n1, C.c_uint8),
                         _p(k2, C.c_int64), _p(n2, C.c_uint8),
                         _p(vals, C.c_double), _p(vn, C.c_uint8),
                         C.c_int64(len(k1)), C.byref(out), C.byref(ng))
    assert st == 0, st
    return [out[i] for i in range(ng.value)]


class OraDecGroup(C.Structure):
    _fields_ = [
        ("key", C.c_int64), ("key_isnull", C.c_int),
        ("count_star", C.c_int64), ("count_v", C.c_int64),
        ("sum_hi", C.c_int64), ("sum_lo", C.c_uint64),
        ("sum_isnull", C.c_int),
    ]

    @property
    def sum128(self):
        return (self.sum_hi << 64) | self.sum_lo


def agg_i64_dec(keys, vals, key_null=None, val_null=None):
    """Exact int128 decimal aggregate (oracle.h ora_agg_i64_dec); returns
    OraDecGroup list sorted by (key_isnull, key); .sum128 is the exact sum."""
    L = lib()
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.int64)
    kn = None if key_null is None else np.ascontiguousarray(key_null, np.uint8)
    vn = None if val_null is None else np.ascontiguousarray(val_null, np.uint8)
    out = C.POINTER(OraDecGroup)()
    ng = C.c_int64(0)
    st = L.ora_agg_i64_dec(_p(keys, C.c_int64), _p(kn, C.c_uint8),
                           _p(vals, C.c_int64), _p(vn, C.c_uint8),
                           C.c_int64(len(keys)), C.byref(out), C.byref(ng))
    assert st == 0, st
    return [out[i] for i in range(ng.value)]


class OraKeyset(C.Structure):
    _fields_ = [("nkeys", C.c_int),
                ("keys", C.POINTER(C.c_int64) * 8),
                ("nulls", C.POINTER(C.c_uint8) * 8)]


class OraAggNGroup(C.Structure):
    _fields_ = [("row_idx", C.c_int64), ("count_star", C.c_int64),
                ("count_v", C.c_int64), ("sum_v", C.c_double),
                ("sum_isnull", C.c_int)]


def _keyset(key_cols, null_cols):
    """key_cols: list of int64 arrays; null_cols: list (entries may be
    None). Returns (OraKeyset, keepalive-list)."""
    ks = OraKeyset()
    ks.nkeys = len(key_cols)
    keep = []
    for c, k in enumerate(key_cols):
        a = np.ascontiguousarray(k, dtype=np.int64)
        keep.append(a)
        ks.keys[c] = a.ctypes.data_as(C.POINTER(C.c_int64))
        nc = None if null_cols is None else null_cols[c]
        if nc is not None:
            na = np.ascontiguousarray(nc, dtype=np.uint8)
            keep.append(na)
            ks.nulls[c] = na.ctypes.data_as(C.POINTER(C.c_uint8))
    return ks, keep


def agg_i64n(key_cols, vals, null_cols=None, val_null=None):
    """N-key group-by (oracle.h ora_agg_i64n): groups carry the defining
    ROW INDEX (representative tuple); sorted by row_idx."""
    L = lib()
    ks, keep = _keyset(key_cols, null_cols)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    vn = None if val_null is None else np.ascontiguousarray(val_null, np.uint8)
    out = C.POINTER(OraAggNGroup)()
    ng = C.c_int64(0)
    st = L.ora_agg_i64n(C.byref(ks), _p(vals, C.c_double), _p(vn, C.c_uint8),
                        C.c_int64(len(vals)), C.byref(out), C.byref(ng))
    assert st == 0, st
    return [out[i] for i in range(ng.value)]


def join_i64n(bkey_cols, pkey_cols, join_type, bnull_cols=None,
              pnull_cols=None):
    L = lib()
    bks, keepb = _keyset(bkey_cols, bnull_cols)
    pks, keepp = _keyset(pkey_cols, pnull_cols)
    nb = len(bkey_cols[0]) if len(bkey_cols) else 0
    npr = len(pkey_cols[0]) if len(pkey_cols) else 0
    ob = C.POINTER(C.c_int64)()
    op = C.POINTER(C.c_int64)()
    n = C.c_int64(0)
    st = L.ora_join_i64n(C.byref(bks), C.c_int64(nb), C.byref(pks),
                         C.c_int64(npr), C.c_int(join_type),
                         C.byref(ob), C.byref(op), C.byref(n))
    assert st == 0, st
    nn = n.value
    bi = np.ctypeslib.as_array(ob, (nn,)).copy() if nn else np.empty(0, np.int64)
    pi = np.ctypeslib.as_array(op, (nn,)).copy() if nn else np.empty(0, np.int64)
    return bi, pi
