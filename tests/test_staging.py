"""Heap-page staging shim tests (otbx_stage_pages, csrc/staging.c): build
PostgreSQL-format heap pages in Python following the public on-disk layout
(PageHeaderData bufpage.h:157, ItemIdData itemid.h:25, HeapTupleHeaderData
htup_details.h:118) and verify the walker deforms them into the same SoA
columns the synthetic generator produces — the provider's BeginCustomScan
staging step, host-side, no GPU needed."""
import ctypes as C
import struct

import numpy as np
import pytest

PAGE = 8192
LP_UNUSED, LP_NORMAL, LP_REDIRECT, LP_DEAD = 0, 1, 2, 3
HEAP_HASNULL = 0x0001


class AttDesc(C.Structure):
    _fields_ = [("attlen", C.c_uint16), ("attalign", C.c_uint16)]


def _lib():
    from opentenbase_amd._lib import lib
    L = lib()
    L.otbx_stage_pages.restype = C.c_int32
    return L


def pack_tuple(values, atts, null_mask=None, natts=None):
    """values: list of (bytes for each non-null attr or None); atts:
    [(len, align)]. Returns the tuple bytes (header + bitmap + data)."""
    n = len(atts) if natts is None else natts
    has_null = null_mask is not None and any(null_mask)
    bitmap = b""
    if has_null:
        nb = (n + 7) // 8
        bits = 0
        for a in range(n):
            if not null_mask[a]:
                bits |= 1 << a  # bit SET = not null (htup_details.h:76)
        bitmap = bits.to_bytes(nb, "little")
    t_hoff = 23 + len(bitmap)
    t_hoff = (t_hoff + 7) & ~7  # MAXALIGN
    data = b""
    off = t_hoff
    for a in range(n):
        if has_null and null_mask[a]:
            continue
        ln, al = atts[a]
        pad = (-off) % al
        data += b"\x00" * pad
        off += pad
        data += values[a]
        off += ln
    infomask2 = n
    infomask = HEAP_HASNULL if has_null else 0
    hdr = struct.pack("<IIIHHHHHB", 2, 0, 0, 0, 0, 0, infomask2, infomask,
                      t_hoff)
    # header is 23 bytes: xmin 4, xmax 4, cid 4, ctid 6 (packed as H H H),
    # infomask2 2, infomask 2, hoff 1  -> the struct above emits 4+4+4+2+2+2
    # +2+2+1 = 23
    assert len(hdr) == 23
    pad = b"\x00" * (t_hoff - 23 - len(bitmap))
    return hdr + bitmap + pad + data


def pack_page(tuples, flags=None):
    """tuples: list of tuple byte strings; flags: per-item lp_flags
    (default LP_NORMAL). Items with non-NORMAL flags get lp_off=0,len=0."""
    nitems = len(tuples)
    flags = flags or [LP_NORMAL] * nitems
    item_ids = []
    body = bytearray(PAGE)
    upper = PAGE
    for t, f in zip(tuples, flags):
        if f != LP_NORMAL:
            item_ids.append(0 | (f << 15) | 0)
            continue
        ln = len(t)
        upper -= ln
        upper &= ~7  # MAXALIGN tuple starts
        body[upper:upper + ln] = t
        item_ids.append(upper | (f << 15) | (ln << 17))
    lower = 24 + 4 * nitems
    assert lower <= upper
    struct.pack_into("<QHHHHHH", body, 0, 0, 0, 0, lower, upper, PAGE,
                     PAGE | 4)
    struct.pack_into("<I", body, 20, 0)
    for i, lp in enumerate(item_ids):
        struct.pack_into("<I", body, 24 + 4 * i, lp)
    return bytes(body)


LI_ATTS = [(8, 8), (8, 8), (8, 8), (8, 8), (8, 8), (1, 1), (1, 1), (4, 4)]
LI_COLS = ["l_orderkey", "l_quantity", "l_extendedprice", "l_discount",
           "l_tax", "l_returnflag", "l_linestatus", "l_shipdate"]
LI_NP = [np.int64, np.float64, np.float64, np.float64, np.float64, np.uint8,
         np.uint8, np.int32]


def lineitem_pages(t, n, rows_per_page=64):
    """Pack oracle-generated lineitem rows into heap pages."""
    li = t["lineitem"]
    pages = []
    for lo in range(0, n, rows_per_page):
        tups = []
        for i in range(lo, min(lo + rows_per_page, n)):
            vals = [np.int64(li["l_orderkey"][i]).tobytes(),
                    np.float64(li["l_quantity"][i]).tobytes(),
                    np.float64(li["l_extendedprice"][i]).tobytes(),
                    np.float64(li["l_discount"][i]).tobytes(),
                    np.float64(li["l_tax"][i]).tobytes(),
                    bytes([li["l_returnflag"][i]]),
                    bytes([li["l_linestatus"][i]]),
                    np.int32(li["l_shipdate"][i]).tobytes()]
            tups.append(pack_tuple(vals, LI_ATTS))
        pages.append(pack_page(tups))
    return b"".join(pages), len(pages)


def stage(pages_bytes, npages, atts, ncols, cap, with_nulls=False):
    L = _lib()
    adesc = (AttDesc * ncols)(*[AttDesc(l, a) for l, a in atts])
    cols = [(C.c_uint8 * (cap * atts[a][0]))() for a in range(ncols)]
    colp = (C.c_void_p * ncols)(*[C.cast(c, C.c_void_p) for c in cols])
    nulls = None
    nullp = None
    if with_nulls:
        nulls = [(C.c_uint8 * cap)() for _ in range(ncols)]
        nullp = (C.POINTER(C.c_uint8) * ncols)(
            *[C.cast(x, C.POINTER(C.c_uint8)) for x in nulls])
    nrows = C.c_int64(0)
    st = L.otbx_stage_pages(C.c_char_p(pages_bytes), C.c_int64(npages),
                            C.c_size_t(PAGE), adesc, C.c_int32(ncols), colp,
                            nullp, C.c_int64(cap), C.byref(nrows))
    return st, nrows.value, cols, nulls


def test_lineitem_page_roundtrip():
    from oracle import oracle_py as ora
    n = 4000
    t = ora.gen_tables(n)
    pages, npages = lineitem_pages(t, n)
    st, nrows, cols, _ = stage(pages, npages, LI_ATTS, 8, n)
    assert st == 0
    assert nrows == n
    for a, (name, dt) in enumerate(zip(LI_COLS, LI_NP)):
        got = np.frombuffer(bytes(cols[a])[: n * LI_ATTS[a][0]], dtype=dt)
        assert np.array_equal(got.view(np.uint8),
                              t["lineitem"][name][:n].view(np.uint8)), name


def test_dead_and_unused_items_skipped():
    atts = [(8, 8), (4, 4)]
    tups = [pack_tuple([np.int64(k).tobytes(), np.int32(k * 10).tobytes()],
                       atts) for k in range(5)]
    page = pack_page(tups, flags=[LP_NORMAL, LP_DEAD, LP_NORMAL, LP_UNUSED,
                                  LP_REDIRECT])
    st, nrows, cols, _ = stage(page, 1, atts, 2, 10)
    assert st == 0
    assert nrows == 2
    keys = np.frombuffer(bytes(cols[0])[:16], dtype=np.int64)
    assert keys.tolist() == [0, 2]


def test_null_bitmap():
    atts = [(8, 8), (8, 8), (4, 4)]
    tups = [
        pack_tuple([np.int64(1).tobytes(), np.float64(1.5).tobytes(),
                    np.int32(7).tobytes()], atts),
        pack_tuple([np.int64(2).tobytes(), None, np.int32(9).tobytes()],
                   atts, null_mask=[False, True, False]),
    ]
    page = pack_page(tups)
    st, nrows, cols, nulls = stage(page, 1, atts, 3, 4, with_nulls=True)
    assert st == 0 and nrows == 2
    v = np.frombuffer(bytes(cols[1])[:16], dtype=np.float64)
    assert v[0] == 1.5 and v[1] == 0.0  # NULL slot zeroed
    assert list(nulls[1][:2]) == [0, 1]
    assert list(nulls[0][:2]) == [0, 0]
    k = np.frombuffer(bytes(cols[0])[:16], dtype=np.int64)
    assert k.tolist() == [1, 2]  # attrs after a NULL shift left in the tuple
    c = np.frombuffer(bytes(cols[2])[:8], dtype=np.int32)
    assert c.tolist() == [7, 9]


def test_null_without_buffer_is_error():
    atts = [(8, 8)]
    page = pack_page([pack_tuple([None], atts, null_mask=[True])])
    st, _, _, _ = stage(page, 1, atts, 1, 4, with_nulls=False)
    assert st == 3  # OTBX_ERR_INVALID: caller declared NOT NULL


def test_dropped_column_trailing_null():
    # tuple written before a column was added: natts=1 < descriptor's 2
    atts = [(8, 8), (4, 4)]
    tup = pack_tuple([np.int64(5).tobytes()], [(8, 8)], natts=1)
    page = pack_page([tup])
    st, nrows, cols, nulls = stage(page, 1, atts, 2, 4, with_nulls=True)
    assert st == 0 and nrows == 1
    assert list(nulls[1][:1]) == [1]  # heap_getattr: missing attr is NULL
    assert np.frombuffer(bytes(cols[0])[:8], dtype=np.int64)[0] == 5


def test_capacity_exceeded_is_error():
    atts = [(8, 8)]
    page = pack_page([pack_tuple([np.int64(k).tobytes()], atts)
                      for k in range(4)])
    st, _, _, _ = stage(page, 1, atts, 1, 2)
    assert st == 3


def test_staging_fuzz_mixed_pages():
    """Randomized pages: variable tuples per page, random dead/unused
    items, random NULL masks, a dropped-column tuple — the walker's output
    must equal a straightforward Python reconstruction."""
    rng = np.random.default_rng(23)
    atts = [(8, 8), (4, 4), (8, 8), (1, 1)]
    ncols = len(atts)
    pages = []
    exp_rows = []
    for _ in range(20):
        tups, flags = [], []
        for _ in range(int(rng.integers(0, 40))):
            flag = int(rng.choice([LP_NORMAL, LP_NORMAL, LP_NORMAL,
                                   LP_DEAD, LP_UNUSED]))
            nm = [bool(rng.random() < 0.2) for _ in range(ncols)]
            vals_raw = [int(rng.integers(-2**40, 2**40)),
                        int(rng.integers(-2**20, 2**20)),
                        int(rng.integers(0, 2**30)),
                        int(rng.integers(0, 256))]
            packs = [np.int64(vals_raw[0]).tobytes(),
                     np.int32(vals_raw[1]).tobytes(),
                     np.int64(vals_raw[2]).tobytes(),
                     bytes([vals_raw[3]])]
            packed = [None if nm[a] else packs[a] for a in range(ncols)]
            tups.append(pack_tuple(packed, atts,
                                   null_mask=nm if any(nm) else None))
            flags.append(flag)
            if flag == LP_NORMAL:
                exp_rows.append((vals_raw, nm))
        pages.append(pack_page(tups, flags=flags))
    blob = b"".join(pages)
    cap = len(exp_rows) + 8
    st, nrows, cols, nulls = stage(blob, len(pages), atts, ncols, cap,
                                   with_nulls=True)
    assert st == 0
    assert nrows == len(exp_rows)
    dts = [np.int64, np.int32, np.int64, np.uint8]
    for a in range(ncols):
        got = np.frombuffer(bytes(cols[a])[: nrows * atts[a][0]],
                            dtype=dts[a])
        for r, (vals_raw, nm) in enumerate(exp_rows):
            assert nulls[a][r] == (1 if nm[a] else 0), (a, r)
            if not nm[a]:
                assert int(got[r]) == int(dts[a](vals_raw[a])), (a, r)


@pytest.mark.gpu
def test_staged_pages_q1_parity():
    """End to end: heap pages → walker → from_host staging → fused Q1 kernel
    == oracle on the same rows (the full provider staging flow)."""
    import torch  # noqa: F401
    from opentenbase_amd import executor as ex
    from oracle import oracle_py as ora
    ex.init_device(0)
    n = 40000
    t = ora.gen_tables(n)
    pages, npages = lineitem_pages(t, n)
    st, nrows, cols, _ = stage(pages, npages, LI_ATTS, 8, n)
    assert st == 0 and nrows == n
    host_cols = {name: np.frombuffer(bytes(cols[a])[: n * LI_ATTS[a][0]],
                                     dtype=dt)
                 for a, (name, dt) in enumerate(zip(LI_COLS, LI_NP))}
    li = ex.GpuLineitem.from_host(host_cols, with_orderkey=False)
    node = ex.GpuQ1PartialAgg(li)
    node.BeginCustomScan()
    rows = node._run()
    exp = ora.q1_partial(t)
    assert len(rows) == len(exp)
    for r, e in zip(rows, exp):
        assert r["count_order"] == e.count_order
        assert abs(r["sum_qty"] - e.sum_qty) <= 1e-9 * abs(e.sum_qty)
        assert abs(r["sum_charge"] - e.sum_charge) <= 1e-9 * abs(e.sum_charge)


def test_null_bitmap_must_fit_under_hoff():
    """Malformed tuple: HEAP_HASNULL set but t_hoff leaves no room for the
    bitmap — the walker must reject it (ST_ERR_INVALID), not read past the
    tuple (a 1..natts/8-byte OOB read before the fix, past the caller's
    buffer when the tuple ends the last page)."""
    atts = [(8, 8), (4, 4)]
    # build a VALID 2-col tuple with a null, then corrupt t_hoff down to 23
    tup = bytearray(pack_tuple([np.int64(7).tobytes(), None], atts,
                               null_mask=[False, True]))
    assert tup[22] == 24  # 23 + 1 bitmap byte, MAXALIGNed
    tup[22] = 23          # t_hoff no longer covers the bitmap
    page = pack_page([bytes(tup)])
    st, nrows, _, _ = stage(page, 1, atts, 2, 8, with_nulls=True)
    assert st == 3  # OTBX_ERR_INVALID
    # the uncorrupted page stages fine
    page_ok = pack_page([pack_tuple([np.int64(7).tobytes(), None], atts,
                                    null_mask=[False, True])])
    st, nrows, cols, nulls = stage(page_ok, 1, atts, 2, 8, with_nulls=True)
    assert st == 0 and nrows == 1 and nulls[1][0] == 1


def test_item_pointer_into_header_rejected():
    """Malformed line pointer aiming into the page header / ItemId array
    (lp_off < pd_upper) must be rejected — PageGetItemIdCareful analog."""
    atts = [(8, 8)]
    page = bytearray(pack_page([pack_tuple([np.int64(1).tobytes()], atts)]))
    # corrupt item 0's lp_off to 24 (inside the ItemId area)
    lp = struct.unpack_from("<I", page, 24)[0]
    lp = (lp & ~0x7FFF) | 24
    struct.pack_into("<I", page, 24, lp)
    st, _, _, _ = stage(bytes(page), 1, atts, 1, 8)
    assert st == 3


def test_garbage_pages_never_crash():
    """Pure adversarial input: random bytes as pages. The walker must
    return a status (OK with plausible rows, or ERR_INVALID) without
    crashing or writing outside the declared buffers — the provider may
    hand it torn or corrupt pages and the shim is the last line before
    a device upload. (hypothesis-style, fixed seeds for determinism)"""
    atts = [(8, 8), (4, 4), (1, 1)]
    rng = np.random.default_rng(99)
    for trial in range(200):
        npg = int(rng.integers(1, 3))
        blob = rng.integers(0, 256, npg * PAGE, dtype=np.uint8)
        if trial % 3 == 0:
            # plant a plausible header so the item walk engages
            lower = int(rng.integers(0, 200))
            upper = int(rng.integers(0, PAGE + 200))
            struct_bytes = bytearray(blob[:PAGE].tobytes())
            import struct as _s
            _s.pack_into("<HH", struct_bytes, 12, lower, upper)
            blob[:PAGE] = np.frombuffer(bytes(struct_bytes), dtype=np.uint8)
        cap = 4096
        st, nrows, _, _ = stage(blob.tobytes(), npg, atts, 3, cap,
                                with_nulls=True)
        assert st in (0, 3)
        if st == 0:
            assert 0 <= nrows <= cap
