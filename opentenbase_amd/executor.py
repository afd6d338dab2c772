"""Host-side mirror of the reference's CustomScan provider contract.

The reference drives a custom executor node through CustomExecMethods
(include/nodes/extensible.h:117-152; nodeCustom.c:31-125):
BeginCustomScan(estate, eflags) → repeated ExecCustomScan(node) returning one
tuple per call (None = done) → ReScanCustomScan → EndCustomScan. The classes
here keep those names, argument meanings and the error convention (OtbxError ↔
ereport(ERROR)) so the parity tests read like the reference's node lifecycle;
INTEGRATION.md shows the C provider these map onto.

Device memory, streams: torch (plumbing only — all compute is in libotbx.so;
there is NO torch/CPU fallback for any operator here).
"""
import ctypes as C

import torch

from ._lib import (CustomerDev, KeysetDev, LineitemDev, OrdersDev,
                   OtbxError, PartDev, call, check, lib)

Q1_SLOT_ORDER = [(b"A", b"F"), (b"A", b"O"), (b"N", b"F"),
                 (b"N", b"O"), (b"R", b"F"), (b"R", b"O")]
Q1_CUTOFF_DEFAULT = 2436   # date '1998-12-01' - interval '90 day'
Q3_DATE_DEFAULT = 1169     # date '1995-03-15'
SEED_DEFAULT = 42


def _stream():
    return C.c_void_p(torch.cuda.current_stream().cuda_stream)


def init_device(device=0):
    torch.cuda.set_device(device)
    call("otbx_init", device)


def device_synchronize():
    torch.cuda.synchronize()


# ---------------- staged tables (device-resident column cache) -------------

def _build_key32(src_tensor):
    """Staging-time compact-key cache (otbx.h otbx_build_key32): int32 copy
    of an i64 key column, or None if any key falls outside [0, 2^31)."""
    n = src_tensor.numel()
    dst = torch.empty(n, dtype=torch.int32, device="cuda")
    ok = C.c_int32(0)
    call("otbx_build_key32", C.c_void_p(src_tensor.data_ptr()), C.c_int64(n),
         C.c_void_p(dst.data_ptr()), C.byref(ok), _stream())
    return dst if ok.value else None


class GpuLineitem:
    COLS = [("l_orderkey", torch.int64), ("l_quantity", torch.float64),
            ("l_extendedprice", torch.float64), ("l_discount", torch.float64),
            ("l_tax", torch.float64), ("l_returnflag", torch.uint8),
            ("l_linestatus", torch.uint8), ("l_shipdate", torch.int32),
            ("l_partkey", torch.int64)]

    def __init__(self, n, with_orderkey=True, with_partkey=False,
                 device="cuda"):
        self.n = n
        self.t = {}
        for name, dt in self.COLS:
            if (name == "l_orderkey" and not with_orderkey) or                (name == "l_partkey" and not with_partkey):
                self.t[name] = None
                continue
            self.t[name] = torch.empty(n, dtype=dt, device=device)
        self.cstruct = LineitemDev(
            n=n, **{k: C.c_void_p(0 if v is None else v.data_ptr())
                    for k, v in self.t.items()})

    @classmethod
    def generate(cls, n_global, rank=0, nranks=1, seed=SEED_DEFAULT,
                 with_orderkey=True, with_partkey=False):
        t = cls(n_global // nranks, with_orderkey=with_orderkey,
                with_partkey=with_partkey)
        call("otbx_gen_lineitem_dev", C.byref(t.cstruct), C.c_uint64(seed),
             C.c_int64(n_global), C.c_uint32(rank), C.c_uint32(nranks),
             _stream())
        t._stage_caches()
        return t

    @classmethod
    def from_host(cls, cols, with_orderkey=True, with_partkey=False):
        """The real otbx_stage_table flow: host columnar
        arrays (the provider's heap→SoA staging output) → device column
        cache via otbx_memcpy_h2d. cols: dict of numpy arrays keyed like
        COLS. Returns the staged table; PCIe-inclusive, one-time."""
        import numpy as np
        n = len(cols["l_shipdate"])
        t = cls(n, with_orderkey=with_orderkey, with_partkey=with_partkey)
        for name, dt in cls.COLS:
            if t.t[name] is None:
                continue
            src = np.ascontiguousarray(cols[name])
            assert src.nbytes == t.t[name].numel() * t.t[name].element_size()
            call("otbx_memcpy_h2d", C.c_void_p(t.t[name].data_ptr()),
                 src.ctypes.data_as(C.c_void_p), C.c_size_t(src.nbytes),
                 _stream())
        call("otbx_stream_sync", _stream())  # host buffers may be freed
        t._stage_caches()
        return t

    def _stage_caches(self):
        """Staging-time derived layouts (built once, outside any timed
        region, like the zone-map metadata): the int32 compact-key cache
        (otbx.h l_orderkey32) and, when partkey is staged, the Q9
        probe-side AoS record cache (otbx.h q9rec)."""
        if self.t.get("l_orderkey") is not None:
            self._okey32 = _build_key32(self.t["l_orderkey"])
            if self._okey32 is not None:
                self.cstruct.l_orderkey32 = C.c_void_p(
                    self._okey32.data_ptr())
        if self.t.get("l_partkey") is not None:
            self._pkey32 = _build_key32(self.t["l_partkey"])
            if self._pkey32 is not None:
                self.cstruct.l_partkey32 = C.c_void_p(
                    self._pkey32.data_ptr())
        if self.t.get("l_partkey") is None or self.t.get("l_orderkey") is None:
            return
        self._q9rec = torch.empty(self.n * 32, dtype=torch.uint8,
                                  device="cuda")
        self.cstruct.q9rec = C.c_void_p(self._q9rec.data_ptr())
        call("otbx_build_q9recs", C.byref(self.cstruct),
             C.c_void_p(self._q9rec.data_ptr()), _stream())

    def bytes_staged(self):
        extra = self._q9rec.numel() if getattr(self, "_q9rec", None) is not None else 0
        return extra + sum(v.numel() * v.element_size()
                           for v in self.t.values() if v is not None)


class GpuOrders:
    def __init__(self, n, device="cuda"):
        self.n = n
        self.t = {
            "o_orderkey": torch.empty(n, dtype=torch.int64, device=device),
            "o_custkey": torch.empty(n, dtype=torch.int64, device=device),
            "o_orderdate": torch.empty(n, dtype=torch.int32, device=device),
            "o_shippriority": torch.empty(n, dtype=torch.int32, device=device),
        }
        self.cstruct = OrdersDev(
            n=n, **{k: C.c_void_p(v.data_ptr()) for k, v in self.t.items()})

    @classmethod
    def generate(cls, n_global, ncust_global, rank=0, nranks=1,
                 seed=SEED_DEFAULT, skew=False):
        t = cls(n_global // nranks)
        call("otbx_gen_orders_dev", C.byref(t.cstruct), C.c_uint64(seed),
             C.c_int64(n_global), C.c_int64(ncust_global), C.c_uint32(rank),
             C.c_uint32(nranks), C.c_int(1 if skew else 0), _stream())
        t._stage_key32()
        return t

    def _stage_key32(self):
        self._okey32 = _build_key32(self.t["o_orderkey"])
        self._ckey32 = _build_key32(self.t["o_custkey"])
        if self._okey32 is not None:
            self.cstruct.o_orderkey32 = C.c_void_p(self._okey32.data_ptr())
        if self._ckey32 is not None:
            self.cstruct.o_custkey32 = C.c_void_p(self._ckey32.data_ptr())


class GpuCustomer:
    def __init__(self, n, device="cuda"):
        self.n = n
        self.t = {
            "c_custkey": torch.empty(n, dtype=torch.int64, device=device),
            "c_mktsegment": torch.empty(n, dtype=torch.uint8, device=device),
        }
        self.cstruct = CustomerDev(
            n=n, **{k: C.c_void_p(v.data_ptr()) for k, v in self.t.items()})

    @classmethod
    def generate(cls, n_global, rank=0, nranks=1, seed=SEED_DEFAULT):
        t = cls(n_global // nranks)
        call("otbx_gen_customer_dev", C.byref(t.cstruct), C.c_uint64(seed),
             C.c_int64(n_global), C.c_uint32(rank), C.c_uint32(nranks),
             _stream())
        return t


class GpuPart:
    """Replicated dimension table (every rank holds all rows — the locator
    'R' / replicated-relation case of the reference's shard layout)."""

    def __init__(self, n, device="cuda"):
        self.n = n
        self.t = {
            "p_partkey": torch.empty(n, dtype=torch.int64, device=device),
            "p_type": torch.empty(n, dtype=torch.uint8, device=device),
        }
        self.cstruct = PartDev(
            n=n, **{k: C.c_void_p(v.data_ptr()) for k, v in self.t.items()})

    @classmethod
    def generate(cls, n_global, seed=SEED_DEFAULT):
        t = cls(n_global)
        call("otbx_gen_part_dev", C.byref(t.cstruct), C.c_uint64(seed),
             C.c_int64(n_global), _stream())
        return t


# ---------------- CustomScan lifecycle ----------------

class CustomScanState:
    """Base lifecycle, mirroring CustomExecMethods (extensible.h:117-152)."""

    def __init__(self):
        self._begun = False

    def BeginCustomScan(self, estate=None, eflags=0):
        self._begun = True
        self._begin(estate, eflags)
        self._rows = None
        self._pos = 0

    def ExecCustomScan(self):
        """One tuple per call; None = end of stream (nodeCustom.c:104)."""
        if not self._begun:
            raise OtbxError(3, "ExecCustomScan before BeginCustomScan")
        if self._rows is None:
            self._rows = self._run()
        if self._pos >= len(self._rows):
            return None
        row = self._rows[self._pos]
        self._pos += 1
        return row

    def ReScanCustomScan(self):
        self._rows = None
        self._pos = 0

    def EndCustomScan(self):
        self._begun = False
        self._end()

    # subclass hooks
    def _begin(self, estate, eflags): pass
    def _run(self): raise NotImplementedError
    def _end(self): pass

    def explain(self):
        """EXPLAIN ANALYZE analog: per-kernel HIP-event timings for this
        node, mirroring the reference's per-plan-node Instrumentation
        (include/executor/instrument.h:45, merged at the CN by
        commands/explain_dist.c). Returns {phase: ms} or None before run."""
        return None


class GpuSeqScanCount(CustomScanState):
    """SeqScan + qual + COUNT(*) (BASELINE config 2). Replaces the
    SeqScan→Agg(count) fragment (execScan.c:140 + nodeAgg.c)."""

    def __init__(self, lineitem, cutoff=Q1_CUTOFF_DEFAULT):
        super().__init__()
        self.li = lineitem
        self.cutoff = cutoff

    def _run(self):
        out = torch.zeros(1, dtype=torch.int64, device="cuda")
        call("otbx_scan_count", C.c_void_p(self.li.t["l_shipdate"].data_ptr()),
             C.c_int64(self.li.n), C.c_int32(self.cutoff),
             C.c_void_p(out.data_ptr()), _stream())
        return [(int(out.cpu().item()),)]


class GpuQ1PartialAgg(CustomScanState):
    """The DN fragment of TPC-H Q1: SeqScan → qual → project → Partial
    HashAgg (AGGSPLIT_INITIAL_SERIAL), one fused kernel. Emits one partial
    group state per call, ordered by (l_returnflag, l_linestatus).

    partial_state_tensors() exposes the dense device buffers the Coordinator
    merge all-gathers (fragment.py)."""

    def __init__(self, lineitem, cutoff=Q1_CUTOFF_DEFAULT):
        super().__init__()
        self.li = lineitem
        self.cutoff = cutoff
        self.kernel_ms = None
        self.sums = None
        self.counts = None

    def _run(self):
        self.sums = torch.empty((6, 5), dtype=torch.float64, device="cuda")
        self.counts = torch.empty(6, dtype=torch.int64, device="cuda")
        ms = C.c_float(0.0)
        call("otbx_q1_partial", C.byref(self.li.cstruct), C.c_int32(self.cutoff),
             C.c_void_p(self.sums.data_ptr()), C.c_void_p(self.counts.data_ptr()),
             _stream(), C.byref(ms))
        self.kernel_ms = ms.value
        return q1_rows_from_state(self.sums, self.counts)

    def partial_state_tensors(self):
        return self.sums, self.counts

    def explain(self):
        if self.kernel_ms is None:
            return None
        return {"k_q1_partial (SeqScan+Qual+Project+PartialAgg, fused)":
                self.kernel_ms}


def q1_rows_from_state(sums, counts):
    """Dense [6,5] sums + [6] counts → partial group rows (host)."""
    s = sums.cpu().numpy() if hasattr(sums, "cpu") else sums
    c = counts.cpu().numpy() if hasattr(counts, "cpu") else counts
    rows = []
    for g, (rf, ls) in enumerate(Q1_SLOT_ORDER):
        if c[g] == 0:
            continue
        rows.append({
            "l_returnflag": rf.decode(), "l_linestatus": ls.decode(),
            "sum_qty": float(s[g][0]), "sum_base_price": float(s[g][1]),
            "sum_disc_price": float(s[g][2]), "sum_charge": float(s[g][3]),
            "sum_disc": float(s[g][4]), "count_order": int(c[g]),
        })
    return rows


def q1_finalize(rows):
    """Finalize Aggregate (CN): avg = Sx/N (float8_avg semantics)."""
    out = []
    for r in rows:
        n = r["count_order"]
        fin = dict(r)
        fin["avg_qty"] = r["sum_qty"] / n
        fin["avg_price"] = r["sum_base_price"] / n
        fin["avg_disc"] = r["sum_disc"] / n
        out.append(fin)
    return out


class GpuQ3Fragment(CustomScanState):
    """The DN fragment of TPC-H Q3: customer⋈orders⋈lineitem + Partial
    HashAgg keyed on l_orderkey. cust_keys: optional device int64 tensor of
    the REPLICATED (post-broadcast) filtered customer keys;
    without it, the local customer shard is filtered in-kernel."""

    NP_DTYPE = [("l_orderkey", "i8"), ("revenue", "f8"),
                ("o_orderdate", "i4"), ("o_shippriority", "i4")]

    def __init__(self, customer, orders, lineitem, segment=0,
                 date=Q3_DATE_DEFAULT, cust_keys=None, k=10):
        super().__init__()
        self.cu, self.od, self.li = customer, orders, lineitem
        self.segment, self.date = segment, date
        self.cust_keys = cust_keys
        self.k = k
        self.kernel_ms = None    # [keyset, orders, probe_agg, compact]
        self.probe_hits = None   # N_probe_hits (roofline formula)
        self.ngroups = None
        self.groups = None       # lazy: fetch_groups()
        self._groups_dev = None

    def _run(self):
        import numpy as np
        L = lib()
        ncust = self.cu.n if self.cust_keys is None else len(self.cust_keys)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_q3_workspace_bytes(C.c_int64(ncust), C.c_int64(self.od.n),
                                        C.c_int64(self.li.n), C.byref(ws_bytes)))
        ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
        cap = self.od.n if self.od.n > 0 else 1
        groups = torch.empty(cap * 24, dtype=torch.uint8, device="cuda")
        ng = torch.zeros(1, dtype=torch.int64, device="cuda")
        stats = torch.zeros(1, dtype=torch.int64, device="cuda")
        ms = (C.c_float * 4)()
        ck = C.c_void_p(self.cust_keys.data_ptr()) if self.cust_keys is not None else None
        nck = C.c_int64(0 if self.cust_keys is None else len(self.cust_keys))
        call("otbx_q3_partial", C.byref(self.cu.cstruct), C.byref(self.od.cstruct),
             C.byref(self.li.cstruct), ck, nck, C.c_uint8(self.segment),
             C.c_int32(self.date), C.c_void_p(ws.data_ptr()),
             C.c_size_t(ws_bytes.value), C.c_void_p(groups.data_ptr()),
             C.c_int64(cap), C.c_void_p(ng.data_ptr()),
             C.c_void_p(stats.data_ptr()), _stream(), ms)
        self.kernel_ms = list(ms)
        self._groups_dev = groups
        self.ngroups = int(ng.cpu().item())
        self.probe_hits = int(stats.cpu().item())
        # GPU top-k pre-selection (LIMIT 10 below the merge); the final
        # ordering of the ≤ few-thousand candidates happens on host
        n = self.ngroups
        if n == 0:
            self.groups = np.empty(0, dtype=np.dtype(self.NP_DTYPE))
            return []
        # retry with a larger cap if the threshold histogram bin holds more
        # candidates than fit (ncand receives the TRUE count; silently
        # truncating could drop real top-k rows)
        cap_cand = 1 << 20
        while True:
            cand = torch.empty(cap_cand * 24, dtype=torch.uint8, device="cuda")
            ncand = torch.zeros(1, dtype=torch.int64, device="cuda")
            hist = torch.empty(16384, dtype=torch.int32, device="cuda")
            call("otbx_topk_by_revenue", C.c_void_p(groups.data_ptr()),
                 C.c_int64(n), C.c_int64(self.k), C.c_void_p(cand.data_ptr()),
                 C.c_int64(cap_cand), C.c_void_p(ncand.data_ptr()),
                 C.c_void_p(hist.data_ptr()), _stream())
            nc = int(ncand.cpu().item())
            if nc <= cap_cand:
                break
            cap_cand = max(cap_cand * 2, nc)
        cands = cand[: nc * 24].cpu().numpy().view(np.dtype(self.NP_DTYPE))
        return [tuple(r) for r in q3_topk(cands, self.k)]

    def explain(self):
        if self.kernel_ms is None:
            return None
        names = ["customer keyset build (Hash build side 1)",
                 "orders filter+probe+insert (HashJoin 1 + Hash build 2)",
                 "lineitem scan-filter + probe + partial agg (HashJoin 2 + PartialAgg)",
                 "group compaction (emit)"]
        return dict(zip(names, self.kernel_ms))

    def fetch_groups(self, ordered=False):
        """D2H copy of ALL partial groups (parity tests / debugging).
        ordered=True returns them fully sorted by (revenue DESC,
        o_orderdate ASC) via the GPU radix sort (the no-LIMIT ORDER BY
        path)."""
        import numpy as np
        if ordered:
            return order_groups(self._groups_dev, self.ngroups)
        if self.groups is None or len(self.groups) != self.ngroups:
            self.groups = self._groups_dev[: self.ngroups * 24].cpu().numpy() \
                .view(np.dtype(self.NP_DTYPE))
        return self.groups


def q3_topk(groups, k=10):
    """ORDER BY revenue DESC, o_orderdate ASC LIMIT k (host-side; the
    coordinator merge-sort analog, execFragment.c:4035)."""
    import numpy as np
    order = np.lexsort((groups["l_orderkey"], groups["o_orderdate"],
                        -groups["revenue"]))
    return groups[order][:k]


def order_groups(groups_dev_u8, n):
    """ORDER BY revenue DESC, o_orderdate ASC over device q3 groups (raw
    24-B-record uint8 tensor) — the full-sort operator.
    Returns a sorted structured numpy array."""
    import numpy as np
    L = lib()
    ws_bytes = C.c_size_t(0)
    check(L.otbx_order_groups_workspace_bytes(C.c_int64(n), C.byref(ws_bytes)))
    ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8, device="cuda")
    out = torch.empty(max(n, 1) * 24, dtype=torch.uint8, device="cuda")
    call("otbx_order_groups", C.c_void_p(groups_dev_u8.data_ptr()),
         C.c_int64(n), C.c_void_p(out.data_ptr()), C.c_void_p(ws.data_ptr()),
         C.c_size_t(ws_bytes.value), _stream())
    return out[: n * 24].cpu().numpy().view(np.dtype(GpuQ3Fragment.NP_DTYPE))


class GpuQ9Fragment(CustomScanState):
    """The DN fragment of the Q9-shaped mix query (BASELINE config 5):
    lineitem ⋈ part[p_type % typemod == typeval] ⋈ orders, partial agg
    keyed on year(o_orderdate) — two HashJoins under a Partial HashAgg
    with a tiny dense group domain (7 years). Emits one partial state row
    per year; the Coordinator merge is elementwise (fragment.py
    merge_q9_partials), exactly the Q1 pattern."""

    YEARS = list(range(1992, 1999))

    def __init__(self, part, orders, lineitem, typemod=17, typeval=0,
                 nranks=1):
        super().__init__()
        self.pt, self.od, self.li = part, orders, lineitem
        self.typemod, self.typeval = typemod, typeval
        self.nranks = nranks
        self.kernel_ms = None
        self.sums = None
        self.counts = None

    def _run(self):
        L = lib()
        ws_bytes = C.c_size_t(0)
        check(L.otbx_q9_workspace_bytes(C.c_int64(self.pt.n),
                                        C.c_int64(self.od.n),
                                        C.c_int64(self.li.n),
                                        C.c_uint32(self.nranks),
                                        C.byref(ws_bytes)))
        ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
        self.sums = torch.empty(7, dtype=torch.float64, device="cuda")
        self.counts = torch.empty(7, dtype=torch.int64, device="cuda")
        ms = C.c_float(0.0)
        call("otbx_q9_partial", C.byref(self.pt.cstruct),
             C.byref(self.od.cstruct), C.byref(self.li.cstruct),
             C.c_uint8(self.typemod), C.c_uint8(self.typeval),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(self.sums.data_ptr()),
             C.c_void_p(self.counts.data_ptr()), _stream(), C.byref(ms))
        self.kernel_ms = ms.value
        return q9_rows_from_state(self.sums, self.counts)

    def partial_state_tensors(self):
        return self.sums, self.counts

    def explain(self):
        if self.kernel_ms is None:
            return None
        return {"k_q9_fused (Scan+2×HashJoin probe+PartialAgg, fused)":
                self.kernel_ms}


def q9_rows_from_state(sums, counts):
    """Dense [7] sums + [7] counts → partial rows ordered by year."""
    s = sums.cpu().numpy() if hasattr(sums, "cpu") else sums
    c = counts.cpu().numpy() if hasattr(counts, "cpu") else counts
    return [{"o_year": 1992 + g, "sum_revenue": float(s[g]),
             "count_rows": int(c[g])}
            for g in range(7) if c[g] != 0]


def partition_by_key(keys):
    """Repartition exchange, GPU half (the 'Distribute results
    by H: col' exchange of make_remotesubplan, createplan.c:8671): returns
    (perm, counts) where perm groups row indices into contiguous per-rank
    segments (owner = key % world) and counts[r] is segment r's length.
    fragment.exchange_rows() moves the gathered segments over RCCL."""
    import torch.distributed as dist
    world = dist.get_world_size() if dist.is_available() and \
        dist.is_initialized() else 1
    n = len(keys)
    perm = torch.empty(n, dtype=torch.int64, device="cuda")
    counts = (C.c_int64 * max(world, 1))()
    call("otbx_partition_by_key", C.c_void_p(keys.data_ptr()), C.c_int64(n),
         C.c_uint32(world), C.c_void_p(perm.data_ptr()), counts, _stream())
    return perm, list(counts)


_GATHER_FN = {torch.int64: "otbx_gather_i64", torch.float64: "otbx_gather_f64",
              torch.int32: "otbx_gather_i32", torch.uint8: "otbx_gather_u8"}


def gather(src, perm):
    """dst[i] = src[perm[i]] via the native gather kernels."""
    dst = torch.empty(len(perm), dtype=src.dtype, device="cuda")
    call(_GATHER_FN[src.dtype], C.c_void_p(src.data_ptr()),
         C.c_void_p(perm.data_ptr()), C.c_int64(len(perm)),
         C.c_void_p(dst.data_ptr()), _stream())
    return dst


class GpuHashAgg(CustomScanState):
    """Composable HashAggregate over i64 keys / f64 values with full NULL
    semantics (execGrouping.c:295 + nodeAgg.c:743). Inputs: device tensors;
    null bitmaps optional uint8 tensors (1 = NULL)."""

    def __init__(self, keys, vals, key_null=None, val_null=None):
        super().__init__()
        self.keys, self.vals = keys, vals
        self.key_null, self.val_null = key_null, val_null

    def _run(self):
        import numpy as np
        L = lib()
        n = len(self.keys)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_agg_i64_workspace_bytes(C.c_int64(n), C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8, device="cuda")
        out = torch.empty(max(n, 1) * 40, dtype=torch.uint8, device="cuda")
        ng = torch.zeros(1, dtype=torch.int64, device="cuda")
        kn = C.c_void_p(self.key_null.data_ptr()) if self.key_null is not None else None
        vn = C.c_void_p(self.val_null.data_ptr()) if self.val_null is not None else None
        call("otbx_agg_i64", C.c_void_p(self.keys.data_ptr()), kn,
             C.c_void_p(self.vals.data_ptr()), vn, C.c_int64(n),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), _stream())
        ngroups = int(ng.cpu().item())
        dt = np.dtype([("key", "i8"), ("count_star", "i8"), ("count_v", "i8"),
                       ("sum_v", "f8"), ("key_isnull", "i4"), ("sum_isnull", "i4")])
        arr = out[: ngroups * 40].cpu().numpy().view(dt).copy()
        arr.sort(order=["key_isnull", "key"])
        return list(arr)


class GpuHashAggDec(CustomScanState):
    """Exact decimal aggregate: scaled-int64 values, int128 sum
    (Int128AggState semantics, numeric.c:5072/:4998/:5365). Bit-exact —
    rows carry .sum128 as a Python int."""

    def __init__(self, keys, vals, key_null=None, val_null=None):
        super().__init__()
        self.keys, self.vals = keys, vals
        self.key_null, self.val_null = key_null, val_null

    def _run(self):
        import numpy as np
        L = lib()
        n = len(self.keys)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_agg_i64_dec_workspace_bytes(C.c_int64(n),
                                                 C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8,
                         device="cuda")
        out = torch.empty(max(n, 1) * 48, dtype=torch.uint8, device="cuda")
        ng = torch.zeros(1, dtype=torch.int64, device="cuda")

        def vp(t):
            return C.c_void_p(t.data_ptr()) if t is not None else None

        call("otbx_agg_i64_dec", vp(self.keys), vp(self.key_null),
             vp(self.vals), vp(self.val_null), C.c_int64(n),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), _stream())
        ngroups = int(ng.cpu().item())
        dt = np.dtype([("key", "i8"), ("count_star", "i8"), ("count_v", "i8"),
                       ("sum_hi", "i8"), ("sum_lo", "u8"),
                       ("key_isnull", "i4"), ("sum_isnull", "i4")])
        arr = out[: ngroups * 48].cpu().numpy().view(dt).copy()
        arr.sort(order=["key_isnull", "key"])
        rows = []
        for r in arr:
            d = {k: int(r[k]) for k in dt.names}
            d["sum128"] = (int(r["sum_hi"]) << 64) | int(r["sum_lo"])
            rows.append(d)
        return rows


def _keyset_dev(key_tensors, null_tensors=None):
    """Build an otbx_keyset (device pointers) from torch tensors."""
    ks = KeysetDev()
    ks.nkeys = len(key_tensors)
    for c, k in enumerate(key_tensors):
        ks.keys[c] = k.data_ptr()
        nt = None if null_tensors is None else null_tensors[c]
        ks.nulls[c] = nt.data_ptr() if nt is not None else None
    return ks


class GpuHashAggN(CustomScanState):
    """N-key (1..8) composable HashAggregate: groups carry the defining
    ROW INDEX (the representative-tuple pattern, execGrouping.c
    firstTuple); the caller reads key values back through the index."""

    def __init__(self, key_tensors, vals, null_tensors=None, val_null=None):
        super().__init__()
        self.keys, self.vals = key_tensors, vals
        self.nulls, self.vn = null_tensors, val_null

    def _run(self):
        import numpy as np
        L = lib()
        n = len(self.vals)
        ks = _keyset_dev(self.keys, self.nulls)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_agg_i64n_workspace_bytes(C.c_int64(n),
                                              C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8,
                         device="cuda")
        out = torch.empty(max(n, 1) * 40, dtype=torch.uint8, device="cuda")
        ng = torch.zeros(1, dtype=torch.int64, device="cuda")
        vn = C.c_void_p(self.vn.data_ptr()) if self.vn is not None else None
        call("otbx_agg_i64n", C.byref(ks),
             C.c_void_p(self.vals.data_ptr()), vn, C.c_int64(n),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), _stream())
        ngroups = int(ng.cpu().item())
        dt = np.dtype([("row_idx", "i8"), ("count_star", "i8"),
                       ("count_v", "i8"), ("sum_v", "f8"),
                       ("sum_isnull", "i4"), ("_pad", "i4")])
        arr = out[: ngroups * 40].cpu().numpy().view(dt).copy()
        arr.sort(order="row_idx")
        return list(arr)


class GpuHashJoinN(CustomScanState):
    """N-key (1..8) HashJoin, all six join types; pair encoding and
    overflow contract as GpuHashJoin."""

    def __init__(self, bkeys, pkeys, join_type=0, bnulls=None, pnulls=None,
                 cap_pairs=None):
        super().__init__()
        self.bkeys, self.pkeys = bkeys, pkeys
        self.bnulls, self.pnulls = bnulls, pnulls
        self.join_type = JOIN_TYPES.get(join_type, join_type) \
            if isinstance(join_type, str) else join_type
        self.cap_pairs = cap_pairs

    def _run(self):
        L = lib()
        nb = len(self.bkeys[0]) if self.bkeys else 0
        npr = len(self.pkeys[0]) if self.pkeys else 0
        bks = _keyset_dev(self.bkeys, self.bnulls)
        pks = _keyset_dev(self.pkeys, self.pnulls)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_join_i64n_workspace_bytes(C.c_int64(nb), C.c_int64(npr),
                                               C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8,
                         device="cuda")
        cap = self.cap_pairs if self.cap_pairs \
            else max(4 * max(nb, npr) + nb + npr, 64)
        ob = torch.empty(cap, dtype=torch.int64, device="cuda")
        op = torch.empty(cap, dtype=torch.int64, device="cuda")
        npairs = torch.zeros(1, dtype=torch.int64, device="cuda")
        call("otbx_join_i64n", C.byref(bks), C.c_int64(nb), C.byref(pks),
             C.c_int64(npr), C.c_int32(self.join_type),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(ob.data_ptr()), C.c_void_p(op.data_ptr()),
             C.c_int64(cap), C.c_void_p(npairs.data_ptr()), _stream())
        n = int(npairs.cpu().item())
        if n > cap:
            raise OtbxError(3, f"join pair overflow: {n} > cap {cap}")
        return list(zip(ob[:n].cpu().numpy().tolist(),
                        op[:n].cpu().numpy().tolist()))


def dec_avg(sum128, count, extra_scale=0):
    """Exact AVG finalizer for the decimal aggregate's int128 state
    (Int128AggState {N, sumX}, numeric.c:5072): returns the scaled-int64
    average round-half-up away from zero — the reference's numeric
    rounding (round_var HALF_ADJUST_ROUND, numeric.c:724,1743,7145 via
    div_var). extra_scale shifts the result scale by 10^extra_scale
    relative to the input scale (e.g. cents in -> tenth-cents out with
    extra_scale=1). Python ints are unbounded, so this is exact for any
    int128 state; the combine phase sums states exactly first. Raises on
    count == 0 (AVG over no rows is NULL at the SQL level — the caller
    checks sum_isnull/count first, nodeAgg.c finalize semantics)."""
    if count <= 0:
        raise OtbxError(3, "dec_avg: count must be positive (NULL AVG is "
                           "decided by the caller)")
    num = sum128 * (10 ** extra_scale)
    if num >= 0:
        return (2 * num + count) // (2 * count)
    return -((2 * (-num) + count) // (2 * count))


JOIN_TYPES = {"inner": 0, "left": 1, "semi": 2, "anti": 3, "right": 4,
              "full": 5}


class GpuHashJoin(CustomScanState):
    """Composable HashJoin on i64 keys (nodeHash.c/nodeHashjoin.c), all six
    join types (HJ_* fill states, nodeHashjoin.c:139-144) and an optional
    second key column (multi-key combine, nodeHash.c:2059). Emits
    (build_idx, probe_idx) pairs, -1 = NULL-fill side (see include/otbx.h);
    result-set parity (order-free). join_type 0/'inner' with a single key
    routes through the optimized partitioned inner-join path."""

    def __init__(self, build_keys, probe_keys, build_null=None, probe_null=None,
                 cap_pairs=None, join_type=0, build_keys2=None,
                 probe_keys2=None, build_null2=None, probe_null2=None):
        super().__init__()
        self.bk, self.pk = build_keys, probe_keys
        self.bn, self.pn = build_null, probe_null
        self.bk2, self.pk2 = build_keys2, probe_keys2
        self.bn2, self.pn2 = build_null2, probe_null2
        self.cap_pairs = cap_pairs
        self.join_type = JOIN_TYPES.get(join_type, join_type) \
            if isinstance(join_type, str) else join_type

    def _run(self):
        L = lib()
        nb, npr = len(self.bk), len(self.pk)
        jt = self.join_type
        two_key = self.bk2 is not None
        ext = jt != 0 or two_key
        ws_bytes = C.c_size_t(0)
        if ext:
            check(L.otbx_join_ext_workspace_bytes(
                C.c_int64(nb), C.c_int64(npr), C.byref(ws_bytes)))
        else:
            check(L.otbx_join_i64_workspace_bytes(
                C.c_int64(nb), C.c_int64(npr), C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8, device="cuda")
        # outer fills can exceed the match count: pairs <= matches + nb + np
        cap = self.cap_pairs if self.cap_pairs \
            else max(4 * max(nb, npr) + nb + npr, 64)
        ob = torch.empty(cap, dtype=torch.int64, device="cuda")
        op = torch.empty(cap, dtype=torch.int64, device="cuda")
        npairs = torch.zeros(1, dtype=torch.int64, device="cuda")

        def vp(t):
            return C.c_void_p(t.data_ptr()) if t is not None else None

        if two_key:
            call("otbx_join_i64x2", vp(self.bk), vp(self.bn), vp(self.bk2),
                 vp(self.bn2), C.c_int64(nb), vp(self.pk), vp(self.pn),
                 vp(self.pk2), vp(self.pn2), C.c_int64(npr), C.c_int32(jt),
                 C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
                 vp(ob), vp(op), C.c_int64(cap), vp(npairs), _stream())
        elif ext:
            call("otbx_join_i64_ext", vp(self.bk), vp(self.bn), C.c_int64(nb),
                 vp(self.pk), vp(self.pn), C.c_int64(npr), C.c_int32(jt),
                 C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
                 vp(ob), vp(op), C.c_int64(cap), vp(npairs), _stream())
        else:
            call("otbx_join_i64", vp(self.bk), vp(self.bn), C.c_int64(nb),
                 vp(self.pk), vp(self.pn), C.c_int64(npr),
                 C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
                 vp(ob), vp(op), C.c_int64(cap), vp(npairs), _stream())
        n = int(npairs.cpu().item())
        if n > cap:
            raise OtbxError(3, f"join pair overflow: {n} > cap {cap}")
        return list(zip(ob[:n].cpu().numpy().tolist(),
                        op[:n].cpu().numpy().tolist()))


class GpuHashAgg2(CustomScanState):
    """Two-key composable HashAggregate (multi-key GROUP BY,
    execGrouping.c:295 with NULL==NULL grouping :525). Returns rows sorted
    by (k1_isnull, k1, k2_isnull, k2) for deterministic comparison."""

    def __init__(self, keys1, keys2, vals, key1_null=None, key2_null=None,
                 val_null=None):
        super().__init__()
        self.k1, self.k2, self.vals = keys1, keys2, vals
        self.n1, self.n2, self.vn = key1_null, key2_null, val_null

    def _run(self):
        import numpy as np
        L = lib()
        n = len(self.k1)
        ws_bytes = C.c_size_t(0)
        check(L.otbx_agg_i64x2_workspace_bytes(C.c_int64(n),
                                               C.byref(ws_bytes)))
        ws = torch.empty(max(ws_bytes.value, 1), dtype=torch.uint8, device="cuda")
        out = torch.empty(max(n, 1) * 56, dtype=torch.uint8, device="cuda")
        ng = torch.zeros(1, dtype=torch.int64, device="cuda")

        def vp(t):
            return C.c_void_p(t.data_ptr()) if t is not None else None

        call("otbx_agg_i64x2", vp(self.k1), vp(self.n1), vp(self.k2),
             vp(self.n2), vp(self.vals), vp(self.vn), C.c_int64(n),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), _stream())
        ngroups = int(ng.cpu().item())
        dt = np.dtype([("key1", "i8"), ("key2", "i8"), ("count_star", "i8"),
                       ("count_v", "i8"), ("sum_v", "f8"),
                       ("key1_isnull", "i4"), ("key2_isnull", "i4"),
                       ("sum_isnull", "i4"), ("_pad", "i4")])
        arr = out[: ngroups * 56].cpu().numpy().view(dt).copy()
        arr.sort(order=["key1_isnull", "key1", "key2_isnull", "key2"])
        return list(arr)
