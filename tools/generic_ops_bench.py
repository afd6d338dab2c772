#!/usr/bin/env python3
"""Scale benchmark of the generic composable operators (otbx_agg_i64 /
otbx_join_i64) — arbitrary-plan building blocks, distinct from the fused
query pipelines."""
import ctypes as C
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from opentenbase_amd import executor as ex  # noqa: E402
from opentenbase_amd._lib import call, lib  # noqa: E402


def bench_agg(n, ngroups):
    g = torch.Generator(device="cuda").manual_seed(1)
    keys = torch.randint(0, ngroups, (n,), dtype=torch.int64, device="cuda",
                         generator=g)
    vals = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
    L = lib()
    ws_bytes = C.c_size_t(0)
    L.otbx_agg_i64_workspace_bytes(C.c_int64(n), C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    out = torch.empty(n * 40, dtype=torch.uint8, device="cuda")
    ng = torch.zeros(1, dtype=torch.int64, device="cuda")
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    best = 1e9
    for _ in range(5):
        t0 = time.time()
        call("otbx_agg_i64", C.c_void_p(keys.data_ptr()), None,
             C.c_void_p(vals.data_ptr()), None, C.c_int64(n),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream)
        torch.cuda.synchronize()
        best = min(best, time.time() - t0)
    print(f"agg  n={n:>11_} groups={ngroups:>9_}: {best*1e3:8.2f} ms "
          f"({n/best/1e9:6.1f} Grows/s), ngroups={int(ng.cpu().item())}")


def bench_join(nb, np_):
    g = torch.Generator(device="cuda").manual_seed(2)
    bk = torch.randint(0, nb, (nb,), dtype=torch.int64, device="cuda",
                       generator=g)
    pk = torch.randint(0, nb, (np_,), dtype=torch.int64, device="cuda",
                       generator=g)
    L = lib()
    ws_bytes = C.c_size_t(0)
    L.otbx_join_i64_workspace_bytes(C.c_int64(nb), C.c_int64(np_),
                                    C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    cap = int(np_ * 2.2)
    ob = torch.empty(cap, dtype=torch.int64, device="cuda")
    op = torch.empty(cap, dtype=torch.int64, device="cuda")
    npairs = torch.zeros(1, dtype=torch.int64, device="cuda")
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    best = 1e9
    for _ in range(5):
        t0 = time.time()
        call("otbx_join_i64", C.c_void_p(bk.data_ptr()), None, C.c_int64(nb),
             C.c_void_p(pk.data_ptr()), None, C.c_int64(np_),
             C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
             C.c_void_p(ob.data_ptr()), C.c_void_p(op.data_ptr()),
             C.c_int64(cap), C.c_void_p(npairs.data_ptr()), stream)
        torch.cuda.synchronize()
        best = min(best, time.time() - t0)
    print(f"join nb={nb:>10_} np={np_:>11_}: {best*1e3:8.2f} ms "
          f"({np_/best/1e9:6.1f} Gprobes/s), pairs={int(npairs.cpu().item())}")


def bench_round2_ops():
    """Generality-tier round-2 operators at scale, timed at the C-ABI
    (device outputs; no host materialization — the executor wrappers
    additionally convert results to Python objects, which dominates at
    100 M-pair scale and is not kernel time). Correctness-first designs;
    numbers are evidence-of-function, not roofline targets."""
    L = lib()
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
    g = torch.Generator(device="cuda").manual_seed(3)
    n = 100_000_000
    k1 = torch.randint(0, 100_000, (n,), dtype=torch.int64, device="cuda",
                       generator=g)
    k2 = torch.randint(0, 50, (n,), dtype=torch.int64, device="cuda",
                       generator=g)
    v = torch.rand(n, dtype=torch.float64, device="cuda", generator=g)
    vi = torch.randint(-10**12, 10**12, (n,), dtype=torch.int64,
                       device="cuda", generator=g)

    def timed(label, fn, units):
        torch.cuda.synchronize()
        best = 1e9
        for _ in range(3):
            t0 = time.time()
            fn()
            torch.cuda.synchronize()
            best = min(best, time.time() - t0)
        print(f"{label}: {best*1e3:8.2f} ms ({units/best/1e9:6.1f} Grows/s)")

    ws_bytes = C.c_size_t(0)
    L.otbx_agg_i64x2_workspace_bytes(C.c_int64(n), C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    out = torch.empty(n * 56, dtype=torch.uint8, device="cuda")
    ng = torch.zeros(1, dtype=torch.int64, device="cuda")
    timed("agg2  n=100M groups=5M       ", lambda: call(
        "otbx_agg_i64x2", C.c_void_p(k1.data_ptr()), None,
        C.c_void_p(k2.data_ptr()), None, C.c_void_p(v.data_ptr()), None,
        C.c_int64(n), C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
        C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream), n)

    L.otbx_agg_i64_dec_workspace_bytes(C.c_int64(n), C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    timed("dec   n=100M groups=100k     ", lambda: call(
        "otbx_agg_i64_dec", C.c_void_p(k1.data_ptr()), None,
        C.c_void_p(vi.data_ptr()), None, C.c_int64(n),
        C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
        C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream), n)

    from opentenbase_amd._lib import KeysetDev
    ks = KeysetDev()
    ks.nkeys = 4
    for c, t in enumerate([k1, k2, k2, k2]):
        ks.keys[c] = t.data_ptr()
    L.otbx_agg_i64n_workspace_bytes(C.c_int64(n), C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    timed("aggn4 n=100M groups=5M       ", lambda: call(
        "otbx_agg_i64n", C.byref(ks), C.c_void_p(v.data_ptr()), None,
        C.c_int64(n), C.c_void_p(ws.data_ptr()), C.c_size_t(ws_bytes.value),
        C.c_void_p(out.data_ptr()), C.c_void_p(ng.data_ptr()), stream), n)

    nb = 10_000_000
    bk = torch.randint(0, nb, (nb,), dtype=torch.int64, device="cuda",
                       generator=g)
    pk = torch.randint(0, 2 * nb, (n,), dtype=torch.int64, device="cuda",
                       generator=g)
    L.otbx_join_ext_workspace_bytes(C.c_int64(nb), C.c_int64(n),
                                    C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    cap = 2 * n
    ob = torch.empty(cap, dtype=torch.int64, device="cuda")
    op = torch.empty(cap, dtype=torch.int64, device="cuda")
    npairs = torch.zeros(1, dtype=torch.int64, device="cuda")
    for jt, name in [(1, "left"), (2, "semi"), (3, "anti")]:
        timed(f"joinx {name:<5} nb=10M np=100M  ", lambda jt=jt: call(
            "otbx_join_i64_ext", C.c_void_p(bk.data_ptr()), None,
            C.c_int64(nb), C.c_void_p(pk.data_ptr()), None, C.c_int64(n),
            C.c_int32(jt), C.c_void_p(ws.data_ptr()),
            C.c_size_t(ws_bytes.value), C.c_void_p(ob.data_ptr()),
            C.c_void_p(op.data_ptr()), C.c_int64(cap),
            C.c_void_p(npairs.data_ptr()), stream), n)


if __name__ == "__main__":
    ex.init_device(0)
    bench_agg(600_000_000, 4)
    bench_agg(600_000_000, 1_000_000)
    bench_agg(600_000_000, 100_000_000)
    bench_join(15_000_000, 600_000_000)
    bench_join(150_000_000, 600_000_000)
    if "--round2-ops" in sys.argv:
        bench_round2_ops()
