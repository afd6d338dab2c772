#!/usr/bin/env python3
"""Isolate the Q3 customer-phase anomaly: k_count_customer_seg reads a
135 MB table in ~0.30 ms (~8x its stream floor) when each step follows
~0.3 ms of host work (top-k merge). Hypothesis: the first kernel after a
host gap pays an inter-step bubble (DVFS/queue ramp), not real kernel
cost. Test: call otbx_q3_partial back-to-back with NO host work between
(phase timings from HIP events inside the C-ABI) vs with a deliberate
host sleep between steps."""
import ctypes as C
import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from opentenbase_amd import executor as ex  # noqa: E402
from opentenbase_amd._lib import call, lib  # noqa: E402


def main():
    ex.init_device(0)
    n = 600_000_000
    li = ex.GpuLineitem.generate(n)
    od = ex.GpuOrders.generate(n // 4, n // 40)
    cu = ex.GpuCustomer.generate(n // 40)
    L = lib()
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)
    ws_bytes = C.c_size_t(0)
    L.otbx_q3_workspace_bytes(C.c_int64(cu.n), C.c_int64(od.n),
                              C.c_int64(li.n), C.byref(ws_bytes))
    ws = torch.empty(ws_bytes.value, dtype=torch.uint8, device="cuda")
    cap = 1 << 26
    groups = torch.empty(cap * 24, dtype=torch.uint8, device="cuda")
    ng = torch.zeros(1, dtype=torch.int64, device="cuda")
    stats = torch.zeros(1, dtype=torch.int64, device="cuda")
    ms = (C.c_float * 4)()

    def step():
        call("otbx_q3_partial", C.byref(cu.cstruct), C.byref(od.cstruct),
             C.byref(li.cstruct), None, C.c_int64(0), C.c_uint8(0),
             C.c_int32(ex.Q3_DATE_DEFAULT), C.c_void_p(ws.data_ptr()),
             C.c_size_t(ws_bytes.value), C.c_void_p(groups.data_ptr()),
             C.c_int64(cap), C.c_void_p(ng.data_ptr()),
             C.c_void_p(stats.data_ptr()), stream, ms)
        return list(ms)

    for mode, gap in [("back_to_back", 0.0), ("gap_1ms", 0.001),
                      ("gap_5ms", 0.005)]:
        phases = [0.0] * 4
        reps = 8
        step()  # warm
        for _ in range(reps):
            if gap:
                torch.cuda.synchronize()
                time.sleep(gap)
            p = step()
            phases = [a + b for a, b in zip(phases, p)]
        torch.cuda.synchronize()
        print(mode, [round(x / reps, 3) for x in phases],
              "(customer, orders, probe, compact) ms", flush=True)


if __name__ == "__main__":
    main()
