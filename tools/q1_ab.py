#!/usr/bin/env python3
"""Within-probe interleaved A/B of the Q1 kernel variants (guide §5.4 rule 24):
N rounds × variants interleaved in one process; reports median/min ms and
checks result parity across variants. Run on the GPU box."""
import ctypes as C
import statistics
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from opentenbase_amd import executor as ex  # noqa: E402
from opentenbase_amd._lib import call  # noqa: E402


def main():
    sf = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 8
    variants = [0, 1, 2]
    ex.init_device(0)
    li = ex.GpuLineitem.generate(sf * 6_000_000, with_orderkey=False)
    torch.cuda.synchronize()
    sums = torch.empty((6, 5), dtype=torch.float64, device="cuda")
    counts = torch.empty(6, dtype=torch.int64, device="cuda")
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)

    def run(v):
        ms = C.c_float(0.0)
        call("otbx_q1_partial_variant", C.byref(li.cstruct), C.c_int32(2436),
             C.c_void_p(sums.data_ptr()), C.c_void_p(counts.data_ptr()),
             stream, C.byref(ms), C.c_int(v))
        return ms.value, sums.cpu().numpy().copy(), counts.cpu().numpy().copy()

    # warmup + parity reference
    ref = run(0)
    for v in variants[1:]:
        _, s, c = run(v)
        assert (c == ref[2]).all(), f"variant {v} count mismatch"
        import numpy as np
        rel = np.abs(s - ref[1]) / np.maximum(np.abs(ref[1]), 1e-300)
        assert rel.max() < 1e-12, f"variant {v} sums differ rel {rel.max()}"

    times = {v: [] for v in variants}
    for r in range(rounds):
        for v in variants:
            times[v].append(run(v)[0])
    bytes_ = li.n * 38
    for v in variants:
        med = statistics.median(times[v])
        mn = min(times[v])
        print(f"variant {v}: median {med:.3f} ms  min {mn:.3f} ms  "
              f"median BW {bytes_ / med / 1e9 * 1e3:.0f} GB/s  "
              f"all {[f'{t:.3f}' for t in times[v]]}")


if __name__ == "__main__":
    main()
