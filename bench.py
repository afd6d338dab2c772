#!/usr/bin/env python3
"""bench.py — measures the BASELINE.json metric (TPC-H Q1/Q3-shaped rows/s +
GB/s) on MI355X.

Default (no flags): N=1 GPU, workload = BASELINE config 3 — TPC-H Q1 at SF100
on one GPU (the largest single-GPU config; the metric's home). One "step" =
one full pass of the offloaded DN fragment (fused scan+filter+partial-agg
kernel over the HBM-resident SF100 lineitem) + the Coordinator merge
(all-gather of partial states + finalize). Inputs are generated on-device
once, before the timed region.

Multi-GPU: one rank per GPU (torch.distributed over RCCL/xGMI), weak scaling
(each rank owns an SF100 shard); value = whole-job rows/s over all ranks with
MAX-over-ranks timing.

The JSON line carries:
  roofline     — dominant kernel (the fused Q1 kernel): achieved algorithmic
                 GB/s (38 B/row × rows ÷ HIP-event kernel time, events on the
                 launch stream inside the C-ABI) vs 8 TB/s HBM peak
                 (MI355X_MICROARCH.md). traffic: measured HBM bytes/launch
                 from a committed rocprofv3 --pmc summary (profiles/), with
                 the gfx950 FETCH_SIZE ×2 correction, else null.
  cpu_baseline — the CPU oracle (kind "port": the reference executor
                 restated, scalar, 1 thread) timed on this host on a bounded
                 sample of the same workload.
"""
import argparse
import ctypes as C
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

Q1_BYTES_PER_ROW = 38      # shipdate 4 + flags 2 + 4×f64
SCAN_BYTES_PER_ROW = 4
HBM_PEAK = 8.0e12          # B/s, spec (MI355X_MICROARCH.md)
SF_DEFAULT = 100
LI_PER_SF = 6_000_000


def log(*a):
    print(*a, file=sys.stderr, flush=True)


def load_traffic(workload, sf):
    p = os.path.join(REPO, "profiles", "roofline_traffic.json")
    if not os.path.exists(p):
        return None
    try:
        with open(p) as f:
            j = json.load(f)
        entries = j if isinstance(j, list) else [j]
        for e in entries:
            if e.get("workload") == workload and e.get("sf_per_gpu") == sf:
                return float(e["traffic_bytes_per_launch"])
    except Exception:
        pass
    return None


def cpu_baseline(query, sample_rows, shards=1):
    """Oracle CLI (scalar port of the reference executor) on a bounded
    sample; returns the cpu_baseline JSON object. shards > 1 runs one CLI
    process per shard concurrently (--rank/--nranks) — the DataNode
    deployment analog (each backend is single-threaded; parallelism in
    production is process-per-shard, SURVEY §8b threading note) — and
    reports whole-job rows/s = global rows / max-over-shards seconds,
    with cores = shards."""
    cli = os.path.join(REPO, "oracle", "oracle_cli")
    # always (re)build: an oracle.c edit with a stale committed binary would
    # silently time old code; make is incremental and cheap
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")],
                   check=True, capture_output=True)
    # generators require rows % nranks == 0 for EVERY sharded table:
    # lineitem = rows, orders = rows/4, customer = rows/40
    sample_rows -= sample_rows % (40 * shards)
    if shards <= 1:
        out = subprocess.run([cli, query, "--rows", str(sample_rows)],
                             check=True, capture_output=True,
                             text=True).stdout
        j = json.loads(out)
        rows_s = j["rows"] / j["seconds"]
        secs = j["seconds"]
        tag = "scalar"
    else:
        procs = [subprocess.Popen(
            [cli, query, "--rows", str(sample_rows), "--rank", str(r),
             "--nranks", str(shards)],
            stdout=subprocess.PIPE, text=True) for r in range(shards)]
        js = []
        for p in procs:
            out, _ = p.communicate()
            assert p.returncode == 0, p.returncode
            js.append(json.loads(out))
        secs = max(j["seconds"] for j in js)
        rows_s = sum(j["rows"] for j in js) / secs
        tag = f"{shards} shard processes"
    return {
        "value": rows_s, "unit": "rows/s", "cores": shards, "kind": "port",
        "sample": f"TPC-H {query.upper()} executor over {sample_rows} "
                  f"synthetic lineitem rows (SF{sample_rows // LI_PER_SF}), "
                  f"{tag}, {secs:.1f}s",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--sf", type=int, default=SF_DEFAULT,
                    help="scale factor PER GPU (weak scaling)")
    ap.add_argument("--workload", default="tpch_q1",
                    choices=["tpch_q1", "scan_count", "tpch_q3",
                             "tpch_q9mix"])
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--skew", action="store_true",
                    help="config-5 skewed distribution keys (hot custkeys)")
    ap.add_argument("--cpu-sample-rows", type=int, default=1_200_000_000,
                    help="~10 s of scalar CPU work on the target host")
    ap.add_argument("--cpu-shards", type=int, default=1,
                    help="oracle baseline processes (DataNode deployment "
                         "analog; cores reported = this)")
    args = ap.parse_args()

    import torch
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if args.gpus != world:
        log(f"note: --gpus {args.gpus} but WORLD_SIZE={world} — rank count "
            "comes from the torchrun environment (contract); running "
            f"{world} rank(s)")
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1:
        import torch.distributed as dist
        # OTBX_DIST_BACKEND=gloo lets the full multi-rank path run on a
        # single-GPU box (CI); production is nccl (= RCCL over xGMI)
        backend = os.environ.get("OTBX_DIST_BACKEND", "nccl")
        local_rank = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(local_rank)
        dist.init_process_group(backend)
    else:
        dist = None

    from opentenbase_amd import executor as ex
    from opentenbase_amd import fragment

    ex.init_device(local_rank)
    stream = C.c_void_p(torch.cuda.current_stream().cuda_stream)

    rows_per_gpu = args.sf * LI_PER_SF
    n_global = rows_per_gpu * world

    t0 = time.time()
    if args.workload == "tpch_q1":
        li = ex.GpuLineitem.generate(n_global, rank=rank, nranks=world,
                                     with_orderkey=False)
        bytes_per_row = Q1_BYTES_PER_ROW
    elif args.workload == "scan_count":
        li = ex.GpuLineitem.generate(n_global, rank=rank, nranks=world,
                                     with_orderkey=False)
        bytes_per_row = SCAN_BYTES_PER_ROW
    elif args.workload == "tpch_q9mix":
        # BASELINE config 5's second shape: lineitem ⋈ part ⋈ orders,
        # GROUP BY year. Streamed probe-side bytes: l_partkey only =
        # 8 B/row (the filter pass); per part-filter hit: survivor id
        # written+read 8 + staged 32-B record (orderkey/price/discount,
        # otbx.h q9rec) + dtab 4 = 44 B/hit.
        li = ex.GpuLineitem.generate(n_global, rank=rank, nranks=world,
                                     with_partkey=True)
        od = ex.GpuOrders.generate(n_global // 4, n_global // 40, rank=rank,
                                   nranks=world, skew=args.skew)
        pt = ex.GpuPart.generate(max(n_global // 30, 1))
        bytes_per_row = 8
    else:  # tpch_q3
        li = ex.GpuLineitem.generate(n_global, rank=rank, nranks=world)
        od = ex.GpuOrders.generate(n_global // 4, n_global // 40, rank=rank,
                                   nranks=world, skew=args.skew)
        cu = ex.GpuCustomer.generate(n_global // 40, rank=rank, nranks=world)
        bytes_per_row = 28  # probe-side ALGORITHMIC bytes (the
        # staged int32 key cache means the kernel physically streams 4 B
        # of key instead of the contract's 8 — frac is vs the contract)
    torch.cuda.synchronize()
    staging_s = time.time() - t0
    log(f"rank {rank}: staged {rows_per_gpu} rows in {staging_s:.1f}s "
        f"({li.bytes_staged() / 1e9:.1f} GB in HBM)")

    kernel_ms_acc = []
    hits_acc = []
    phase_ms = []

    def step():
        if args.workload == "tpch_q1":
            node = ex.GpuQ1PartialAgg(li)
            node.BeginCustomScan()
            node._rows = node._run()          # fused kernel
            kernel_ms_acc.append(node.kernel_ms)
            s, c = node.partial_state_tensors()
            rows = fragment.merge_q1_partials(s, c)   # CN merge (collective)
            return fragment.finalize_q1(rows)
        elif args.workload == "scan_count":
            out = torch.zeros(1, dtype=torch.int64, device="cuda")
            from opentenbase_amd._lib import call
            call("otbx_scan_count", C.c_void_p(li.t["l_shipdate"].data_ptr()),
                 C.c_int64(li.n), C.c_int32(2436), C.c_void_p(out.data_ptr()),
                 stream)
            return int(out.cpu().item())
        elif args.workload == "tpch_q9mix":
            node = ex.GpuQ9Fragment(pt, od, li, nranks=world)
            node.BeginCustomScan()
            node._rows = node._run()
            kernel_ms_acc.append(node.kernel_ms)
            s, c = node.partial_state_tensors()
            hits_acc.append(int(c.sum().item()))  # rows surviving both joins
            return fragment.merge_q9_partials(s, c)
        else:
            bcast = None
            if world > 1:
                import ctypes as CT
                from opentenbase_amd._lib import call
                keys = torch.empty(cu.n, dtype=torch.int64, device="cuda")
                nk = torch.zeros(1, dtype=torch.int64, device="cuda")
                call("otbx_filter_customer", CT.byref(cu.cstruct), CT.c_uint8(0),
                     CT.c_void_p(keys.data_ptr()), CT.c_void_p(nk.data_ptr()),
                     stream)
                local = keys[: int(nk.cpu().item())]
                bcast = fragment.broadcast_customer_keys(local).to("cuda")
            node = ex.GpuQ3Fragment(cu, od, li, cust_keys=bcast)
            node.BeginCustomScan()
            node._rows = node._run()
            # probe+partial-agg kernel is the roofline-dominant phase
            kernel_ms_acc.append(node.kernel_ms[2])
            hits_acc.append(node.probe_hits)
            phase_ms.append(node.kernel_ms)
            import numpy as np
            cands = np.array(node._rows, dtype=np.dtype(node.NP_DTYPE))
            return fragment.merge_q3_topk(cands, 10)

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    kernel_ms_acc.clear()
    hits_acc.clear()
    phase_ms.clear()

    barrier_sync()
    t_start = time.time()
    for _ in range(args.steps):
        step()
    barrier_sync()
    elapsed = time.time() - t_start

    # MAX over ranks
    if dist is not None:
        dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())

    ms_per_step = elapsed / args.steps * 1000.0
    total_rows = rows_per_gpu * world
    value = total_rows / (elapsed / args.steps)
    gbps = value * bytes_per_row / 1e9

    kmean_ms = sum(kernel_ms_acc) / len(kernel_ms_acc) if kernel_ms_acc else None
    roofline = None
    if kmean_ms:
        if args.workload == "tpch_q3":
            # probe kernel: 28 B/row streamed + 64 B per probe hit (§8d)
            hits = hits_acc[-1]
            algo_bytes = 28 * rows_per_gpu + 64 * hits
        elif args.workload == "tpch_q9mix":
            # filter streams 8 B/row; per hit: id write+read 8 + 32-B
            # staged record + 4-B dtab = 44 B.
            # NOTE: the filter is bound by random-gather
            # REQUEST throughput (uniform partkeys), not HBM bytes — frac
            # vs the byte roofline understates it; the component microbench
            # (tools/microbench/q9_gather_ab) gives the real floor.
            hits = hits_acc[-1]
            algo_bytes = 8 * rows_per_gpu + 44 * hits
        else:
            algo_bytes = rows_per_gpu * bytes_per_row
        achieved = algo_bytes / (kmean_ms / 1e3)  # B/s, per GPU
        traffic = load_traffic(args.workload, args.sf)
        roofline = {
            "bound": "hbm",
            "achieved": achieved / 1e9,
            "peak": HBM_PEAK / 1e9,
            "unit": "GB/s",
            "frac": achieved / HBM_PEAK,
            "traffic": traffic,
        }

    cpu = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline \
            and args.workload in ("tpch_q1", "tpch_q3", "tpch_q9mix"):
        log("running cpu_baseline (oracle port, 1 core)...")
        q = {"tpch_q1": "q1", "tpch_q3": "q3",
             "tpch_q9mix": "q9"}[args.workload]
        # join executors are ~4x slower per row: smaller sample
        rows = args.cpu_sample_rows if q == "q1" else args.cpu_sample_rows // 4
        cpu = cpu_baseline(q, rows, shards=args.cpu_shards)

    if rank == 0:
        result = {
            "metric": "rows/s (TPC-H Q1 SF100-per-GPU, scan→filter→partial "
                      "hash-agg→merge)" if args.workload == "tpch_q1"
                      else f"rows/s ({args.workload})",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            # BASELINE.md publishes no reference numbers; its baseline IS the
            # measured CPU-executor restatement (BASELINE.md "CPU-baseline
            # plan"), so vs_baseline = GPU value / measured cpu_baseline
            "vs_baseline": (value / cpu["value"]) if cpu else None,
            "dtype": "f64",
            "data": "synthetic (dbgen-shaped, seed 42, generated on-device)",
            "gb_per_s_scanned": gbps,
            "config": {
                "workload": args.workload,
                "sf_per_gpu": args.sf,
                "rows_per_gpu": rows_per_gpu,
                "bytes_per_row": bytes_per_row,
                "parallelism": f"dp{world} (1 shard/GPU, RCCL merge)",
                # staged compact-key caches active (int32 orderkey/custkey)
                "key32": bool(getattr(li.cstruct, "l_orderkey32", None))
                if args.workload == "tpch_q3" else None,
                # cold-cache cost: one-time on-device staging,
                # outside the timed region; the timed steps are hot-cache
                "staging_s": round(staging_s, 3),
                # Q3 only: per-phase HIP-event ms of the last step
                # [customer build, orders side, probe window, compact]
                "phase_ms": [round(x, 3) for x in phase_ms[-1]]
                if phase_ms else None,
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(result), flush=True)

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
