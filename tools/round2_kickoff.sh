#!/bin/bash
# Round-2 first GPU call: run the prepared microbench
# variants and re-baseline the generic ops, writing everything under
# gpurun_out/ for triage. Usage:
#   /usr/local/graft/bin/gpurun --timeout 600 -- 'bash tools/round2_kickoff.sh'
set -x
mkdir -p gpurun_out

# (re)build the microbenches if the binaries are absent — they are
# git-ignored; normally the in-tree binaries travel with the snapshot
for mb in scatter_ab append_ab; do
  [ -x "tools/microbench/$mb" ] || \
    hipcc --offload-arch=gfx950 -O3 "tools/microbench/$mb.hip" \
          -o "tools/microbench/$mb"
done

# 1. Scatter A/B: v7 (2-row vectorized tile sort) and v8 (padded segments,
#    no pre-count) vs the shipped v5 — fold the winner into
#    k_tile_scatter1/2 (v8 also needs the slack+retry notes in §8b).
timeout 120 ./tools/microbench/scatter_ab | tee gpurun_out/r2_scatter_ab.txt

# 2. Append plateau: v6 (tile-staged compaction) vs the shipped per-wave
#    staged appenders (v2/v4) — if v6 wins, it drops into
#    k_ord_filter_date (Q3) and k_q9_filter (Q9-mix).
timeout 120 ./tools/microbench/append_ab | tee gpurun_out/r2_append_ab.txt

# 3. Current generic-op baselines on this box (tile default ON), for
#    apples-to-apples before any kernel edits.
timeout 180 python tools/generic_ops_bench.py 2>&1 | tee gpurun_out/r2_ops_baseline.txt

# 4. Quick suite sanity on the fresh box.
timeout 700 python -m pytest tests -q -m gpu -x 2>&1 | tail -3 | tee gpurun_out/r2_suite.txt

# 5. All-cores CPU baseline (report shard-per-process totals
#    next to the 1-core scalar number; cores stated in the object).
timeout 300 python -c "
import sys; sys.argv=['bench']
import importlib.util, os, json
spec = importlib.util.spec_from_file_location('bench', 'bench.py'); b = importlib.util.module_from_spec(spec); spec.loader.exec_module(b)
import multiprocessing as mp
for w in (1, 32, mp.cpu_count()):
    cb = b.cpu_baseline('q1', 600_000_000 if w>1 else 150_000_000, shards=w)
    print(json.dumps(cb))
" | tee gpurun_out/r2_cpu_baseline_cores.txt
