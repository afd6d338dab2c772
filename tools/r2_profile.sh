#!/bin/bash
# Round-2 profiling sweep: kernel traces + PMC FETCH_SIZE/WRITE_SIZE for
# Q1/Q3/Q9 (separate --pmc passes per the gpurun counter-collection rule;
# never combined with trace domains). Summaries via tools/rocpd_summary.py.
set -x
export TMPDIR=/tmp
mkdir -p gpurun_out

run_prof() { # $1 tag, $2 rocprof args, $3 bench args
  local tag="$1" pargs="$2" bargs="$3"
  mkdir -p "gpurun_out/p_${tag}"
  timeout 300 rocprofv3 $pargs -d "gpurun_out/p_${tag}" -o runc -- \
      python bench.py $bargs --no-cpu-baseline \
      > "gpurun_out/p_${tag}.log" 2>&1
  echo "${tag}=$?"
  python tools/rocpd_summary.py gpurun_out/p_${tag}/*.db \
      --json "gpurun_out/p_${tag}.sum.json" \
      > "gpurun_out/p_${tag}.sum.txt" 2>&1
  rm -rf "gpurun_out/p_${tag}"   # the .db files are large; keep summaries
}

for wl in tpch_q1 tpch_q3 tpch_q9mix; do
  short=${wl#tpch_}
  run_prof "${short}_trace" "--kernel-trace --stats" \
           "--workload $wl --steps 7 --warmup 2"
  run_prof "${short}_fetch" "--pmc FETCH_SIZE" \
           "--workload $wl --steps 5 --warmup 2"
  run_prof "${short}_write" "--pmc WRITE_SIZE" \
           "--workload $wl --steps 5 --warmup 2"
done

for f in gpurun_out/p_*_trace.sum.txt; do echo "== $f"; head -8 "$f"; done
for f in gpurun_out/p_*_fetch.sum.txt gpurun_out/p_*_write.sum.txt; do
  echo "== $f"; head -10 "$f"
done
