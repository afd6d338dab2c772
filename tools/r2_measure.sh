#!/bin/bash
# Round-2 final measurement sweep (driver-visible Q3/Q9
# records WITH cpu_baseline + vs_baseline, plus all-cores CPU figures and
# SF300/skew robustness). Writes JSON under gpurun_out/ for commit into
# profiles/.
set -x
mkdir -p gpurun_out

# 1. headline benches with CPU baselines (scalar, 1 core)
timeout 300 python bench.py --steps 20 --warmup 5 \
    > gpurun_out/m_q1.json 2> gpurun_out/m_q1.err; echo q1=$?
timeout 300 python bench.py --workload tpch_q3 --steps 15 --warmup 4 \
    > gpurun_out/m_q3.json 2> gpurun_out/m_q3.err; echo q3=$?
timeout 300 python bench.py --workload tpch_q9mix --steps 15 --warmup 4 \
    > gpurun_out/m_q9.json 2> gpurun_out/m_q9.err; echo q9=$?

# 2. all-cores CPU baselines (one oracle process per shard; SURVEY §8d both
#    per-shard and all-cores totals). 32 shards ≈ the host's sweet spot
#    (r2_cpu_baseline_cores.txt); report nproc too.
timeout 600 python -c "
import importlib.util, json, multiprocessing as mp, sys
sys.argv = ['bench']
spec = importlib.util.spec_from_file_location('bench', 'bench.py')
b = importlib.util.module_from_spec(spec); spec.loader.exec_module(b)
out = {'nproc': mp.cpu_count(), 'runs': []}
for q, rows in [('q1', 600_000_000), ('q3', 240_000_000),
                ('q9', 240_000_000)]:
    for w in (1, 32):
        cb = b.cpu_baseline(q, rows if w > 1 else rows // 4, shards=w)
        cb['query'] = q
        out['runs'].append(cb)
        print(json.dumps(cb), flush=True)
json.dump(out, open('gpurun_out/m_cpu_allcores.json', 'w'), indent=1)
" 2>&1 | tail -8

# 3. SF300 robustness (3x contract size per GPU)
for wl in tpch_q1 tpch_q3 tpch_q9mix; do
  timeout 300 python bench.py --workload $wl --sf 300 --steps 5 --warmup 2 \
      --no-cpu-baseline > gpurun_out/m_${wl}_sf300.json \
      2> gpurun_out/m_${wl}_sf300.err; echo ${wl}_sf300=$?
done

# 4. config-5 skewed distribution keys (20% hot custkeys -> 80% of orders)
timeout 300 python bench.py --workload tpch_q3 --skew --steps 10 --warmup 3 \
    --no-cpu-baseline > gpurun_out/m_q3_skew.json \
    2> gpurun_out/m_q3_skew.err; echo q3_skew=$?

python3 - <<'EOF'
import json
for f in ["m_q1", "m_q3", "m_q9", "m_tpch_q1_sf300", "m_tpch_q3_sf300",
          "m_tpch_q9mix_sf300", "m_q3_skew"]:
    try:
        j = json.load(open(f"gpurun_out/{f}.json"))
        cb = j.get("cpu_baseline") or {}
        print(f, round(j["ms_per_step"], 3), "ms",
              round(j["value"] / 1e9, 1), "Grows/s",
              "vs_cpu", round(j["vs_baseline"], 1) if j.get("vs_baseline")
              else None, "cpu", round(cb.get("value", 0) / 1e6, 1), "M/s")
    except Exception as e:
        print(f, "ERR", e)
EOF
