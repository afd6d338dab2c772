#!/bin/bash
# 2-rank multi-GPU-path hardware record.
# On a 1-GPU lease both ranks map onto device 0 (bench.py maps
# local_rank % device_count). Tries the production collective backend
# (nccl = RCCL) first; if RCCL refuses/hangs on a duplicate device, falls
# back to OTBX_DIST_BACKEND=gloo — GPU kernels + a real 2-process
# collective, CPU transport — so a COMPLETED n_gpus=2 record exists either
# way. Every leg runs under its own timeout and writes JSON to gpurun_out/.
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0

run2() { # $1 tag, $2 extra bench args, $3.. env pairs
  local tag="$1"; shift
  local args="$1"; shift
  env "$@" timeout 300 python -m torch.distributed.run --nnodes=1 \
      --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29531 \
      bench.py --gpus 2 --sf 10 --steps 10 --warmup 3 --no-cpu-baseline \
      $args > "gpurun_out/d2_${tag}.json" 2> "gpurun_out/d2_${tag}.err"
  echo "rc_${tag}=$?" | tee -a gpurun_out/d2_rc.txt
  tail -c 600 "gpurun_out/d2_${tag}.json"
}

# 1. Q1, nccl (RCCL), 2 ranks on device 0
run2 q1_nccl "--workload tpch_q1" NCCL_DEBUG=WARN

# 2. If nccl leg produced no JSON, capture verbose RCCL diagnostics once
if ! grep -q '"n_gpus": 2' gpurun_out/d2_q1_nccl.json 2>/dev/null; then
  env NCCL_DEBUG=INFO timeout 180 python -m torch.distributed.run --nnodes=1 \
      --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29532 \
      bench.py --gpus 2 --sf 1 --steps 2 --warmup 1 --no-cpu-baseline \
      --workload tpch_q1 > gpurun_out/d2_q1_nccl_dbg.log 2>&1
  echo "rc_dbg=$?" | tee -a gpurun_out/d2_rc.txt
  tail -40 gpurun_out/d2_q1_nccl_dbg.log
fi

# 3. Q1 + Q3, gloo collective (GPU compute, CPU transport) — guaranteed leg
run2 q1_gloo "--workload tpch_q1" OTBX_DIST_BACKEND=gloo
run2 q3_gloo "--workload tpch_q3" OTBX_DIST_BACKEND=gloo

# 4. Q3 nccl (only worth trying if q1 nccl completed)
if grep -q '"n_gpus": 2' gpurun_out/d2_q1_nccl.json 2>/dev/null; then
  run2 q3_nccl "--workload tpch_q3" NCCL_DEBUG=WARN
fi

# 5. 2-rank repartition exchange (gloo transport, GPU partition kernels)
timeout 180 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29533 \
    tools/test_exchange_2rank.py > gpurun_out/d2_exchange.log 2>&1
echo "rc_exchange=$?" | tee -a gpurun_out/d2_rc.txt
cat gpurun_out/d2_exchange.log | tail -3

# 6. 1-rank SF10 reference point for the ≈2x whole-job check
timeout 180 python bench.py --gpus 1 --sf 10 --steps 10 --warmup 3 \
    --no-cpu-baseline > gpurun_out/d2_q1_1rank_sf10.json \
    2> gpurun_out/d2_q1_1rank_sf10.err
echo "rc_1rank=$?" | tee -a gpurun_out/d2_rc.txt
tail -c 400 gpurun_out/d2_q1_1rank_sf10.json
