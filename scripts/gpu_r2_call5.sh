#!/bin/bash
# Round-2 GPU call 5: dgrad race debug, thread-mode retry (seq-LSTM gated
# to T>1), and default-candidate benches.
set -x
mkdir -p gpurun_out/r2c5
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c5/summary.txt

# 1) dgrad structural/race debug
timeout 420 python scripts/debug_dgrad3.py > gpurun_out/r2c5/dgrad.log 2>&1
echo "dgrad rc=$?" | tee -a $S
grep -E "run-to-run|total bad|with-sync" gpurun_out/r2c5/dgrad.log | tee -a $S

run_bench () {
  name=$1; shift
  timeout 480 python bench.py --steps 15 --warmup 6 "$@" \
    > gpurun_out/r2c5/bench_$name.log 2>&1
  rc=$?
  line=$(grep -o '{\"metric.*}' gpurun_out/r2c5/bench_$name.log | tail -1)
  echo "bench_$name rc=$rc :: $line" | tee -a $S
}
# 2) thread-mode inference retry
run_bench thr_eager_e128
run_bench thr_graph_e256 --use-graph 1 --envs-per-actor 256
# 3) process-mode fallback confirm (the proven 539k config)
run_bench proc_graph_e256 --use-graph 1 --envs-per-actor 256 --inference-worker process
cat $S
