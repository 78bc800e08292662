#!/bin/bash
# Round-2 GPU call 4: thread-mode inference validation (kills the
# cross-process hipGraph fault class), conv oracle re-check vs CPU
# references, per-op conv timings, default-candidate benches.
set -x
mkdir -p gpurun_out/r2c4
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c4/summary.txt

# 1) conv oracles vs CPU references
SCALERL_EXPERIMENTAL=1 timeout 600 python -m pytest \
  tests/test_conv_experimental.py tests/test_conv_backward_experimental.py \
  -m gpu -q > gpurun_out/r2c4/conv_tests.log 2>&1
echo "conv_tests rc=$? :: $(tail -1 gpurun_out/r2c4/conv_tests.log)" | tee -a $S

# 2) per-op conv timings (native vs MIOpen) at N=20736
timeout 600 python scripts/conv_kernel_bench.py \
  > gpurun_out/r2c4/conv_ops.log 2>&1
echo "conv_ops rc=$?" | tee -a $S
cat gpurun_out/r2c4/conv_ops.log | grep conv | tee -a $S

run_bench () {
  name=$1; shift
  timeout 480 python bench.py --steps 15 --warmup 6 "$@" \
    > gpurun_out/r2c4/bench_$name.log 2>&1
  rc=$?
  line=$(grep -o '{\"metric.*}' gpurun_out/r2c4/bench_$name.log | tail -1)
  echo "bench_$name rc=$rc :: $line" | tee -a $S
}
# 3) thread-mode inference (now default) benches
run_bench thr_eager_e128
run_bench thr_graph_e128 --use-graph 1
run_bench thr_graph_e256 --use-graph 1 --envs-per-actor 256
run_bench thr_graph_e256_db --use-graph 1 --envs-per-actor 256 --double-buffer 1
cat $S
