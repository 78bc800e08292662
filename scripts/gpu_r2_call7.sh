#!/bin/bash
# Final r2 GPU call: learning evidence with enough updates + fp32 point.
set -x
mkdir -p gpurun_out/r2c7
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c7/summary.txt

export MIOPEN_FIND_MODE=1
timeout 300 python scripts/learning_evidence.py --iters 600 \
  --learning-rate 6e-4 --entropy-cost 0.003 \
  > gpurun_out/r2c7/learn.log 2>&1
echo "learn rc=$? :: $(grep -o '{.*}' gpurun_out/r2c7/learn.log | tail -1)" | tee -a $S

# fp32 point: default (hybrid) MIOpen find — exhaustive fp32 find blew the
# 420 s budget in r2c6; disclose the find mode with the number
unset MIOPEN_FIND_MODE
timeout 260 python bench.py --steps 8 --warmup 3 --dtype fp32 \
  > gpurun_out/r2c7/bench_fp32.log 2>&1
echo "fp32 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c7/bench_fp32.log | tail -1)" | tee -a $S
cat $S
