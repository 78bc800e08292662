#!/bin/bash
# Round-3 bootstrap call (pre-baked at the end of r2): validate the v3
# conv kernels, A/B them per-op, get the deferred fp32 point and the
# gamma=0.5 GPU learning curve.  ~20 min.
set -x
mkdir -p gpurun_out/r3c1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r3c1/summary.txt

# 1) v3 kernel oracles (dgrad stride-decomposed, wgrad/fwd panel-staged)
SCALERL_EXPERIMENTAL=1 MIOPEN_FIND_MODE=1 timeout 600 python -m pytest \
  tests/test_conv_experimental.py tests/test_conv_backward_experimental.py \
  -m gpu -q > gpurun_out/r3c1/conv_tests.log 2>&1
echo "conv_tests rc=$? :: $(tail -1 gpurun_out/r3c1/conv_tests.log)" | tee -a $S

# 2) per-op A/B incl. v3 rows (drives the swap decision)
MIOPEN_FIND_MODE=1 timeout 600 python scripts/conv_kernel_bench.py \
  > gpurun_out/r3c1/conv_ops.log 2>&1
echo "conv_ops rc=$?" | tee -a $S
grep conv gpurun_out/r3c1/conv_ops.log | tee -a $S

# 3) GPU-scale learning at gamma=0.5 (script default since r2c7 analysis)
MIOPEN_FIND_MODE=1 timeout 480 python scripts/learning_evidence.py \
  --iters 300 > gpurun_out/r3c1/learn.log 2>&1
echo "learn rc=$? :: $(grep -o '{.*}' gpurun_out/r3c1/learn.log | tail -1)" | tee -a $S

# 4) fp32 point: pre-warm the fp32 MIOpen find OUTSIDE the bench timeout
#    (r2 attempts died inside find), then the bench with a warm cache
export MIOPEN_FIND_MODE=1
export MIOPEN_USER_DB_PATH=/tmp/scalerl_miopen
mkdir -p $MIOPEN_USER_DB_PATH
timeout 600 python scripts/learner_micro.py --batch-size 256 --steps 4 \
  --warmup 2 --dtype fp32 > gpurun_out/r3c1/micro_fp32.log 2>&1
echo "micro_fp32 rc=$? :: $(tail -1 gpurun_out/r3c1/micro_fp32.log)" | tee -a $S
timeout 480 python bench.py --steps 10 --warmup 4 --dtype fp32 \
  > gpurun_out/r3c1/bench_fp32.log 2>&1
echo "fp32 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r3c1/bench_fp32.log | tail -1)" | tee -a $S
cat $S
