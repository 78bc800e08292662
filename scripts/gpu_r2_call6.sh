#!/bin/bash
# Round-2 GPU call 6: conv oracle re-run (wrapper use-after-free fixed),
# eager-vs-graph default decision, kept perf numbers, rocprof CSV.
set -x
mkdir -p gpurun_out/r2c6
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c6/summary.txt

# 1) conv oracles (wrapper fixed — expect green)
SCALERL_EXPERIMENTAL=1 timeout 420 python -m pytest \
  tests/test_conv_experimental.py tests/test_conv_backward_experimental.py \
  -m gpu -q > gpurun_out/r2c6/conv_tests.log 2>&1
echo "conv_tests rc=$? :: $(tail -1 gpurun_out/r2c6/conv_tests.log)" | tee -a $S

# 2) eager E256 (graph-vs-eager default decision; graph numbers exist)
timeout 420 python bench.py --steps 15 --warmup 6 \
  > gpurun_out/r2c6/bench_eager_e256.log 2>&1
echo "eager_e256 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c6/bench_eager_e256.log | tail -1)" | tee -a $S

# 3) GPU-scale learning evidence (bf16, bench shapes)
timeout 420 python scripts/learning_evidence.py --iters 120 \
  > gpurun_out/r2c6/learn.log 2>&1
echo "learn rc=$? :: $(grep -o '{.*}' gpurun_out/r2c6/learn.log | tail -1)" | tee -a $S

# 4) fp32 bench point
timeout 420 python bench.py --steps 10 --warmup 4 --dtype fp32 \
  > gpurun_out/r2c6/bench_fp32.log 2>&1
echo "fp32 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c6/bench_fp32.log | tail -1)" | tee -a $S

# 5) configs 2/4/5 numbers
timeout 420 python scripts/bench_a3c.py --steps 25 --warmup 8 \
  > gpurun_out/r2c6/a3c.log 2>&1
echo "a3c rc=$? :: $(grep -o '{.*}' gpurun_out/r2c6/a3c.log | tail -1)" | tee -a $S
timeout 420 python scripts/bench_apex.py > gpurun_out/r2c6/apex.log 2>&1
echo "apex rc=$? :: $(tail -1 gpurun_out/r2c6/apex.log)" | tee -a $S
timeout 420 python scripts/bench_ddppo.py > gpurun_out/r2c6/ddppo.log 2>&1
echo "ddppo rc=$? :: $(tail -1 gpurun_out/r2c6/ddppo.log)" | tee -a $S

# 6) rocprof kernel-stats CSV with the hand-written kernels on the path
cd /tmp && export TMPDIR=/tmp
SCALERL_NATIVE_CONV=1 timeout 420 rocprofv3 --kernel-trace --stats \
  --output-format csv -d /tmp/prof -o micro \
  -- python /root/repo/scripts/learner_micro.py --batch-size 256 --steps 10 --warmup 4 \
  > /root/repo/gpurun_out/r2c6/rocprof.log 2>&1
echo "rocprof rc=$?" | tee -a /root/repo/$S
for f in $(find /tmp/prof -name '*stats*.csv' | head -3); do
  head -45 "$f" > /root/repo/gpurun_out/r2c6/$(basename $f)
done
cat /root/repo/$S
