#!/bin/bash
# Round-2 GPU call 6: kept perf numbers for the non-flagship configs +
# learning evidence + fp32 point + rocprof kernel-stats CSV.
set -x
mkdir -p gpurun_out/r2c6
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c6/summary.txt

# 1) GPU-scale learning evidence (bf16, bench shapes)
timeout 480 python scripts/learning_evidence.py --iters 120 \
  > gpurun_out/r2c6/learn.log 2>&1
echo "learn rc=$? :: $(grep -o '{.*}' gpurun_out/r2c6/learn.log | tail -1)" | tee -a $S

# 2) fp32 bench point
timeout 480 python bench.py --steps 12 --warmup 5 --dtype fp32 \
  > gpurun_out/r2c6/bench_fp32.log 2>&1
echo "fp32 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c6/bench_fp32.log | tail -1)" | tee -a $S

# 3) BASELINE config 2 (A3C 42x42, 16 CPU actors + 1 GPU learner)
timeout 480 python scripts/bench_a3c.py --steps 25 --warmup 8 \
  > gpurun_out/r2c6/a3c.log 2>&1
echo "a3c rc=$? :: $(grep -o '{.*}' gpurun_out/r2c6/a3c.log | tail -1)" | tee -a $S

# 4) Ape-X (config 4 shape, 1 rank)
timeout 480 python scripts/bench_apex.py \
  > gpurun_out/r2c6/apex.log 2>&1
echo "apex rc=$? :: $(tail -1 gpurun_out/r2c6/apex.log)" | tee -a $S

# 5) DD-PPO (config 5 shape, 1 rank)
timeout 480 python scripts/bench_ddppo.py \
  > gpurun_out/r2c6/ddppo.log 2>&1
echo "ddppo rc=$? :: $(tail -1 gpurun_out/r2c6/ddppo.log)" | tee -a $S

# 6) rocprof kernel stats CSV (hand-written kernels on the hot path)
cd /tmp && export TMPDIR=/tmp
SCALERL_NATIVE_CONV=1 timeout 480 rocprofv3 --kernel-trace --stats \
  --output-format csv -d /tmp/prof -o micro \
  -- python /root/repo/scripts/learner_micro.py --batch-size 256 --steps 10 --warmup 4 \
  > /root/repo/gpurun_out/r2c6/rocprof.log 2>&1
echo "rocprof rc=$?" | tee -a /root/repo/$S
for f in $(find /tmp/prof -name '*stats*.csv' | head -3); do
  head -40 "$f" > /root/repo/gpurun_out/r2c6/$(basename $f)
done
ls /tmp/prof >> /root/repo/$S
cat /root/repo/$S
