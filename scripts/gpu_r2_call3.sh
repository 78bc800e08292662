#!/bin/bash
# Round-2 GPU call 3: v2 conv kernel validation + graph-capture fix +
# feeding sweeps (the learner is data-starved: dequeue+gather = 82%).
set -x
mkdir -p gpurun_out/r2c3
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c3/summary.txt

# 1) v2 conv oracle tests (fwd + wgrad/dgrad)
SCALERL_EXPERIMENTAL=1 timeout 600 python -m pytest \
  tests/test_conv_experimental.py tests/test_conv_backward_experimental.py \
  -m gpu -q > gpurun_out/r2c3/conv_tests.log 2>&1
echo "conv_tests rc=$? :: $(tail -1 gpurun_out/r2c3/conv_tests.log)" | tee -a $S

# 2) micro A/B: MIOpen vs native v2 convs
timeout 420 python scripts/learner_micro.py --batch-size 256 --steps 20 --warmup 6 \
  > gpurun_out/r2c3/micro_base.log 2>&1
echo "micro_base rc=$? :: $(tail -1 gpurun_out/r2c3/micro_base.log)" | tee -a $S
SCALERL_NATIVE_CONV=1 timeout 420 python scripts/learner_micro.py \
  --batch-size 256 --steps 20 --warmup 6 > gpurun_out/r2c3/micro_conv.log 2>&1
echo "micro_conv rc=$? :: $(tail -1 gpurun_out/r2c3/micro_conv.log)" | tee -a $S

run_bench () {
  name=$1; shift
  timeout 480 python bench.py --steps 15 --warmup 6 "$@" \
    > gpurun_out/r2c3/bench_$name.log 2>&1
  rc=$?
  line=$(grep -o '{\"metric.*}' gpurun_out/r2c3/bench_$name.log | tail -1)
  deq=$(grep -E 'dequeue|total:' gpurun_out/r2c3/bench_$name.log | tr '\n' ' ')
  echo "bench_$name rc=$rc :: $line :: $deq" | tee -a $S
}
# 3) graph fix validation (per-step LSTM inside capture)
run_bench graph --use-graph 1
# 4) feeding sweeps (graph + bf16 inference)
export SCALERL_INF_BF16=1
run_bench g_bf16      --use-graph 1
run_bench g_bf16_a32  --use-graph 1 --num-actors 32
run_bench g_bf16_e256 --use-graph 1 --envs-per-actor 256
cat $S
