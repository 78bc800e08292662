#!/bin/bash
# Round-2 GPU call 1: hardware-validate the three gated perf paths and A/B
# them on the learner micro.  Everything bounded by `timeout`; outputs under
# gpurun_out/r2c1/.
set -x
mkdir -p gpurun_out/r2c1
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0

# 1) gated oracle tests: conv fwd+bwd, lstm (incl. seq path), td/ppo
SCALERL_EXPERIMENTAL=1 timeout 600 python -m pytest \
  tests/test_conv_experimental.py tests/test_conv_backward_experimental.py \
  tests/test_lstm.py tests/test_td.py tests/test_vtrace.py \
  -m gpu -q > gpurun_out/r2c1/gated_tests.log 2>&1
echo "gated tests rc=$?" | tee -a gpurun_out/r2c1/summary.txt

# 2) learner micro A/B at B=256 (the bench batch)
run_micro () {
  name=$1; shift
  timeout 420 env "$@" python scripts/learner_micro.py \
    --batch-size 256 --steps 20 --warmup 6 \
    > gpurun_out/r2c1/micro_$name.log 2>&1
  echo "micro_$name rc=$? :: $(tail -1 gpurun_out/r2c1/micro_$name.log)" \
    | tee -a gpurun_out/r2c1/summary.txt
}
run_micro base NOOP=1
run_micro seq SCALERL_LSTM_SEQ=1
run_micro conv SCALERL_NATIVE_CONV=1
run_micro conv_seq SCALERL_NATIVE_CONV=1 SCALERL_LSTM_SEQ=1

# 3) hipGraph learner step under the full pipeline (inference worker
#    coexistence — the round-1 HSA fault repro)
timeout 600 python bench.py --steps 12 --warmup 6 --use-graph 1 \
  > gpurun_out/r2c1/bench_graph.log 2>&1
echo "bench_graph rc=$? :: $(grep -o '{.*}' gpurun_out/r2c1/bench_graph.log | tail -1)" \
  | tee -a gpurun_out/r2c1/summary.txt

# 4) graph + native conv + seq-lstm combined micro-ish bench run
timeout 600 env SCALERL_NATIVE_CONV=1 SCALERL_LSTM_SEQ=1 \
  python bench.py --steps 12 --warmup 6 --use-graph 1 \
  > gpurun_out/r2c1/bench_graph_native.log 2>&1
echo "bench_graph_native rc=$? :: $(grep -o '{.*}' gpurun_out/r2c1/bench_graph_native.log | tail -1)" \
  | tee -a gpurun_out/r2c1/summary.txt

cat gpurun_out/r2c1/summary.txt
