#!/bin/bash
# Round-2 GPU call 2: dgrad3 debug, new-default bench A/Bs (seq LSTM,
# graph-before-worker, bf16 inference, actor sweep), rocprof kernel stats.
set -x
mkdir -p gpurun_out/r2c2
export MIOPEN_FIND_MODE=1
export HSA_ENABLE_IPC_MODE_LEGACY=0
S=gpurun_out/r2c2/summary.txt

# 0) quick numerics re-check with seq LSTM now default
timeout 420 python -m pytest tests/test_lstm.py tests/test_vtrace.py -m gpu -q \
  > gpurun_out/r2c2/lstm_tests.log 2>&1
echo "lstm_tests rc=$? :: $(tail -1 gpurun_out/r2c2/lstm_tests.log)" | tee -a $S

# 1) dgrad3 structural debug
timeout 300 python scripts/debug_dgrad3.py > gpurun_out/r2c2/dgrad3.log 2>&1
echo "dgrad3 rc=$?" | tee -a $S

run_bench () {
  name=$1; shift
  timeout 480 env "$@" python bench.py --steps 15 --warmup 6 \
    > gpurun_out/r2c2/bench_$name.log 2>&1
  echo "bench_$name rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c2/bench_$name.log | tail -1)" | tee -a $S
}
# 2) new defaults, eager
run_bench eager NOOP=1
# 3) graph capture before worker
timeout 480 python bench.py --steps 15 --warmup 6 --use-graph 1 \
  > gpurun_out/r2c2/bench_graph.log 2>&1
echo "bench_graph rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c2/bench_graph.log | tail -1)" | tee -a $S
# 4) + bf16 inference worker
timeout 480 env SCALERL_INF_BF16=1 python bench.py --steps 15 --warmup 6 --use-graph 1 \
  > gpurun_out/r2c2/bench_graph_infbf16.log 2>&1
echo "bench_graph_infbf16 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c2/bench_graph_infbf16.log | tail -1)" | tee -a $S
# 5) actor sweep at the faster learner
timeout 480 env SCALERL_INF_BF16=1 python bench.py --steps 15 --warmup 6 --use-graph 1 --num-actors 32 \
  > gpurun_out/r2c2/bench_a32.log 2>&1
echo "bench_a32 rc=$? :: $(grep -o '{\"metric.*}' gpurun_out/r2c2/bench_a32.log | tail -1)" | tee -a $S
nproc >> $S

# 6) rocprof kernel stats of the learner micro (guides conv optimization)
cd /tmp && export TMPDIR=/tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof -o micro \
  -- python /root/repo/scripts/learner_micro.py --batch-size 256 --steps 12 --warmup 4 \
  > /root/repo/gpurun_out/r2c2/rocprof_micro.log 2>&1
echo "rocprof rc=$?" | tee -a /root/repo/$S
find /tmp/prof -name '*stats*' | head -5 >> /root/repo/$S
for f in $(find /tmp/prof -name '*kernel_stats*.csv' | head -2); do
  head -40 "$f" > /root/repo/gpurun_out/r2c2/$(basename $f)
done
cat /root/repo/$S
