set -x
mkdir -p gpurun_out
timeout 240 python -m pytest tests/test_impala_integration.py -m gpu -x -q > gpurun_out/pytest_gpu4.log 2>&1; echo "ITEST=$?"; tail -3 gpurun_out/pytest_gpu4.log
# learner ceiling first (single process)
timeout 240 python scripts/learner_micro.py --batch-size 32 > gpurun_out/micro_b32.log 2>&1; echo "M32=$?"; cat gpurun_out/micro_b32.log
timeout 240 python scripts/learner_micro.py --batch-size 64 > gpurun_out/micro_b64.log 2>&1; echo "M64=$?"; cat gpurun_out/micro_b64.log
timeout 240 python scripts/learner_micro.py --batch-size 128 > gpurun_out/micro_b128.log 2>&1; echo "M128=$?"; cat gpurun_out/micro_b128.log
# rocprof the learner micro (single process, clean kernel stats)
export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o micro -- python scripts/learner_micro.py --steps 20 --warmup 8 > gpurun_out/micro_prof.log 2>&1; echo "PROF=$?"
find gpurun_out/prof -type f 2>/dev/null | head
# full-pipeline sweep at bigger env counts
run() { name=$1; shift; timeout 240 python bench.py --steps 25 --warmup 8 "$@" > gpurun_out/bench_$name.log 2>&1; echo "BENCH_$name=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_$name.log | head -1; grep -A9 "learner timings" gpurun_out/bench_$name.log | head -11; }
run gpu16x64b64 --inference gpu --num-actors 16 --envs-per-actor 64 --batch-size 64
run gpu24x64b64 --inference gpu --num-actors 24 --envs-per-actor 64 --batch-size 64
run gpu16x128b128 --inference gpu --num-actors 16 --envs-per-actor 128 --batch-size 128
