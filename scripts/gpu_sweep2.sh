set -x
mkdir -p gpurun_out
# fail-fast sanity: tiny gpu-inference run
timeout 180 python -m pytest tests/test_impala_integration.py -m gpu -x -q > gpurun_out/pytest_gpu3.log 2>&1; echo "ITEST=$?"; tail -3 gpurun_out/pytest_gpu3.log
run() { name=$1; shift; timeout 240 python bench.py --steps 25 --warmup 8 "$@" > gpurun_out/bench_$name.log 2>&1; echo "BENCH_$name=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_$name.log | head -1; grep -A8 "learner timings" gpurun_out/bench_$name.log | head -10; }
run gpu8x16 --inference gpu --num-actors 8 --envs-per-actor 16
run gpu16x16 --inference gpu --num-actors 16 --envs-per-actor 16
run gpu16x32 --inference gpu --num-actors 16 --envs-per-actor 32 --batch-size 32
run gpu12x64 --inference gpu --num-actors 12 --envs-per-actor 64 --batch-size 64
