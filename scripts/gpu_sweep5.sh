set -x
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_impala_integration.py -m gpu -x -q > gpurun_out/pytest_gpu5.log 2>&1; echo "ITEST=$?"; tail -3 gpurun_out/pytest_gpu5.log
run() { name=$1; shift; timeout 280 python bench.py --steps 40 --warmup 10 "$@" > gpurun_out/bench_$name.log 2>&1; echo "BENCH_$name=$?"; grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_$name.log | head -2; grep -A10 "learner timings" gpurun_out/bench_$name.log | head -12; }
run g16x128b128 --inference gpu --num-actors 16 --envs-per-actor 128 --batch-size 128
run g24x128b256 --inference gpu --num-actors 24 --envs-per-actor 128 --batch-size 256
run g32x128b256 --inference gpu --num-actors 32 --envs-per-actor 128 --batch-size 256
run g32x128b512 --inference gpu --num-actors 32 --envs-per-actor 128 --batch-size 512
