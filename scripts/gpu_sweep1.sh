set -x
mkdir -p gpurun_out
timeout 240 python -m pytest tests/test_impala_integration.py -m gpu -x -q > gpurun_out/pytest_gpu2.log 2>&1; echo "ITEST=$?"; tail -3 gpurun_out/pytest_gpu2.log
run() { name=$1; shift; timeout 300 python bench.py --steps 25 --warmup 8 "$@" > gpurun_out/bench_$name.log 2>&1; echo "BENCH_$name=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_$name.log | head -1; }
run cpu64  --inference cpu --num-actors 64
run gpu8x16 --inference gpu --num-actors 8 --envs-per-actor 16
run gpu16x16 --inference gpu --num-actors 16 --envs-per-actor 16
run gpu16x32 --inference gpu --num-actors 16 --envs-per-actor 32 --batch-size 32
run gpu12x64 --inference gpu --num-actors 12 --envs-per-actor 64 --batch-size 64
export TMPDIR=/tmp
(cd /tmp && true)
timeout 300 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o prof1 -- python bench.py --steps 10 --warmup 5 --inference gpu --num-actors 12 --envs-per-actor 16 > gpurun_out/bench_prof.log 2>&1; echo "PROF=$?"
find gpurun_out/prof -type f | head -10
