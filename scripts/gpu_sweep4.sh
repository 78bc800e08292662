set -x
mkdir -p gpurun_out
timeout 200 python scripts/learner_micro.py --batch-size 32 > gpurun_out/micro_b32.log 2>&1; echo "M32=$?"; cat gpurun_out/micro_b32.log
timeout 200 python scripts/learner_micro.py --batch-size 128 > gpurun_out/micro_b128.log 2>&1; echo "M128=$?"; cat gpurun_out/micro_b128.log
timeout 200 python scripts/learner_micro.py --batch-size 128 --use-lstm 0 > gpurun_out/micro_b128_nolstm.log 2>&1; echo "M128NL=$?"; cat gpurun_out/micro_b128_nolstm.log
export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o micro -- python scripts/learner_micro.py --steps 15 --warmup 8 --batch-size 128 > gpurun_out/micro_prof.log 2>&1; echo "PROF=$?"
find gpurun_out/prof -type f 2>/dev/null | head -5
run() { name=$1; shift; timeout 240 python bench.py --steps 30 --warmup 8 "$@" > gpurun_out/bench_$name.log 2>&1; echo "BENCH_$name=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_$name.log | head -1; grep -A10 "learner timings" gpurun_out/bench_$name.log | head -12; }
run pf16x64b64 --inference gpu --num-actors 16 --envs-per-actor 64 --batch-size 64
run pf16x128b128 --inference gpu --num-actors 16 --envs-per-actor 128 --batch-size 128
run pf24x128b128 --inference gpu --num-actors 24 --envs-per-actor 128 --batch-size 128
run pf24x128b256 --inference gpu --num-actors 24 --envs-per-actor 128 --batch-size 256
