set -x
mkdir -p gpurun_out
timeout 280 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_final.log 2>&1; echo "PYTEST=$?"; tail -3 gpurun_out/pytest_gpu_final.log
timeout 260 python bench.py --steps 30 --warmup 8 > gpurun_out/bench_default.log 2>&1; echo "BENCH=$?"
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_default.log | head -2
export TMPDIR=/tmp
timeout 260 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o micro -- python scripts/learner_micro.py --steps 15 --warmup 6 --batch-size 256 > gpurun_out/micro_prof.log 2>&1; echo "PROF=$?"; tail -1 gpurun_out/micro_prof.log
find gpurun_out/prof -name "*.csv" | head -5
for f in $(find gpurun_out/prof -name "*stats*.csv" | head -2); do echo "== $f"; head -15 "$f"; done
