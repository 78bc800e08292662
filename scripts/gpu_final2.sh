mkdir -p gpurun_out
rm -rf gpurun_out/prof gpurun_out/*.log 2>/dev/null
timeout 280 python bench.py --steps 25 --warmup 8 > gpurun_out/bench_default2.log 2>&1; echo "BENCH=$?"
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_default2.log | head -2
grep -A10 "learner timings" gpurun_out/bench_default2.log | head -12
