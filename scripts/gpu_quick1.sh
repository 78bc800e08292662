set -x
mkdir -p gpurun_out
timeout 200 python -m pytest tests/test_impala_integration.py::test_impala_gpu_learner_end_to_end -m gpu -x -q > gpurun_out/pytest_g1.log 2>&1; echo "T1=$?"; tail -2 gpurun_out/pytest_g1.log
timeout 280 python bench.py --steps 40 --warmup 10 --inference gpu --num-actors 24 --envs-per-actor 128 --batch-size 256 > gpurun_out/bench_graph_b256.log 2>&1; echo "B=$?"
grep -o '"value": [0-9.]*\|"ms_per_step": [0-9.]*' gpurun_out/bench_graph_b256.log | head -2
grep -A10 "learner timings" gpurun_out/bench_graph_b256.log | head -12
