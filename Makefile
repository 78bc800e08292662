# Developer shortcuts (the driver uses bench.py / __graft_entry__.py directly)

.PHONY: build test test-gpu bench micro clean

build:
	python -c "from scalerl_amd.ops import build_kernels; build_kernels()"

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --steps 30 --warmup 10

micro:
	python scripts/learner_micro.py --batch-size 128

clean:
	rm -f scalerl_amd/ops/_hip_ops.so
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true
