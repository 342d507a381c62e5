# Convenience targets (the canonical build is python -m byteps_amd.ops.build)

.PHONY: build test test-gpu bench clean

build:
	python -m byteps_amd.ops.build

test: build
	python -m pytest tests/ -x -q -m "not gpu"

test-gpu: build
	python -m pytest tests/ -x -q -m gpu

bench: build
	python bench.py --steps 30 --warmup 10

clean:
	rm -f byteps_amd/ops/_core.so
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true
