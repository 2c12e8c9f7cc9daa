# Convenience targets (see README.md / MIGRATING.md for details)

.PHONY: build test test-gpu bench sweep soak clean

build:
	python -m feddrift_amd.ops.build

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:                      # on an MI355X box
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --steps 1000 --warmup 200

sweep:                         # client-count throughput ladder (GPU)
	python scripts/bench_sweep.py

clean:
	rm -rf feddrift_amd/ops/hip/_build
