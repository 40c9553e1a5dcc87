# MI355X gpushare device plugin
PY ?= python3

.PHONY: build test test-gpu bench bench-all scale lint image clean

build:
	$(PY) -m gpushare_amd.native.build

test: build
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests/ -q -m gpu

bench: build
	$(PY) bench.py --gpus 1 --steps 10 --warmup 2

bench-all: build            # all six BASELINE configs
	$(PY) benchmarks/run_all.py

scale: build                # co-location + inventory-scale experiments
	$(PY) benchmarks/colocation_fairness.py
	$(PY) benchmarks/inventory_scale.py

lint:
	$(PY) -m compileall -q gpushare_amd tests bench.py __graft_entry__.py
	$(PY) tools/lint.py

image:
	docker build -t gpushare/amd-device-plugin:latest .

clean:
	rm -f gpushare_amd/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +
