# Developer entry points (reference Makefile:8-10)
PY ?= python

.PHONY: build test test-gpu standalone bench lint

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -q -m gpu

standalone:
	bash tests/run_tests.sh

bench:
	$(PY) bench.py --steps 10 --warmup 3

accuracy:
	$(PY) benchmarks/accuracy.py
