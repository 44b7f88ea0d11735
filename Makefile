# kubegpu-amd build targets (reference analog: /root/reference/Makefile).
PY ?= python

.PHONY: all native test test-gpu bench clean

all: native

native:
	$(PY) -m kubegpu_amd.build_native

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 20 --warmup 5

clean:
	rm -rf kubegpu_amd/csrc/bin kubegpu_amd/_ext kubegpu_amd/_schedcore*.so
	find . -name __pycache__ -type d -exec rm -rf {} +
