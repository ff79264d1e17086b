# dragnet_amd developer targets (the reference's Makefile analog:
# `make test` there runs catest -a; here the pytest suites).

PY ?= python3

.PHONY: all build test test-gpu soak bench clean

all: build

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

soak:
	$(PY) tools_dev/soak_cpu.py

bench:
	$(PY) bench.py --steps 10 --warmup 3

clean:
	rm -rf build dragnet_amd/ops/_dragnet_hip*.so \
	    dragnet_amd/index/_csink*.so dragnet_amd/index/_points*.so \
	    dragnet_amd/ops/hip/*_hip.hip
