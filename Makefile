# dragnet_amd build/test entry points (the reference's make test analog)
.PHONY: build test test-gpu bench lint

build:
	PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 10 --warmup 3

lint:
	python -m pyflakes dragnet_amd bench.py __graft_entry__.py || true
