# Convenience targets (the reference drives everything through make +
# build.sh; here the Python/HIP build is one in-tree extension).
PY ?= python

.PHONY: build test test-gpu bench clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 30 --warmup 8

clean:
	rm -rf build lightctr_amd/ops/csrc/*_hip.hip lightctr_amd/ops/csrc/*_hip.cpp
