# Build/test entry points (the reference drives everything through its
# Makefile too). `make build` cross-compiles the gfx950 extension on any box.
PY ?= python

.PHONY: build test test-gpu bench clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) -m dgl_operator_amd.csrc.build

test: build
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests -q -m gpu

bench: build
	$(PY) bench.py --gpus 1 --steps 30 --warmup 10

clean:
	rm -rf dgl_operator_amd/_C.so dgl_operator_amd/csrc/.obj
