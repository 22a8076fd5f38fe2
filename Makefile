# Developer convenience targets.
PY ?= python

.PHONY: build test test-gpu bench-cpu lint clean

build:            ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:             ## CPU test suite (gloo multi-process included)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:         ## on the MI355X box
	$(PY) -m pytest tests -q -m gpu

bench-cpu:        ## plumbing check of the driver contract
	$(PY) bench.py --device cpu --steps 2 --warmup 1 --batch 4

clean:
	rm -rf build out ddlbench_amd/ops/_hip_ops*.so
