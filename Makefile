# Convenience targets (reference parity: Makefile build/test targets,
# SURVEY.md §2 row 19). Everything is plain python underneath.
PY ?= python

.PHONY: build test test-par test-gpu bench manager clean

build:          ## gfx950 HIP kernels + C++ native ext, in-tree
	$(PY) setup.py build_ext --inplace

test:           ## CPU suite (139 tests, incl. gloo multi-process)
	$(PY) -m pytest tests -q -m "not gpu"

test-par:       ## CPU suite, 4-way parallel
	$(PY) -m pytest tests -q -m "not gpu" -n 4

test-gpu:       ## on an MI355X box
	$(PY) -m pytest tests -q -m gpu

bench:          ## driver benchmark contract (CPU smoke off-GPU)
	$(PY) bench.py

manager:        ## run the control plane
	$(PY) -m datatunerx_amd.api.manager

clean:
	rm -rf build datatunerx_amd/**/__pycache__ .pytest_cache
