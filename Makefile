# Convenience targets (the reference shipped an SDK Makefile; the real build
# is build_ext.py -> hipcc --offload-arch=gfx950, .so in-tree).
PY ?= python

.PHONY: ext test test-gpu bench clean

ext:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) build_ext.py

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --steps 30 --warmup 5

clean:
	rm -f cuda_gmm_mpi_amd/ops/_gmm_hip.so
	rm -rf build __pycache__ .pytest_cache
