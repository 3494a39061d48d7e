# Reference analog: the BlueFog Makefile (mpirun -np 4 pytest targets).
# No MPI here — the suite spawns its own multi-process worlds.

PY ?= python

.PHONY: build test test-gpu bench bench-kernels examples clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 20 --warmup 5

bench-kernels:
	$(PY) bench_kernels.py

examples:
	./bfrun -np 2 $(PY) examples/pytorch_average_consensus.py
	./bfrun -np 2 $(PY) examples/pytorch_optimization.py --method exact_diffusion --iters 60

clean:
	rm -rf build bluefog_amd/*.so bluefog_amd/__pycache__ tests/__pycache__
