# Build / test / bench entry points (parity with the reference's Makefile,
# /root/reference/Makefile:1-15, minus the Docker wrapper — this image ships
# ROCm + PyTorch already).

PY ?= python

.PHONY: build test test-gpu bench clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) -c 'import __graft_entry__ as g; g.build()'

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 500 --warmup 100

clean:
	rm -rf sparkflow_amd/ops/_build build *.egg-info
