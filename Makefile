# Build / test / bench entry points (parity with the reference's Makefile,
# /root/reference/Makefile:1-15, minus the Docker wrapper — this image ships
# ROCm + PyTorch already).

PY ?= python

.PHONY: build test test-gpu bench ci cluster-demo clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) -c 'import __graft_entry__ as g; g.build()'

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 500 --warmup 100

# CI entry (parity: reference .travis.yml:16-18 runs the Spark test
# suite in docker): full CPU test run INCLUDING the executed Spark-ML
# suite (real pyspark if installed, bundled localspark otherwise).
ci: build test

# one-command multi-process cluster demo (parity: reference
# docker-compose.yml:1-22 Spark master + worker cluster) — N spawned
# executor processes run fit/transform/pipeline-save-load end to end,
# with fit driving the torch.distributed PS star.
cluster-demo:
	$(PY) examples/cluster_demo.py --executors 2

clean:
	rm -rf sparkflow_amd/ops/_build build *.egg-info
