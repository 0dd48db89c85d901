# Build/test entry points (the reference ships a Makefile + Docker/conda CI,
# reference Makefile:1-14, .github/workflows/test.yml; here the "build" is the
# in-tree gfx950 HIP extension and the tests are pytest).

PY ?= python

.PHONY: build test test-gpu bench clean

build:
	$(PY) -c "from sparktorch_amd.ops.build import build_extension; print(build_extension(verbose=True))"

test:
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -x -q -m gpu

bench:
	$(PY) bench.py --steps 20 --warmup 8

clean:
	rm -rf sparktorch_amd/ops/build sparktorch_amd/ops/*.so .pytest_cache
