# Build/test/bench entry points (reference Makefile:46-230 analogue).
# The gfx950 HIP extension builds in-tree so the .so ships with the repo
# snapshot to GPU machines.

PYTHON ?= python3
ARCH ?= gfx950

.PHONY: build test test-gpu bench bench-sim bench-router serve clean

build:
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace

test:
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests/ -x -q -m gpu

bench:
	$(PYTHON) bench.py

bench-sim:
	$(PYTHON) -m llm_instance_gateway_amd.sim.main --routing-type all

bench-router:
	$(PYTHON) benchmarks/bench_router.py

serve:
	$(PYTHON) -m llm_instance_gateway_amd.server --config examples/gateway.yaml

clean:
	rm -rf build llm_instance_gateway_amd/_C*.so
