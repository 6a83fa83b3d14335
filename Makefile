# Build/test/bench entry points (reference Makefile:46-230 analogue).
# The gfx950 HIP extension builds in-tree so the .so ships with the repo
# snapshot to GPU machines.

PYTHON ?= python3
ARCH ?= gfx950

.PHONY: build build-debug test test-gpu test-native-debug bench bench-sim bench-router serve clean

build:
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace

# Debug extension build: host-side index/range validation in the bindings
# (LIG_DEBUG) — catches bad slot/unit/block-table values before a kernel
# dereferences them.  Separate --build-temp: setuptools only tracks source
# mtimes, so sharing objects with `build` would silently reuse the other
# configuration's .o files.  Run `make build` afterwards to restore the
# release .so in-tree (the snapshot ships whatever is in-tree).
build-debug:
	LIG_DEBUG=1 PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace --build-temp build/debug-temp

# Sanitizer pass over the native router core (production logic under
# ASan+UBSan; SURVEY.md §5 names this gap in the reference).
test-native-debug:
	g++ -std=c++17 -O1 -g -fsanitize=address,undefined -fno-omit-frame-pointer \
	  llm_instance_gateway_amd/router/csrc/router_core_test.cpp \
	  -o build/router_core_test
	./build/router_core_test

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
