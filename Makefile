# Build/test entry points — counterpart of the reference Makefile
# (Makefile:42-116): `build` is gated on `test` the way the reference
# gates docker-build on its envtest run.
PYTHON ?= python3
ARCH ?= gfx950

.PHONY: all test test-gpu build bench clean install lint

all: build

# CPU test suite (the driver's per-round gate); no GPU required
test:
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

# GPU numerics/e2e suite — run on an MI355X box
test-gpu:
	$(PYTHON) -m pytest tests/ -x -q -m gpu

# in-tree gfx950 extension build, gated on the CPU tests
build: test
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace

# extension build without the test gate (CI images that ran tests already)
build-only:
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace

bench:
	$(PYTHON) bench.py --gpus 1

install: build
	$(PYTHON) -m pip install -e .

clean:
	rm -rf build/ torch_on_k8s_amd/ops/_C*.so torch_on_k8s_amd/ops/csrc/*_hip.*
