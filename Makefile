# cro-amd build/test entrypoints (reference Makefile parity where applicable)

PYTHON ?= python3
HIPCC ?= /opt/rocm/bin/hipcc

.PHONY: all build test test-gpu bench manifests lint docker-build bundle clean

all: build

# Compile the gfx950 HIP extensions in-tree (cross-compiles without a GPU).
build:
	$(PYTHON) -m cro_amd.hip.build --force

# CPU test suite (the driver's round gate). GPU tests are marked `gpu`.
test:
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests/ -x -q -m gpu

bench:
	$(PYTHON) bench.py --steps 20 --warmup 5

# Regenerate CRD manifests (controller-gen analog); CI asserts no drift.
manifests:
	$(PYTHON) -m cro_amd.api.v1alpha1.crd config/crd/bases

lint:
	$(PYTHON) -m compileall -q cro_amd tests bench.py __graft_entry__.py
	$(PYTHON) tools/lint_imports.py

docker-build:
	docker build -t cro-amd-operator:latest .

# OLM bundle (operator-sdk `make bundle` analog, offline): assembles the
# CSV base + CRDs + samples into bundle/ in registry+v1 layout.
bundle: manifests
	$(PYTHON) tools/make_bundle.py bundle

clean:
	rm -f cro_amd/hip/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +
