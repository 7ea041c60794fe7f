# Developer entry points (the reference's Makefile equivalents).
PY ?= python3

.PHONY: build test test-gpu bench crds installer docker

build:  ## compile the gfx950 HIP extension in-tree
	$(PY) -m arks_amd.ops.build

test:   ## CPU suite (engine, scheduler, server, gateway, control plane)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:  ## kernel numerics + engine e2e (needs an MI355X)
	$(PY) -m pytest tests -q -m gpu

bench:  ## the driver-contract benchmark (one GPU)
	$(PY) bench.py --steps 16 --warmup 4

crds:   ## regenerate deploy/crds from the pydantic API types
	$(PY) scripts/gen_crds.py

installer:  ## bundle deploy/ into dist/ single-apply manifests
	$(PY) scripts/build_installer.py

docker:
	docker build -t arks-amd/runtime:latest .
