# Developer entry points (the reference's Makefile equivalents).
PY ?= python3

.PHONY: build test test-gpu test-e2e bench crds installer docker

build:  ## compile the gfx950 HIP extension in-tree
	$(PY) -m arks_amd.ops.build

test:   ## CPU suite (engine, scheduler, server, gateway, control plane)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:  ## kernel numerics + engine e2e (needs an MI355X)
	$(PY) -m pytest tests -q -m gpu

test-e2e:  ## full-stack e2e: local sockets always; kind cluster when available
	$(PY) -m pytest tests/test_e2e_local.py -q
	@command -v kind >/dev/null 2>&1 && bash scripts/e2e_kind.sh || echo "kind not installed: skipped the real-cluster leg"

bench:  ## the driver-contract benchmark (one GPU)
	$(PY) bench.py --steps 16 --warmup 4

crds:   ## regenerate deploy/crds from the pydantic API types
	$(PY) scripts/gen_crds.py

installer:  ## bundle deploy/ into dist/ single-apply manifests
	$(PY) scripts/build_installer.py

docker:
	docker build -t arks-amd/runtime:latest .
