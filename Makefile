# Developer entrypoints (the reference drives these through a
# Taskfile; same targets, make syntax).  Python 3.10 + ROCm 7.x
# toolchain assumed on PATH.

PY ?= python3

.PHONY: help build test test-gpu generate sync bench serve smoke \
        precommit clean

help:            ## list targets
	@grep -E '^[a-z-]+:.*##' $(MAKEFILE_LIST) | \
	  awk -F':.*## ' '{printf "  %-12s %s\n", $$1, $$2}'

build:           ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:            ## CPU test suite (no GPU needed)
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu:        ## GPU test suite (MI355X)
	$(PY) -m pytest tests/ -x -q -m gpu

generate:        ## regenerate spec-derived files (constants, docs, openapi.yaml)
	$(PY) tools/generate.py

sync:            ## refresh community metadata tables (needs network)
	$(PY) tools/community_sync.py

bench:           ## driver-contract benchmark, 1 GPU, short run
	$(PY) bench.py --steps 30 --warmup 10

serve:           ## run the gateway (env-configured)
	$(PY) -m inference_gateway_amd

smoke:           ## build + one tiny GPU forward (driver smoke)
	$(PY) __graft_entry__.py smoke

precommit:       ## the checks CI runs (tests + codegen drift)
	bash scripts/pre-commit-check.sh

clean:           ## drop build artifacts (keeps the committed sources)
	rm -rf build/ inference_gateway_amd/ops/_kernels*.so \
	  inference_gateway_amd/ops/csrc/*_hip.hip
