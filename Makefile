# Developer entry points (parity: reference Makefile targets).

PY ?= python3

.PHONY: build manifests test test-gpu test-integration bench images lint

build:            ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

manifests:        ## regenerate CRD YAML from the api types (controller-gen analog)
	$(PY) -m runbooks_amd.api.crd config/crd/bases

test:             ## CPU suite (unit + envtest-style integration, gloo multi-proc)
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:         ## GPU kernel numerics + engine tests (MI355X box)
	$(PY) -m pytest tests/ -q -m gpu

test-integration: ## controller integration tests only
	$(PY) -m pytest tests/test_controller_integration.py tests/test_client_cli.py -q

bench:            ## flagship single-GPU bench (driver contract)
	$(PY) bench.py --gpus 1

images:           ## build all workload container images
	docker build -f images/base.Dockerfile -t runbooks-amd/base:latest .
	for i in model-loader dataset-loader trainer server notebook; do \
	  docker build -t runbooks-amd/$$i:latest images/$$i; done

installation-manifests: manifests  ## render install bundles
	kubectl kustomize config/install-kind > install/kind/manifests.yaml || true
	kubectl kustomize config/install-gcp > install/gcp/manifests.yaml || true

# API reference docs (reference: crd-ref-docs target, Makefile:204-214)
docs: ## regenerate docs/api.md from the CRD dataclasses
	python scripts/gen_api_docs.py
