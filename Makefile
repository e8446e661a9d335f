# FusionInfer-AMD build/test entry points (reference Makefile parity,
# SURVEY.md §2.1 #17, adapted: hipcc instead of go build, pytest instead
# of envtest).

PY ?= python

.PHONY: all build test test-gpu bench crd render-samples clean

all: build

build:            ## compile the gfx950 HIP extension in-tree
	$(PY) -m fusioninfer_amd.ops.build

test:             ## CPU test suite (runs anywhere)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:         ## GPU test suite (MI355X box)
	$(PY) -m pytest tests -q -m gpu

bench:            ## flagship serving benchmark (1 GPU)
	$(PY) bench.py --steps 64 --warmup 16

bench-kernels:    ## kernel microbenchmarks (GPU)
	$(PY) tools/bench_kernels.py all

lint:             ## offline linter (reference: golangci-lint, 16 linters)
	$(PY) tools/lint.py

IMG_REGISTRY ?= fusioninfer-amd
TAG ?= latest

docker-build:     ## build engine / EPP / controller images (reference
	## Makefile docker-build; needs a docker daemon)
	docker build -f docker/Dockerfile.engine -t $(IMG_REGISTRY)/engine:$(TAG) .
	docker build -f docker/Dockerfile.epp -t $(IMG_REGISTRY)/epp:$(TAG) .
	docker build -f docker/Dockerfile.controller -t $(IMG_REGISTRY)/controller:$(TAG) .

build-installer:  ## render dist/install.yaml from config/ (reference
	## build-installer; kubectl kustomize when present, python fallback)
	mkdir -p dist
	@if command -v kubectl >/dev/null 2>&1; then \
		kubectl kustomize config/default > dist/install.yaml; \
	else \
		$(PY) tools/render_installer.py > dist/install.yaml; \
	fi
	@echo wrote dist/install.yaml

deploy:           ## apply the deploy stack to the current cluster
	kubectl apply -f dist/install.yaml

run:              ## run the manager locally against an in-process store
	$(PY) -m fusioninfer_amd.controlplane run

crd:              ## print the InferenceService CRD
	$(PY) -m fusioninfer_amd.controlplane crd

render-samples:   ## reconcile + render every sample InferenceService
	for f in config/samples/*.yaml; do \
		echo "== $$f"; $(PY) -m fusioninfer_amd.controlplane render $$f; \
	done

serve:            ## OpenAI-compatible server on :8000
	$(PY) -m fusioninfer_amd.server --model Qwen3-8B

clean:
	rm -f fusioninfer_amd/ops/_C.so
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true
