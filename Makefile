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
