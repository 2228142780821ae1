# WVA-AMD build/test targets (reference Makefile analog)
PYTHON ?= python3

.PHONY: test test-gpu build bench lint crd clean

test:              ## CPU test suite (no GPU required)
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-gpu:          ## GPU numerics suite (MI355X)
	$(PYTHON) -m pytest tests/ -q -m gpu

build:             ## build the gfx950 HIP extension in-tree
	$(PYTHON) -m wva_amd.ops.build

bench:             ## headline benchmark (1 GPU; CPU fallback profile without one)
	$(PYTHON) bench.py --gpus 1 --steps 24 --warmup 6

crd:               ## regenerate the CRD manifest
	$(PYTHON) -m wva_amd.api.crd > deploy/crd/llmd.ai_variantautoscalings.yaml

clean:
	rm -rf wva_amd/ops/build wva_amd/ops/csrc/*_hip.hip .pytest_cache

chart-render: ## render the Helm chart (all features on) with the in-repo renderer
	python scripts/render_chart.py --set hpa.enabled=true --set vllmService.enabled=true --set inferno.enabled=true
