# Convenience targets (the reference shipped a Makefile for its data
# helper; here the extensions build through setup.py).

PY ?= python

.PHONY: build build-rot-v2 test test-gpu bench smoke clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

# A/B build: bank-model-solved conflict-free LDS layouts (see docs/DESIGN.md)
build-rot-v2:
	RELORA_AMD_ROT_V2=1 PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 10 --warmup 3

smoke:
	$(PY) -c "import __graft_entry__; __graft_entry__.build(); __graft_entry__.smoke()"

clean:
	rm -rf build relora_amd/ops/_relora_hip*.so relora_amd/data/_index_helpers*.so
