# Convenience targets — the commands the docs reference, in one place.

PY ?= python

.PHONY: build test test-gpu bench zoo infer clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --steps 30 --warmup 10

zoo:
	for m in mobilenet_v1 mobilenet_v2 mobilenet_v3_large \
	         mobilenet_v3_small mnasnet_a1; do \
	  $(PY) bench.py --model $$m --steps 30 --warmup 10; done

infer:
	$(PY) tools/bench_infer.py --graph

clean:
	rm -rf build yet_another_mobilenet_series_amd/ops/csrc/*_hip.hip \
	       yet_another_mobilenet_series_amd/ops/csrc/*_hip.cpp
