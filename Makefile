# Convenience targets (reference-repo Makefile parity).
PY ?= python3

.PHONY: build test test-gpu race bench smoke clean

build:
	$(PY) -c "from llm_d_kv_cache_amd._build import build_all; build_all()"

test: build
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests -q -m gpu

# reference `make unit-test-race` parity: TSan over the native control plane
race:
	bash tools/tsan_check.sh

bench: build
	$(PY) bench.py --steps 12 --warmup 3

smoke: build
	$(PY) -c "import __graft_entry__ as g; g.smoke(); print('smoke ok')"

clean:
	rm -f llm_d_kv_cache_amd/_kvcore.so llm_d_kv_cache_amd/_kvoffload.so
