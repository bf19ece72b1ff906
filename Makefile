# anovos_amd build/test entry points (reference Makefile parity).
.PHONY: build test test-gpu demo bench clean

build:
	python -c "from anovos_amd.ops.hip.build import build; build(verbose=True)"

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

demo:
	./run_anovos_demo.sh

bench:
	python bench.py --steps 5 --warmup 2

clean:
	rm -rf anovos_amd/ops/hip/build report_stats output intermediate_data stats mlruns
