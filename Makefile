# Developer entry points (reference analog: Makefile)
PY ?= python3

test:            ## CPU test suite (driver-equivalent)
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu:        ## GPU tier (run on an MI355X box)
	$(PY) -m pytest tests/ -x -q -m gpu

build:           ## compile the gfx950 HIP kernel library in-tree
	$(PY) mcp_context_forge_amd/ops/build.py --force

serve:           ## run the HTTP gateway
	$(PY) -m mcp_context_forge_amd serve

bench:           ## flagship 1-GPU benchmark
	$(PY) bench.py --steps 20 --warmup 5

bench-cpu:
	$(PY) bench.py --no-gpu --steps 3 --warmup 1 --requests-per-step 256

load:            ## hey-rig analog against a running gateway
	$(PY) loadtest/load_rpc.py --n 10000 --c 200

smoke:
	$(PY) __graft_entry__.py && $(PY) __graft_entry__.py smoke

.PHONY: test test-gpu build serve bench bench-cpu load smoke

mutation:        ## mutation-testing sweep over core modules (writes profiles/mutation_report.txt)
	python tools/mutation_check.py --stride 5
