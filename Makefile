# rbg-mi355x developer entry points
PY ?= python

.PHONY: build test test-gpu bench serve stress clean

build:            ## compile the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:             ## CPU suite (tiny model, gloo)
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:         ## MI355X suite (kernel numerics + GPU serving e2e)
	$(PY) -m pytest tests/ -q -m gpu

bench:            ## driver-contract benchmark, 1 GPU
	$(PY) bench.py --steps 30 --warmup 10

serve:            ## run the node daemon
	$(PY) -m rbg_amd.cli.daemon --run-root /tmp/rbg --persist-dir /tmp/rbg-state

stress:           ## controller stress (create/update/delete @ QPS)
	$(PY) tools/stress.py --groups 10 --qps 5

clean:
	rm -rf build rbg_amd/ops/*.so
