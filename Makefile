# Convenience targets (the driver uses __graft_entry__.py / bench.py directly)
PY ?= python

.PHONY: build test test-gpu bench sweep clean

build:
	$(PY) -m adapcc_amd.ops.build

test: build
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests -q -m gpu

bench: build
	$(PY) bench.py --steps 10 --warmup 3

sweep: build
	$(PY) -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
	    --master-addr 127.0.0.1 benchmarks/allreduce_sweep.py \
	    --transports native,pg

clean:
	rm -f adapcc_amd/_core.so adapcc_amd/ops/.build_stamp
	find . -name __pycache__ -type d | xargs rm -rf
