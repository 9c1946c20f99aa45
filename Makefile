# Developer targets (parity role of the reference's Makefile).
PY ?= python

.PHONY: build test test-gpu bench smoke all-systems docker

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) -m stoix_amd.ops.build

test:
	$(PY) -m pytest tests -x -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -x -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 12 --warmup 3

bench-8:
	$(PY) -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
	    --master-addr 127.0.0.1 bench.py --gpus 8 --steps 12 --warmup 3

smoke:
	$(PY) __graft_entry__.py smoke

all-systems:
	bash bash_scripts/run-algorithms.sh

docker:
	docker build -t stoix_amd .
