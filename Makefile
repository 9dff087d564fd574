# Reference-parity entry points (reference Makefile:1-21 is an env
# installer; here the targets drive the MI355X-native build/test/bench).

PY ?= python

.PHONY: build test test-gpu bench train clean

build:
	$(PY) -c "import __graft_entry__ as g; g.build()"

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench: build
	$(PY) bench.py --gpus 1 --steps 50 --warmup 10

train: build
	$(PY) run.py -n=DDP_warmup

clean:
	rm -rf ddp_tricks_amd/ops/build ddp_tricks_amd/ops/_hip_ops.so
