# agentainer-amd build/test targets (reference Makefile parity)
PY ?= python

.PHONY: build test test-gpu test-crash test-persistence test-network test-all bench verify

build:
	$(PY) -m agentainer_amd.ops.build

test:
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -q -m gpu

test-crash:
	scripts/tests/test-crash-replay.sh

test-persistence:
	scripts/tests/test-persistence.sh

test-network:
	scripts/tests/test-network-isolation.sh

test-all: test test-crash test-persistence test-network

bench:
	$(PY) bench.py

verify:
	@$(PY) -c "import torch, agentainer_amd; print('torch', torch.__version__, 'cuda', torch.cuda.is_available())"
	@which hipcc && hipcc --version | head -1
