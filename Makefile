# agentainer-amd build/test targets (reference Makefile parity)
PY ?= python

.PHONY: build build-asan install test test-gpu test-gpu-serialized test-sanitize test-crash test-persistence test-network test-all bench verify

build:
	$(PY) -m agentainer_amd.ops.build

# host-AddressSanitized binding layer (SURVEY.md §5 sanitizer row);
# run via LD_PRELOAD=$$(hipcc -print-file-name=libclang_rt.asan-x86_64.so)
build-asan:
	AGENTAINER_ASAN=1 $(PY) -m agentainer_amd.ops.build --force

install:
	$(PY) -m pip install --no-build-isolation --no-deps -e .

test:
	$(PY) -m pytest tests/ -q -m "not gpu"

# CPU suite under Python dev mode (faulthandler on, warnings surfaced,
# dealloc checks) — the per-round sanitizer pass
test-sanitize:
	$(PY) -X dev -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -q -m gpu

# serialized-kernel GPU pass: AMD_SERIALIZE_KERNEL=3 makes every kernel
# launch synchronous so a faulting kernel is attributed exactly
test-gpu-serialized:
	AMD_SERIALIZE_KERNEL=3 $(PY) -m pytest tests/ -q -m gpu

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
