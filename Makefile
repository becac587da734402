# substratus-amd dev loop (reference Makefile:98-156 equivalents)

PY ?= python

.PHONY: build
build:  ## build the gfx950 HIP extension in-tree
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

.PHONY: test
test:  ## CPU test suite (unit + integration + system; no GPU needed)
	$(PY) -m pytest tests/ -q -m "not gpu"

.PHONY: test-unit
test-unit:
	$(PY) -m pytest tests/test_controllers.py tests/test_quant_cpu.py -q

.PHONY: test-integration
test-integration:  ## controllers against the in-memory API + gRPC SCI
	$(PY) -m pytest tests/test_controllers.py tests/test_sci.py tests/test_cli.py -q

.PHONY: test-system
test-system:  ## end-to-end Model+Server -> completion request (CPU)
	$(PY) -m pytest tests/test_system_local.py -q

.PHONY: test-gpu
test-gpu:  ## kernel parity + engine tests (run on MI355X)
	$(PY) -m pytest tests/ -q -m gpu

.PHONY: test-gpu-sanitize
test-gpu-sanitize:  ## GPU tests under serialized launches + uncached allocator (sanitizer tier)
	bash scripts/sanitize_gpu.sh

.PHONY: release
release:  ## dist/: sdist + wheel + container-tools tarball + checksums (goreleaser analog)
	bash scripts/release.sh

.PHONY: dev
dev: build  ## rebuild the extension and restart the local platform (skaffold-loop analog)
	-$(PY) -m substratus_amd.cli.main down
	$(PY) -m substratus_amd.cli.main up

.PHONY: bench
bench: build
	$(PY) bench.py

.PHONY: bench-train
bench-train: build
	$(PY) bench.py --mode train

.PHONY: crds
crds:  ## regenerate config/crd/bases
	$(PY) scripts/gen_crds.py

.PHONY: up
up:  ## start the local platform daemon
	$(PY) -m substratus_amd.kube.apiserver

install:  ## console scripts (sub, nbwatch) into the current environment
	pip install -e . --no-deps --no-build-isolation
