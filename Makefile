# amd-k8s-operator-libs build/test targets
# (the reference's Makefile drives go test/lint/envtest; these are the
# Python/HIP equivalents)

PY ?= python3
HIPCC ?= /opt/rocm/bin/hipcc

.PHONY: all build test test-par test-gpu test-real-apiserver bench demo lint coverage clean

all: build test

# Cross-compile the native gfx950 GPU validator in-tree (no GPU required).
build:
	$(PY) k8s_operator_libs_amd/native/build.py

# CPU test suite (the envtest-style suites run against the in-memory and
# HTTP mini apiservers).
test:
	$(PY) -m pytest tests/ -q -m "not gpu" --timeout 300

# CPU suite under pytest-xdist (4 workers): shakes out cross-test
# interference and shared-state races.
test-par:
	$(PY) -m pytest tests/ -q -m "not gpu" -n 4 --timeout 300

# GPU test suite: requires an MI355X with ROCm.
test-gpu:
	$(PY) -m pytest tests/ -q -m gpu --timeout 600

# Conformance surface against a REAL kube-apiserver + etcd (the reference's
# envtest strategy, upgrade_suit_test.go:86-93 / Makefile:76-78).  Binaries
# are discovered via $$KUBEBUILDER_ASSETS / $$TEST_ASSET_KUBE_APISERVER +
# $$TEST_ASSET_ETCD / /usr/local/kubebuilder/bin / $$PATH; alternatively
# point $$CONFORMANCE_URL (+$$CONFORMANCE_TOKEN) at a disposable cluster
# (e.g. kind).  Fails loudly when no real substrate is available so it can't
# silently pass on the mini-apiserver alone — see docs/testing.md.
test-real-apiserver:
	@$(PY) -c "from k8s_operator_libs_amd.testing.envtest import find_assets; \
	import os, sys; \
	ok = find_assets() is not None or bool(os.environ.get('CONFORMANCE_URL')); \
	sys.exit(0 if ok else (print('ERROR: no kube-apiserver/etcd binaries found and CONFORMANCE_URL unset.', \
	'Install envtest assets (setup-envtest use -p path) and export KUBEBUILDER_ASSETS,', \
	'or point CONFORMANCE_URL at a disposable cluster.', file=sys.stderr) or 1))"
	$(PY) -m pytest tests/test_conformance.py -q --timeout 600

# Flagship benchmark (BASELINE config #3).
bench:
	$(PY) bench.py --steps 10 --warmup 2

# Watch a full 8-node rolling upgrade against the mini-apiserver.
demo:
	$(PY) examples/amdgpu_upgrade_operator.py --demo

# Real linting (golangci-lint stand-in): bytecode sanity + the in-repo AST
# linter (tools/lint.py — unused imports, mutable defaults, bare excepts,
# placeholder-less f-strings, print-in-library, missing docstrings, ...;
# ruff/mypy are not installable in the offline image).
lint:
	$(PY) -m compileall -q k8s_operator_libs_amd tests examples tools bench.py __graft_entry__.py
	$(PY) tools/lint.py

coverage:
	$(PY) -m pytest tests/ -q -m "not gpu" --timeout 300 \
		--cov=k8s_operator_libs_amd --cov-report=term 2>/dev/null \
		|| $(PY) -m pytest tests/ -q -m "not gpu" --timeout 300

clean:
	rm -f k8s_operator_libs_amd/native/_gpu_validator.so
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true
