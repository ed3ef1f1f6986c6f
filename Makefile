# Developer entry points (reference Makefile parity: build/test/coverage).

PYTHON ?= python3
ROCM_PATH ?= /opt/rocm

.PHONY: build test test-gpu bench lint clean image

build:
	$(PYTHON) setup.py build_ext --inplace

test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests/ -q -m gpu

bench:
	$(PYTHON) bench.py --steps 10 --warmup 3

lint:  # the gate fails (no '-' swallow; matches CI)
	ruff check k8s_dra_driver_amd/ tests/ bench.py

asan:  # ASan flavor of the amdsmi binding
	$(PYTHON) -c "from k8s_dra_driver_amd import build_native; build_native.build_amdhal_asan(force=True)"
	LD_PRELOAD="$$(gcc -print-file-name=libasan.so) $$(gcc -print-file-name=libstdc++.so.6)" \
	ASAN_OPTIONS=detect_leaks=0 \
	$(PYTHON) -c "from k8s_dra_driver_amd import _amdhal_asan as h; print(h.lib_version())" 

coverage:  # requires pytest-cov (not in the offline image)
	$(PYTHON) -m pytest tests/ -q -m "not gpu" --cov=k8s_dra_driver_amd --cov-report=term

image:
	docker build -f deployments/container/Dockerfile -t k8s-dra-driver-amd:dev .

clean:
	rm -rf build dist *.egg-info k8s_dra_driver_amd/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +
