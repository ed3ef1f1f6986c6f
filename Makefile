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

lint:
	-ruff check k8s_dra_driver_amd/ tests/ bench.py

coverage:  # requires pytest-cov (not in the offline image)
	$(PYTHON) -m pytest tests/ -q -m "not gpu" --cov=k8s_dra_driver_amd --cov-report=term

image:
	docker build -f deployments/container/Dockerfile -t k8s-dra-driver-amd:dev .

clean:
	rm -rf build dist *.egg-info k8s_dra_driver_amd/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +
