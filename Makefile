# Ergonomics parity with the reference Makefile (build/test/bench targets).

PYTHON ?= python
ARCH ?= gfx950

.PHONY: build test test-gpu bench bench-micro soak churn lint clean

build:
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) setup.py build_ext --inplace

test:
	$(PYTHON) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests -q -m gpu

bench:
	$(PYTHON) bench.py --gpus 1 --steps 10 --warmup 3

bench-micro:
	$(PYTHON) scripts/bench_micro.py

soak:
	$(PYTHON) scripts/soak.py --seconds 180

churn:
	$(PYTHON) scripts/churn_check.py

lint:
	$(PYTHON) -m compileall -q llmd_kvcache_amd examples scripts services bench.py __graft_entry__.py

clean:
	rm -rf build llmd_kvcache_amd/ops/_kvidx_C*.so llmd_kvcache_amd/ops/csrc/*_hip.hip

IMAGE ?= llmd-kvcache-amd:latest
BASE_IMAGE ?= rocm/pytorch:latest

.PHONY: image container-smoke
image:
	docker build -t $(IMAGE) --build-arg BASE=$(BASE_IMAGE) .

container-smoke:
	BASE=$(BASE_IMAGE) IMG=$(IMAGE) bash scripts/container_smoke.sh

.PHONY: asan
asan:
	g++ -std=c++17 -O1 -g -fsanitize=address,undefined \
	    -fno-sanitize-recover=all -o build/asan_check scripts/asan_check.cc \
	    && ./build/asan_check
