VERSION ?= v0.1.0
IMAGE   ?= vgpu-amd

.PHONY: all native test test-gpu bench docker helm-package clean

all: native

# Native enforcement artifacts, cross-compiled for gfx950.
native:
	$(MAKE) -C k8s_device_plugin_amd/csrc

test:
	python -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -x -q -m gpu

bench:
	python bench.py --gpus 1

bench-density:
	python benchmarks/density_bench.py --pods 10 --prewarm --arbitrate

bench-sched:
	python benchmarks/sched_bench.py --pods 32 --gpus 8

docker:
	docker build -f docker/Dockerfile -t $(IMAGE):$(VERSION) .

helm-package:
	helm package charts/vgpu-amd

clean:
	$(MAKE) -C k8s_device_plugin_amd/csrc clean
