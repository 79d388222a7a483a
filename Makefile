# MI355X-native k8s device plugin + node labeller
IMAGE_DP ?= rocm/k8s-device-plugin-mi355x
IMAGE_NL ?= rocm/k8s-node-labeller-mi355x
TAG ?= $(shell git describe --always --dirty 2>/dev/null || echo dev)

.PHONY: all native test test-gpu bench images dp-image labeller-image ubi-dp-image ubi-labeller-image helm clean

all: native

native:
	PYTORCH_ROCM_ARCH=gfx950 python3 -m k8s_device_plugin_amd.native.build

test:
	python3 -m pytest tests/ -q -m "not gpu"

test-gpu:
	python3 -m pytest tests/ -q -m gpu

bench:
	python3 bench.py --gpus 1 --steps 500 --warmup 50

images: dp-image labeller-image ubi-dp-image ubi-labeller-image

dp-image:
	docker build -f deploy/Dockerfile -t $(IMAGE_DP):$(TAG) .

labeller-image:
	docker build -f deploy/labeller.Dockerfile -t $(IMAGE_NL):$(TAG) .

ubi-dp-image:
	docker build -f deploy/ubi-dp.Dockerfile -t $(IMAGE_DP):$(TAG)-ubi .

ubi-labeller-image:
	docker build -f deploy/ubi-labeller.Dockerfile -t $(IMAGE_NL):$(TAG)-ubi .

helm:
	helm lint deploy/helm/amd-gpu

clean:
	rm -f k8s_device_plugin_amd/native/*.so
	find . -name __pycache__ -type d -exec rm -rf {} +

# ASAN + TSAN over the native-server matrix; logs in profiles/sanitizers/
sanitize:
	bash scripts/run_sanitizers.sh all

# 10-minute live-daemon soak (native server; use SOAK_ARGS to customize)
soak:
	python3 scripts/soak.py --minutes 10 --pulse 2 --deep-every 3 $(SOAK_ARGS)
