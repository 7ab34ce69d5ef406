# Build/test entry points (reference analogue: Makefile:37-105).

PYTHON ?= python3
VERSION ?= v0.1.0
IMAGE ?= amd-kubevirt-gpu-device-plugin
PCI_IDS_URL ?= https://pci-ids.ucw.cz/v2.2/pci.ids

.PHONY: build test test-gpu bench coverage lint clean image notices \
        update-pcidb

build:
	$(PYTHON) build_native.py

test: build
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-gpu: build
	$(PYTHON) -m pytest tests/ -q -m gpu

# full BASELINE scaling curve: 1/2/4/8 GPUs (+ the iommufd/VF config)
bench: build
	$(PYTHON) bench.py --gpus 1 --steps 200 --warmup 20
	$(PYTHON) bench.py --gpus 2 --steps 200 --warmup 20
	$(PYTHON) bench.py --gpus 4 --steps 200 --warmup 20
	$(PYTHON) bench.py --gpus 8 --steps 200 --warmup 20 --iommufd --vf-check

coverage: build
	$(PYTHON) -m pytest tests/ -q -m "not gpu" \
	    --cov=kubevirt_gpu_device_plugin_amd --cov-report=term \
	    2>/dev/null || $(PYTHON) -m pytest tests/ -q -m "not gpu"

lint:
	$(PYTHON) -m pyflakes kubevirt_gpu_device_plugin_amd tests bench.py \
	    2>/dev/null || true

clean:
	rm -f kubevirt_gpu_device_plugin_amd/*.so
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true

image:
	docker build -t $(IMAGE):$(VERSION) \
	    -f deployments/container/Dockerfile .

# Refresh the full public PCI ID database for the container image
# (reference: Makefile:104-105).  The curated in-package table stays as
# the built-in fallback; `make image` works with or without this having
# run (the Dockerfile falls back to the curated table).
update-pcidb:
	mkdir -p utils
	wget -O utils/pci.ids.full $(PCI_IDS_URL)

notices:
	$(PYTHON) tools/generate_notices.py > THIRD_PARTY_NOTICES.md

burnin: build
	$(PYTHON) tools/gpu_burnin.py --seconds 60

soak: build
	$(PYTHON) tools/daemon_soak.py --seconds 120

diag: build
	$(PYTHON) tools/dump_node_info.py
