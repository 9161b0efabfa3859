# InstaSlice-AMD build/test driver (reference analog: Makefile).
PYTHON ?= python3
IMG_PREFIX ?= instaslice-amd
ROCM_ARCH ?= gfx950

.PHONY: all native test test-gpu bench scenarios crd lint docker-build \
        install deploy undeploy clean

all: native

native:  ## compile partitiond + payload + instaslice-stored, in-tree
	PYTORCH_ROCM_ARCH=$(ROCM_ARCH) $(PYTHON) build_native.py

test:    ## CPU test suite (no GPU needed)
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-gpu:  ## GPU tier (run on a MI355X box)
	$(PYTHON) -m pytest tests/ -q -m gpu

bench:   ## flagship benchmark, single GPU
	$(PYTHON) bench.py

scenarios:  ## BASELINE.json measurement configs 1-5 (fake-SMI)
	$(PYTHON) -m benchmarks.scenarios

lint:    ## static checks (ruff if present, else a strict compile pass)
	@if command -v ruff >/dev/null 2>&1; then 	    ruff check instaslice_amd tests benchmarks bench.py; 	else 	    $(PYTHON) -m compileall -q instaslice_amd tests benchmarks bench.py 	        __graft_entry__.py build_native.py && echo "compileall OK (ruff unavailable)"; 	fi

crd:     ## regenerate the CRD manifest from api/crd.py
	$(PYTHON) -m instaslice_amd.api.crd > \
	    config/crd/bases/inference.codeflare.dev_instaslices.yaml

docker-build:  ## controller + daemonset + payload images
	docker build -f Dockerfile.controller -t $(IMG_PREFIX)-controller:latest .
	docker build -f Dockerfile.daemonset  -t $(IMG_PREFIX)-daemonset:latest .
	docker build -f Dockerfile.payload    -t $(IMG_PREFIX)-payload:latest .

install:  ## CRDs into the cluster
	kubectl apply -f config/crd/bases/inference.codeflare.dev_instaslices.yaml

deploy: install  ## full stack
	kubectl apply -k config/default

undeploy:
	kubectl delete -k config/default --ignore-not-found=true

clean:
	rm -f instaslice_amd/smi/_partitiond*.so instaslice_amd/ops/_payload*.so
	rm -f instaslice_amd/bin/partitiond instaslice_amd/bin/instaslice-payload
	find . -name __pycache__ -type d -exec rm -rf {} + 2>/dev/null || true
