# gpu-docker-api-amd — build / test / run
# (replaces the reference's dual nvidia/mock Go build tags, Makefile:25-48:
#  here the mock flavor is a runtime config, not a separate binary)

PY ?= python3
HIPCC ?= /opt/rocm/bin/hipcc
ARCH ?= gfx950

.PHONY: all native test test-gpu bench run run-mock clean openapi parity run-docker-sim run-etcd-sim

all: native

native:            ## build HIP/C++ components in-tree (gfx950)
	$(PY) -m gpu_docker_api_amd.ops.build

test:              ## CPU test suite (no GPU needed)
	$(PY) -m pytest tests/ -x -q -m "not gpu"

test-gpu:          ## GPU-gated tests (run on an MI355X box)
	$(PY) -m pytest tests/ -x -q -m gpu

bench:             ## flagship latency benchmark, 1 GPU
	$(PY) bench.py --gpus 1

run:               ## daemon against real dockerd + amdsmi inventory
	$(PY) -m gpu_docker_api_amd --runtime docker --inventory auto

run-proc:          ## daemon with the native process runtime (no dockerd)
	$(PY) -m gpu_docker_api_amd --runtime proc --inventory auto

run-mock:          ## daemon in full-mock mode (CPU-only dev box)
	$(PY) -m gpu_docker_api_amd --runtime mock --inventory mock

openapi:           ## regenerate api/openapi.json from the live app
	$(PY) scripts/export_openapi.py

parity:            ## prove route/body parity vs the reference spec
	$(PY) scripts/openapi_parity.py

run-docker-sim:    ## dev engine: dockerd-compatible API over a unix socket
	$(PY) -m gpu_docker_api_amd.testing.dockerd_sim --socket /tmp/gda-dockerd.sock --data /tmp/gda-engine

run-etcd-sim:      ## dev etcd: v3 JSON gateway on TCP
	$(PY) -m gpu_docker_api_amd.state.etcd_fake --port 2379

clean:
	rm -rf gpu_docker_api_amd/ops/*.so csrc/bin .state merges
