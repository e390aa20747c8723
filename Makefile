# cea_amd build — native shim + HIP kernels built IN-TREE so the .so files
# travel to GPU nodes with the source snapshot.
# Parity with the reference Makefile targets (test/format/vet/presubmit,
# /root/reference/Makefile:15-35) adapted to the Python+C++ toolchain.

ROCM ?= /opt/rocm
HIPCC ?= $(ROCM)/bin/hipcc
CXX ?= g++

SMI_SO := cea_amd/amdsmi/libceaamd_smi.so
GPU_SO := cea_amd/ops/libceaamd_gpu.so
RCCL_BENCH := cea_amd/bin/all_reduce_perf
FAKE_SMI_SO := tests/_build/libamd_smi.so

.PHONY: all smi gpu rcclbench fake-smi examples test gputest lint presubmit clean

all: smi gpu rcclbench

smi: $(SMI_SO)

# test-only fake libamd_smi, LD_PRELOADed so CPU tests drive the REAL shim
fake-smi: $(FAKE_SMI_SO)

$(FAKE_SMI_SO): csrc/fake_amdsmi.cpp
	mkdir -p tests/_build
	$(CXX) -O2 -std=c++17 -Wall -shared -fPIC -pthread \
	  -I$(ROCM)/include $< -o $@

$(SMI_SO): csrc/amdsmi_shim.cpp
	$(CXX) -O2 -std=c++17 -Wall -shared -fPIC -pthread \
	  -I$(ROCM)/include $< -o $@ \
	  -L$(ROCM)/lib -lamd_smi -Wl,-rpath,$(ROCM)/lib

gpu: $(GPU_SO)

$(GPU_SO): csrc/gpu_ops.hip
	$(HIPCC) --offload-arch=gfx950 -O3 -std=c++17 -Wall -shared -fPIC $< -o $@

test:
	python3 -m pytest tests/ -x -q -m "not gpu"

gputest:
	python3 -m pytest tests/ -x -q -m gpu

rcclbench: $(RCCL_BENCH)

$(RCCL_BENCH): csrc/all_reduce_perf.cpp
	mkdir -p cea_amd/bin
	$(HIPCC) --offload-arch=gfx950 -O3 -std=c++17 -Wall $< -o $@ \
	  -I$(ROCM)/include -L$(ROCM)/lib -lrccl -Wl,-rpath,$(ROCM)/lib

lint:
	python3 -m compileall -q cea_amd cmd tests bench.py __graft_entry__.py
	python3 build_tools/boilerplate.py

examples:
	$(MAKE) -C example/cu-fencing

presubmit: lint all examples test

clean:
	rm -f $(SMI_SO) $(GPU_SO) $(RCCL_BENCH)
	$(MAKE) -C example/cu-fencing clean

# container image targets — parity /root/reference/Makefile:46-93
# (the reference builds amd64+arm64 presubmit images; MI355X nodes are
# amd64-only, but CI cross-build parity is kept via buildx)
IMAGE ?= cea-amd/gpu-device-plugin
TAG ?= $(shell cat VERSION 2>/dev/null || echo dev)
ALL_ARCH ?= amd64 arm64

image:
	docker build -t $(IMAGE):$(TAG) .

push: image
	docker push $(IMAGE):$(TAG)

# cross-arch single-platform build (reference: container-% at Makefile:52-60)
image-%:
	docker buildx build --platform linux/$* -t $(IMAGE)-$*:$(TAG) --load .

# multi-arch manifest build+push (reference: container-multi-arch + push-multi-arch,
# Makefile:62-93).  The runtime image is only exercised on amd64 (MI355X);
# arm64 exists for CI build parity.
image-multi-arch:
	docker buildx build \
	  --platform $(shell echo "$(ALL_ARCH)" | sed 's/ /,linux\//; s/^/linux\//') \
	  -t $(IMAGE):$(TAG) .

push-multi-arch:
	docker buildx build --push \
	  --platform $(shell echo "$(ALL_ARCH)" | sed 's/ /,linux\//; s/^/linux\//') \
	  -t $(IMAGE):$(TAG) .
