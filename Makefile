# ec-mi355x — build everything (CPU oracle, GPU core, plugin harness)
HIPCC    ?= hipcc
GPU_ARCH ?= gfx950
HIPFLAGS ?= --offload-arch=$(GPU_ARCH) -O3 -std=c++17 -fPIC
CXX      ?= g++
CXXFLAGS ?= -O2 -std=c++17 -fPIC -Wall

all: oracle core

oracle:
	$(MAKE) -C oracle

core: ceph_amd/libec_mi355x_core.so

ceph_amd/libec_mi355x_core.so: ceph_amd/csrc/ec_core.hip ceph_amd/csrc/gf.cpp ceph_amd/csrc/gf.h include/ec_mi355x.h
	$(HIPCC) $(HIPFLAGS) -shared ceph_amd/csrc/ec_core.hip ceph_amd/csrc/gf.cpp -o $@

clean:
	$(MAKE) -C oracle clean
	rm -f ceph_amd/libec_mi355x_core.so

.PHONY: all oracle core clean
