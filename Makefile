# ec-mi355x — build everything (CPU oracle, GPU core, plugin harness)
HIPCC    ?= hipcc
GPU_ARCH ?= gfx950
HIPFLAGS ?= --offload-arch=$(GPU_ARCH) -O3 -std=c++17 -fPIC
CXX      ?= g++
CXXFLAGS ?= -O2 -std=c++17 -fPIC -Wall
HARNESS   = ceph_amd/harness
ROCM     ?= /opt/rocm

all: oracle core harness

oracle:
	$(MAKE) -C oracle

core: ceph_amd/libec_mi355x_core.so

ceph_amd/libec_mi355x_core.so: ceph_amd/csrc/ec_core.hip ceph_amd/csrc/gf.cpp ceph_amd/csrc/gf.h include/ec_mi355x.h
	$(HIPCC) $(HIPFLAGS) -shared ceph_amd/csrc/ec_core.hip ceph_amd/csrc/gf.cpp -o $@

# ---- plugin harness (standalone mirror of the reference's registry layer;
#      plugins resolve base-class/registry symbols from the host binary at
#      dlopen, like real Ceph plugins do — hence -rdynamic on binaries) ----
HARNESS_HDRS = $(HARNESS)/ec_types.h $(HARNESS)/erasure_code.h $(HARNESS)/erasure_code_plugin.h

harness: $(HARNESS)/ec_benchmark $(HARNESS)/registry_selftest $(HARNESS)/ec_non_regression \
         $(HARNESS)/libec_mi355x.so $(HARNESS)/libec_oracle.so $(HARNESS)/libec_lrc.so $(HARNESS)/libec_shec.so $(HARNESS)/libec_clay.so \
         $(HARNESS)/libec_fix_missing_version.so $(HARNESS)/libec_fix_bad_version.so \
         $(HARNESS)/libec_fix_missing_init.so $(HARNESS)/libec_fix_fail_init.so \
         $(HARNESS)/libec_fix_no_register.so

$(HARNESS)/ec_benchmark: $(HARNESS)/ec_benchmark.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -rdynamic $(HARNESS)/ec_benchmark.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

$(HARNESS)/registry_selftest: $(HARNESS)/registry_selftest.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -rdynamic $(HARNESS)/registry_selftest.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

# non-regression corpus tool (mirror of ceph_erasure_code_non_regression)
$(HARNESS)/ec_non_regression: $(HARNESS)/ec_non_regression.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -rdynamic $(HARNESS)/ec_non_regression.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

# product plugin: links the HIP core; undefined harness symbols resolve from
# the loading binary (RTLD_NOW), mirroring real Ceph plugin linkage
# harness objects are linked in so the .so also loads standalone (python
# ctypes); when loaded by ec_benchmark, symbol interposition resolves the
# registry to the host binary's copy (one registry instance, like real Ceph)
$(HARNESS)/libec_mi355x.so: $(HARNESS)/plugin_mi355x.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc ceph_amd/libec_mi355x_core.so $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared $(HARNESS)/plugin_mi355x.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc \
	  -L ceph_amd -lec_mi355x_core '-Wl,-rpath,$$ORIGIN/..' -L $(ROCM)/lib -lamdhip64 -ldl -o $@

# test fixture plugin (CPU, oracle-backed)
$(HARNESS)/libec_oracle.so: $(HARNESS)/plugin_oracle.cc oracle/ec_ref.c oracle/ec_ref.h $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared $(HARNESS)/plugin_oracle.cc oracle/ec_ref.c $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

# SHEC plugin (shingled EC; byte work on the standard mi355x kernels)
$(HARNESS)/libec_shec.so: $(HARNESS)/plugin_shec.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc ceph_amd/libec_mi355x_core.so $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared $(HARNESS)/plugin_shec.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc \
	  -L ceph_amd -lec_mi355x_core '-Wl,-rpath,$$ORIGIN/..' -L $(ROCM)/lib -lamdhip64 -ldl -o $@

# Clay coupled-layer plugin (composes registry sub-codecs)
$(HARNESS)/libec_clay.so: $(HARNESS)/plugin_clay.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared $(HARNESS)/plugin_clay.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

# LRC layered-composition plugin (host-side dispatch over registry
# sub-plugins; default sub-plugin mi355x, 'oracle' for CPU tests)
$(HARNESS)/libec_lrc.so: $(HARNESS)/plugin_lrc.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared $(HARNESS)/plugin_lrc.cc $(HARNESS)/erasure_code.cc $(HARNESS)/erasure_code_plugin.cc -ldl -o $@

# registry failure-mode fixtures
$(HARNESS)/libec_fix_missing_version.so: $(HARNESS)/plugin_fixture.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared -DFIXTURE_MISSING_VERSION $< -o $@
$(HARNESS)/libec_fix_bad_version.so: $(HARNESS)/plugin_fixture.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared -DFIXTURE_BAD_VERSION $< -o $@
$(HARNESS)/libec_fix_missing_init.so: $(HARNESS)/plugin_fixture.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared -DFIXTURE_MISSING_INIT $< -o $@
$(HARNESS)/libec_fix_fail_init.so: $(HARNESS)/plugin_fixture.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared -DFIXTURE_FAIL_INIT $< -o $@
$(HARNESS)/libec_fix_no_register.so: $(HARNESS)/plugin_fixture.cc $(HARNESS_HDRS)
	$(CXX) $(CXXFLAGS) -shared -DFIXTURE_NO_REGISTER $< -o $@

clean:
	$(MAKE) -C oracle clean
	rm -f ceph_amd/libec_mi355x_core.so $(HARNESS)/*.so $(HARNESS)/ec_benchmark $(HARNESS)/registry_selftest $(HARNESS)/ec_non_regression

.PHONY: all oracle core harness clean
