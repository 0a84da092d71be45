# Build for the MI355X-native VictoriaLogs block-scan engine.
#
#   make            -> oracle/liboracle.so (CPU) + victorialogs_amd/libvlogsql.so (HIP, gfx950)
#   make oracle     -> CPU-only pieces (no hipcc needed)
#
# The HIP library cross-compiles for gfx950 without a GPU; .so files are
# built in-tree so they travel with the gpurun snapshot.

CXX      ?= g++
HIPCC    ?= hipcc
ARCH     ?= gfx950
CXXFLAGS ?= -O3 -std=c++17 -fPIC -Wall
HIPFLAGS ?= -O3 -std=c++17 -fPIC --offload-arch=$(ARCH)

CORE_SRCS := $(wildcard victorialogs_amd/csrc/core/*.cpp)
CORE_OBJS := $(patsubst victorialogs_amd/csrc/core/%.cpp,build/core/%.o,$(CORE_SRCS))

all: oracle hip rowops emu

oracle: oracle/liboracle.so

hip: victorialogs_amd/libvlogsql.so

# host build of the per-row device code (scan_rowops.h) for CPU-side
# differential fuzzing against the oracle (tests/test_rowops_fuzz.py)
rowops: tools/host_rowops/librowops.so

# CPU emulation build of the WHOLE product pipeline (real staging + real
# per-row device code, HIP stubbed): tests/test_emu_pipeline.py loads it via
# VQL_LIB to run the parity batteries and the 1386 reference fixtures end to
# end without a GPU.  TEST INFRASTRUCTURE — never shipped as the product.
emu: tools/host_emu/libvlogsql_emu.so

tools/host_emu/libvlogsql_emu.so: victorialogs_amd/csrc/vql_api.cpp \
		tools/host_emu/emu_kernels.cpp tools/host_emu/hip/hip_runtime.h \
		victorialogs_amd/csrc/hip/scan_rowops.h \
		victorialogs_amd/csrc/hip/scan_types.h $(CORE_OBJS)
	$(CXX) $(CXXFLAGS) -shared -Itools/host_emu -Ivictorialogs_amd/csrc \
		victorialogs_amd/csrc/vql_api.cpp tools/host_emu/emu_kernels.cpp \
		$(CORE_OBJS) -o $@ -ldl -lpthread

tools/host_rowops/librowops.so: tools/host_rowops/harness.cpp \
		victorialogs_amd/csrc/hip/scan_rowops.h \
		victorialogs_amd/csrc/hip/scan_types.h $(CORE_OBJS)
	$(CXX) $(CXXFLAGS) -shared -Ivictorialogs_amd/csrc/core \
		tools/host_rowops/harness.cpp $(CORE_OBJS) -o $@ -ldl -lpthread

build/core/%.o: victorialogs_amd/csrc/core/%.cpp $(wildcard victorialogs_amd/csrc/core/*.h) victorialogs_amd/csrc/core/unicode_ranges.inc
	@mkdir -p build/core
	$(CXX) $(CXXFLAGS) -c $< -o $@

build/oracle/%.o: oracle/%.cpp oracle/oracle_filter.h $(wildcard victorialogs_amd/csrc/core/*.h)
	@mkdir -p build/oracle
	$(CXX) $(CXXFLAGS) -c $< -o $@

oracle/liboracle.so: $(CORE_OBJS) build/oracle/oracle_filter.o build/oracle/oracle_api.o
	$(CXX) -shared $^ -o $@ -ldl -lpthread

build/hip/scan_kernels.o: victorialogs_amd/csrc/hip/scan_kernels.hip \
		victorialogs_amd/csrc/hip/scan_types.h \
		victorialogs_amd/csrc/hip/scan_rowops.h \
		victorialogs_amd/csrc/core/parse_float.h \
		victorialogs_amd/csrc/core/ryu.h \
		victorialogs_amd/csrc/core/xxhash64.h \
		victorialogs_amd/csrc/core/unicode_ranges.inc
	@mkdir -p build/hip
	$(HIPCC) $(HIPFLAGS) -x hip -c $< -o $@

build/hip/vql_api.o: victorialogs_amd/csrc/vql_api.cpp victorialogs_amd/csrc/hip/scan_types.h $(wildcard victorialogs_amd/csrc/core/*.h)
	@mkdir -p build/hip
	$(HIPCC) $(HIPFLAGS) -Ivictorialogs_amd/csrc -c $< -o $@

victorialogs_amd/libvlogsql.so: $(CORE_OBJS) build/hip/scan_kernels.o build/hip/vql_api.o
	$(HIPCC) $(HIPFLAGS) -shared $^ -o $@ -ldl -lpthread

clean:
	rm -rf build oracle/liboracle.so victorialogs_amd/libvlogsql.so

.PHONY: all oracle hip clean
