# edgehip — MI355X-native FUSE object-storage data path.
# Host C++ builds with g++; HIP device code builds with hipcc for gfx950 ONLY
# (no CUDA shims, no multi-arch fat binaries). A GPU-less box builds everything
# (hipcc cross-compiles) and runs the host engine; engine=gpu fails loudly
# there by design.
#
#   make             -> bin/edgehip (CLI) + lib/libedgehip.so (ctypes API)
#   make test-bins   -> standalone GPU kernel-test binaries (run via gpurun)

ROCM     ?= /opt/rocm
HIPCC    ?= $(ROCM)/bin/hipcc
CXX      ?= g++
ARCH     ?= gfx950
CXXFLAGS ?= -O2 -g -std=c++17 -Wall -Wextra -fPIC -pthread -MMD -MP -I$(ROCM)/include -D__HIP_PLATFORM_AMD__
HIPFLAGS ?= -O3 -std=c++17 --offload-arch=$(ARCH) -fPIC -MMD -MP
LDFLAGS  ?= -L$(ROCM)/lib -lamdhip64 -lrccl -lroctx64 -Wl,-rpath,$(ROCM)/lib -pthread

HOST_SRCS := src/obs.cc src/bufpool.cc src/conf.cc src/manifest.cc src/blake3.cc src/lz4.cc \
             src/cdc.cc src/gf256.cc src/journal.cc src/store.cc src/obj.cc src/vfs.cc src/fuse.cc \
             src/engine/engine.cc src/dist/dist.cc
HIP_SRCS  := src/engine/gpu_engine.hip src/kernels/fingerprint.hip \
             src/kernels/lz4.hip src/kernels/rs_gf256.hip \
             src/kernels/checksum.hip src/kernels/memops.hip \
             src/kernels/dedup_index.hip src/kernels/cdc.hip
HOST_OBJS := $(HOST_SRCS:.cc=.o)
HIP_OBJS  := $(HIP_SRCS:.hip=.o)
OBJS      := $(HOST_OBJS) $(HIP_OBJS)

.DEFAULT_GOAL := all
.PHONY: all clean test-bins tsan asan

# Sanitizer flavors: host C++ only (kernels unchanged).
#   make asan  -> bin/edgehip-asan      make tsan -> bin/edgehip-tsan
tsan:
	$(MAKE) clean >/dev/null
	$(MAKE) CXXFLAGS="-O1 -g -std=c++17 -fPIC -pthread -MMD -MP -I$(ROCM)/include -D__HIP_PLATFORM_AMD__ -fsanitize=thread" \
	        LDFLAGS="$(LDFLAGS) -fsanitize=thread" bin/edgehip lib/libedgehip.so
	@mkdir -p tsan
	mv bin/edgehip tsan/edgehip-tsan
	mv lib/libedgehip.so tsan/libedgehip-tsan.so
	$(MAKE) clean >/dev/null

asan:
	$(MAKE) clean >/dev/null
	$(MAKE) CXXFLAGS="-O1 -g -std=c++17 -fPIC -pthread -MMD -MP -I$(ROCM)/include -D__HIP_PLATFORM_AMD__ -fsanitize=address" \
	        LDFLAGS="$(LDFLAGS) -fsanitize=address" bin/edgehip lib/libedgehip.so
	@mkdir -p tsan
	mv bin/edgehip tsan/edgehip-asan
	mv lib/libedgehip.so tsan/libedgehip-asan.so
	$(MAKE) clean >/dev/null

all: bin/edgehip lib/libedgehip.so bin/kernbench bin/engine_bench

bin/engine_bench: bench/engine_bench.o lib/libedgehip.so
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) -o $@ bench/engine_bench.o -Llib -ledgehip -Wl,-rpath,'$$ORIGIN/../lib' $(LDFLAGS)

bin/kernbench: bench/kernbench.o $(HIP_OBJS:src/engine/gpu_engine.o=) src/gf256.o src/obs.o src/bufpool.o
	@mkdir -p bin
	$(HIPCC) -o $@ bench/kernbench.o src/kernels/fingerprint.o src/kernels/lz4.o \
	    src/kernels/rs_gf256.o src/kernels/checksum.o src/kernels/memops.o \
	    src/kernels/dedup_index.o src/kernels/cdc.o src/gf256.o src/obs.o src/bufpool.o -L$(ROCM)/lib -lroctx64 -lpthread

bin/edgehip: $(OBJS) src/main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) -o $@ $^ $(LDFLAGS)

lib/libedgehip.so: $(OBJS) src/capi.o src/capi_gpu.o
	@mkdir -p lib
	$(CXX) $(CXXFLAGS) -shared -o $@ $^ $(LDFLAGS)

%.o: %.cc
	$(CXX) $(CXXFLAGS) -c -o $@ $<

%.o: %.hip
	$(HIPCC) $(HIPFLAGS) -c -o $@ $<

clean:
	rm -f $(OBJS) src/main.o src/capi.o src/capi_gpu.o bench/kernbench.o bench/kernbench.d bench/engine_bench.o bench/engine_bench.d
	rm -f $(OBJS:.o=.d) src/main.d src/capi.d src/capi_gpu.d
	rm -rf bin lib

-include $(OBJS:.o=.d) src/main.d src/capi.d src/capi_gpu.d
