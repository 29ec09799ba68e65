# ucc_amd build: everything compiled by hipcc for gfx950, objects shared
# between the standalone C library and the python extension. In-tree .so
# outputs (they travel to the GPU box with the repo snapshot).

HIPCC    ?= hipcc
ARCH     ?= gfx950
PYTHON   ?= python3
PY_EXT   := $(shell $(PYTHON) -c "import sysconfig;print(sysconfig.get_config_var('EXT_SUFFIX'))")
PY_INC   := $(shell $(PYTHON) -m pybind11 --includes)

CXXFLAGS := --offload-arch=$(ARCH) -O3 -std=c++17 -fPIC -Wall -Wextra \
            -Wno-unused-parameter -MMD -MP -DUCC_AMD_HAS_HIP -DUCC_AMD_HAS_TL_CDNA4 -DUCC_AMD_HAS_TL_RCCL
LDFLAGS  := -shared -fPIC -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

BUILD := build

# library sources (no python dependency)
LIB_SRCS := $(wildcard src/utils/*.cc) \
            $(wildcard src/core/*.cc) \
            $(wildcard src/schedule/*.cc) \
            $(wildcard src/coll_score/*.cc) \
            $(wildcard src/coll_patterns/*.cc) \
            $(wildcard src/cl/*.cc) \
            $(wildcard src/mc/*.cc) \
            $(wildcard src/ec/*.cc) \
            $(wildcard src/topo/*.cc) \
            $(wildcard src/tl/self/*.cc) \
            $(wildcard src/tl/shm/*.cc) \
            $(wildcard src/tl/tcp/*.cc) \
            $(wildcard src/tl/cdna4/*.cc) \
            $(wildcard src/tl/rccl/*.cc)
KERNEL_SRCS := $(wildcard src/ec/kernels/*.hip) $(wildcard src/tl/cdna4/kernels/*.hip)

LIB_OBJS := $(patsubst %.cc,$(BUILD)/%.o,$(LIB_SRCS)) \
            $(patsubst %.hip,$(BUILD)/%.o,$(KERNEL_SRCS))
BIND_OBJ := $(BUILD)/src/bind/module.o

MODULE := ucc_amd/_core$(PY_EXT)
SHLIB  := $(BUILD)/libucc_amd.so

PERFTEST := build/ucc_perftest
INFO     := build/ucc_info
NTESTS   := build/test_generic_dt build/test_obj_size build/test_mt

all: $(MODULE) $(SHLIB) $(PERFTEST) $(INFO) $(NTESTS)

$(BUILD)/tests/%.o: tests/native/%.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

build/test_generic_dt: $(BUILD)/tests/test_generic_dt.o $(LIB_OBJS)
	$(HIPCC) $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

build/test_obj_size: $(BUILD)/tests/test_obj_size.o $(LIB_OBJS)
	$(HIPCC) $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

build/test_mt: $(BUILD)/tests/test_mt.o $(LIB_OBJS)
	$(HIPCC) $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

$(BUILD)/tools/%.o: tools/%.cc tools/shm_oob.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(PERFTEST): $(BUILD)/tools/perftest.o $(LIB_OBJS)
	$(HIPCC) $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

$(INFO): $(BUILD)/tools/info.o $(LIB_OBJS)
	$(HIPCC) $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

$(BUILD)/%.o: %.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BIND_OBJ): src/bind/module.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(PY_INC) -c $< -o $@

$(MODULE): $(LIB_OBJS) $(BIND_OBJ)
	$(HIPCC) $(LDFLAGS) $^ -o $@ -lrt

$(SHLIB): $(LIB_OBJS)
	$(HIPCC) $(LDFLAGS) $^ -o $@ -lrt

clean:
	rm -rf $(BUILD) ucc_amd/_core*.so

check: all
	python3 -m pytest tests/ -x -q -m "not gpu"

.PHONY: all clean check

# auto-generated header dependencies (-MMD)
-include $(LIB_OBJS:.o=.d) $(BIND_OBJ:.o=.d)

# ---------------------------------------------------------------- ASAN
# Host AddressSanitizer build of the full library + native tests
# (device instrumentation unsupported on gfx950 without xnack; the
# option is auto-dropped for device code). `make asan` builds and runs.
ASAN_FLAGS := -O1 -g -fsanitize=address -fno-omit-frame-pointer
ASAN_OBJS  := $(patsubst %.cc,$(BUILD)/asan/%.o,$(LIB_SRCS)) \
              $(patsubst %.hip,$(BUILD)/asan/%.o,$(KERNEL_SRCS))

$(BUILD)/asan/%.o: %.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) -Wno-option-ignored -c $< -o $@

$(BUILD)/asan/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) -Wno-option-ignored -c $< -o $@

$(BUILD)/asan/tools/%.o: tools/%.cc tools/shm_oob.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) -Wno-option-ignored -c $< -o $@

build/asan_perftest: $(BUILD)/asan/tools/perftest.o $(ASAN_OBJS)
	$(HIPCC) -fsanitize=address $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

$(BUILD)/asan/tests/%.o: tests/native/%.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) -Wno-option-ignored -c $< -o $@

build/asan_generic_dt: $(BUILD)/asan/tests/test_generic_dt.o $(ASAN_OBJS)
	$(HIPCC) -fsanitize=address $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

build/asan_obj_size: $(BUILD)/asan/tests/test_obj_size.o $(ASAN_OBJS)
	$(HIPCC) -fsanitize=address $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

.PHONY: asan
asan: build/asan_generic_dt build/asan_obj_size build/asan_perftest
	ASAN_OPTIONS=detect_leaks=1 ./build/asan_obj_size
	ASAN_OPTIONS=detect_leaks=1 ./build/asan_generic_dt
	for cl in allreduce bcast alltoall alltoallv reduce_scatter barrier; do \
	  ASAN_OPTIONS=detect_leaks=1 ./build/asan_perftest -c $$cl -j 3 -b 8 -e 65536 -n 2 -w 1 || exit 1; done
	for cl in allreduce gather allgatherv; do \
	  ASAN_OPTIONS=detect_leaks=1 UCC_FAKE_NODE_SPLIT=2 ./build/asan_perftest -c $$cl -j 4 -b 8 -e 16384 -n 2 -w 1 || exit 1; done
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TL_TCP_SLIDING_MIN=65536 UCC_TL_TCP_SLIDING_WINDOW=16384 ./build/asan_perftest -c allreduce -j 3 -b 65536 -e 262144 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=bcast:@dbt:99 ./build/asan_perftest -c bcast -j 5 -b 8192 -e 131072 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=bcast:@knomial:99 UCC_TL_TCP_KN_RADIX=3 ./build/asan_perftest -c bcast -j 6 -b 8 -e 4096 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=alltoallv:@hybrid:99 UCC_TL_TCP_A2AV_HYBRID_THRESH=512 ./build/asan_perftest -c alltoallv -j 5 -b 64 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allgather:@bruck:99 ./build/asan_perftest -c allgather -j 5 -b 64 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allgather:@neighbor:99 UCC_TL_TCP_AG_NEIGHBOR_MIN=0 ./build/asan_perftest -c allgather -j 6 -b 64 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allgather:@sparbit:99 ./build/asan_perftest -c allgather -j 7 -b 64 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allgather:@knomial:99 UCC_TL_TCP_KN_RADIX=3 ./build/asan_perftest -c allgather -j 7 -b 64 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allreduce:@dbt:99 ./build/asan_perftest -c allreduce -j 6 -b 8192 -e 131072 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=reduce_scatterv:@ring:99 UCC_TL_TCP_RS_RING_MIN=0 ./build/asan_perftest -c allgatherv -j 5 -b 1024 -e 32768 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=allreduce:@knomial:99 UCC_TL_TCP_KN_RADIX=3 ./build/asan_perftest -c allreduce -j 7 -b 8 -e 16384 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=gather:@knomial:99,scatter:@knomial:99 UCC_TL_TCP_KN_RADIX=3 ./build/asan_perftest -c gather -j 6 -b 64 -e 8192 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=reduce_scatter:@ring:99 UCC_TL_TCP_RS_RING_MIN=256 ./build/asan_perftest -c reduce_scatter -j 5 -b 1024 -e 65536 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_TL_SHM_ENABLE=0 UCC_TUNE=reduce_scatter:@knomial:99 ./build/asan_perftest -c reduce_scatter -j 7 -b 256 -e 65536 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_FAKE_SOCKET_SPLIT=3 UCC_TL_SHM_CHUNK_SIZE=4096 ./build/asan_perftest -c bcast -j 6 -b 1024 -e 65536 -n 2 -w 1
	ASAN_OPTIONS=detect_leaks=1 UCC_FAKE_SOCKET_SPLIT=3 UCC_TL_SHM_CHUNK_SIZE=4096 ./build/asan_perftest -c allreduce -j 6 -b 1024 -e 65536 -n 2 -w 1

# ---------------------------------------------------------------- TSAN
# Host ThreadSanitizer build + the native MT test (lock-free progress
# queue, shm slot counters, concurrent multi-team driving).
TSAN_FLAGS := -O1 -g -fsanitize=thread -fno-omit-frame-pointer
TSAN_OBJS  := $(patsubst %.cc,$(BUILD)/tsan/%.o,$(LIB_SRCS)) \
              $(patsubst %.hip,$(BUILD)/tsan/%.o,$(KERNEL_SRCS))

$(BUILD)/tsan/%.o: %.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(TSAN_FLAGS) -Wno-option-ignored -c $< -o $@

$(BUILD)/tsan/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(TSAN_FLAGS) -Wno-option-ignored -c $< -o $@

$(BUILD)/tsan/tests/%.o: tests/native/%.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(TSAN_FLAGS) -Wno-option-ignored -c $< -o $@

build/tsan_mt: $(BUILD)/tsan/tests/test_mt.o $(TSAN_OBJS)
	$(HIPCC) -fsanitize=thread $^ -o $@ -lrt -L/opt/rocm/lib -lrccl -Wl,-rpath,/opt/rocm/lib

.PHONY: tsan
tsan: build/tsan_mt
	TSAN_OPTIONS="halt_on_error=1" ./build/tsan_mt
	TSAN_OPTIONS="halt_on_error=1" UCC_TL_SHM_ENABLE=0 ./build/tsan_mt
