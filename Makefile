# mlsl_amd build: C++17 core + CDNA4 (gfx950) HIP kernels + RCCL, built with
# hipcc into an in-tree shared library the Python ctypes binding loads.
# No MPI, no CUDA, no external deps beyond ROCm.

HIPCC      ?= hipcc
GPU_ARCH   ?= gfx950
BUILD      := build
LIB        := mlsl_amd/libmlsl_amd.so
SELFTEST   := $(BUILD)/schedule_selftest

CXXFLAGS   := -O3 -std=c++17 -fPIC -Wall -Wextra -Wno-unused-parameter \
              --offload-arch=$(GPU_ARCH) -I/opt/rocm/include -MMD -MP
LDFLAGS    := -shared -fPIC -L/opt/rocm/lib -lrccl -lamdhip64 -pthread

CSRC := \
    mlsl_amd/csrc/core/log.cpp \
    mlsl_amd/csrc/core/config.cpp \
    mlsl_amd/csrc/core/sysinfo.cpp \
    mlsl_amd/csrc/core/signals.cpp \
    mlsl_amd/csrc/comm/schedule.cpp \
    mlsl_amd/csrc/comm/quant.cpp \
    mlsl_amd/csrc/comm/device_pool.cpp \
    mlsl_amd/csrc/comm/bootstrap.cpp \
    mlsl_amd/csrc/comm/mesh.cpp \
    mlsl_amd/csrc/comm/group.cpp \
    mlsl_amd/csrc/comm/request.cpp \
    mlsl_amd/csrc/comm/engine.cpp \
    mlsl_amd/csrc/comm/context.cpp \
    mlsl_amd/csrc/comm/p2p_transport.cpp \
    mlsl_amd/csrc/comm/device_comm.cpp \
    mlsl_amd/csrc/dl/environment.cpp \
    mlsl_amd/csrc/dl/rma.cpp \
    mlsl_amd/csrc/dl/session.cpp \
    mlsl_amd/csrc/bind/c_api.cpp \
    mlsl_amd/csrc/bind/ops_api.cpp

HIPSRC := mlsl_amd/csrc/hip/kernels.hip

OBJS := $(CSRC:%.cpp=$(BUILD)/%.o) $(HIPSRC:%.hip=$(BUILD)/%.o)

.PHONY: all lib selftest apitest benchlat samples quantplugin clean test

all: lib selftest apitest e2e benchlat samples quantplugin

lib: $(LIB)

$(LIB): $(OBJS)
	$(HIPCC) $(OBJS) $(LDFLAGS) -o $@

$(BUILD)/%.o: %.cpp
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -x hip -c $< -o $@

selftest: $(SELFTEST)

$(SELFTEST): $(BUILD)/mlsl_amd/csrc/tests/schedule_selftest.o $(BUILD)/mlsl_amd/csrc/comm/schedule.o $(BUILD)/mlsl_amd/csrc/core/log.o
	@mkdir -p $(dir $@)
	$(HIPCC) $^ -L/opt/rocm/lib -lamdhip64 -pthread -o $@

APITEST := $(BUILD)/api_selftest

apitest: $(APITEST)

$(APITEST): $(BUILD)/mlsl_amd/csrc/tests/api_selftest.o $(LIB)
	@mkdir -p $(dir $@)
	$(HIPCC) $< -Lmlsl_amd -lmlsl_amd -Wl,-rpath,'$$ORIGIN/../mlsl_amd' -pthread -o $@

E2E := $(BUILD)/mlsl_e2e

e2e: $(E2E)

$(E2E): $(BUILD)/mlsl_amd/csrc/tests/mlsl_e2e.o $(LIB)
	@mkdir -p $(dir $@)
	$(HIPCC) $< -Lmlsl_amd -lmlsl_amd -Wl,-rpath,'$$ORIGIN/../mlsl_amd' -pthread -o $@

BENCHLAT := $(BUILD)/bench_latency

benchlat: $(BENCHLAT)

$(BENCHLAT): $(BUILD)/mlsl_amd/csrc/tests/bench_latency.o $(LIB)
	@mkdir -p $(dir $@)
	$(HIPCC) $< -Lmlsl_amd -lmlsl_amd -Wl,-rpath,'$$ORIGIN/../mlsl_amd' -pthread -o $@

SAMPLES := $(BUILD)/mlsl_sample $(BUILD)/cmlsl_sample

samples: $(SAMPLES)

quantplugin: $(BUILD)/libquant_plugin.so

$(BUILD)/libquant_plugin.so: samples/quant_plugin.c
	@mkdir -p $(dir $@)
	gcc -O2 -shared -fPIC -o $@ $< -lm

$(BUILD)/mlsl_sample: samples/mlsl_sample.cpp $(LIB)
	@mkdir -p $(dir $@)
	$(HIPCC) -O2 -std=c++17 -Imlsl_amd/csrc/include samples/mlsl_sample.cpp \
	    -Lmlsl_amd -lmlsl_amd -Wl,-rpath,'$$ORIGIN/../mlsl_amd' -o $@

$(BUILD)/cmlsl_sample: samples/cmlsl_sample.c $(LIB)
	@mkdir -p $(dir $@)
	gcc -O2 -Imlsl_amd/csrc/include samples/cmlsl_sample.c \
	    -Lmlsl_amd -lmlsl_amd -Wl,-rpath,'$$ORIGIN/../mlsl_amd' -o $@

# Host-logic sanitizer build: the schedule algebra is pure C++ (no HIP), so
# it compiles with g++ + ASan/UBSan — catches indexing/overflow bugs in the
# collective schedules (the reference had no sanitizer integration).
ASAN := $(BUILD)/schedule_selftest_asan

asan: $(ASAN)
	$(ASAN)

$(ASAN): mlsl_amd/csrc/tests/schedule_selftest.cpp mlsl_amd/csrc/comm/schedule.cpp mlsl_amd/csrc/core/log.cpp
	@mkdir -p $(dir $@)
	g++ -O1 -g -std=c++17 -fsanitize=address,undefined -fno-omit-frame-pointer \
	    $^ -pthread -o $@

# Full-stack ASan: the API selftest (planner+engine+mesh, TCP world
# matrix) host-compiled with g++ + ASan/UBSan; device kernels stubbed.
ASAN_API := $(BUILD)/api_selftest_asan
ASAN_SRC := $(CSRC) mlsl_amd/csrc/tests/api_selftest.cpp             mlsl_amd/csrc/tests/asan_kernel_stubs.cpp

asan-api: $(ASAN_API)

$(ASAN_API): $(ASAN_SRC)
	@mkdir -p $(dir $@)
	g++ -O1 -g -std=c++17 -D__HIP_PLATFORM_AMD__ -I/opt/rocm/include 	    -fsanitize=address,undefined -fno-omit-frame-pointer 	    $(ASAN_SRC) -L/opt/rocm/lib -lrccl -lamdhip64 -pthread -o $@

TSAN_API := $(BUILD)/api_selftest_tsan

tsan-api: $(TSAN_API)

$(TSAN_API): $(ASAN_SRC)
	@mkdir -p $(dir $@)
	g++ -O1 -g -std=c++17 -D__HIP_PLATFORM_AMD__ -I/opt/rocm/include \
	    -fsanitize=thread -fno-omit-frame-pointer \
	    $(ASAN_SRC) -L/opt/rocm/lib -lrccl -lamdhip64 -pthread -o $@

# Concurrent-submit race test (multi-producer MPSC ring) under TSan.
TSAN_SUBMIT := $(BUILD)/submit_race_tsan
TSAN_SUBMIT_SRC := $(CSRC) mlsl_amd/csrc/tests/submit_race_selftest.cpp \
    mlsl_amd/csrc/tests/asan_kernel_stubs.cpp

tsan-submit: $(TSAN_SUBMIT)

$(TSAN_SUBMIT): $(TSAN_SUBMIT_SRC)
	@mkdir -p $(dir $@)
	g++ -O1 -g -std=c++17 -D__HIP_PLATFORM_AMD__ -I/opt/rocm/include \
	    -fsanitize=thread -fno-omit-frame-pointer \
	    $(TSAN_SUBMIT_SRC) -L/opt/rocm/lib -lrccl -lamdhip64 -pthread -o $@

test: all samples
	$(SELFTEST)
	python -m pytest tests/ -x -q -m "not gpu"

clean:
	rm -rf $(BUILD) $(LIB)

-include $(OBJS:.o=.d)
-include $(BUILD)/mlsl_amd/csrc/tests/schedule_selftest.d
-include $(BUILD)/mlsl_amd/csrc/tests/api_selftest.d
-include $(BUILD)/mlsl_amd/csrc/tests/bench_latency.d
-include $(BUILD)/mlsl_amd/csrc/tests/mlsl_e2e.d
