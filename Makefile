# faabric-mi355x build: C++20 core + gfx950 HIP kernels + pybind11 module.
# Everything is compiled with hipcc (amdclang++) so HIP-using translation
# units and host-only ones share one toolchain; device code targets gfx950
# ONLY (MI355X/CDNA4) — no multi-arch fatbins, no CUDA paths.

HIPCC ?= /opt/rocm/bin/hipcc
GPU_ARCH ?= gfx950

PY := python3
PY_INC := $(shell $(PY) -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYBIND_INC := $(shell $(PY) -c "import pybind11; print(pybind11.get_include())")
EXT_SUFFIX := $(shell $(PY) -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")

BUILD := build
TARGET := faabric_amd/_core$(EXT_SUFFIX)

CXXFLAGS := -O2 -g -std=c++20 -fPIC -Wall -Wno-unused-function \
            -Icpp/include -I$(PY_INC) -I$(PYBIND_INC) \
            --offload-arch=$(GPU_ARCH) -fvisibility=hidden -MMD -MP
LDFLAGS := -shared -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread

CPP_SRCS := $(wildcard cpp/src/*.cpp) $(wildcard cpp/bindings/*.cpp)
HIP_SRCS := $(wildcard cpp/hip/*.hip)

CPP_OBJS := $(patsubst cpp/%.cpp,$(BUILD)/%.o,$(CPP_SRCS))
HIP_OBJS := $(patsubst cpp/%.hip,$(BUILD)/%.o,$(HIP_SRCS))
OBJS := $(CPP_OBJS) $(HIP_OBJS)

# Build the module AND the example binaries: the test suite runs the
# binaries, and a stale build/ after a core change fails confusingly
all: $(TARGET) examples

$(BUILD)/%.o: cpp/%.cpp
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/%.o: cpp/%.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -x hip -c $< -o $@

$(TARGET): $(OBJS)
	$(HIPCC) $(OBJS) $(LDFLAGS) -o $@

# C++ example binaries (reference examples/ + planner_server parity)
CORE_OBJS = $(filter-out $(BUILD)/bindings/%,$(OBJS))
EXAMPLE_BINS = $(BUILD)/check $(BUILD)/planner_server $(BUILD)/server \
               $(BUILD)/selftest $(BUILD)/is_app_migratable \
               $(BUILD)/disttest
examples: $(TARGET) $(EXAMPLE_BINS)

# Sanitizer sweeps (reference CI parity: Address/Thread sanitised suites,
# .github/workflows/tests.yml). Host-only compile: sanitizers don't apply
# to device code.
SAN_SRCS := $(CPP_SRCS_CORE) examples/selftest.cpp
CPP_SRCS_CORE := $(wildcard cpp/src/*.cpp)

asan-check:
	@mkdir -p $(BUILD)/asan
	$(HIPCC) -O1 -g -std=c++20 -fsanitize=address -fno-omit-frame-pointer \
	    -Icpp/include --offload-arch=$(GPU_ARCH) \
	    $(CPP_SRCS_CORE) cpp/hip/snapshot_kernels.hip examples/selftest.cpp \
	    -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread \
	    -o $(BUILD)/asan/selftest
	FAABRIC_PORT_OFFSET=6600 $(BUILD)/asan/selftest

tsan-check:
	@mkdir -p $(BUILD)/tsan
	$(HIPCC) -O1 -g -std=c++20 -fsanitize=thread \
	    -Icpp/include --offload-arch=$(GPU_ARCH) \
	    $(CPP_SRCS_CORE) cpp/hip/snapshot_kernels.hip examples/selftest.cpp \
	    -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread \
	    -o $(BUILD)/tsan/selftest
	FAABRIC_PORT_OFFSET=6700 TSAN_OPTIONS="report_bugs=1" \
	    $(BUILD)/tsan/selftest

# Multi-process dist scenarios (cross-process transport, leader
# collectives, live MPI migration) under each sanitizer — the reference
# CI sanitizes its dist suite, not just unit tests
asan-dist:
	@mkdir -p $(BUILD)/asan
	$(HIPCC) -O1 -g -std=c++20 -fsanitize=address -fno-omit-frame-pointer \
	    -Icpp/include --offload-arch=$(GPU_ARCH) \
	    $(CPP_SRCS_CORE) cpp/hip/snapshot_kernels.hip examples/disttest.cpp \
	    -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread \
	    -o $(BUILD)/asan/disttest
	DISTTEST_BASE_OFFSET=6200 DISTTEST_STOP_FILE=$(BUILD)/asan/stop \
	    $(BUILD)/asan/disttest

tsan-dist:
	@mkdir -p $(BUILD)/tsan
	$(HIPCC) -O1 -g -std=c++20 -fsanitize=thread \
	    -Icpp/include --offload-arch=$(GPU_ARCH) \
	    $(CPP_SRCS_CORE) cpp/hip/snapshot_kernels.hip examples/disttest.cpp \
	    -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread \
	    -o $(BUILD)/tsan/disttest
	DISTTEST_BASE_OFFSET=6400 DISTTEST_STOP_FILE=$(BUILD)/tsan/stop \
	    TSAN_OPTIONS="report_bugs=1" $(BUILD)/tsan/disttest

.PHONY: asan-check tsan-check asan-dist tsan-dist

$(BUILD)/%: $(BUILD)/examples/%.o $(CORE_OBJS)
	$(HIPCC) $< $(CORE_OBJS) \
	    -L/opt/rocm/lib -lrccl -lamdhip64 -lz -l:libzstd.so.1 -pthread -o $@

$(BUILD)/examples/%.o: examples/%.cpp
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

clean:
	rm -rf $(BUILD) $(TARGET)

-include $(OBJS:.o=.d)

.PHONY: all clean examples
