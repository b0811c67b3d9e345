# gloo_amd build: host C++ with g++, HIP device code with hipcc
# (gfx950 only). Produces the in-tree python extension gloo_amd/_C*.so
# and (later) the benchmark binary.

PYTHON ?= python3
HIPCC ?= /opt/rocm/bin/hipcc
CXX ?= g++
ROCM ?= /opt/rocm

EXT_SUFFIX := $(shell $(PYTHON) -c "import sysconfig;print(sysconfig.get_config_var('EXT_SUFFIX'))")
PY_INC := $(shell $(PYTHON) -c "import sysconfig;print(sysconfig.get_paths()['include'])")
PYBIND_INC := $(shell $(PYTHON) -c "import pybind11;print(pybind11.get_include())")

TARGET := gloo_amd/_C$(EXT_SUFFIX)
BENCH := bin/gloo_amd_bench

CXXFLAGS := -O3 -g -std=c++17 -fPIC -Wall -Wextra -Wno-unused-parameter \
  -MMD -MP -pthread -Icsrc -I$(ROCM)/include -D__HIP_PLATFORM_AMD__=1

# make SANITIZE=thread (or address) for instrumented builds
ifdef SANITIZE
CXXFLAGS += -fsanitize=$(SANITIZE) -fno-omit-frame-pointer
LDFLAGS_EXTRA := -fsanitize=$(SANITIZE)
endif
HIPCCFLAGS := -O3 -std=c++17 -fPIC -MMD -MP --offload-arch=gfx950 -Icsrc

CC_SRCS := $(shell find csrc -name '*.cc' ! -path 'csrc/bindings/*' ! -path 'csrc/bench/*')
HIP_SRCS := $(shell find csrc -name '*.hip' 2>/dev/null)
BIND_SRCS := $(shell find csrc/bindings -name '*.cc')

CC_OBJS := $(patsubst csrc/%.cc,build/%.o,$(CC_SRCS))
HIP_OBJS := $(patsubst csrc/%.hip,build/%.hip.o,$(HIP_SRCS))
BIND_OBJS := $(patsubst csrc/%.cc,build/%.o,$(BIND_SRCS))

LDFLAGS := -L$(ROCM)/lib -lamdhip64 -lroctx64 -lssl -lcrypto -luv -pthread $(LDFLAGS_EXTRA)

all: $(TARGET) $(BENCH)

build/%.o: csrc/%.cc
	@mkdir -p $(dir $@)
	$(CXX) $(CXXFLAGS) -I$(PY_INC) -I$(PYBIND_INC) -c $< -o $@

build/%.hip.o: csrc/%.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(HIPCCFLAGS) -c $< -o $@

$(TARGET): $(CC_OBJS) $(HIP_OBJS) $(BIND_OBJS)
	$(CXX) -shared -o $@ $^ $(LDFLAGS)

clean:
	rm -rf build $(TARGET) $(BENCH)

-include $(shell find build -name "*.d" 2>/dev/null)

.PHONY: all clean

bench: $(BENCH)

BENCH_OBJS := build/bench/main.o

$(BENCH): $(CC_OBJS) $(HIP_OBJS) $(BENCH_OBJS)
	@mkdir -p bin
	$(CXX) -o $@ $^ $(LDFLAGS)

build/bench/%.o: csrc/bench/%.cc
	@mkdir -p $(dir $@)
	$(CXX) $(CXXFLAGS) -c $< -o $@

examples: $(CC_OBJS) $(HIP_OBJS)
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) examples/example_allreduce.cc $(CC_OBJS) $(HIP_OBJS) -o bin/example_allreduce $(LDFLAGS)
	$(CXX) $(CXXFLAGS) examples/example_v2.cc $(CC_OBJS) $(HIP_OBJS) -o bin/example_v2 $(LDFLAGS)

.PHONY: examples

# Race-detection stress harness (see csrc/bench/race_stress.cc).
RACE_OBJS := build/bench/race_stress.o
race_stress: bin/race_stress
bin/race_stress: $(RACE_OBJS) $(CC_OBJS) $(HIP_OBJS)
	@mkdir -p bin
	$(CXX) -o $@ $^ $(LDFLAGS)
.PHONY: race_stress
