# Build the xps core extension in-tree (the .so travels to the GPU box).
HIPCC ?= hipcc
ARCH ?= gfx950
PYEXT := $(shell python3-config --extension-suffix)
PYINC := $(shell python3 -m pybind11 --includes)
# Link against torch's bundled HIP runtime when torch is installed: torch
# ships its own libamdhip64.so (soname without .so.7). If we linked the
# system .so.7, a process importing both would run TWO HIP runtimes and
# pointers could not cross (memory faults). With -L<torch/lib> first,
# -lamdhip64 resolves to torch's copy and the dynamic loader shares one
# runtime; on a torch-less box the rpath falls back to /opt/rocm/lib.
TORCHLIB := $(shell python3 -c "import torch, os; print(os.path.join(os.path.dirname(torch.__file__), 'lib'))" 2>/dev/null)
CXXFLAGS := -O3 -std=c++17 -fPIC -Wall -Wno-unused-function --offload-arch=$(ARCH) $(PYINC)
LDFLAGS := -shared -fPIC
# `make ASAN=1` — address-sanitized build (ps-lite Makefile:54-56 parity)
ifeq ($(ASAN),1)
CXXFLAGS += -fsanitize=address -fno-omit-frame-pointer -g
LDFLAGS += -fsanitize=address
endif
# `make TSAN=1 cppbench` — thread-sanitized C++ bench (beyond the
# reference, which has no TSAN config): run build/bench_kv for a
# race-checked end-to-end cluster
ifeq ($(TSAN),1)
CXXFLAGS += -fsanitize=thread -fno-omit-frame-pointer -g
LDFLAGS += -fsanitize=thread
endif
ifneq ($(TORCHLIB),)
LDFLAGS += -L$(TORCHLIB) -Wl,-rpath,$(TORCHLIB)
endif

SRCS := csrc/env.cc csrc/wire.cc csrc/tcp.cc csrc/van.cc csrc/postoffice.cc \
        csrc/customer.cc csrc/resender.cc csrc/hip_util.cc csrc/hip_pool.cc csrc/host_pool.cc \
        csrc/gpu_plane.cc csrc/shm_ring.cc csrc/server_handlers.cc csrc/ps.cc csrc/pybind.cc
HIPSRCS := $(wildcard csrc/*.hip)
OBJS := $(SRCS:%.cc=build/%.o) $(HIPSRCS:%.hip=build/%.hip.o)

TARGET := ps_lite_amd/_core$(PYEXT)

all: $(TARGET)

build/%.o: %.cc csrc/*.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

build/%.hip.o: %.hip csrc/*.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(TARGET): $(OBJS)
	$(HIPCC) $(LDFLAGS) $(OBJS) -o $@

clean:
	rm -rf build $(TARGET)

# pure-C++ benchmark (no Python): build/bench_kv
CORE_OBJS := $(filter-out build/csrc/pybind.o,$(OBJS))
build/bench_kv.o: examples/cpp/bench_kv.cc csrc/*.h
	$(HIPCC) $(CXXFLAGS) -c examples/cpp/bench_kv.cc -o build/bench_kv.o
CPPBENCH_LD := -O3 --offload-arch=$(ARCH)
ifeq ($(ASAN),1)
CPPBENCH_LD += -fsanitize=address
endif
ifeq ($(TSAN),1)
CPPBENCH_LD += -fsanitize=thread
endif
cppbench: $(CORE_OBJS) build/bench_kv.o
	$(HIPCC) $(CPPBENCH_LD) build/bench_kv.o $(CORE_OBJS) \
	    -o build/bench_kv -lpthread

# self-contained lint gate (reference parity: tests/lint.py + make lint)
lint:
	python3 scripts/lint.py

# the local CI gate: build + lint + the CPU test suite (the GPU suite
# needs an MI355X: `python -m pytest tests -m gpu` there)
check: all lint
	python3 -m pytest tests -q -m "not gpu"

.PHONY: all clean cppbench lint check
