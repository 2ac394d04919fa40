# Build parsec_amd/_core.so (C++/HIP extension, gfx950-only).
# hipcc cross-compiles device code without a GPU; the .so is built in-tree
# so it travels to GPU boxes with the source snapshot.

HIPCC      ?= hipcc
ARCH       ?= gfx950
PYINC      := $(shell python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PBINC      := $(shell python3 -c "import pybind11; print(pybind11.get_include())")
ROCM       ?= /opt/rocm

CXXFLAGS   := -O3 -std=c++17 -fPIC --offload-arch=$(ARCH) \
              -I$(PYINC) -I$(PBINC) -Isrc -I$(ROCM)/include \
              -Wall -Wno-unused-function -fvisibility=hidden
LDFLAGS    := -shared -L$(ROCM)/lib -lrocblas -lrocsolver -lrccl

SRCS       := src/common.cpp src/runtime.cpp src/device_gpu.cpp src/comm.cpp \
              src/rccl_comm.cpp src/capi.cpp src/profiling.cpp src/dtd.cpp src/kernels_blas.cpp src/kernels_qr.cpp src/kernels_lu.cpp src/kernels_bf16.cpp src/kernels_panel.cpp src/kernels_reshape.cpp src/kernels_qr_bcgs.cpp \
              src/kernels_hip.cpp src/gpu_graph.cpp src/pins_modules.cpp src/pybind.cpp
OBJS       := $(SRCS:src/%.cpp=build/%.o)
COBJS      := $(filter-out build/pybind.o,$(OBJS))
TARGET     := parsec_amd/_core.so
CLIB       := parsec_amd/libparsec_amd.so

all: $(TARGET) $(CLIB)

build/%.o: src/%.cpp src/*.hpp | build
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

build:
	mkdir -p build

$(TARGET): $(OBJS)
	$(HIPCC) $(OBJS) $(LDFLAGS) -o $@

# Pure C/C++ runtime library (no Python bindings): the linkable surface
# for standalone C programs, like the reference's libparsec.
$(CLIB): $(COBJS)
	$(HIPCC) $(COBJS) $(LDFLAGS) -o $@

clean:
	rm -rf build $(TARGET) $(CLIB)

.PHONY: all clean

# Convenience targets
test: all
	python3 -m pytest tests -q -m "not gpu"

gpu-test: all
	python3 -m pytest tests -q -m gpu

# CI entry: build + CPU suite (what a runner without a GPU can check).
check: all
	python3 -c "import __graft_entry__ as g; g.build()"
	python3 -m pytest tests -q -m "not gpu"

bench: all
	python3 bench.py --steps 3 --warmup 1

.PHONY: test gpu-test bench
