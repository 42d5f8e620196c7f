# grapehip build — hipcc cross-compiles for gfx950 (no GPU needed to build).
HIPCC ?= hipcc
GPU_ARCH ?= gfx950
PYINC := $(shell python3 -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PBINC := $(shell python3 -c "import pybind11; print(pybind11.get_include())")

CXXFLAGS := -O3 -std=c++17 -fPIC -Wall -Wno-unused-function \
            -I$(PYINC) -I$(PBINC) -Icpp
HIPFLAGS := --offload-arch=$(GPU_ARCH)
LDFLAGS := -shared -L/opt/rocm/lib -lrccl

WITH_HIP ?= 1
ifeq ($(WITH_HIP),1)
CXXFLAGS += -DGRAPEHIP_WITH_HIP
HIP_OBJS := build/gpu_engine.o
else
HIP_OBJS :=
endif

# Address-sanitized host build for the CPU engine (reference WITH_ASAN,
# CMakeLists:15,87-90): make clean && make WITH_ASAN=1 WITH_HIP=0
WITH_ASAN ?= 0
ifeq ($(WITH_ASAN),1)
CXXFLAGS += -fsanitize=address -fno-omit-frame-pointer -g
LDFLAGS += -fsanitize=address
endif

TARGET := grapehip/_core.so

all: $(TARGET)

build:
	mkdir -p build

build/net.o: cpp/core/net.cpp cpp/core/net.hpp | build
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

build/bindings.o: cpp/bindings.cpp cpp/core/*.hpp cpp/apps/*.hpp $(wildcard cpp/hip/*.hpp) | build
	$(HIPCC) $(CXXFLAGS) $(HIPFLAGS) -c $< -o $@

build/gpu_engine.o: cpp/hip/gpu_engine.hip cpp/hip/*.hpp cpp/core/*.hpp | build
	$(HIPCC) $(CXXFLAGS) $(HIPFLAGS) -c $< -o $@

$(TARGET): build/net.o build/bindings.o $(HIP_OBJS)
	$(HIPCC) $(HIPFLAGS) $^ $(LDFLAGS) -o $@

clean:
	rm -rf build $(TARGET)

.PHONY: all clean
