# Makefile — builds the native engine (host C++ + gfx950 HIP kernels), the
# Python extension (in-tree, so it travels to GPU boxes with the snapshot)
# and the sboxgates CLI binary.
#
#   make -j            # everything
#   make ext           # just the Python extension
#   make cli           # just the CLI
#
# hipcc cross-compiles gfx950 without a GPU present.

HIPCC      ?= hipcc
GPU_ARCH   ?= gfx950
PYTHON     ?= python3

SRC        := sboxgates_amd/csrc
BUILD      := build
INC        := -I$(SRC)/include
PY_INC     := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYBIND_INC := $(shell $(PYTHON) -c "import pybind11; print(pybind11.get_include())")
EXT_SUFFIX := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")

CXXFLAGS   := -O3 -mavx2 -std=c++20 -fPIC -Wall -Wextra -Wno-unused-parameter -MMD -MP $(INC)
HIPFLAGS   := --offload-arch=$(GPU_ARCH)
LDFLAGS    := -shared -fPIC

HOST_SRCS  := $(wildcard $(SRC)/host/*.cpp)
HOST_OBJS  := $(patsubst $(SRC)/host/%.cpp,$(BUILD)/host_%.o,$(HOST_SRCS))
HIP_OBJS   := $(BUILD)/kernels.o
EXT        := sboxgates_amd/_core$(EXT_SUFFIX)
CLI        := bin/sboxgates

all: ext cli

ext: $(EXT)
cli: $(CLI)

$(BUILD):
	mkdir -p $(BUILD) bin

$(BUILD)/host_%.o: $(SRC)/host/%.cpp | $(BUILD)
	$(HIPCC) -x c++ $(CXXFLAGS) -c $< -o $@

$(BUILD)/kernels.o: $(SRC)/hip/kernels.hip | $(BUILD)
	$(HIPCC) $(CXXFLAGS) $(HIPFLAGS) -c $< -o $@

$(BUILD)/bindings.o: $(SRC)/bindings.cpp | $(BUILD)
	$(HIPCC) -x c++ $(CXXFLAGS) -I$(PY_INC) -I$(PYBIND_INC) -c $< -o $@

$(BUILD)/cli_main.o: $(SRC)/cli/main.cpp | $(BUILD)
	$(HIPCC) -x c++ $(CXXFLAGS) -c $< -o $@

$(EXT): $(HOST_OBJS) $(HIP_OBJS) $(BUILD)/bindings.o
	$(HIPCC) $(LDFLAGS) $^ -o $@

$(CLI): $(HOST_OBJS) $(HIP_OBJS) $(BUILD)/cli_main.o | $(BUILD)
	$(HIPCC) $^ -o $@

clean:
	rm -rf $(BUILD) bin $(EXT)

-include $(BUILD)/*.d

.PHONY: all ext cli clean

# Sanitizer self-tests: host engine under ASan+UBSan / TSan (g++; the HIP
# runtime is stubbed out — CPU scan paths only).
SAN_SRCS := $(HOST_SRCS) $(SRC)/test/gpu_stub.cpp $(SRC)/test/selftest.cpp

asan: | $(BUILD)
	g++ -O1 -g -std=c++20 -fsanitize=address,undefined -fno-omit-frame-pointer \
	  $(INC) $(SAN_SRCS) -o $(BUILD)/selftest_asan -lpthread
	$(BUILD)/selftest_asan

tsan: | $(BUILD)
	g++ -O1 -g -std=c++20 -fsanitize=thread $(INC) $(SAN_SRCS) \
	  -o $(BUILD)/selftest_tsan -lpthread
	$(BUILD)/selftest_tsan

.PHONY: asan tsan
