# migbm build: C++ host core (g++/OpenMP) + HIP gfx950 kernels (hipcc) -> lib_migbm.so
# The .so is placed in lightgbm_amd/lib/ where the Python package's libpath finds it.

CXX      ?= g++
HIPCC    ?= /opt/rocm/bin/hipcc
ROCM     ?= /opt/rocm

CXXFLAGS  = -O3 -std=c++17 -fPIC -fopenmp -Wall -Wextra -Wno-unused-parameter \
            -Icpp/include -MMD -MP
HIPFLAGS  = --offload-arch=gfx950 -O3 -std=c++17 -fPIC -Icpp/include -Wall \
            -Wno-unused-parameter -MMD -MP
LDFLAGS   = -shared -fopenmp -L$(ROCM)/lib -Wl,-rpath,$(ROCM)/lib

BUILD    := build
LIBDIR   := lightgbm_amd/lib
TARGET   := $(LIBDIR)/lib_migbm.so

HOST_SRCS := $(wildcard cpp/src/*.cpp)
HIP_SRCS  := $(wildcard cpp/src/hip/*.hip.cpp)

HOST_OBJS := $(patsubst cpp/src/%.cpp,$(BUILD)/%.o,$(HOST_SRCS))
HIP_OBJS  := $(patsubst cpp/src/hip/%.hip.cpp,$(BUILD)/hip/%.o,$(HIP_SRCS))

# link with hipcc when HIP objects exist (pulls in amdhip64 + device code)
ifeq ($(strip $(HIP_OBJS)),)
  LINKER = $(CXX)
  EXTRA_LIBS =
else
  LINKER = $(HIPCC)
  EXTRA_LIBS = -lamdhip64 -lrccl
endif

CLI := lightgbm_amd/bin/migbm

all: $(TARGET) $(CLI)

$(CLI): cpp/cli/main.cpp $(TARGET)
	mkdir -p lightgbm_amd/bin
	$(CXX) $(CXXFLAGS) cpp/cli/main.cpp -Llightgbm_amd/lib -l_migbm \
	  -Wl,-rpath,'$$ORIGIN/../lib' -o $(CLI)

$(BUILD):
	mkdir -p $(BUILD) $(BUILD)/hip $(LIBDIR)

$(BUILD)/%.o: cpp/src/%.cpp | $(BUILD)
	$(CXX) $(CXXFLAGS) -c $< -o $@

$(BUILD)/hip/%.o: cpp/src/hip/%.hip.cpp | $(BUILD)
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

$(TARGET): $(HOST_OBJS) $(HIP_OBJS) | $(BUILD)
	$(LINKER) $(LDFLAGS) $(HOST_OBJS) $(HIP_OBJS) $(EXTRA_LIBS) -o $@

# ---- ASAN lane: host-only (no HIP) CLI under AddressSanitizer.
# Usage: make asan && ./build_asan/migbm_asan config=examples/binary_classification/train.conf
ASAN_BUILD := build_asan
ASAN_FLAGS := -O1 -g -std=c++17 -fopenmp -fsanitize=address -fno-omit-frame-pointer \
              -Icpp/include -DMIGBM_NO_HIP
ASAN_CLI   := $(ASAN_BUILD)/migbm_asan

asan: $(ASAN_CLI)

$(ASAN_CLI): $(HOST_SRCS) cpp/cli/main.cpp
	mkdir -p $(ASAN_BUILD)
	$(CXX) $(ASAN_FLAGS) $(HOST_SRCS) cpp/cli/main.cpp -o $(ASAN_CLI)

# ---- UBSAN lane (host-only CLI under UndefinedBehaviorSanitizer)
UBSAN_FLAGS := -O1 -g -std=c++17 -fopenmp -fsanitize=undefined -fno-omit-frame-pointer \
               -Icpp/include -DMIGBM_NO_HIP
ubsan: $(HOST_SRCS) cpp/cli/main.cpp
	mkdir -p build_ubsan
	$(CXX) $(UBSAN_FLAGS) $(HOST_SRCS) cpp/cli/main.cpp -o build_ubsan/migbm_ubsan

# ---- TSAN lane (host-only CLI under ThreadSanitizer; OpenMP races in the
# histogram/partition paths surface here). Built with amdclang++/libomp: its
# runtime carries TSAN annotations, gcc's libgomp does not (false positives).
TSAN_CXX   := /opt/rocm/lib/llvm/bin/clang++
TSAN_FLAGS := -O1 -g -std=c++17 -fopenmp -fsanitize=thread -fno-omit-frame-pointer \
              -Icpp/include -DMIGBM_NO_HIP
tsan: $(HOST_SRCS) cpp/cli/main.cpp
	mkdir -p build_tsan
	$(TSAN_CXX) $(TSAN_FLAGS) $(HOST_SRCS) cpp/cli/main.cpp -o build_tsan/migbm_tsan

clean:
	rm -rf $(BUILD) $(TARGET) $(CLI) $(ASAN_BUILD) build_ubsan build_tsan

-include $(BUILD)/*.d $(BUILD)/hip/*.d

.PHONY: all clean asan ubsan tsan
