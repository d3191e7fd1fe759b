# qrack_amd native build — hipcc, gfx950 only (MI355X / CDNA4).
# Host-only translation units still compile with hipcc for flag consistency;
# device code lives in csrc/hip/*.hip.

HIPCC      ?= hipcc
GPU_ARCH   ?= gfx950
PYTHON     ?= python3
EXT_SUFFIX := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")
PYBIND_INC := $(shell $(PYTHON) -m pybind11 --includes)

BUILD      := build
TARGET     := qrack_amd/_qrack$(EXT_SUFFIX)

CXXFLAGS   := -O3 -std=c++17 -fPIC -Wno-unused-result -DQRACK_AMD_ENABLE_HIP \
              -DQRACK_AMD_HIP_ENGINE --offload-arch=$(GPU_ARCH) -Icsrc $(PYBIND_INC)
LDFLAGS    := -shared

CPP_SRCS   := $(wildcard csrc/*.cpp) $(wildcard csrc/common/*.cpp) $(wildcard csrc/hip/*.cpp)
HIP_SRCS   := $(wildcard csrc/hip/*.hip)
OBJS       := $(patsubst csrc/%.cpp,$(BUILD)/%.o,$(CPP_SRCS)) \
              $(patsubst csrc/hip/%.hip,$(BUILD)/hip/%.o,$(HIP_SRCS))

all: $(TARGET)

$(BUILD)/%.o: csrc/%.cpp $(wildcard csrc/*.hpp) $(wildcard csrc/*.inc) $(wildcard csrc/common/*.hpp) $(wildcard csrc/hip/*.hpp)
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/hip/%.o: csrc/hip/%.hip $(wildcard csrc/*.hpp) $(wildcard csrc/common/*.hpp) $(wildcard csrc/hip/*.hpp)
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -x hip -c $< -o $@

$(TARGET): $(OBJS)
	$(HIPCC) $(LDFLAGS) $(OBJS) -o $@

clean:
	rm -rf $(BUILD) qrack_amd/_qrack*.so

.PHONY: all clean

# sanitizer lane: CPU-only battery under ASAN+UBSAN (no GPU needed)
asan:
	g++ -O1 -g -std=c++17 -fsanitize=address,undefined -fno-omit-frame-pointer \
	    -Icsrc csrc/qinterface.cpp csrc/qengine_cpu.cpp csrc/qengine_sparse.cpp \
	    csrc/qstabilizer.cpp csrc/qstabilizerhybrid.cpp csrc/qunit.cpp csrc/qbdt.cpp \
	    csrc/qpager.cpp csrc/qfactory.cpp csrc/qengine_turboquant.cpp csrc/common/parallel_for.cpp \
	    tools/asan_smoke.cpp -o build/asan_smoke -lpthread
	./build/asan_smoke

# full-suite ASAN lane: CPU-only extension instrumented with ASan+UBSan,
# whole pytest CPU battery runs against it (catches heap bugs in every
# layer incl. the C ABI / pinvoke compat surfaces)
asan-suite:
	@mkdir -p build/asan_ext/qrack_amd
	g++ -O1 -g -shared -fPIC -std=c++17 -fsanitize=address -fno-omit-frame-pointer \
	    -Icsrc $(PYBIND_INC) \
	    csrc/bindings.cpp csrc/capi.cpp csrc/qinterface.cpp csrc/qengine_cpu.cpp \
	    csrc/qengine_sparse.cpp csrc/qengine_turboquant.cpp csrc/qstabilizer.cpp \
	    csrc/qstabilizerhybrid.cpp csrc/qunit.cpp csrc/qbdt.cpp csrc/qpager.cpp \
	    csrc/qfactory.cpp csrc/common/parallel_for.cpp \
	    -o build/asan_ext/qrack_amd/_qrack$(EXT_SUFFIX) -lpthread
	cp qrack_amd/*.py build/asan_ext/qrack_amd/
	cd build/asan_ext && \
	  LD_PRELOAD="$$(g++ -print-file-name=libasan.so) $$(g++ -print-file-name=libstdc++.so.6)" \
	  PYTHONMALLOC=malloc ASAN_OPTIONS=detect_leaks=0 PYTHONPATH=.:$(CURDIR) \
	  timeout 2400 $(PYTHON) -m pytest $(CURDIR)/tests -x -q -m "not gpu" -p no:cacheprovider

# TSAN lane: exercises the thread-pool ParallelFor under the race detector
# (the reference has no sanitizer lanes; the thread-safety
# contract — one engine instance per thread — is enforced here instead)
tsan:
	g++ -O1 -g -std=c++17 -fsanitize=thread -fno-omit-frame-pointer \
	    -Icsrc csrc/qinterface.cpp csrc/qengine_cpu.cpp csrc/qengine_sparse.cpp \
	    csrc/qstabilizer.cpp csrc/qstabilizerhybrid.cpp csrc/qunit.cpp csrc/qbdt.cpp \
	    csrc/qpager.cpp csrc/qfactory.cpp csrc/qengine_turboquant.cpp csrc/common/parallel_for.cpp \
	    tools/asan_smoke.cpp -o build/tsan_smoke -lpthread
	./build/tsan_smoke
