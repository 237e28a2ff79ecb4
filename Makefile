# Builds the hipstore native core: the oim_amd._hipstore Python module
# (in-tree, travels to the GPU box) and the hipstored daemon.
# hipcc cross-compiles gfx950 without a GPU present.

ROCM ?= /opt/rocm
HIPCC := $(ROCM)/bin/hipcc
ARCH ?= gfx950
PYTHON ?= python3

PY_INC := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYBIND_INC := $(shell $(PYTHON) -c "import pybind11; print(pybind11.get_include())")
EXT_SUFFIX := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")

CXXFLAGS := -O3 -std=c++17 -fPIC -Wall -Wextra -Wno-unused-parameter \
            -Inative/include
HIPFLAGS := --offload-arch=$(ARCH)
LDLIBS := -L$(ROCM)/lib -lrccl

B := native/build
HDRS := $(wildcard native/include/hipstore/*.h)

CORE_OBJS := $(B)/json.o $(B)/bdev.o $(B)/crc32c.o $(B)/rpc_server.o \
             $(B)/methods.o $(B)/nbd.o $(B)/ublk.o $(B)/composite.o $(B)/vhost.o $(B)/vhost_master.o \
             $(B)/nvmf_common.o $(B)/nvmf_target.o $(B)/nvmf_initiator.o \
             $(B)/rados_client.o $(B)/rados_cluster.o \
             $(B)/gpu.o

.PHONY: all clean
all: oim_amd/_hipstore$(EXT_SUFFIX) bin/hipstored

$(B):
	mkdir -p $(B)

$(B)/%.o: native/src/%.cpp $(HDRS) | $(B)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(B)/gpu.o: native/src/gpu.hip $(HDRS) | $(B)
	$(HIPCC) $(CXXFLAGS) $(HIPFLAGS) -c $< -o $@

$(B)/pybind.o: native/src/pybind.cpp $(HDRS) | $(B)
	$(HIPCC) $(CXXFLAGS) -fvisibility=hidden -I$(PY_INC) -I$(PYBIND_INC) -c $< -o $@

oim_amd/_hipstore$(EXT_SUFFIX): $(CORE_OBJS) $(B)/pybind.o
	$(HIPCC) $(HIPFLAGS) -shared $^ $(LDLIBS) -o $@

bin/hipstored: $(CORE_OBJS) $(B)/main.o
	mkdir -p bin
	$(HIPCC) $(HIPFLAGS) $^ $(LDLIBS) -o $@

# AddressSanitizer build of the daemon: run the protocol-heavy tests
# against it for leak/UB checking:
#   make asan && TEST_HIPSTORED_BINARY=bin/hipstored-asan \
#     python -m pytest tests/test_vhost.py tests/test_nvmf.py -m "not gpu"
ASAN_FLAGS := -O1 -g -fsanitize=address -fno-omit-frame-pointer
ASAN_OBJS := $(patsubst $(B)/%.o,$(B)/asan/%.o,$(CORE_OBJS)) $(B)/asan/main.o

$(B)/asan:
	mkdir -p $(B)/asan

$(B)/asan/%.o: native/src/%.cpp $(HDRS) | $(B)/asan
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) -c $< -o $@

$(B)/asan/gpu.o: native/src/gpu.hip $(HDRS) | $(B)/asan
	$(HIPCC) $(CXXFLAGS) $(ASAN_FLAGS) $(HIPFLAGS) -c $< -o $@

.PHONY: asan
asan: $(ASAN_OBJS)
	mkdir -p bin
	$(HIPCC) $(HIPFLAGS) -fsanitize=address $^ $(LDLIBS) -o bin/hipstored-asan

clean:
	rm -rf $(B) bin oim_amd/_hipstore*.so

.PHONY: test gpu-test bench
test: all
	$(PYTHON) -m pytest tests -q -m "not gpu"

gpu-test: all
	$(PYTHON) -m pytest tests -q -m gpu

bench: all
	$(PYTHON) bench.py --steps 20 --warmup 5
