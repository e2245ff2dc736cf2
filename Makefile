# nezha — MI355X-native build.
# Everything is compiled with hipcc as HIP C++ for gfx950; the built
# artifacts live in build/ (gitignored; ships to the GPU box via the gpurun
# snapshot since they are not gpurun-ignored).

HIPCC    ?= hipcc
ARCH     ?= gfx950
BUILD    ?= build

CXXFLAGS := -O3 -std=c++17 -fPIC -I include -I . --offload-arch=$(ARCH) \
            -Wall -Wno-unused-function -x hip -MMD -MP
LDFLAGS  := -L/opt/rocm/lib -lrccl -lz

HOST_SRCS := src/core/tensor.cc src/core/ops.cc src/cpu/cpu_ops.cc \
             src/util/util.cc src/util/prof.cc src/ckpt/npz.cc src/nn/modules.cc src/nn/optim.cc \
             src/models/models.cc src/train/trainer.cc src/train/ddp.cc \
             src/comm/rccl_comm.cc src/comm/fake_comm.cc src/comm/tcp_comm.cc \
             src/capi.cc
HIP_SRCS  := src/hip/elementwise.hip src/hip/gemm.hip src/hip/gemm_f32.hip src/hip/attention.hip

OBJS := $(HOST_SRCS:%.cc=$(BUILD)/%.o) $(HIP_SRCS:%.hip=$(BUILD)/%.o)

LIB := $(BUILD)/libnezha.so

CMDS := $(BUILD)/nezha-train $(BUILD)/nezha-bench $(BUILD)/nezha-info
TESTS_CPP := $(BUILD)/test_core $(BUILD)/test_ddp_fake $(BUILD)/test_modules $(BUILD)/test_ddp_fake_gpu

.PHONY: all lib cmds tests clean
all: lib cmds tests
lib: $(LIB)
cmds: $(CMDS)
tests: $(TESTS_CPP)

$(BUILD)/%.o: %.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(LIB): $(OBJS)
	$(HIPCC) -shared -fPIC $(OBJS) $(LDFLAGS) -o $@

$(BUILD)/nezha-%: cmd/%.cc $(LIB)
	$(HIPCC) $(CXXFLAGS) $< -o $@ -L$(BUILD) -lnezha $(LDFLAGS) -Wl,-rpath,'$$ORIGIN'

$(BUILD)/test_%: tests/cpp/test_%.cc $(LIB)
	$(HIPCC) $(CXXFLAGS) $< -o $@ -L$(BUILD) -lnezha $(LDFLAGS) -Wl,-rpath,'$$ORIGIN'

-include $(OBJS:.o=.d)

# Sanitizer tier: host-side ASan+UBSan build of the
# library + C++ tests in build-asan/ (device code is uninstrumented —
# gfx950 has no xnack+ ASan here; the -Woption-ignored warning is expected).
SAN_BUILD := build-asan
SAN_FLAGS := -fsanitize=address,undefined -fno-omit-frame-pointer -Wno-option-ignored
SAN_OBJS := $(HOST_SRCS:%.cc=$(SAN_BUILD)/%.o) $(HIP_SRCS:%.hip=$(SAN_BUILD)/%.o)

$(SAN_BUILD)/%.o: %.cc
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(SAN_FLAGS) -c $< -o $@

$(SAN_BUILD)/%.o: %.hip
	@mkdir -p $(dir $@)
	$(HIPCC) $(CXXFLAGS) $(SAN_FLAGS) -c $< -o $@

$(SAN_BUILD)/libnezha.so: $(SAN_OBJS)
	$(HIPCC) -shared -fPIC $(SAN_FLAGS) $(SAN_OBJS) $(LDFLAGS) -o $@

$(SAN_BUILD)/test_%: tests/cpp/test_%.cc $(SAN_BUILD)/libnezha.so
	$(HIPCC) $(CXXFLAGS) $(SAN_FLAGS) $< -o $@ -L$(SAN_BUILD) -lnezha $(LDFLAGS) -Wl,-rpath,'$$ORIGIN'

# perf yardstick (vendor lib, never linked into the product lib — C7)
$(BUILD)/yardstick: tools/yardstick.cc
	@mkdir -p $(BUILD)
	$(HIPCC) -O2 -std=c++17 -x hip --offload-arch=$(ARCH) $< -o $@ -L/opt/rocm/lib -lhipblaslt

.PHONY: yardstick
yardstick: $(BUILD)/yardstick

.PHONY: tests-asan
tests-asan: $(SAN_BUILD)/test_core $(SAN_BUILD)/test_ddp_fake
	$(SAN_BUILD)/test_core
	$(SAN_BUILD)/test_ddp_fake

clean:
	rm -rf $(BUILD) $(SAN_BUILD)

.PHONY: test help
test: all  ## run the CPU test tiers (C++ binaries + pytest)
	./build/test_core && ./build/test_modules && ./build/test_ddp_fake
	python3 -m pytest tests -q -m "not gpu"

help:  ## list targets
	@echo "targets: all (lib+cmds+tests), lib, cmds, tests, test (run CPU tiers),"
	@echo "         tests-asan (ASan/UBSan tier), yardstick (hipBLASLt compare), clean"
	@echo "GPU runs go through gpurun: see tools/gpurun/*.sh and docs/perf/README.md"
