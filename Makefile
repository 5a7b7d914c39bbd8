# Makefile — builds the MI355X-native core: the _hpk Python extension and the
# standalone miniapp binaries. Everything cross-compiles for gfx950 with hipcc
# (no GPU needed to build).
#
#   make -j        # extension + binaries
#   make ext       # just the Python extension
#   make bins      # just the standalone binaries

HIPCC      ?= /opt/rocm/bin/hipcc
GPU_ARCH   ?= gfx950
PYTHON     ?= python3

NATIVE     := hpc_patterns_amd/native
BUILD      := build
BIN        := bin

PY_INC     := $(shell $(PYTHON) -c "import sysconfig; print(sysconfig.get_paths()['include'])")
PYBIND_INC := $(shell $(PYTHON) -c "import pybind11; print(pybind11.get_include())")
EXT_SUFFIX := .so

CXXFLAGS   := -O3 -std=c++17 -fPIC --offload-arch=$(GPU_ARCH) -I$(NATIVE) -Wall
LDEXTRA    ?=
LDFLAGS    := $(LDEXTRA) -L/opt/rocm/lib -lrocm_smi64 -lrocprofiler-sdk-roctx -lhsa-runtime64

LIB_SRCS   := $(NATIVE)/kernels.hip $(NATIVE)/gemm.hip $(NATIVE)/conc.hip $(NATIVE)/topo.hip $(NATIVE)/ipc.hip $(NATIVE)/trace.hip $(NATIVE)/sdma.hip $(NATIVE)/staged.hip
LIB_OBJS   := $(patsubst $(NATIVE)/%.hip,$(BUILD)/%.o,$(LIB_SRCS))

EXT_SO     := hpc_patterns_amd/_hpk$(EXT_SUFFIX)
BINARIES   := $(BIN)/hpk_conc $(BIN)/hpk_topology $(BIN)/hpk_allreduce $(BIN)/hpk_p2p $(BIN)/hpk_interop $(BIN)/hpk_membench

# Real-MPI twins (MPICH 3.3.2 ships in /opt/conda — not GPU-aware, so the
# miniapps use pinned-direct / staged-device buffer modes; see PARITY.md).
# rpath order matters: the system libstdc++ must shadow conda's old one.
MPI_HOME   ?= /opt/conda
HAVE_MPI   := $(wildcard $(MPI_HOME)/include/mpi.h)
MPI_CFLAGS := -I$(MPI_HOME)/include
# libmpi by absolute path (NOT -L$(MPI_HOME)/lib: that would let the linker
# resolve libstdc++ from conda's old copy); rpath keeps system dirs first
# so the same shadowing cannot happen at runtime.
MPI_LD     := $(MPI_HOME)/lib/libmpi.so -Wl,-rpath,/usr/lib/x86_64-linux-gnu \
              -Wl,-rpath,/opt/rocm/lib -Wl,-rpath,$(MPI_HOME)/lib
ifneq ($(HAVE_MPI),)
BINARIES   += $(BIN)/hpk_mpi_allreduce $(BIN)/hpk_mpi_p2p
endif

.PHONY: all ext bins clean
all: ext bins
ext: $(EXT_SO)
bins: $(BINARIES)

$(BUILD)/%.o: $(NATIVE)/%.hip $(NATIVE)/include/hpk.h | $(BUILD)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BUILD)/ext.o: $(NATIVE)/ext.cpp $(NATIVE)/include/hpk.h | $(BUILD)
	$(HIPCC) $(CXXFLAGS) -I$(PY_INC) -I$(PYBIND_INC) -c $< -o $@

$(EXT_SO): $(BUILD)/ext.o $(LIB_OBJS)
	$(HIPCC) --offload-arch=$(GPU_ARCH) -shared $^ -o $@ $(LDFLAGS)

$(BUILD)/%_main.o: cpp/%_main.cpp $(NATIVE)/include/hpk.h | $(BUILD)
	$(HIPCC) $(CXXFLAGS) -c $< -o $@

$(BIN)/hpk_conc: $(BUILD)/conc_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS)

$(BIN)/hpk_topology: $(BUILD)/topology_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS)

$(BIN)/hpk_allreduce: $(BUILD)/allreduce_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS) -lrccl

$(BIN)/hpk_p2p: $(BUILD)/p2p_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS) -lrccl

$(BIN)/hpk_interop: $(BUILD)/interop_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS) -lrccl

$(BIN)/hpk_membench: $(BUILD)/membench_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS)

# dedicated rule (NOT a target-specific CXXFLAGS append: `make asan`
# overrides CXXFLAGS on the command line, which would drop the append) —
# the shorter-stem pattern beats the generic %_main.o rule
$(BUILD)/mpi_%_main.o: cpp/mpi_%_main.cpp $(NATIVE)/include/hpk.h | $(BUILD)
	$(HIPCC) $(CXXFLAGS) $(MPI_CFLAGS) -c $< -o $@

$(BIN)/hpk_mpi_allreduce: $(BUILD)/mpi_allreduce_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS) $(MPI_LD)

$(BIN)/hpk_mpi_p2p: $(BUILD)/mpi_p2p_main.o $(LIB_OBJS) | $(BIN)
	$(HIPCC) --offload-arch=$(GPU_ARCH) $^ -o $@ $(LDFLAGS) $(MPI_LD)

$(BUILD):
	mkdir -p $(BUILD)
$(BIN):
	mkdir -p $(BIN)

# Race/memory-error hunting build: host-side ASan on the
# binaries. Run on a GPU box with ASAN_OPTIONS=detect_leaks=0 (the HIP
# runtime intentionally holds allocations).
asan:
	$(MAKE) clean-bins
	$(MAKE) CXXFLAGS="$(CXXFLAGS) -fsanitize=address -g1" LDEXTRA="-fsanitize=address" bins

clean-bins:
	rm -rf $(BIN) $(filter %_main.o,$(wildcard $(BUILD)/*.o))

test:
	$(PYTHON) -m pytest tests -q -m "not gpu"

# full GPU battery (tests + performance floors + miniapps) — needs an MI355X
gpu-test:
	bash scripts/ci_gpu.sh

clean:
	rm -rf $(BUILD) $(BIN) $(EXT_SO)

.PHONY: asan clean-bins test gpu-test
