# Top-level build: GPU apply-path library (libgra.so, gfx950) + CPU oracle.
# Built IN-TREE so the .so files travel with the gpurun snapshot.
HIPCC ?= hipcc
ARCH ?= gfx950
HIPFLAGS ?= -O3 -std=c++17 --offload-arch=$(ARCH) -fPIC -Wall

CSRC := rocksplicator_amd/csrc
OBJS := build/engine.o build/host_store.o build/builder.o build/gen.o

all: rocksplicator_amd/libgra.so oracle/libwb_oracle.so

build:
	mkdir -p build

build/engine.o: $(CSRC)/engine.hip $(CSRC)/wb_format.h $(CSRC)/host_store.h $(CSRC)/snappy.h include/rocksplicator_gpu.h | build
	$(HIPCC) $(HIPFLAGS) -x hip -c $< -o $@

build/host_store.o: $(CSRC)/host_store.cpp $(CSRC)/host_store.h $(CSRC)/wb_format.h | build
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

build/builder.o: $(CSRC)/builder.cpp $(CSRC)/wb_format.h include/rocksplicator_gpu.h | build
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

build/gen.o: $(CSRC)/gen.cpp $(CSRC)/wb_format.h $(CSRC)/snappy.h include/rocksplicator_gpu.h | build
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

rocksplicator_amd/libgra.so: $(OBJS)
	$(HIPCC) $(HIPFLAGS) -shared $(OBJS) -o $@

oracle/libwb_oracle.so: oracle/wb_oracle.c oracle/wb_oracle.h
	$(MAKE) -C oracle

# auxiliary binaries (C++ chain test, streaming bench, kernel microbenches,
# DbWrapper-adapter compile proof)
tools: build/test_cpp_chain build/bench_stream build/micro_copy build/micro_snappy build/seam_compile_check

# compile-proof of INTEGRATION.md's GpuApplyDbWrapper against the exact
# DbWrapper virtual seam (mock restatement of db_wrapper.h:6-15); plain g++
# on purpose — the adapter is host C++ a maintainer builds without hipcc
build/seam_compile_check: tools/mock_seam/compile_check.cpp tools/mock_seam/gpu_apply_db_wrapper.h tools/mock_seam/rocksdb_replicator/db_wrapper.h include/rocksplicator_gpu.h rocksplicator_amd/libgra.so | build
	g++ -O2 -std=c++17 -Wall -Wextra $< -Itools/mock_seam -Iinclude -Lrocksplicator_amd -lgra -Wl,-rpath,'$$ORIGIN/../rocksplicator_amd' -o $@
	./build/seam_compile_check

build/test_cpp_chain: scripts/test_cpp_chain.cpp include/rocksplicator_replicator.hpp include/rocksplicator_gpu.h rocksplicator_amd/libgra.so | build
	$(HIPCC) $(HIPFLAGS) $< -Iinclude -Lrocksplicator_amd -lgra -Wl,-rpath,'$$ORIGIN/../rocksplicator_amd' -o $@

build/bench_stream: scripts/bench_stream.cpp include/rocksplicator_gpu.h rocksplicator_amd/libgra.so | build
	$(HIPCC) $(HIPFLAGS) $< -Lrocksplicator_amd -lgra -Wl,-rpath,'$$ORIGIN/../rocksplicator_amd' -o $@

build/micro_copy: scripts/micro_copy.hip | build
	$(HIPCC) $(HIPFLAGS) -x hip $< -o $@

build/micro_snappy: scripts/micro_snappy.hip | build
	$(HIPCC) $(HIPFLAGS) -x hip $< -o $@

clean:
	rm -rf build rocksplicator_amd/libgra.so
	$(MAKE) -C oracle clean

.PHONY: all clean
