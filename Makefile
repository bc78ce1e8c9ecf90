# Build for the MI355X-native DBSP hot path.
#   make oracle  - CPU oracle (test infrastructure + bench cpu_baseline)
#   make gen     - CPU Nexmark generator library
#   make hip     - HIP kernels + engine for gfx950 (cross-compiles without a GPU)
#   make all     - everything
CXX      ?= g++
HIPCC    ?= hipcc
ARCH     ?= gfx950
CXXFLAGS ?= -O2 -std=c++17 -Wall -fPIC
HIPFLAGS ?= -O3 -std=c++17 --offload-arch=$(ARCH) -fPIC

PKG = database-stream-processor_amd

all: oracle gen hip

oracle: oracle/liboracle_dbsp.so
oracle/liboracle_dbsp.so: oracle/zset_oracle.cpp include/dbsp_hip.h
	$(CXX) $(CXXFLAGS) -shared oracle/zset_oracle.cpp -o $@

gen: $(PKG)/libdbsp_gen.so
$(PKG)/libdbsp_gen.so: $(PKG)/csrc/gen_lib.cpp $(PKG)/csrc/nexmark_gen.hpp include/dbsp_hip.h
	$(CXX) $(CXXFLAGS) -shared $(PKG)/csrc/gen_lib.cpp -o $@

hip: $(PKG)/libdbsp_hip.so
$(PKG)/libdbsp_hip.so: $(PKG)/csrc/kernels.hip $(PKG)/csrc/engine.cpp $(PKG)/csrc/nexmark_gen.hpp include/dbsp_hip.h
	$(HIPCC) $(HIPFLAGS) -shared $(PKG)/csrc/kernels.hip $(PKG)/csrc/engine.cpp \
	    -L/opt/rocm/lib -lrccl -o $@

clean:
	rm -f oracle/liboracle_dbsp.so $(PKG)/libdbsp_gen.so $(PKG)/libdbsp_hip.so

.PHONY: all oracle gen hip clean
