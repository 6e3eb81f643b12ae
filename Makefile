# Build everything native: the gfx950 compute library, the mgp module .so's,
# the mgp host mock, and the oracle (+_ref where /root/reference exists).
# `__graft_entry__.build()` drives this via `make all`.

HIPCC := hipcc
GXX := g++
ARCH := --offload-arch=gfx950
HIPFLAGS := $(ARCH) -O3 -std=c++17 -fPIC -Wall -Werror -Wno-unused-function
LIBDIR := memgraph_amd/lib

MGX_SRCS := memgraph_amd/csrc/mgx_api.cpp memgraph_amd/csrc/comm.cpp
MGX_HIP_SRCS := memgraph_amd/csrc/graph_build.hip memgraph_amd/csrc/pagerank.hip \
                memgraph_amd/csrc/pronline.hip memgraph_amd/csrc/katz_online.hip \
                memgraph_amd/csrc/labelrankt.hip memgraph_amd/csrc/leiden.hip \
                memgraph_amd/csrc/wcc.hip memgraph_amd/csrc/katz.hip \
                memgraph_amd/csrc/betweenness.hip \
                memgraph_amd/csrc/louvain.hip
MGX_OBJS := $(MGX_SRCS:%.cpp=build/%.o) $(MGX_HIP_SRCS:%.hip=build/%.o)

MODULES := pagerank katz_centrality community_detection weakly_connected_components betweenness_centrality pagerank_online katz_centrality_online community_detection_online leiden_community_detection

all: $(LIBDIR)/libmgx_analytics.so modules mock oracle

build/%.o: %.cpp memgraph_amd/csrc/mgx_internal.h include/mgx_analytics.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

build/%.o: %.hip memgraph_amd/csrc/mgx_internal.h include/mgx_analytics.h include/mgx_graphgen.h
	@mkdir -p $(dir $@)
	$(HIPCC) $(HIPFLAGS) -c $< -o $@

$(LIBDIR)/libmgx_analytics.so: $(MGX_OBJS)
	@mkdir -p $(LIBDIR)
	$(HIPCC) $(ARCH) -shared -o $@ $(MGX_OBJS) -L/opt/rocm/lib -lrccl

# The drop-in module .so's: stem names fixed by the reference's module
# loader (module.cpp:1412-1428). Host C++ only; they link the compute lib.
.PHONY: modules
modules: $(MODULES:%=$(LIBDIR)/modules/%.so)

$(LIBDIR)/modules/%.so: memgraph_amd/modules/%_module.cpp memgraph_amd/modules/module_common.hpp include/mgx_mgp.h $(LIBDIR)/libmgx_analytics.so
	@mkdir -p $(LIBDIR)/modules
	$(GXX) -O2 -std=c++17 -fPIC -Wall -shared -o $@ $< \
	  -Iinclude -L$(LIBDIR) -lmgx_analytics '-Wl,-rpath,$$ORIGIN/..'

.PHONY: mock
mock: tests/mock/libmgp_mock.so

tests/mock/libmgp_mock.so: tests/mock/mgp_mock.cpp include/mgx_mgp.h
	$(GXX) -O2 -std=c++17 -fPIC -Wall -shared -o $@ $< -Iinclude

.PHONY: oracle
oracle:
	$(MAKE) -C oracle

.PHONY: clean
clean:
	rm -rf build $(LIBDIR) tests/mock/libmgp_mock.so
	$(MAKE) -C oracle clean
