# libbigstitch — MI355X (gfx950) build. Built in-tree so the .so travels
# with the repo snapshot to the GPU box (see __graft_entry__.build()).
HIPCC ?= hipcc
ARCH ?= gfx950
CXXFLAGS = --offload-arch=$(ARCH) -O3 -std=c++17 -fPIC -Wall

LIB = bigstitcher_spark_amd/libbigstitch.so

all: $(LIB) cli

$(LIB): bigstitcher_spark_amd/csrc/bigstitch.hip include/bigstitch.h
	$(HIPCC) $(CXXFLAGS) -shared -Iinclude $< -o $@

resource-report: bigstitcher_spark_amd/csrc/bigstitch.hip
	$(HIPCC) $(CXXFLAGS) -Rpass-analysis=kernel-resource-usage -c $< -o /tmp/bs_rr.o

clean:
	rm -f $(LIB)

.PHONY: all clean resource-report

# ---- host surface: N5/XML/SpimData + CLI binaries ----
HOSTDIR = bigstitcher_spark_amd/csrc/host
HOSTOBJS = $(HOSTDIR)/bs_json.o $(HOSTDIR)/bs_n5.o $(HOSTDIR)/bs_zarr.o $(HOSTDIR)/bs_xml.o $(HOSTDIR)/bs_spimdata.o
BINDIR = bigstitcher_spark_amd/bin
CXX_HOST = g++
HOSTFLAGS = -O2 -std=c++17 -fPIC -Wall

$(HOSTDIR)/%.o: $(HOSTDIR)/%.cpp $(HOSTDIR)/%.h
	$(CXX_HOST) $(HOSTFLAGS) -c $< -o $@

cli: $(BINDIR)/stitching $(BINDIR)/create-fusion-container $(BINDIR)/affine-fusion $(BINDIR)/solver $(BINDIR)/resave

$(BINDIR)/stitching: $(HOSTDIR)/cli_stitching.cpp $(HOSTOBJS) $(LIB)
	@mkdir -p $(BINDIR)
	$(CXX_HOST) $(HOSTFLAGS) -Iinclude $< $(HOSTOBJS) -o $@ -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib

$(BINDIR)/create-fusion-container: $(HOSTDIR)/cli_container.cpp $(HOSTOBJS)
	@mkdir -p $(BINDIR)
	$(CXX_HOST) $(HOSTFLAGS) -Iinclude $< $(HOSTOBJS) -o $@ -lz -l:libzstd.so.1

$(BINDIR)/affine-fusion: $(HOSTDIR)/cli_fusion.cpp $(HOSTOBJS) $(LIB)
	@mkdir -p $(BINDIR)
	$(CXX_HOST) $(HOSTFLAGS) -Iinclude $< $(HOSTOBJS) -o $@ -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib

$(BINDIR)/solver: $(HOSTDIR)/cli_solver.cpp $(HOSTOBJS)
	@mkdir -p $(BINDIR)
	$(CXX_HOST) $(HOSTFLAGS) -Iinclude $< $(HOSTOBJS) -o $@ -lz -l:libzstd.so.1

$(BINDIR)/resave: $(HOSTDIR)/cli_resave.cpp $(HOSTOBJS) $(LIB)
	@mkdir -p $(BINDIR)
	$(CXX_HOST) $(HOSTFLAGS) -Iinclude $< $(HOSTOBJS) -o $@ -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib

clean-cli:
	rm -f $(HOSTOBJS) $(BINDIR)/*

.PHONY: cli clean-cli
