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

# ---- host-side AddressSanitizer build (sanitizer config)
# Builds the GPU-free host binaries (create-fusion-container, solver)
# with ASAN and runs the CPU host tests against them:
#   make asan && BS_BIN=bigstitcher_spark_amd/bin-asan python -m pytest \
#       tests/test_cli_host.py -q -m "not gpu"
ASANDIR = bigstitcher_spark_amd/bin-asan
ASANFLAGS = -O1 -g -std=c++17 -fsanitize=address -fno-omit-frame-pointer -Wall

asan:
	@mkdir -p $(ASANDIR)
	$(CXX_HOST) $(ASANFLAGS) -Iinclude $(HOSTDIR)/cli_container.cpp $(HOSTDIR)/bs_json.cpp $(HOSTDIR)/bs_n5.cpp $(HOSTDIR)/bs_zarr.cpp $(HOSTDIR)/bs_xml.cpp $(HOSTDIR)/bs_spimdata.cpp -o $(ASANDIR)/create-fusion-container -lz -l:libzstd.so.1
	$(CXX_HOST) $(ASANFLAGS) -Iinclude $(HOSTDIR)/cli_solver.cpp $(HOSTDIR)/bs_json.cpp $(HOSTDIR)/bs_n5.cpp $(HOSTDIR)/bs_zarr.cpp $(HOSTDIR)/bs_xml.cpp $(HOSTDIR)/bs_spimdata.cpp -o $(ASANDIR)/solver -lz -l:libzstd.so.1
	$(CXX_HOST) $(ASANFLAGS) -Iinclude $(HOSTDIR)/cli_stitching.cpp $(HOSTDIR)/bs_json.cpp $(HOSTDIR)/bs_n5.cpp $(HOSTDIR)/bs_zarr.cpp $(HOSTDIR)/bs_xml.cpp $(HOSTDIR)/bs_spimdata.cpp -o $(ASANDIR)/stitching -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib
	$(CXX_HOST) $(ASANFLAGS) -Iinclude $(HOSTDIR)/cli_fusion.cpp $(HOSTDIR)/bs_json.cpp $(HOSTDIR)/bs_n5.cpp $(HOSTDIR)/bs_zarr.cpp $(HOSTDIR)/bs_xml.cpp $(HOSTDIR)/bs_spimdata.cpp -o $(ASANDIR)/affine-fusion -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib
	$(CXX_HOST) $(ASANFLAGS) -Iinclude $(HOSTDIR)/cli_resave.cpp $(HOSTDIR)/bs_json.cpp $(HOSTDIR)/bs_n5.cpp $(HOSTDIR)/bs_zarr.cpp $(HOSTDIR)/bs_xml.cpp $(HOSTDIR)/bs_spimdata.cpp -o $(ASANDIR)/resave -Lbigstitcher_spark_amd -lbigstitch -lz -l:libzstd.so.1 -L/opt/rocm/lib -lamdhip64 -Wl,-rpath,'$$ORIGIN/..' -Wl,-rpath,/opt/rocm/lib

.PHONY: asan
