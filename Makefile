# libbigstitch — MI355X (gfx950) build. Built in-tree so the .so travels
# with the repo snapshot to the GPU box (see __graft_entry__.build()).
HIPCC ?= hipcc
ARCH ?= gfx950
CXXFLAGS = --offload-arch=$(ARCH) -O3 -std=c++17 -fPIC -Wall

LIB = bigstitcher_spark_amd/libbigstitch.so

all: $(LIB)

$(LIB): bigstitcher_spark_amd/csrc/bigstitch.hip include/bigstitch.h
	$(HIPCC) $(CXXFLAGS) -shared -Iinclude $< -o $@

resource-report: bigstitcher_spark_amd/csrc/bigstitch.hip
	$(HIPCC) $(CXXFLAGS) -Rpass-analysis=kernel-resource-usage -c $< -o /tmp/bs_rr.o

clean:
	rm -f $(LIB)

.PHONY: all clean resource-report
