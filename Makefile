# Build the MI355X POST engine (libpost_hip.so, gfx950) and the CPU oracle.
# hipcc cross-compiles gfx950 without a GPU; the in-tree .so travels to the
# GPU box with the gpurun snapshot.
HIPCC   ?= hipcc
ARCH    ?= gfx950
HIPFLAGS ?= --offload-arch=$(ARCH) -O3 -std=c++17 -fPIC -Wall

ENGINE  := go-spacemesh_amd/libpost_hip.so
CSRC    := go-spacemesh_amd/csrc

all: $(ENGINE) oracle

SRCS_ENGINE := $(CSRC)/kernels.hip $(CSRC)/engine.cpp $(CSRC)/crypto_host.cpp
HDRS_ENGINE := $(CSRC)/post_common.h $(CSRC)/kernel_args.h \
               $(CSRC)/crypto_host.h include/spacemesh_post.h

$(ENGINE): $(SRCS_ENGINE) $(HDRS_ENGINE)
	$(HIPCC) $(HIPFLAGS) -shared $(SRCS_ENGINE) -o $@

# A/B variant: non-temporal scratch access (POST_ENGINE_LIB selects)
go-spacemesh_amd/libpost_hip_nt.so: $(SRCS_ENGINE) $(HDRS_ENGINE)
	$(HIPCC) $(HIPFLAGS) -DPOSTE_NT=1 -shared $(SRCS_ENGINE) -o $@

nt: go-spacemesh_amd/libpost_hip_nt.so
.PHONY: nt

# A/B variant: scan ILP depth 4
go-spacemesh_amd/libpost_hip_scan4.so: $(SRCS_ENGINE) $(HDRS_ENGINE)
	$(HIPCC) $(HIPFLAGS) -DPOSTE_SCAN_ILP=4 -shared $(SRCS_ENGINE) -o $@
scan4: go-spacemesh_amd/libpost_hip_scan4.so
.PHONY: scan4

# A/B variant: scan ILP depth 1
go-spacemesh_amd/libpost_hip_scan1.so: $(SRCS_ENGINE) $(HDRS_ENGINE)
	$(HIPCC) $(HIPFLAGS) -DPOSTE_SCAN_ILP=1 -shared $(SRCS_ENGINE) -o $@
scan1: go-spacemesh_amd/libpost_hip_scan1.so
.PHONY: scan1

oracle:
	$(MAKE) -C oracle

clean:
	rm -f $(ENGINE)
	$(MAKE) -C oracle clean

.PHONY: all oracle clean
