# Build the ft_sgemm CLI binary and the torch extension (gfx950 only).
HIPCC ?= hipcc
ARCH  ?= gfx950
ROCM  ?= /opt/rocm

all: gen cli ext

gen:
	python3 csrc/codegen/gen_kernels.py

cli: bin/ft_sgemm

# filter out PyTorch-hipify byproducts (kernel_*_hip.hip copies)
KERNEL_TUS := $(filter-out %_hip.hip, $(wildcard csrc/generated/kernel_*.hip))

bin/ft_sgemm: csrc/cli_main.hip csrc/dispatch.hip csrc/rocblas_path.hip csrc/ft_kernels.hpp csrc/tier_launch.hpp csrc/ft_core.h csrc/generated/tile_params.h $(KERNEL_TUS)
	mkdir -p bin
	$(HIPCC) -x hip --offload-arch=$(ARCH) -O3 -std=c++17 -Icsrc $(FTFLAGS) \
	  csrc/cli_main.hip csrc/dispatch.hip csrc/rocblas_path.hip $(KERNEL_TUS) \
	  -L$(ROCM)/lib -lrocblas -lroctx64 -o $@

# Paranoid-sync build for the race check (tools/race_check.sh): every
# async-staging site gets an immediate full vmcnt/lgkm drain + barrier;
# outputs must be BIT-IDENTICAL to the normal build (same per-thread
# arithmetic order) — any difference indicates a staging race.
cli-paranoid:
	mkdir -p bin
	$(HIPCC) -x hip --offload-arch=$(ARCH) -O3 -std=c++17 -Icsrc -DFT_PARANOID \
	  csrc/cli_main.hip csrc/dispatch.hip csrc/rocblas_path.hip $(KERNEL_TUS) \
	  -L$(ROCM)/lib -lrocblas -lroctx64 -o bin/ft_sgemm_paranoid

ext:
	# ninja does not track header deps for hipcc sources: rebuild .hip TUs
	# whenever a csrc header is newer than the built extension
	@SO=$$(ls -t ft_sgemm_amd/_C*.so 2>/dev/null | head -1); \
	if [ -n "$$SO" ] && [ -n "$$(find csrc \( -name '*.hpp' -o -name '*.h' \) -newer $$SO 2>/dev/null)" ]; then touch csrc/*.hip csrc/generated/*.hip; fi
	PYTORCH_ROCM_ARCH=$(ARCH) python3 setup.py build_ext --inplace

clean:
	rm -rf bin build ft_sgemm_amd/_C*.so

.PHONY: all gen cli ext clean
