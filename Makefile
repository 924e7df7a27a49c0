# Build the ft_sgemm CLI binary and the torch extension (gfx950 only).
HIPCC ?= hipcc
ARCH  ?= gfx950
ROCM  ?= /opt/rocm

all: gen cli ext

gen:
	python3 csrc/codegen/gen_kernels.py

cli: bin/ft_sgemm

bin/ft_sgemm: csrc/cli_main.hip csrc/dispatch.hip csrc/rocblas_path.hip csrc/ft_kernels.hpp csrc/ft_core.h csrc/generated/tile_params.h
	mkdir -p bin
	$(HIPCC) -x hip --offload-arch=$(ARCH) -O3 -std=c++17 -Icsrc \
	  csrc/cli_main.hip csrc/dispatch.hip csrc/rocblas_path.hip \
	  -L$(ROCM)/lib -lrocblas -o $@

ext:
	PYTORCH_ROCM_ARCH=$(ARCH) python3 setup.py build_ext --inplace

clean:
	rm -rf bin build ft_sgemm_amd/_C*.so

.PHONY: all gen cli ext clean
