#!/usr/bin/env bash
# Reference-parity entry point (kernel/ft_sgemm/code_gen/gen.sh:1-13).
set -e
cd "$(dirname "$0")"
python3 gen_kernels.py
