#!/bin/bash
# Build tools/lt_probe (hipBLASLt vs pcnn_deep_gemm probe).  hipcc must
# get the .cpp and the prebuilt kernel .o in separate compile/link steps
# (passing a .o alongside a .cpp makes hipcc parse the ELF as source).
set -e
cd "$(dirname "$0")/.."
hipcc --offload-arch=gfx950 -O3 -c tools/lt_probe.cpp -o tools/lt_probe.o
hipcc --offload-arch=gfx950 tools/lt_probe.o csrc/hip/conv_kernels.o \
    -L/opt/rocm/lib -lhipblaslt -o tools/lt_probe
