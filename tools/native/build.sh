#!/bin/bash
# Build the standalone native scorer for gfx950 (cross-compiles without a GPU).
set -e
cd "$(dirname "$0")/../.."
/opt/rocm/bin/hipcc --offload-arch=gfx950 -O3 -std=c++17 -ffp-contract=off \
  tools/native/ifa_score.cpp \
  isolation_forest_amd/ops/hip/forest_kernels.hip \
  -Iisolation_forest_amd/ops/hip -lz -o tools/native/ifa_score
echo "built tools/native/ifa_score"
