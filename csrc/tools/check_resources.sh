#!/bin/bash
# Per-kernel VGPR/SGPR/LDS/occupancy report (no torch headers needed):
#   bash csrc/tools/check_resources.sh csrc/fmha.hip
set -e
SRC=${1:-csrc/fmha.hip}
hipcc --offload-arch=gfx950 -O3 -std=c++17 -DVITFSDP_KERNELS_ONLY \
  -Rpass-analysis=kernel-resource-usage -c "$SRC" -o /tmp/vitfsdp_ra.o
