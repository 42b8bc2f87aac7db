#!/bin/bash
# Device-AddressSanitizer compile of every HIP kernel TU (SURVEY.md §5
# "race detection / sanitizers": the reference has none; our CI-lite
# at least proves the kernels are ASAN-instrumentable).
#
#   bash csrc/tools/sanitize_build.sh          # compile-check all TUs
#
# To RUN instrumented kernels on a GPU box, the target needs xnack
# (HSA_XNACK=1) and an ASAN-enabled runtime; this script only performs
# the instrumentation build, which catches OOB patterns the compiler
# can prove and keeps the TUs compatible with -fsanitize=address.
set -e
cd "$(dirname "$0")/../.."
for SRC in csrc/*.hip; do
  case "$SRC" in *_hip.hip) continue;; esac   # skip build-generated hipify artifacts
  echo "== $SRC =="
  hipcc --offload-arch=gfx950:xnack+ -fsanitize=address -g -O1 \
    -std=c++17 -DVITFSDP_KERNELS_ONLY -c "$SRC" -o /tmp/vitfsdp_asan.o
done
echo "ASAN build OK"
