#!/bin/bash
# Round-2 opening A/B sweep on one MI355X (run via gpurun; ~15-20 min):
#   /usr/local/graft/bin/gpurun --timeout 1500 -- 'bash benchmarks/ab_sweep.sh'
# Prints one labeled JSON line per variant; all knobs measured within
# the SAME box so the ±5% box-to-box clock variance cancels.
# Multi-GPU sweeps (P2P algos, prefetch depth under real RCCL) follow
# docs/SCALING.md and need the driver's 8-GPU node, not this script.
set -u
cd "$(dirname "$0")/.."
STEPS=${STEPS:-8}
WARM=${WARM:-3}
MODEL=${MODEL:-vit10b}

run() {
  local label="$1"; shift
  echo "=== $label"
  timeout 420 env "$@" python bench.py --model "$MODEL" \
      --steps "$STEPS" --warmup "$WARM" ${EXTRA:-} 2>&1 | tail -1
}

run baseline
EXTRA="--fuse_residual" run fuse_residual
EXTRA="--grad_ckpt_blocks 16" run partial_ckpt_16
EXTRA="--fuse_residual --grad_ckpt_blocks 16" run fused_plus_partial
run wgrad_mode2 VITFSDP_NATIVE_WGRAD=2
EXTRA="--fuse_residual" run fused_plus_wgrad2 VITFSDP_NATIVE_WGRAD=2
