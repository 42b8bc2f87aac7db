#!/usr/bin/env python3
"""CLI entry point — drop-in replacement for the reference's
run_vit_training.py (same 29 flags and defaults, reference
run_vit_training.py:327-364), launching one process per local MI355X.

Usage (single node, all visible GPUs):
    python3 run_vit_training.py --fake_data [flags]
or under torchrun:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 run_vit_training.py --fake_data [flags]
"""

from vit_10b_fsdp_example_amd.cli import parse_args
from vit_10b_fsdp_example_amd.launch import spawn
from vit_10b_fsdp_example_amd.train import main


if __name__ == "__main__":
    cfg = parse_args()
    spawn(main, args=(cfg,))
