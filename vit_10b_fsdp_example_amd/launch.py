"""Per-node process launcher.

Capability parity with xmp.spawn (reference run_vit_training.py:364):
one worker process per local GPU, with rank/topology environment set up
the torchrun way (RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR /
MASTER_PORT), RCCL process-group init happening in the worker via
dist.init_distributed().

Behavior:
  * if WORLD_SIZE is already in the environment (launched by
    torch.distributed.run / torchrun), run the worker inline;
  * else spawn one process per visible GPU (or run inline when <2 GPUs),
    mirroring xmp.spawn's one-process-per-device model on a single node.

Multi-node: use torchrun --nnodes=N (the reference's xla_dist SSH
fan-out has no single-node equivalent to replicate; torchrun covers it).
"""

import os

import torch
import torch.multiprocessing as mp


def _worker(local_rank, world_size, fn, args, master_port):
    os.environ["RANK"] = str(local_rank)
    os.environ["LOCAL_RANK"] = str(local_rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = os.environ.get("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(master_port)
    fn(*args)


def spawn(fn, args=()):
    """Run fn(*args) once per local device (or inline under torchrun /
    single device)."""
    if "WORLD_SIZE" in os.environ:
        fn(*args)
        return
    n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    if n_gpus <= 1:
        fn(*args)
        return
    master_port = int(os.environ.get("MASTER_PORT", "29500"))
    mp.start_processes(
        _worker,
        args=(n_gpus, fn, args, master_port),
        nprocs=n_gpus,
        start_method="spawn",
        join=True,
    )
