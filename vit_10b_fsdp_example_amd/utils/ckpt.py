"""Per-rank sharded checkpoint save/load.

Capability parity with the reference (utils.py:24-43 + xm.save):
  * every rank writes its own shard file ``epoch_{e}_rank_{r}.ckpt``
    (master_only=False path, reference run_vit_training.py:297-299),
  * checkpoint dict keys: {"model", "shard_metadata", "optimizer",
    "lr_scheduler"} — kept byte-identical so downstream tooling and the
    consolidation CLI stay compatible,
  * tensors are copied to host before torch.save (xm.save moves XLA
    tensors to CPU; here we .cpu() so the file never holds device refs).
"""

import os

import torch

from .. import dist as xdist


def _to_cpu(obj):
    if torch.is_tensor(obj):
        return obj.detach().cpu()
    if isinstance(obj, dict):
        return {k: _to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        t = type(obj)
        return t(_to_cpu(v) for v in obj)
    return obj


def save_ckpt(ckpt_path, model, optimizer, lr_scheduler, master_only=True):
    """Save a (possibly sharded) checkpoint.

    With master_only=False every rank writes its own file (FSDP shard
    checkpoints); with master_only=True only rank 0 writes (small /
    non-sharded runs), matching xm.save's contract.
    """
    get_meta = getattr(model, "get_shard_metadata", None)
    ckpt = {
        "model": _to_cpu(model.state_dict()),
        # "shard_metadata" enables offline consolidation via
        # `python3 -m vit_10b_fsdp_example_amd.consolidate_sharded_ckpts`
        "shard_metadata": get_meta() if get_meta is not None else None,
        "optimizer": _to_cpu(optimizer.state_dict()),
        "lr_scheduler": _to_cpu(lr_scheduler.state_dict()),
    }
    if master_only and xdist.get_rank() != 0:
        return
    os.makedirs(os.path.dirname(os.path.abspath(ckpt_path)), exist_ok=True)
    torch.save(ckpt, ckpt_path)
    print(f"checkpoint saved to {ckpt_path}\n", end="", flush=True)


def load_ckpt(ckpt_path, model, optimizer=None, lr_scheduler=None):
    """Load a per-rank shard checkpoint saved by save_ckpt (reference
    utils.py:37-43: torch.load on CPU + three load_state_dict calls)."""
    assert os.path.exists(ckpt_path), f"checkpoint not found: {ckpt_path}"
    ckpt = torch.load(ckpt_path, map_location="cpu", weights_only=False)
    model.load_state_dict(ckpt["model"])
    if optimizer is not None:
        optimizer.load_state_dict(ckpt["optimizer"])
    if lr_scheduler is not None:
        lr_scheduler.load_state_dict(ckpt["lr_scheduler"])
    print(f"resumed from checkpoint {ckpt_path}\n", end="", flush=True)
    return ckpt
