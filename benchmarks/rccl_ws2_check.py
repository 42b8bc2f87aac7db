#!/usr/bin/env python3
"""RCCL ws=2 validation on a single MI355X: run the
FSDP engine through the real `nccl` (=RCCL) backend with both ranks on
the same GPU — dual communicators, async all_gather_into_tensor,
reduce_scatter_tensor, and the batch_isend_irecv P2P algorithms — and
check the training trajectory against the ws=1 run of the same global
batch.

Launch (on a GPU box):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --master-port 29511 \
        benchmarks/rccl_ws2_check.py [--steps 4] [--ag p2p] [--rs p2p]

Single-process reference (prints the expected losses):
    python benchmarks/rccl_ws2_check.py --ws1

Both ranks map to cuda:0 (dist.py takes local_rank % device_count), so
this runs inside a 1-GPU gpurun lease.  Rank 0 prints one JSON line:
{"losses": [...], "gnorms": [...], "backend": "nccl", "algos": ...}.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run(steps, ws1=False):
    import torch

    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd.models import build_fsdp_vit_model
    from vit_10b_fsdp_example_amd.ops import CrossEntropyLoss, FusedAdamW
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args([
        "--fake_data", "--image_size", "224", "--patch_size", "14",
        "--embed_dim", "640", "--num_heads", "4", "--num_blocks", "3",
        "--num_classes", "100", "--batch_size", "8", "--num_workers", "0",
    ])
    device = xdist.init_distributed()
    world = xdist.get_world_size()
    rank = xdist.get_rank()
    torch.manual_seed(1234)
    model = build_fsdp_vit_model(cfg, device, compute_dtype=torch.bfloat16)
    loss_fn = CrossEntropyLoss()
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.1)

    gen = torch.Generator().manual_seed(55)
    losses, gnorms = [], []
    for _ in range(steps):
        gx = torch.randn(8, 3, 224, 224, generator=gen)
        gy = torch.randint(0, 100, (8,), generator=gen)
        per = 8 // world
        x = gx[rank * per:(rank + 1) * per].to(device, torch.bfloat16)
        y = gy[rank * per:(rank + 1) * per].to(device)
        loss = loss_fn(model(x), y)
        loss.backward()
        gn = model.clip_grad_norm_(1.0, defer_scale=True)
        opt.step()
        opt.zero_grad(set_to_none=True)
        mean_loss = (
            xdist.mesh_reduce("loss", float(loss.detach()), sum) / world
        )
        losses.append(round(mean_loss, 4))
        gnorms.append(round(float(gn.detach()), 4))
    if rank == 0:
        print(json.dumps({
            "losses": losses,
            "gnorms": gnorms,
            "world": world,
            "backend": (
                torch.distributed.get_backend()
                if torch.distributed.is_initialized() else "none"
            ),
            "algos": {
                "ag": os.environ.get("VITFSDP_AG_ALGO", "allgather"),
                "rs": os.environ.get("VITFSDP_RS_ALGO", "reducescatter"),
            },
        }), flush=True)
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()




def loopback():
    """RCCL cannot place two ranks on one GPU ("Duplicate GPU detected",
    RCCL 2.26 — gpurun_out/ws2_full.log), so inside a 1-GPU lease this
    exercises the REAL nccl(=RCCL) backend at ws=1: init, dual
    communicator creation (CommContext), and the exact collective calls
    + message sizes of the ViT-10B FSDP step (async
    all_gather_into_tensor of the 629 MB bf16 unit payload,
    reduce_scatter_tensor, scalar all-reduce) issued directly on the
    groups (the engine's own ws=1 path short-circuits them)."""
    import time

    import torch
    import torch.distributed as dist

    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.parallel import CommContext

    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29529")
    device = xdist.init_distributed()
    assert dist.is_initialized() and dist.get_backend() == "nccl"
    CommContext.reset()
    ctx = CommContext.get()
    assert ctx.gather_group is not None and ctx.reduce_group is not None

    unit = 314_639_360  # ViT-10B block params
    shard = torch.randn(unit, device=device).to(torch.bfloat16)
    full = torch.empty(unit, device=device, dtype=torch.bfloat16)
    out = {}
    t0 = time.time()
    for _ in range(3):
        w = dist.all_gather_into_tensor(full, shard, group=ctx.gather_group,
                                        async_op=True)
        w.wait()
    torch.cuda.synchronize()
    out["allgather_ms"] = round((time.time() - t0) / 3 * 1e3, 2)
    assert torch.equal(full, shard)

    grads = torch.randn(unit, device=device).to(torch.bfloat16)
    red = torch.empty(unit, device=device, dtype=torch.bfloat16)
    t0 = time.time()
    for _ in range(3):
        w = dist.reduce_scatter_tensor(red, grads, group=ctx.reduce_group,
                                       async_op=True)
        w.wait()
    torch.cuda.synchronize()
    out["reducescatter_ms"] = round((time.time() - t0) / 3 * 1e3, 2)
    assert torch.equal(red, grads)

    s = torch.ones((), device=device)
    dist.all_reduce(s, group=ctx.reduce_group)
    out["scalar_allreduce"] = float(s)
    print(json.dumps({"loopback": out, "backend": "nccl(RCCL)"}), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--ws1", action="store_true",
                    help="single-process reference run")
    ap.add_argument("--loopback", action="store_true",
                    help="ws=1 through the real RCCL backend (1-GPU lease)")
    ap.add_argument("--ag", default=None, choices=["allgather", "p2p"])
    ap.add_argument("--rs", default=None, choices=["reducescatter", "p2p"])
    a = ap.parse_args()
    if a.ag:
        os.environ["VITFSDP_AG_ALGO"] = a.ag
    if a.rs:
        os.environ["VITFSDP_RS_ALGO"] = a.rs
    if a.ws1:
        for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
                  "MASTER_PORT"):
            os.environ.pop(k, None)
    if a.loopback:
        loopback()
    else:
        run(a.steps, ws1=a.ws1)
