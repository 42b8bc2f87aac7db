"""Checkpoint round-trip + consolidation:
save per-rank shard ckpts -> consolidate -> load into an UNWRAPPED model
-> outputs match the sharded model's."""

import os

import numpy as np
import pytest
import torch

from tests.utils_mp import run_multiprocess

TINY = [
    "--fake_data", "--image_size", "32", "--patch_size", "4",
    "--embed_dim", "64", "--num_heads", "4", "--num_blocks", "3",
    "--num_classes", "10", "--batch_size", "8", "--num_workers", "0",
]


def _build(cfg, device, dtype=torch.float32):
    import torch
    from vit_10b_fsdp_example_amd.models import build_fsdp_vit_model

    torch.manual_seed(1234)
    return build_fsdp_vit_model(cfg, device, compute_dtype=dtype)


def _train_save(rank, world_size, ckpt_dir):
    import torch
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.ops import CrossEntropyLoss, FusedAdamW
    from vit_10b_fsdp_example_amd.utils import (
        get_warmup_cosine_scheduler, save_ckpt,
    )
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args(TINY)
    device = xdist.init_distributed()
    model = _build(cfg, device)
    loss_fn = CrossEntropyLoss()
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.1)
    sched = get_warmup_cosine_scheduler(opt, 10, 100)
    gen = torch.Generator().manual_seed(3)
    for _ in range(2):
        x = torch.randn(4, 3, 32, 32, generator=gen)
        y = torch.randint(0, 10, (4,), generator=gen)
        loss_fn(model(x), y).backward()
        model.clip_grad_norm_(1.0)
        opt.step()
        sched.step()
        opt.zero_grad(set_to_none=True)
    path = os.path.join(ckpt_dir, f"epoch_1_rank_{rank}.ckpt")
    save_ckpt(path, model, opt, sched, master_only=False)
    # also return a deterministic eval output for comparison
    torch.manual_seed(9)
    x = torch.randn(4, 3, 32, 32)
    model.eval()
    with torch.no_grad():
        out = model(x)
    return out.numpy()


def _resume_and_eval(rank, world_size, ckpt_dir):
    import torch
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.ops import CrossEntropyLoss, FusedAdamW
    from vit_10b_fsdp_example_amd.utils import (
        get_warmup_cosine_scheduler, load_ckpt,
    )
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args(TINY)
    device = xdist.init_distributed()
    model = _build(cfg, device)
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.1)
    sched = get_warmup_cosine_scheduler(opt, 10, 100)
    path = os.path.join(ckpt_dir, f"epoch_1_rank_{rank}.ckpt")
    load_ckpt(path, model, opt, sched)
    torch.manual_seed(9)
    x = torch.randn(4, 3, 32, 32)
    model.eval()
    with torch.no_grad():
        out = model(x)
    return out.numpy()


def test_shard_ckpt_roundtrip_ws2(tmp_path):
    ckpt_dir = str(tmp_path)
    outs = run_multiprocess(_train_save, world_size=2, args=(ckpt_dir,))
    np.testing.assert_allclose(outs[0], outs[1], rtol=1e-6, atol=1e-7)
    resumed = run_multiprocess(_resume_and_eval, world_size=2, args=(ckpt_dir,))
    for r in resumed:
        np.testing.assert_allclose(r, outs[0], rtol=1e-6, atol=1e-7)


def test_consolidation_ws2(tmp_path):
    ckpt_dir = str(tmp_path)
    outs = run_multiprocess(_train_save, world_size=2, args=(ckpt_dir,))

    from vit_10b_fsdp_example_amd.consolidate_sharded_ckpts import (
        consolidate_files,
    )
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd.models import FSDPViTModel

    save_path = os.path.join(ckpt_dir, "consolidated.ckpt")
    full = consolidate_files(
        os.path.join(ckpt_dir, "epoch_1_rank_"), ".ckpt", save_path
    )
    assert os.path.exists(save_path)

    cfg = parse_args(TINY)
    torch.manual_seed(0)
    plain = FSDPViTModel(
        image_size=cfg.image_size, patch_size=cfg.patch_size,
        embed_dim=cfg.embed_dim, num_heads=cfg.num_heads,
        num_blocks=cfg.num_blocks, mlp_ratio=cfg.mlp_ratio,
        pos_dropout=0.0, mlp_dropout=0.0, att_dropout=0.0,
        num_classes=cfg.num_classes,
        grad_ckpt_wrap=lambda m: m, fsdp_wrap=lambda m: m,
    )
    missing, unexpected = plain.load_state_dict(full["model"], strict=True)
    assert not missing and not unexpected

    torch.manual_seed(9)
    x = torch.randn(4, 3, 32, 32)
    plain.eval()
    with torch.no_grad():
        out = plain(x)
    np.testing.assert_allclose(out.numpy(), outs[0], rtol=1e-5, atol=1e-6)


def test_resume_trajectory_equality(tmp_path):
    """Training N steps, checkpointing, then continuing must produce the
    same losses as an uninterrupted run (optimizer + scheduler state
    round-trip through the per-rank shard checkpoint)."""
    import torch
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.ops import CrossEntropyLoss, FusedAdamW
    from vit_10b_fsdp_example_amd.utils import (
        get_warmup_cosine_scheduler, load_ckpt, save_ckpt,
    )
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args(TINY)
    device = xdist.init_distributed()

    def make():
        model = _build(cfg, device)
        opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.1)
        sched = get_warmup_cosine_scheduler(opt, 5, 50)
        return model, opt, sched

    def steps(model, opt, sched, gen, n):
        loss_fn = CrossEntropyLoss()
        losses = []
        for _ in range(n):
            x = torch.randn(4, 3, 32, 32, generator=gen)
            y = torch.randint(0, 10, (4,), generator=gen)
            loss = loss_fn(model(x), y)
            loss.backward()
            model.clip_grad_norm_(1.0)
            opt.step()
            sched.step()
            opt.zero_grad(set_to_none=True)
            losses.append(float(loss.detach()))
        return losses

    # uninterrupted: 4 + 2 steps
    model, opt, sched = make()
    gen = torch.Generator().manual_seed(11)
    first = steps(model, opt, sched, gen, 4)
    path = str(tmp_path / "epoch_1_rank_0.ckpt")
    save_ckpt(path, model, opt, sched, master_only=False)
    cont = steps(model, opt, sched, gen, 2)

    # resumed run: fresh objects, load, same remaining data
    CommContext.reset()
    model2, opt2, sched2 = make()
    load_ckpt(path, model2, opt2, sched2)
    gen2 = torch.Generator().manual_seed(11)
    _ = [  # replay the first 4 batches to align the data stream
        (torch.randn(4, 3, 32, 32, generator=gen2),
         torch.randint(0, 10, (4,), generator=gen2))
        for _ in range(4)
    ]
    resumed = steps(model2, opt2, sched2, gen2, 2)
    np.testing.assert_allclose(resumed, cont, rtol=1e-5, atol=1e-6)
    assert first[0] > 0  # sanity
