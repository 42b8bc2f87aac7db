"""FSDP numerics: the sharded engine must reproduce the unsharded
(plain-DDP) training trajectory (N-rank FSDP loss/grad-norm
trajectory == single-rank unsharded trajectory on a tiny ViT, gloo/CPU).
"""

import numpy as np
import pytest
import torch

from tests.utils_mp import run_multiprocess

TINY = [
    "--fake_data", "--image_size", "32", "--patch_size", "4",
    "--embed_dim", "64", "--num_heads", "4", "--num_blocks", "3",
    "--num_classes", "10", "--batch_size", "8", "--num_workers", "0",
]


def _run_trajectory(rank, world_size, extra, steps=4, dropout=False):
    import torch
    from vit_10b_fsdp_example_amd.cli import parse_args
    from vit_10b_fsdp_example_amd import dist as xdist
    from vit_10b_fsdp_example_amd.models import build_fsdp_vit_model
    from vit_10b_fsdp_example_amd.ops import CrossEntropyLoss, FusedAdamW
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    args = TINY + extra
    if dropout:
        args += ["--mlp_dropout", "0.1", "--pos_dropout", "0.1"]
    cfg = parse_args(args)
    device = xdist.init_distributed()
    torch.manual_seed(1234)
    model = build_fsdp_vit_model(cfg, device, compute_dtype=torch.float32)
    loss_fn = CrossEntropyLoss()
    opt = FusedAdamW(model.parameters(), lr=1e-3, weight_decay=0.1)
    losses, gnorms = [], []
    torch.manual_seed(777)
    # deterministic global batch split across ranks: every rank draws the
    # same global batch and takes its slice
    gen = torch.Generator().manual_seed(55)
    for _ in range(steps):
        gx = torch.randn(8, 3, 32, 32, generator=gen)
        gy = torch.randint(0, 10, (8,), generator=gen)
        per = 8 // world_size
        x = gx[rank * per:(rank + 1) * per]
        y = gy[rank * per:(rank + 1) * per]
        out = model(x)
        loss = loss_fn(out, y)
        loss.backward()
        if cfg.run_without_fsdp:
            xdist.reduce_gradients(opt)
            gn = torch.nn.utils.clip_grad_norm_(list(model.parameters()), 1.0)
        else:
            gn = model.clip_grad_norm_(1.0)
        opt.step()
        opt.zero_grad(set_to_none=True)
        # cross-rank mean loss for comparison across world sizes
        mean_loss = xdist.mesh_reduce("loss", loss.item(), sum) / world_size
        losses.append(mean_loss)
        gnorms.append(float(gn))
    return losses, gnorms


def _single_rank_reference(extra=(), steps=4, dropout=False):
    return _run_trajectory(0, 1, list(extra), steps=steps, dropout=dropout)


def test_ws1_fsdp_matches_ddp_modes():
    """All FSDP modes match the unsharded baseline at world_size=1
    (fp32, dropout active so the grad-ckpt RNG path is exercised)."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"], dropout=True)
    for extra in ([], ["--no_grad_ckpt"], ["--no_reshard_after_forward"],
                  ["--shard_on_cpu"], ["--flatten_parameters"],
                  ["--grad_ckpt_blocks", "1"]):
        l, g = _single_rank_reference(extra, dropout=True)
        np.testing.assert_allclose(l, ref_l, rtol=1e-5, atol=1e-6, err_msg=str(extra))
        np.testing.assert_allclose(g, ref_g, rtol=1e-5, atol=1e-6, err_msg=str(extra))


def test_ws2_fsdp_matches_single_rank():
    """2-rank FSDP (gloo) reproduces the single-rank unsharded trajectory:
    same global batch, loss and grad-norm sequence."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory, world_size=2, args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws2_ddp_matches_single_rank():
    """2-rank plain-DDP baseline (reduce_gradients path) reproduces the
    single-rank trajectory."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(
        _run_trajectory, world_size=2, args=(["--run_without_fsdp"],)
    )
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws2_fsdp_grad_ckpt_and_noreshard():
    """ws=2 with grad ckpt off / reshard off still matches."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    for extra in (["--no_grad_ckpt"], ["--no_reshard_after_forward"],
                  ["--shard_on_cpu"]):
        results = run_multiprocess(_run_trajectory, world_size=2, args=(extra,))
        for l, g in results:
            np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
            np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws4_fsdp_matches_single_rank():
    """4-rank FSDP (gloo) reproduces the single-rank trajectory — more
    ranks exercise padding (shard sizes) and collective ordering harder."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory, world_size=4, args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def _run_trajectory_prefetch2(rank, world_size, extra):
    # widen the async-gather window (VITFSDP_PREFETCH_DEPTH=2): the
    # collective ORDER must stay uniform across ranks and numerics
    # unchanged — this is the 8-GPU overlap knob (ROADMAP item 1)
    import vit_10b_fsdp_example_amd.parallel.fsdp as fsdp_mod

    fsdp_mod._PREFETCH_DEPTH = 2
    return _run_trajectory(rank, world_size, extra)


def test_ws2_fsdp_prefetch_depth2():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory_prefetch2, world_size=2,
                               args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def _run_trajectory_p2p_gather(rank, world_size, extra):
    # direct one-shot all-gather (batched isend/irecv into full-buffer
    # slices) instead of the library collective — the xGMI full-mesh
    # algorithm candidate from SURVEY §5 (VITFSDP_AG_ALGO=p2p)
    import vit_10b_fsdp_example_amd.parallel.comm as comm_mod

    comm_mod._AG_ALGO = "p2p"
    return _run_trajectory(rank, world_size, extra)


def test_ws2_fsdp_p2p_allgather():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory_p2p_gather, world_size=2,
                               args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws4_fsdp_p2p_allgather():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory_p2p_gather, world_size=4,
                               args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def _run_trajectory_p2p_both(rank, world_size, extra):
    # both collectives on the one-shot P2P algorithms at once
    import vit_10b_fsdp_example_amd.parallel.comm as comm_mod

    comm_mod._AG_ALGO = "p2p"
    comm_mod._RS_ALGO = "p2p"
    return _run_trajectory(rank, world_size, extra)


def test_ws2_fsdp_p2p_reduce_scatter():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory_p2p_both, world_size=2,
                               args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws4_fsdp_p2p_both_algos():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory_p2p_both, world_size=4,
                               args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws1_fuse_residual_matches_baseline():
    """--fuse_residual (deferred-residual block interface, every add
    fused into the next LN) is numerically identical to the default
    interface — same sums, same rounding points in fp32."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"],
                                          dropout=True)
    for extra in (["--fuse_residual"],
                  ["--fuse_residual", "--no_grad_ckpt"],
                  ["--fuse_residual", "--no_reshard_after_forward"]):
        l, g = _single_rank_reference(extra, dropout=True)
        np.testing.assert_allclose(l, ref_l, rtol=1e-5, atol=1e-6,
                                   err_msg=str(extra))
        np.testing.assert_allclose(g, ref_g, rtol=1e-5, atol=1e-6,
                                   err_msg=str(extra))


def test_ws2_fuse_residual_matches_single_rank():
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory, world_size=2,
                               args=(["--fuse_residual"],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_ws8_fsdp_matches_single_rank():
    """8-rank FSDP (gloo) — the exact rank count of the driver's
    round-end scaling run; exercises the full 8-way collective schedule
    and shard padding (batch 8 -> 1 image/rank)."""
    ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    results = run_multiprocess(_run_trajectory, world_size=8, args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


PAD = [
    "--fake_data", "--image_size", "32", "--patch_size", "4",
    "--embed_dim", "20", "--num_heads", "4", "--num_blocks", "2",
    "--num_classes", "13", "--batch_size", "8", "--num_workers", "0",
]


def _run_trajectory_pad(rank, world_size, extra):
    # root-unit total is 2573 params (odd): at ws=2 the flat shard is
    # PADDED, exercising the zero-tail in gather and reduce-scatter
    global TINY
    saved, TINY[:] = TINY[:], PAD
    try:
        return _run_trajectory(rank, world_size, extra)
    finally:
        TINY[:] = saved


def test_ws2_fsdp_padded_shards():
    saved, TINY[:] = TINY[:], PAD
    try:
        ref_l, ref_g = _single_rank_reference(["--run_without_fsdp"])
    finally:
        TINY[:] = saved
    results = run_multiprocess(_run_trajectory_pad, world_size=2, args=([],))
    for l, g in results:
        np.testing.assert_allclose(l, ref_l, rtol=1e-4, atol=1e-5)
        np.testing.assert_allclose(g, ref_g, rtol=1e-4, atol=1e-5)


def test_determinism_same_seed_same_trajectory():
    """Two identical single-rank runs produce bit-identical loss and
    grad-norm sequences (seeded init, data and dropout)."""
    a_l, a_g = _single_rank_reference([], dropout=True)
    b_l, b_g = _single_rank_reference([], dropout=True)
    assert a_l == b_l and a_g == b_g
