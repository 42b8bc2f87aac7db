"""End-to-end trainer smoke: the full train() path (dataset build, FSDP
model, optimizer, schedule, logging closures, checkpoint save, eval) on
a tiny fake-data config, single process on CPU."""

import glob
import os

from vit_10b_fsdp_example_amd.cli import parse_args


def test_train_e2e_tiny(tmp_path, capsys):
    from vit_10b_fsdp_example_amd.train import main
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args([
        "--fake_data", "--image_size", "16", "--patch_size", "4",
        "--embed_dim", "32", "--num_heads", "2", "--num_blocks", "2",
        "--num_classes", "10", "--batch_size", "4", "--num_workers", "0",
        "--num_epochs", "1", "--ckpt_epoch_interval", "1",
        "--test_epoch_interval", "1", "--log_step_interval", "1",
        "--warmup_steps", "2", "--max_steps_per_epoch", "3",
        "--ckpt_dir", str(tmp_path),
    ])
    # shrink the fake dataset so the final "epoch" is quick: we cap steps
    # via --max_steps_per_epoch, but eval still runs over the val split —
    # patch lengths down for the test
    import vit_10b_fsdp_example_amd.data.datasets as ds

    orig_train, orig_val = ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN
    ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = 16, 8
    try:
        main(cfg)
    finally:
        ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = orig_train, orig_val

    out = capsys.readouterr().out
    assert "training begins" in out
    assert "loss:" in out and "sec/iter:" in out
    assert "accuracy on val:" in out
    assert "training completed" in out
    ckpts = glob.glob(os.path.join(str(tmp_path), "epoch_1_rank_0.ckpt"))
    assert len(ckpts) == 1


def test_train_resume_epoch(tmp_path, capsys):
    """--resume_epoch N: train() loads epoch_N_rank_r.ckpt and continues
    at epoch N+1 (reference resume semantics, run_vit_training.py)."""
    from vit_10b_fsdp_example_amd.train import main
    from vit_10b_fsdp_example_amd.parallel import CommContext

    base = [
        "--fake_data", "--image_size", "16", "--patch_size", "4",
        "--embed_dim", "32", "--num_heads", "2", "--num_blocks", "2",
        "--num_classes", "10", "--batch_size", "4", "--num_workers", "0",
        "--ckpt_epoch_interval", "1", "--test_epoch_interval", "99",
        "--log_step_interval", "1", "--warmup_steps", "2",
        "--max_steps_per_epoch", "2", "--ckpt_dir", str(tmp_path),
    ]
    import vit_10b_fsdp_example_amd.data.datasets as ds

    orig_train, orig_val = ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN
    ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = 8, 4
    try:
        CommContext.reset()
        main(parse_args(base + ["--num_epochs", "1"]))
        assert os.path.exists(os.path.join(str(tmp_path), "epoch_1_rank_0.ckpt"))
        capsys.readouterr()

        CommContext.reset()
        main(parse_args(base + ["--num_epochs", "2", "--resume_epoch", "1"]))
    finally:
        ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = orig_train, orig_val

    out = capsys.readouterr().out
    assert "starting epoch 2" in out
    assert "starting epoch 1" not in out  # resumed, not restarted
    assert os.path.exists(os.path.join(str(tmp_path), "epoch_2_rank_0.ckpt"))


def test_cli_entry_subprocess(tmp_path):
    """The run_vit_training.py entry (parse -> spawn -> train) as a real
    subprocess, inline single-process path."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "run_vit_training.py",
         "--fake_data", "--image_size", "16", "--patch_size", "4",
         "--embed_dim", "32", "--num_heads", "2", "--num_blocks", "1",
         "--num_classes", "10", "--batch_size", "4", "--num_workers", "0",
         "--num_epochs", "1", "--max_steps_per_epoch", "2",
         "--warmup_steps", "1", "--test_epoch_interval", "99",
         "--ckpt_dir", str(tmp_path)],
        capture_output=True, text=True, cwd=repo, timeout=600,
        env={**os.environ, "VITFSDP_FAKE_LEN": "8"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    assert "training completed" in res.stdout
    assert os.path.exists(os.path.join(str(tmp_path), "epoch_1_rank_0.ckpt"))


def test_train_e2e_fuse_residual(tmp_path, capsys):
    """Full train() + eval + checkpoint under the deferred-residual
    block interface (--fuse_residual)."""
    from vit_10b_fsdp_example_amd.train import main
    from vit_10b_fsdp_example_amd.parallel import CommContext

    CommContext.reset()
    cfg = parse_args([
        "--fake_data", "--fuse_residual", "--image_size", "16",
        "--patch_size", "4", "--embed_dim", "32", "--num_heads", "2",
        "--num_blocks", "2", "--num_classes", "10", "--batch_size", "4",
        "--num_workers", "0", "--num_epochs", "1",
        "--ckpt_epoch_interval", "1", "--test_epoch_interval", "1",
        "--log_step_interval", "1", "--warmup_steps", "2",
        "--max_steps_per_epoch", "3", "--ckpt_dir", str(tmp_path),
    ])
    import vit_10b_fsdp_example_amd.data.datasets as ds

    orig_train, orig_val = ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN
    ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = 16, 8
    try:
        main(cfg)
    finally:
        ds.IMAGENET_TRAIN_LEN, ds.IMAGENET_VAL_LEN = orig_train, orig_val

    out = capsys.readouterr().out
    assert "accuracy on val:" in out and "training completed" in out
    assert os.path.exists(os.path.join(str(tmp_path), "epoch_1_rank_0.ckpt"))


def test_trainer_torchrun_ws2(tmp_path):
    """run_vit_training.py under real torchrun at ws=2 (gloo): epoch
    loop + async logger mesh_reduces + per-rank checkpoints + eval,
    across two actual processes."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    import socket

    with socket.socket() as sock:
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         "run_vit_training.py",
         "--fake_data", "--image_size", "16", "--patch_size", "4",
         "--embed_dim", "32", "--num_heads", "2", "--num_blocks", "1",
         "--num_classes", "10", "--batch_size", "4", "--num_workers", "0",
         "--num_epochs", "1", "--max_steps_per_epoch", "2",
         "--warmup_steps", "1", "--log_step_interval", "1",
         "--test_epoch_interval", "1", "--ckpt_dir", str(tmp_path)],
        capture_output=True, text=True, cwd=repo, timeout=900,
        env={**os.environ, "VITFSDP_FAKE_LEN": "8"},
    )
    assert res.returncode == 0, res.stderr[-3000:]
    assert "training completed" in res.stdout
    assert "accuracy on val:" in res.stdout
    for r in (0, 1):
        assert os.path.exists(
            os.path.join(str(tmp_path), f"epoch_1_rank_{r}.ckpt")
        ), r


def test_trainer_with_wgrad_dispatch_mode(tmp_path):
    """VITFSDP_NATIVE_WGRAD=2 in the full trainer: the backward-scoped
    NativeWgradMode must be harmless end-to-end (on CPU it intercepts
    nothing — the cuda gate — but the mode still wraps every backward
    through checkpoint recompute and the FSDP hooks)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "run_vit_training.py",
         "--fake_data", "--image_size", "16", "--patch_size", "4",
         "--embed_dim", "32", "--num_heads", "2", "--num_blocks", "1",
         "--num_classes", "10", "--batch_size", "4", "--num_workers", "0",
         "--num_epochs", "1", "--max_steps_per_epoch", "2",
         "--warmup_steps", "1", "--test_epoch_interval", "99",
         "--ckpt_dir", str(tmp_path)],
        capture_output=True, text=True, cwd=repo, timeout=600,
        env={**os.environ, "VITFSDP_FAKE_LEN": "8",
             "VITFSDP_NATIVE_WGRAD": "2"},
    )
    assert res.returncode == 0, res.stderr[-2000:]
    assert "training completed" in res.stdout
