"""CLI parity: the reference's 29 flags must be accepted with identical
defaults (reference run_vit_training.py:327-363)."""

from vit_10b_fsdp_example_amd.cli import parse_args

REFERENCE_DEFAULTS = {
    "data_dir": "/datasets/imagenet-1k",
    "fake_data": False,
    "num_workers": 4,
    "ckpt_dir": "/tmp/vit_fsdp",
    "resume_epoch": 0,
    "ckpt_epoch_interval": 10,
    "test_epoch_interval": 10,
    "log_step_interval": 20,
    "image_size": 224,
    "patch_size": 14,
    "embed_dim": 5120,
    "num_heads": 32,
    "num_blocks": 32,
    "mlp_ratio": 4.0,
    "pos_dropout": 0.0,
    "att_dropout": 0.0,
    "mlp_dropout": 0.0,
    "num_classes": 1000,
    "batch_size": 1024,
    "num_epochs": 300,
    "lr": 1e-3,
    "weight_decay": 0.1,
    "clip_grad_norm": 1.0,
    "warmup_steps": 10000,
    "grad_ckpt": True,
    "reshard_after_forward": True,
    "flatten_parameters": False,
    "run_without_fsdp": False,
    "shard_on_cpu": False,
}


def test_defaults_match_reference():
    cfg = vars(parse_args([]))
    for key, val in REFERENCE_DEFAULTS.items():
        assert key in cfg, f"missing reference flag dest: {key}"
        assert cfg[key] == val, f"{key}: {cfg[key]} != reference default {val}"


def test_store_false_flags():
    cfg = parse_args(["--no_grad_ckpt", "--no_reshard_after_forward"])
    assert cfg.grad_ckpt is False
    assert cfg.reshard_after_forward is False


def test_store_true_flags():
    cfg = parse_args(
        ["--fake_data", "--flatten_parameters", "--run_without_fsdp",
         "--shard_on_cpu"]
    )
    assert cfg.fake_data and cfg.flatten_parameters
    assert cfg.run_without_fsdp and cfg.shard_on_cpu


def test_value_flags_parse():
    cfg = parse_args(
        ["--embed_dim", "1024", "--num_heads", "16", "--num_blocks", "24",
         "--batch_size", "256", "--lr", "3e-4"]
    )
    assert cfg.embed_dim == 1024 and cfg.num_heads == 16
    assert cfg.num_blocks == 24 and cfg.batch_size == 256
    assert abs(cfg.lr - 3e-4) < 1e-12
