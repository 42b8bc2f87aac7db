"""The consolidation CLI entry (python -m ..., reference utils.py:28-29
parity) must work as a subprocess over real shard files."""

import os
import subprocess
import sys

import torch

from tests.utils_mp import run_multiprocess
from tests.test_ckpt import _train_save


def test_cli_subprocess(tmp_path):
    ckpt_dir = str(tmp_path)
    run_multiprocess(_train_save, world_size=2, args=(ckpt_dir,))
    out = os.path.join(ckpt_dir, "full.ckpt")
    res = subprocess.run(
        [sys.executable, "-m",
         "vit_10b_fsdp_example_amd.consolidate_sharded_ckpts",
         "--ckpt_prefix", os.path.join(ckpt_dir, "epoch_1_rank_"),
         "--ckpt_suffix", ".ckpt", "--save_path", out],
        capture_output=True, text=True, cwd=os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))),
    )
    assert res.returncode == 0, res.stderr
    full = torch.load(out, map_location="cpu", weights_only=False)
    assert "model" in full
    assert any(k.endswith("attn.qkv.weight") for k in full["model"])


def test_missing_rank_file_fails_loudly(tmp_path):
    """Consolidating with a rank file absent must abort, not silently
    produce a half-assembled model."""
    import pytest

    ckpt_dir = str(tmp_path)
    run_multiprocess(_train_save, world_size=2, args=(ckpt_dir,))
    os.remove(os.path.join(ckpt_dir, "epoch_1_rank_1.ckpt"))

    from vit_10b_fsdp_example_amd.consolidate_sharded_ckpts import (
        consolidate_files,
    )

    with pytest.raises(AssertionError, match="rank files"):
        consolidate_files(os.path.join(ckpt_dir, "epoch_1_rank_"), ".ckpt",
                          os.path.join(ckpt_dir, "full.ckpt"))
    assert not os.path.exists(os.path.join(ckpt_dir, "full.ckpt"))
