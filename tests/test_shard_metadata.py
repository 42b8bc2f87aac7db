"""Property test of the FSDP flat-shard layout + metadata
(get_shard_metadata is the contract consolidation relies on): for
arbitrary module shapes, the recorded (name, shape, numel, offset)
entries must exactly tile the flat parameter and reconstruct every
original tensor."""

import torch
import torch.nn as nn
from hypothesis import given, settings
from hypothesis import strategies as st

from vit_10b_fsdp_example_amd.parallel import (
    CommContext, FullyShardedDataParallel as FSDP,
)


@settings(max_examples=15, deadline=None)
@given(
    dims=st.lists(st.integers(min_value=1, max_value=9), min_size=1,
                  max_size=4),
    bias=st.booleans(),
)
def test_layout_tiles_and_reconstructs(dims, bias):
    CommContext.reset()
    torch.manual_seed(0)
    layers, prev = [], 3
    for d in dims:
        layers.append(nn.Linear(prev, d, bias=bias))
        prev = d
    model = nn.Sequential(*layers)
    originals = {
        name: p.detach().clone() for name, p in model.named_parameters()
    }

    fsdp = FSDP(model, compute_dtype=torch.float32)
    meta = fsdp.get_shard_metadata()
    assert meta["world_size"] == 1 and meta["rank"] == 0
    info = meta["shard_info"]["flat_param"]

    flat = fsdp.state_dict()["flat_param"]
    assert flat.numel() == info["padded_numel"] >= info["total_numel"]

    covered = 0
    end_prev = 0
    for entry in info["params"]:
        assert entry["offset"] == end_prev  # contiguous, ordered tiling
        end_prev = entry["offset"] + entry["numel"]
        covered += entry["numel"]
        got = flat.narrow(0, entry["offset"], entry["numel"]).view(
            entry["shape"]
        )
        torch.testing.assert_close(got, originals[entry["name"]])
    assert covered == info["total_numel"]
    assert set(e["name"] for e in info["params"]) == set(originals)


def test_fsdp_rejects_direct_double_wrap():
    import pytest

    CommContext.reset()
    inner = FSDP(nn.Linear(4, 4), compute_dtype=torch.float32)
    with pytest.raises(ValueError, match="do not wrap"):
        FSDP(inner, compute_dtype=torch.float32)


def test_fsdp_rejects_paramless_module():
    import pytest

    CommContext.reset()
    with pytest.raises(AssertionError, match="no parameters"):
        FSDP(nn.ReLU(), compute_dtype=torch.float32)


def test_nested_wrap_prunes_inner_units():
    """A parent FSDP unit must NOT re-shard parameters already owned by
    a nested FSDP child (the _collect walk prunes them)."""
    CommContext.reset()
    child = FSDP(nn.Linear(4, 4), compute_dtype=torch.float32)
    parent_mod = nn.Sequential(child, nn.Linear(4, 2))
    parent = FSDP(parent_mod, compute_dtype=torch.float32)
    # parent's own flat param covers only the outer Linear(4,2): 4*2+2
    assert parent._total_numel == 10
    assert child._total_numel == 20
