"""Offline consolidation of per-rank FSDP shard checkpoints.

Capability parity with the CLI documented in the reference
(utils.py:28-29: `python3 -m torch_xla.distributed.fsdp.
consolidate_sharded_ckpts`):

    python3 -m vit_10b_fsdp_example_amd.consolidate_sharded_ckpts \
        --ckpt_prefix /tmp/vit_fsdp/epoch_300_rank_ \
        --ckpt_suffix .ckpt \
        --save_path  /tmp/vit_fsdp/consolidated.ckpt

Reads every rank's `{prefix}{rank}{suffix}` file, reassembles each FSDP
unit's flat parameter from its shards using the stored shard_metadata,
strips padding, splits it back into the original parameters with their
original fully-qualified names, and writes a single checkpoint whose
"model" entry loads directly into an *unwrapped* FSDPViTModel.
"""

import argparse

import torch


def consolidate_state_dicts(rank_ckpts):
    """Merge a list of per-rank checkpoint dicts (ordered or unordered;
    metadata carries the rank) into one full model state_dict."""
    assert len(rank_ckpts) > 0
    metas = [c["shard_metadata"] for c in rank_ckpts]
    assert all(m is not None for m in metas), "checkpoints carry no shard_metadata"
    world_size = metas[0]["world_size"]
    assert len(rank_ckpts) == world_size, (
        f"need all {world_size} rank files, got {len(rank_ckpts)}"
    )
    by_rank = [None] * world_size
    for ckpt in rank_ckpts:
        by_rank[ckpt["shard_metadata"]["rank"]] = ckpt
    assert all(c is not None for c in by_rank), "duplicate or missing rank files"

    shard_info = metas[0]["shard_info"]
    full_sd = {}
    for key, info in shard_info.items():
        shards = [by_rank[r]["model"][key] for r in range(world_size)]
        flat = torch.cat([s.reshape(-1) for s in shards])
        assert flat.numel() == info["padded_numel"], (
            f"{key}: flat numel {flat.numel()} != padded {info['padded_numel']}"
        )
        for p in info["params"]:
            full_sd[p["name"]] = (
                flat.narrow(0, p["offset"], p["numel"]).view(p["shape"]).clone()
            )

    # non-sharded entries (buffers) pass through from rank 0, with
    # wrapper name segments stripped
    from .parallel.fsdp import FullyShardedDataParallel as FSDP

    for key, val in by_rank[0]["model"].items():
        if key in shard_info:
            continue
        full_sd[FSDP._clean_name(key)] = val
    return full_sd


def consolidate_files(ckpt_prefix, ckpt_suffix, save_path):
    import glob
    import re

    pattern = f"{ckpt_prefix}*{ckpt_suffix}"
    paths = sorted(glob.glob(pattern))
    assert paths, f"no checkpoints match {pattern}"
    rank_ckpts = []
    for path in paths:
        m = re.match(
            re.escape(ckpt_prefix) + r"(\d+)" + re.escape(ckpt_suffix), path
        )
        if not m:
            continue
        rank_ckpts.append(torch.load(path, map_location="cpu", weights_only=False))
    full_sd = consolidate_state_dicts(rank_ckpts)
    out = {"model": full_sd}
    torch.save(out, save_path)
    print(f"consolidated {len(rank_ckpts)} shard ckpts -> {save_path}", flush=True)
    return out


def main():
    parser = argparse.ArgumentParser(
        description="consolidate per-rank FSDP shard checkpoints into one "
        "full model checkpoint"
    )
    parser.add_argument(
        "--ckpt_prefix", type=str, required=True,
        help="path prefix before the rank number, e.g. /ckpts/epoch_10_rank_",
    )
    parser.add_argument("--ckpt_suffix", type=str, default=".ckpt")
    parser.add_argument(
        "--save_path", type=str, required=True,
        help="output path of the consolidated checkpoint",
    )
    args = parser.parse_args()
    consolidate_files(args.ckpt_prefix, args.ckpt_suffix, args.save_path)


if __name__ == "__main__":
    main()
