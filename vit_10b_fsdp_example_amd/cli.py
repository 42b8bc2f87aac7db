"""CLI definition — exact flag/default parity with the reference
(run_vit_training.py:327-363: 29 flags, defaults = ViT-10B).  New,
framework-specific flags are added at the end and never repurpose
reference flags.
"""

import argparse


def build_arg_parser():
    parser = argparse.ArgumentParser()
    # data / io (reference run_vit_training.py:329-336)
    parser.add_argument("--data_dir", type=str, default="/datasets/imagenet-1k")
    parser.add_argument("--fake_data", action="store_true", dest="fake_data")
    parser.add_argument("--num_workers", type=int, default=4)
    parser.add_argument("--ckpt_dir", type=str, default="/tmp/vit_fsdp")
    parser.add_argument("--resume_epoch", type=int, default=0)
    parser.add_argument("--ckpt_epoch_interval", type=int, default=10)
    parser.add_argument("--test_epoch_interval", type=int, default=10)
    parser.add_argument("--log_step_interval", type=int, default=20)

    # model — defaults are the 10-billion-parameter ViT
    # (reference run_vit_training.py:339-348)
    parser.add_argument("--image_size", type=int, default=224)
    parser.add_argument("--patch_size", type=int, default=14)
    parser.add_argument("--embed_dim", type=int, default=5120)
    parser.add_argument("--num_heads", type=int, default=32)
    parser.add_argument("--num_blocks", type=int, default=32)
    parser.add_argument("--mlp_ratio", type=float, default=4.0)
    parser.add_argument("--pos_dropout", type=float, default=0.0)
    parser.add_argument("--att_dropout", type=float, default=0.0)
    parser.add_argument("--mlp_dropout", type=float, default=0.0)
    parser.add_argument("--num_classes", type=int, default=1000)

    # optimization + FSDP (reference run_vit_training.py:351-361)
    parser.add_argument("--batch_size", type=int, default=1024)
    parser.add_argument("--num_epochs", type=int, default=300)
    parser.add_argument("--lr", type=float, default=1e-3)
    parser.add_argument("--weight_decay", type=float, default=0.1)
    parser.add_argument("--clip_grad_norm", type=float, default=1.0)
    parser.add_argument("--warmup_steps", type=int, default=10000)
    parser.add_argument("--no_grad_ckpt", action="store_false", dest="grad_ckpt")
    parser.add_argument(
        "--no_reshard_after_forward", action="store_false",
        dest="reshard_after_forward",
    )
    parser.add_argument(
        "--flatten_parameters", action="store_true", dest="flatten_parameters"
    )
    parser.add_argument(
        "--run_without_fsdp", action="store_true", dest="run_without_fsdp"
    )
    parser.add_argument("--shard_on_cpu", action="store_true", dest="shard_on_cpu")

    # ---- framework-specific additions (not in the reference) ----
    parser.add_argument(
        "--dtype", type=str, default="auto", choices=["auto", "bf16", "fp32"],
        help="FSDP compute dtype: bf16 gather/compute with fp32 master "
        "shards, or full fp32 (reference numerics). auto = bf16 on GPU, "
        "fp32 on CPU.",
    )
    parser.add_argument(
        "--max_steps_per_epoch", type=int, default=0,
        help="stop each epoch after N steps (0 = full epoch); for smoke "
        "tests and benchmarking",
    )
    parser.add_argument(
        "--grad_ckpt_blocks", type=int, default=-1,
        help="number of transformer blocks to wrap with activation "
        "checkpointing (-1 = all when --grad_ckpt is on). With 288 GB of "
        "HBM3E many configurations fit without recomputing every block; "
        "the gradients are identical either way, only recompute work "
        "changes.",
    )
    parser.add_argument(
        "--fuse_residual", action="store_true", dest="fuse_residual",
        help="deferred-residual block interface: blocks exchange "
        "(hidden, stream) pairs and every residual add is fused into "
        "the following LayerNorm kernel (identical math/rounding; no "
        "standalone elementwise add kernels remain)",
    )
    parser.add_argument(
        "--profile", action="store_true", dest="profile",
        help="profile a few early steps with torch.profiler (CPU+GPU "
        "kernel timeline) and write a chrome trace + a top-kernel table "
        "to --ckpt_dir",
    )
    return parser


def parse_args(argv=None):
    return build_arg_parser().parse_args(argv)
