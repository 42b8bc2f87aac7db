from .vit import (
    PatchEmbed,
    Attention,
    Mlp,
    Block,
    FSDPViTModel,
    build_fsdp_vit_model,
    init_vit_weights,
    count_vit_params,
)

__all__ = [
    "PatchEmbed",
    "Attention",
    "Mlp",
    "Block",
    "FSDPViTModel",
    "build_fsdp_vit_model",
    "init_vit_weights",
    "count_vit_params",
]
