from ._extension import ext, has_ext, use_hip
from .layernorm import LayerNorm, layer_norm, fused_add_layer_norm
from .attention import attention, attention_qkv, math_attention
from .cross_entropy import CrossEntropyLoss, cross_entropy
from .adamw import FusedAdamW
from .multi_tensor import local_sqnorm, scale_
from .linear import (
    NativeLinear, NativeWgradMode, TunedGemmMode, gemm_dispatch_context,
    wgrad_backward_context,
)

__all__ = [
    "ext",
    "has_ext",
    "use_hip",
    "LayerNorm",
    "layer_norm",
    "fused_add_layer_norm",
    "attention",
    "attention_qkv",
    "math_attention",
    "CrossEntropyLoss",
    "cross_entropy",
    "FusedAdamW",
    "local_sqnorm",
    "scale_",
    "NativeLinear",
    "NativeWgradMode",
    "TunedGemmMode",
    "gemm_dispatch_context",
    "wgrad_backward_context",
]
