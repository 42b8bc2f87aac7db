"""Multi-head attention core (kernel K4 in SURVEY.md §2D).

The reference gets its attention from the timm Block: per-head
softmax(Q K^T * d^-0.5) V with optional attention dropout
(run_vit_training.py:134-141).  For the 10B config head_dim = 5120/32 =
160 — off the usual 64/128 flash fast paths — so the GPU path is our
own CDNA4 flash-style kernel (MFMA 16x16x32 bf16 tiles, LDS-staged K/V,
online softmax, O(T) memory; backward recomputes the forward tiles).

No Triton, no aotriton SDPA: the CPU/no-ext fallback is an explicit
math composition (matmul + softmax), which is also the numerics
reference for the kernel tests.
"""

import warnings

import torch

from ._extension import ext, use_hip

_DROPOUT_WARNED = False


def _warn_dropout_fallback():
    """--att_dropout > 0 routes to the O(T^2) math composition (the
    flash kernel has no in-kernel RNG yet).  The 10B training recipe
    uses att_dropout 0.0 (reference run_vit_training.py:346 default),
    so this is off the measured path — but it must never be a silent
    30x attention slowdown + O(T^2) memory change."""
    global _DROPOUT_WARNED
    if not _DROPOUT_WARNED:
        _DROPOUT_WARNED = True
        warnings.warn(
            "attention dropout > 0: falling back from the flash kernel "
            "to the explicit-math attention path (O(T^2) memory, "
            "slower). The reference recipe uses --att_dropout 0.",
            stacklevel=3,
        )


def math_attention(q, k, v, scale=None, dropout_p=0.0, training=False):
    """Explicit-math reference path: q,k,v [B, H, T, D]."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    scores = torch.matmul(q, k.transpose(-2, -1)) * scale
    # softmax in fp32 for bf16 inputs (matches the kernel's fp32 online
    # softmax accumulation)
    probs = torch.softmax(scores.float(), dim=-1).to(q.dtype)
    if dropout_p > 0.0 and training:
        probs = torch.nn.functional.dropout(probs, p=dropout_p)
    return torch.matmul(probs, v)


class _FlashAttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        o, lse = ext().fmha_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = ext().fmha_bwd(do.contiguous(), q, k, v, o, lse, ctx.scale)
        return dq, dk, dv, None


def attention(q, k, v, scale=None, dropout_p=0.0, training=False):
    """Attention core on [B, H, T, D] tensors.

    GPU: our flash-style HIP kernel (dropout_p must be 0 there for now —
    the 10B recipe uses att_dropout 0.0; nonzero dropout falls back to
    the math path with a warning-free explicit mask).
    CPU: math composition.
    """
    if scale is None:
        scale = q.shape[-1] ** -0.5
    use_kernel = (
        dropout_p == 0.0
        and q.dtype in (torch.bfloat16, torch.float16)
        and use_hip(q, k, v)
        and hasattr(ext(), "fmha_fwd")
    )
    if use_kernel:
        return _FlashAttentionFn.apply(
            q.contiguous(), k.contiguous(), v.contiguous(), scale
        )
    if dropout_p > 0.0 and training and use_hip(q):
        _warn_dropout_fallback()
    return math_attention(q, k, v, scale, dropout_p, training)


class _FlashAttentionQkvFn(torch.autograd.Function):
    """Zero-copy attention on the fused qkv projection [B,T,3,H,D]:
    the kernels read q/k/v through strides and write O (and dqkv) in the
    [B,T,E] layout the surrounding Linears use — no permute/contiguous
    copies on either side of the attention core."""

    @staticmethod
    def forward(ctx, qkv, num_heads, scale):
        o, lse = ext().fmha_fwd_qkv(qkv, num_heads, scale)
        ctx.save_for_backward(qkv, o, lse)
        ctx.num_heads = num_heads
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        dqkv = ext().fmha_bwd_qkv(
            do.contiguous(), qkv, o, lse, ctx.num_heads, ctx.scale
        )
        return dqkv, None, None


def attention_qkv(qkv, num_heads, scale=None, dropout_p=0.0, training=False):
    """Attention on the fused qkv projection.

    qkv: [B, T, 3, H, D] (a free reshape view of the qkv Linear output);
    returns [B, T, H*D].  GPU bf16 uses the strided flash kernels; the
    fallback path permutes to [B, H, T, D] and runs the math composition.
    """
    B, T, three, H, D = qkv.shape
    assert three == 3 and H == num_heads
    if scale is None:
        scale = D ** -0.5
    use_kernel = (
        dropout_p == 0.0
        and qkv.dtype in (torch.bfloat16,)
        and use_hip(qkv)
        and hasattr(ext(), "fmha_fwd_qkv")
    )
    if use_kernel:
        return _FlashAttentionQkvFn.apply(qkv.contiguous(), num_heads, scale)
    if dropout_p > 0.0 and training and use_hip(qkv):
        _warn_dropout_fallback()
    q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
    o = math_attention(q, k, v, scale, dropout_p, training)
    return o.transpose(1, 2).reshape(B, T, H * D)
