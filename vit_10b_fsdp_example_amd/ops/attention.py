"""Multi-head attention core (kernel K4 in SURVEY.md §2D).

The reference gets its attention from the timm Block: per-head
softmax(Q K^T * d^-0.5) V with optional attention dropout
(run_vit_training.py:134-141).  For the 10B config head_dim = 5120/32 =
160 — off the usual 64/128 flash fast paths — so the GPU path is our
own CDNA4 flash-style kernel (MFMA 16x16x32 bf16 tiles, LDS-staged K/V,
online softmax, O(T) memory; backward recomputes the forward tiles).

Attention dropout runs IN-KERNEL (round 2): the mask is a stateless
integer hash of (seed, global q row, k column) applied to the
post-softmax P on the PV path, regenerated identically by the backward
kernels — no mask storage, O(T) memory preserved
(csrc/fmha.hip dropout_hash; the seed is drawn from torch's CPU RNG so
torch.manual_seed reproduces runs).

No Triton, no aotriton SDPA: the CPU/no-ext fallback is an explicit
math composition (matmul + softmax), which is also the numerics
reference for the kernel tests.
"""

import torch

from ._extension import ext, use_hip


def dropout_mask_reference(seed, bh_q, k, p):
    """Bit-exact python reproduction of csrc/fmha.hip dropout_hash for
    the GPU tests: returns the keep mask (bool) for global row indices
    ``bh_q`` [rows] x key columns ``k`` [cols] at probability p."""
    M = 0xFFFFFFFF
    qg = bh_q.to(torch.int64).reshape(-1, 1)
    kg = k.to(torch.int64).reshape(1, -1)
    x = (int(seed) ^ ((qg * 0x9E3779B9) & M) ^ ((kg * 0x85EBCA6B) & M)) & M
    x = x ^ (x >> 16)
    x = (x * 0x7FEB352D) & M
    x = x ^ (x >> 15)
    x = (x * 0x846CA68B) & M
    x = x ^ (x >> 16)
    thresh = min(int(p * 4294967296.0), 4294967295)
    return x >= thresh


def _draw_seed():
    # CPU RNG so torch.manual_seed makes dropout reproducible, without
    # a device sync
    return int(torch.randint(0, 2**31 - 1, (1,)).item())


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
    def forward(ctx, q, k, v, scale, p_drop, seed):
        o, lse = ext().fmha_fwd(q, k, v, scale, p_drop, seed)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        ctx.p_drop = p_drop
        ctx.seed = seed
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = ext().fmha_bwd(
            do.contiguous(), q, k, v, o, lse, ctx.scale, ctx.p_drop, ctx.seed
        )
        return dq, dk, dv, None, None, None


def attention(q, k, v, scale=None, dropout_p=0.0, training=False):
    """Attention core on [B, H, T, D] tensors.

    GPU: our flash-style HIP kernel, attention dropout in-kernel.
    CPU: math composition.
    """
    if scale is None:
        scale = q.shape[-1] ** -0.5
    use_kernel = (
        q.dtype in (torch.bfloat16, torch.float16)
        and use_hip(q, k, v)
        and hasattr(ext(), "fmha_fwd")
    )
    if use_kernel:
        p = dropout_p if training else 0.0
        seed = _draw_seed() if p > 0.0 else 0
        return _FlashAttentionFn.apply(
            q.contiguous(), k.contiguous(), v.contiguous(), scale, p, seed
        )
    return math_attention(q, k, v, scale, dropout_p, training)


class _FlashAttentionQkvFn(torch.autograd.Function):
    """Zero-copy attention on the fused qkv projection [B,T,3,H,D]:
    the kernels read q/k/v through strides and write O (and dqkv) in the
    [B,T,E] layout the surrounding Linears use — no permute/contiguous
    copies on either side of the attention core."""

    @staticmethod
    def forward(ctx, qkv, num_heads, scale, p_drop, seed):
        o, lse = ext().fmha_fwd_qkv(qkv, num_heads, scale, p_drop, seed)
        ctx.save_for_backward(qkv, o, lse)
        ctx.num_heads = num_heads
        ctx.scale = scale
        ctx.p_drop = p_drop
        ctx.seed = seed
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        dqkv = ext().fmha_bwd_qkv(
            do.contiguous(), qkv, o, lse, ctx.num_heads, ctx.scale,
            ctx.p_drop, ctx.seed,
        )
        return dqkv, None, None, None, None


def attention_qkv(qkv, num_heads, scale=None, dropout_p=0.0, training=False):
    """Attention on the fused qkv projection.

    qkv: [B, T, 3, H, D] (a free reshape view of the qkv Linear output);
    returns [B, T, H*D].  GPU bf16 uses the strided flash kernels with
    in-kernel attention dropout; the fallback path permutes to
    [B, H, T, D] and runs the math composition.
    """
    B, T, three, H, D = qkv.shape
    assert three == 3 and H == num_heads
    if scale is None:
        scale = D ** -0.5
    use_kernel = (
        qkv.dtype in (torch.bfloat16,)
        and use_hip(qkv)
        and hasattr(ext(), "fmha_fwd_qkv")
    )
    if use_kernel:
        p = dropout_p if training else 0.0
        seed = _draw_seed() if p > 0.0 else 0
        return _FlashAttentionQkvFn.apply(
            qkv.contiguous(), num_heads, scale, p, seed
        )
    q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
    o = math_attention(q, k, v, scale, dropout_p, training)
    return o.transpose(1, 2).reshape(B, T, H * D)
