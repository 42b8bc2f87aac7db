"""Vision Transformer built from our own modules over MI355X-native ops.

Capability parity with the reference's FSDPViTModel
(run_vit_training.py:99-162) and the timm 0.4.12 modules it pulls in
(PatchEmbed / Block, run_vit_training.py:14-19), re-designed for the
hardware:

  * PatchEmbed is a pure GEMM, not a convolution: patches do not overlap
    (kernel == stride), so the im2col is a free reshape/permute and the
    projection is a [B*T, 3*p*p] x [3*p*p, E] hipBLASLt GEMM — no MIOpen
    conv path.
  * LayerNorm / attention core / loss run in hand-written CDNA4 HIP
    kernels on GPU (see ops/), eager torch on CPU.
  * The FSDP and gradient-checkpoint wrappers are injected callables
    (grad_ckpt_wrap, fsdp_wrap), preserving the reference's one real
    architectural seam (run_vit_training.py:118-119,145,194-199): the
    same model class runs with or without sharding.

Weight init matches the reference *behavior*: timm 0.4.12's
_init_vit_weights is a single-module (non-recursive) function, and the
reference calls it on container modules where it matches no isinstance
branch — a documented no-op.  So blocks keep framework
default init and only pos_embed gets trunc_normal(std=0.02).  Pass
recursive=True to init_vit_weights for the "intended" behavior.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import dist as xdist
from ..ops import (  # noqa: F401
    LayerNorm, NativeLinear, attention_qkv, cross_entropy,
    fused_add_layer_norm,
)


def init_vit_weights(module, recursive=False):
    """timm-0.4.12-compatible init (see module docstring).

    Single-module semantics by default: only has an effect when `module`
    itself is Linear or LayerNorm.  The reference calls this on
    PatchEmbed / Block containers, where it is a no-op —
    we replicate that exactly so loss curves correspond.
    """
    def _apply(m):
        if isinstance(m, nn.Linear):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, (nn.LayerNorm, LayerNorm)):
            nn.init.ones_(m.weight)
            nn.init.zeros_(m.bias)

    if recursive:
        module.apply(_apply)
    else:
        _apply(module)


class PatchEmbed(nn.Module):
    """Non-overlapping patch projection as a single GEMM.

    Equivalent to timm's Conv2d(kernel=patch, stride=patch) + flatten +
    transpose (reference run_vit_training.py:124): the conv weight
    [E, 3, p, p] flattened over (c, ph, pw) equals our Linear weight
    [E, 3*p*p], and the input patch extraction is a reshape/permute.
    """

    def __init__(self, img_size=224, patch_size=14, in_chans=3, embed_dim=768):
        super().__init__()
        assert img_size % patch_size == 0, "image size must divide by patch size"
        self.img_size = img_size
        self.patch_size = patch_size
        self.in_chans = in_chans
        self.grid_size = img_size // patch_size
        self.num_patches = self.grid_size * self.grid_size
        self.proj = nn.Linear(in_chans * patch_size * patch_size, embed_dim)

    def forward(self, x):
        B, C, H, W = x.shape
        p, g = self.patch_size, self.grid_size
        # [B, C, g, p, g, p] -> [B, g, g, C, p, p] -> [B, T, C*p*p]
        x = x.reshape(B, C, g, p, g, p).permute(0, 2, 4, 1, 3, 5)
        x = x.reshape(B, g * g, C * p * p)
        return self.proj(x)


class Attention(nn.Module):
    """Pre-projection multi-head attention matching timm 0.4.12's module
    (fused qkv Linear with bias, per-head scaled dot product, attention
    dropout, output projection + dropout)."""

    def __init__(self, dim, num_heads, qkv_bias=True, attn_drop=0.0, proj_drop=0.0):
        super().__init__()
        assert dim % num_heads == 0
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.scale = self.head_dim ** -0.5
        # NativeLinear: weight gradients route through the csrc wgrad
        # kernel where the shape qualifies (ops/linear.py)
        self.qkv = NativeLinear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop_p = attn_drop
        self.proj = NativeLinear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x):
        B, T, E = x.shape
        # [B, T, 3, H, d] is a free view of the fused projection; the
        # attention core reads q/k/v through strides and returns [B, T, E]
        # directly (no permute/contiguous copies around the kernel)
        qkv = self.qkv(x).reshape(B, T, 3, self.num_heads, self.head_dim)
        o = attention_qkv(
            qkv,
            self.num_heads,
            scale=self.scale,
            dropout_p=self.attn_drop_p,
            training=self.training,
        )
        return self.proj_drop(self.proj(o))


class Mlp(nn.Module):
    """fc1 -> GELU -> dropout -> fc2 -> dropout (timm Mlp)."""

    def __init__(self, dim, hidden_dim, drop=0.0):
        super().__init__()
        self.fc1 = NativeLinear(dim, hidden_dim)
        self.fc2 = NativeLinear(hidden_dim, dim)
        self.drop = nn.Dropout(drop)
        # register the MLP dims so the GEMM dispatch router can fuse the
        # GELU into the fc1/fc2 hipBLASLt epilogues (VITFSDP_FUSED_GELU)
        from ..ops.linear import configure_gelu_fusion

        configure_gelu_fusion(dim, hidden_dim)

    def forward(self, x):
        x = self.drop(F.gelu(self.fc1(x)))
        return self.drop(self.fc2(x))


class Block(nn.Module):
    """Pre-LN transformer block: x + Attn(LN1(x)); x + MLP(LN2(x))
    (timm Block as configured at reference run_vit_training.py:134-141;
    drop_path unused there, so none here).

    Two interfaces, identical math:
      * default: tensor -> tensor, with the attention residual add fused
        into norm2 (one ln_add kernel) and the MLP residual as a plain
        add;
      * deferred residual (Megatron-style, --fuse_residual): input and
        output are (hidden, stream) pairs and the MLP residual add is
        fused into the NEXT block's norm1 — no standalone elementwise
        add kernels remain anywhere in the block stack.  The bf16
        rounding of every sum matches the plain add exactly (ln_add
        rounds the fp32 sum to bf16 the same way torch's add does).
    """

    def __init__(self, dim, num_heads, mlp_ratio=4.0, qkv_bias=True,
                 drop=0.0, attn_drop=0.0, deferred_residual=False):
        super().__init__()
        self.deferred_residual = deferred_residual
        self.norm1 = LayerNorm(dim, eps=1e-6)
        self.attn = Attention(
            dim, num_heads, qkv_bias=qkv_bias, attn_drop=attn_drop, proj_drop=drop
        )
        self.norm2 = LayerNorm(dim, eps=1e-6)
        self.mlp = Mlp(dim, int(dim * mlp_ratio), drop=drop)

    def forward(self, x):
        if self.deferred_residual:
            return self._forward_deferred(x)
        attn_out = self.attn(self.norm1(x))
        # fused residual add + norm2 (one kernel for x+attn_out and its LN)
        x, normed = fused_add_layer_norm(
            x, attn_out, self.norm2.weight, self.norm2.bias, self.norm2.eps
        )
        x = x + self.mlp(normed)
        return x

    def _forward_deferred(self, x):
        if isinstance(x, (tuple, list)):
            hidden, stream = x
            # previous block's MLP residual add, fused into our norm1
            s0, y0 = fused_add_layer_norm(
                stream, hidden, self.norm1.weight, self.norm1.bias,
                self.norm1.eps,
            )
        else:  # first block: nothing pending
            s0, y0 = x, self.norm1(x)
        attn_out = self.attn(y0)
        s1, y1 = fused_add_layer_norm(
            s0, attn_out, self.norm2.weight, self.norm2.bias, self.norm2.eps
        )
        return self.mlp(y1), s1


class FSDPViTModel(nn.Module):
    """ViT with nested FSDP + gradient checkpointing injected per block
    (reference run_vit_training.py:99-162): PatchEmbed + learned
    pos-embed (no CLS token) + N blocks + final LN + mean-pool + linear
    head (mean pooling per arXiv 2106.04560)."""

    def __init__(
        self,
        image_size,
        patch_size,
        embed_dim,
        num_heads,
        num_blocks,
        mlp_ratio,
        pos_dropout,
        mlp_dropout,
        att_dropout,
        num_classes,
        grad_ckpt_wrap,
        fsdp_wrap,
        fuse_residual=False,
    ):
        super().__init__()
        self.fuse_residual = fuse_residual

        self.patch_embed = PatchEmbed(
            img_size=image_size, patch_size=patch_size, in_chans=3,
            embed_dim=embed_dim,
        )
        init_vit_weights(self.patch_embed)
        num_patches = self.patch_embed.num_patches
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches, embed_dim))
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        self.pos_drop = nn.Dropout(pos_dropout)

        blocks = []
        for idx in range(num_blocks):
            block = Block(
                dim=embed_dim,
                num_heads=num_heads,
                mlp_ratio=mlp_ratio,
                qkv_bias=True,
                drop=mlp_dropout,
                attn_drop=att_dropout,
                deferred_residual=fuse_residual,
            )
            # init BEFORE wrapping: FSDP shards the weights at wrap time
            init_vit_weights(block)
            # grad-ckpt wrapper goes INSIDE the FSDP wrapper so the
            # backward recompute reuses the gathered full params
            # (reference run_vit_training.py:143-145)
            block = fsdp_wrap(grad_ckpt_wrap(block))
            blocks.append(block)
            xdist.master_print(f"built ViT block {idx}")
        self.blocks = nn.Sequential(*blocks)

        self.norm = LayerNorm(embed_dim, eps=1e-6)
        init_vit_weights(self.norm)
        self.head = nn.Linear(embed_dim, num_classes)

    def forward(self, image):
        x = self.patch_embed(image) + self.pos_embed
        x = self.pos_drop(x)
        x = self.blocks(x)
        if self.fuse_residual:
            # last block's pending MLP residual add fuses into the
            # final LayerNorm
            hidden, stream = x
            _, normed = fused_add_layer_norm(
                stream, hidden, self.norm.weight, self.norm.bias,
                self.norm.eps,
            )
        else:
            normed = self.norm(x)
        # mean pooling over the patch sequence instead of a [CLS] token
        logits = self.head(torch.mean(normed, dim=1))
        return logits


def count_vit_params(image_size, patch_size, embed_dim, num_blocks, mlp_ratio,
                     num_classes, in_chans=3):
    """Exact parameter count for the architecture (used by tests; the
    10B default config gives 10,077,917,160 — SURVEY.md §2D)."""
    e = embed_dim
    hid = int(e * mlp_ratio)
    t = (image_size // patch_size) ** 2
    per_block = (
        e * 3 * e + 3 * e
        + e * e + e
        + e * hid + hid
        + hid * e + e
        + 4 * e
    )
    patch_embed = (in_chans * patch_size * patch_size) * e + e
    pos_embed = t * e
    final_norm = 2 * e
    head = e * num_classes + num_classes
    return num_blocks * per_block + patch_embed + pos_embed + final_norm + head


def build_fsdp_vit_model(cfg, device, compute_dtype=torch.float32):
    """Create the ViT with nested FSDP and gradient checkpointing
    (reference run_vit_training.py:165-200).

    Wrap policy mirrors the reference: per-block
    fsdp_wrap(grad_ckpt_wrap(block)), then a root fsdp_wrap over the
    whole model (no grad-ckpt at the root).  --run_without_fsdp swaps
    both wrappers for identity/to(device) — the plain-DDP baseline.
    """
    from ..parallel import FullyShardedDataParallel as FSDP
    from ..parallel import checkpoint_module

    def fsdp_wrap(module):
        if cfg.run_without_fsdp:
            return module.to(device)
        return FSDP(
            module if cfg.shard_on_cpu else module.to(device),
            reshard_after_forward=cfg.reshard_after_forward,
            flatten_parameters=cfg.flatten_parameters,
            compute_dtype=compute_dtype,
            device=device,
            shard_on_cpu=cfg.shard_on_cpu,
        )

    # --grad_ckpt_blocks N: checkpoint only the first N blocks (identical
    # gradients; recompute is pure overhead wherever activations fit in
    # the 288 GB of HBM3E).  -1 = every block, the reference behavior.
    ckpt_limit = getattr(cfg, "grad_ckpt_blocks", -1)
    ckpt_count = {"i": 0}

    def grad_ckpt_wrap(module):
        if not cfg.grad_ckpt:
            return module
        idx = ckpt_count["i"]
        ckpt_count["i"] += 1
        if 0 <= ckpt_limit <= idx:
            return module
        return checkpoint_module(module)

    model = FSDPViTModel(
        image_size=cfg.image_size,
        patch_size=cfg.patch_size,
        embed_dim=cfg.embed_dim,
        num_heads=cfg.num_heads,
        num_blocks=cfg.num_blocks,
        mlp_ratio=cfg.mlp_ratio,
        pos_dropout=cfg.pos_dropout,
        mlp_dropout=cfg.mlp_dropout,
        att_dropout=cfg.att_dropout,
        num_classes=cfg.num_classes,
        grad_ckpt_wrap=grad_ckpt_wrap,
        fsdp_wrap=fsdp_wrap,
        fuse_residual=getattr(cfg, "fuse_residual", False),
    )
    # root wrap without grad-ckpt (reference run_vit_training.py:197-199)
    model = fsdp_wrap(model)
    return model
