"""Model structure / numerics on CPU: patch-embed GEMM == conv2d, shapes,
exact parameter counts (incl. the 10B config), attention math path."""

import numpy as np
import torch
import torch.nn.functional as F

from vit_10b_fsdp_example_amd.models import (
    Block, FSDPViTModel, PatchEmbed, count_vit_params,
)
from vit_10b_fsdp_example_amd.ops import math_attention


def test_patch_embed_equals_conv2d():
    """Our GEMM patch-embed must equal timm's Conv2d(k=p, s=p) + flatten
    + transpose formulation exactly."""
    torch.manual_seed(0)
    pe = PatchEmbed(img_size=28, patch_size=7, in_chans=3, embed_dim=32)
    x = torch.randn(2, 3, 28, 28)
    ours = pe(x)
    conv_w = pe.proj.weight.reshape(32, 3, 7, 7)
    ref = F.conv2d(x, conv_w, pe.proj.bias, stride=7)
    ref = ref.flatten(2).transpose(1, 2)
    np.testing.assert_allclose(
        ours.detach().numpy(), ref.detach().numpy(), rtol=1e-5, atol=1e-5
    )
    assert pe.num_patches == 16


def test_param_count_formula():
    torch.manual_seed(0)
    model = FSDPViTModel(
        image_size=32, patch_size=4, embed_dim=64, num_heads=4, num_blocks=3,
        mlp_ratio=4.0, pos_dropout=0.0, mlp_dropout=0.0, att_dropout=0.0,
        num_classes=10, grad_ckpt_wrap=lambda m: m, fsdp_wrap=lambda m: m,
    )
    actual = sum(p.numel() for p in model.parameters())
    formula = count_vit_params(32, 4, 64, 3, 4.0, 10)
    assert actual == formula


def test_10b_param_count():
    """The default config is the 10-billion-parameter ViT
    (reference README.md:3; exact value derived in SURVEY.md §2D)."""
    assert count_vit_params(224, 14, 5120, 32, 4.0, 1000) == 10_077_917_160


def test_forward_shapes():
    torch.manual_seed(0)
    model = FSDPViTModel(
        image_size=32, patch_size=4, embed_dim=64, num_heads=4, num_blocks=2,
        mlp_ratio=4.0, pos_dropout=0.0, mlp_dropout=0.0, att_dropout=0.0,
        num_classes=10, grad_ckpt_wrap=lambda m: m, fsdp_wrap=lambda m: m,
    )
    out = model(torch.randn(3, 3, 32, 32))
    assert out.shape == (3, 10)


def test_math_attention_vs_naive():
    """math_attention equals an explicit per-head loop."""
    torch.manual_seed(0)
    B, H, T, D = 2, 3, 8, 16
    q, k, v = (torch.randn(B, H, T, D) for _ in range(3))
    out = math_attention(q, k, v)
    ref = torch.empty_like(out)
    for b in range(B):
        for h in range(H):
            s = (q[b, h] @ k[b, h].T) * (D ** -0.5)
            ref[b, h] = torch.softmax(s, dim=-1) @ v[b, h]
    np.testing.assert_allclose(out.numpy(), ref.numpy(), rtol=1e-5, atol=1e-6)


def test_block_grad_ckpt_equivalence():
    """checkpoint_module(block) produces identical outputs and gradients
    to the plain block, including with dropout (RNG preserved)."""
    from vit_10b_fsdp_example_amd.parallel import checkpoint_module

    torch.manual_seed(0)
    block = Block(dim=32, num_heads=4, drop=0.2, attn_drop=0.1)
    ck = checkpoint_module(block)
    x = torch.randn(2, 8, 32, requires_grad=True)

    torch.manual_seed(42)
    y1 = block(x)
    g1 = torch.autograd.grad(y1.sum(), [x] + list(block.parameters()))

    torch.manual_seed(42)
    y2 = ck(x)
    g2 = torch.autograd.grad(y2.sum(), [x] + list(block.parameters()))

    np.testing.assert_allclose(
        y1.detach().numpy(), y2.detach().numpy(), rtol=1e-6, atol=1e-7
    )
    for a, b in zip(g1, g2):
        np.testing.assert_allclose(a.numpy(), b.numpy(), rtol=1e-5, atol=1e-6)


def test_block_deferred_residual_equivalence():
    """The deferred-residual pair interface computes exactly the default
    interface's math across a 2-block chain (the pending add is applied
    by the NEXT block / the final consumer)."""
    torch.manual_seed(1)
    blocks = [Block(dim=32, num_heads=4) for _ in range(2)]
    deferred = [Block(dim=32, num_heads=4, deferred_residual=True)
                for _ in range(2)]
    for d, b in zip(deferred, blocks):
        d.load_state_dict(b.state_dict())

    x = torch.randn(2, 8, 32)
    ref = blocks[1](blocks[0](x))

    h, r = deferred[0](x)
    h, r = deferred[1]((h, r))
    out = h + r  # final pending add
    np.testing.assert_allclose(out.detach().numpy(), ref.detach().numpy(),
                               rtol=1e-5, atol=1e-6)
