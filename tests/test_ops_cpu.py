"""CPU fallback paths of the op layer: these are the numerics oracles
the GPU kernels are tested against, so they must themselves match plain
torch compositions."""

import numpy as np
import torch

from vit_10b_fsdp_example_amd.ops import (
    CrossEntropyLoss, LayerNorm, NativeLinear, attention_qkv,
    fused_add_layer_norm, layer_norm, local_sqnorm, scale_,
)


def test_layer_norm_cpu_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(4, 8, 32)
    ln = LayerNorm(32, eps=1e-6)
    ref = torch.nn.functional.layer_norm(x, (32,), ln.weight, ln.bias, 1e-6)
    np.testing.assert_allclose(ln(x).detach(), ref.detach(), rtol=1e-6)


def test_fused_add_ln_cpu():
    torch.manual_seed(1)
    x, r = torch.randn(2, 4, 16), torch.randn(2, 4, 16)
    w, b = torch.randn(16), torch.randn(16)
    s, y = fused_add_layer_norm(x, r, w, b, 1e-6)
    np.testing.assert_allclose(s, x + r, rtol=1e-6)
    ref = torch.nn.functional.layer_norm(x + r, (16,), w, b, 1e-6)
    np.testing.assert_allclose(y.detach(), ref.detach(), rtol=1e-5, atol=1e-6)


def test_cross_entropy_cpu():
    torch.manual_seed(2)
    logits = torch.randn(8, 10)
    target = torch.randint(0, 10, (8,))
    ours = CrossEntropyLoss()(logits, target)
    ref = torch.nn.functional.cross_entropy(logits, target)
    assert abs(float(ours) - float(ref)) < 1e-6


def test_attention_qkv_cpu_math():
    torch.manual_seed(3)
    B, T, H, D = 2, 8, 2, 16
    qkv = torch.randn(B, T, 3, H, D)
    out = attention_qkv(qkv, H)
    q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
    s = (q @ k.transpose(-2, -1)) * (D ** -0.5)
    ref = (torch.softmax(s, dim=-1) @ v).transpose(1, 2).reshape(B, T, H * D)
    np.testing.assert_allclose(out.detach(), ref.detach(), rtol=1e-5,
                               atol=1e-6)


def test_native_linear_cpu_is_stock():
    torch.manual_seed(4)
    lin = NativeLinear(8, 16)
    x = torch.randn(3, 8, requires_grad=True)
    y = lin(x)
    y.sum().backward()
    ref = torch.nn.Linear(8, 16)
    with torch.no_grad():
        ref.weight.copy_(lin.weight)
        ref.bias.copy_(lin.bias)
    xr = x.detach().clone().requires_grad_(True)
    ref(xr).sum().backward()
    np.testing.assert_allclose(lin.weight.grad, ref.weight.grad, rtol=1e-6)
    np.testing.assert_allclose(x.grad, xr.grad, rtol=1e-6)


def test_sqnorm_scale_cpu():
    ts = [torch.randn(7), torch.randn(13)]
    ref = sum(float(t.pow(2).sum()) for t in ts)
    assert abs(float(local_sqnorm(ts)) - ref) < 1e-5
    before = [t.clone() for t in ts]
    scale_(ts, 0.25)
    for t, b in zip(ts, before):
        np.testing.assert_allclose(t, b * 0.25, rtol=1e-6)
