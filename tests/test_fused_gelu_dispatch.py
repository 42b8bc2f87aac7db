"""Fused-MLP-GELU dispatch routing (ops/linear.py TunedGemmMode): CPU
harness with a mocked extension implementing csrc/fgemm.hip's
fwd_gemm_gelu semantics (exact-erf GELU + bias, returns the
pre-activation) in plain torch, covering the caching / early-stop logic
without a GPU.  Because the fusion is erf-exact, fused results must
match the STOCK eager MLP bit-for-bit up to GEMM rounding — no tanh
tolerance involved.  GPU counterpart: tests/test_gpu_kernels.py.
"""

import pytest
import torch
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from vit_10b_fsdp_example_amd.ops import linear as linmod

TOK, D, HID = 8, 4, 16


@pytest.fixture()
def fused_env(monkeypatch):
    calls = {"gelu": 0}

    def handler(a, b, idx, bias):
        assert idx == "gelu"
        pre = a @ b + bias
        calls["gelu"] += 1
        return F.gelu(pre), pre

    monkeypatch.setenv("VITFSDP_FUSED_GELU", "1")
    monkeypatch.setenv("VITFSDP_FGEMM_MIN_TILES", "0")
    monkeypatch.setattr(linmod, "_FGEMM_MIN_TILES", 0)
    monkeypatch.setitem(linmod._GELU_CFG, "d", D)
    monkeypatch.setitem(linmod._GELU_CFG, "hid", HID)
    return calls, handler


def _mlp_params(seed=0):
    g = torch.Generator().manual_seed(seed)
    w1 = torch.randn(HID, D, generator=g, requires_grad=True)
    b1 = torch.randn(HID, generator=g, requires_grad=True)
    w2 = torch.randn(D, HID, generator=g, requires_grad=True)
    b2 = torch.randn(D, generator=g, requires_grad=True)
    return w1, b1, w2, b2


def _mode(handler):
    return linmod.TunedGemmMode(table={}, native_wgrad=False, handler=handler)


def test_fc1_shape_gate(monkeypatch):
    """Without the test seam, _is_fc1_fwd enforces the fgemm kernel's
    shape/grid constraints (M%256, N%256, K%64, min tiles)."""
    monkeypatch.setitem(linmod._GELU_CFG, "d", 128)
    monkeypatch.setitem(linmod._GELU_CFG, "hid", 512)
    monkeypatch.setattr(linmod, "_FGEMM_MIN_TILES", 2)
    m = linmod.TunedGemmMode(table={}, native_wgrad=False)
    assert m._is_fc1_fwd(("T", "N", 512, 256, 128))
    assert not m._is_fc1_fwd(("T", "N", 512, 8, 128))  # tok % 256 != 0
    assert not m._is_fc1_fwd(("N", "N", 512, 256, 128))  # wrong ops
    assert not m._is_fc1_fwd(("T", "N", 512, 256, 100))  # K % 64 != 0
    monkeypatch.setattr(linmod, "_FGEMM_MIN_TILES", 256)
    assert not m._is_fc1_fwd(("T", "N", 512, 256, 128))  # grid too small


def test_fused_mlp_matches_stock(fused_env):
    calls, handler = fused_env
    w1, b1, w2, b2 = _mlp_params()
    x = torch.randn(TOK, D, requires_grad=True)

    with _mode(handler) as m:
        yf = F.linear(F.gelu(F.linear(x, w1, b1)), w2, b2)
        yf.pow(2).sum().backward()
    assert m.gelu_hits == 1
    assert calls["gelu"] == 1

    xr = x.detach().clone().requires_grad_(True)
    w1r, b1r = (t.detach().clone().requires_grad_(True) for t in (w1, b1))
    w2r, b2r = (t.detach().clone().requires_grad_(True) for t in (w2, b2))
    yr = F.linear(F.gelu(F.linear(xr, w1r, b1r)), w2r, b2r)
    yr.pow(2).sum().backward()

    assert torch.allclose(yf, yr, atol=1e-6)
    for got, ref in [(x, xr), (w1, w1r), (b1, b1r), (w2, w2r), (b2, b2r)]:
        assert torch.allclose(got.grad, ref.grad, atol=1e-5), got.shape


def test_fused_mlp_under_checkpoint(fused_env):
    """Non-reentrant checkpointing: the recompute re-runs the fused fc1
    (gelu included, free via the epilogue) and early-stop still skips
    fc2's forward; gradients match the stock eager run exactly."""
    calls, handler = fused_env
    w1, b1, w2, b2 = _mlp_params(seed=3)
    x = torch.randn(TOK, D, requires_grad=True)

    def block(t):
        return F.linear(F.gelu(F.linear(t, w1, b1)), w2, b2)

    with _mode(handler) as m:
        y = checkpoint(block, x, use_reentrant=False)
        y.pow(2).sum().backward()
    assert calls["gelu"] == 2  # forward + recompute
    assert m.gelu_hits == 2

    xr = x.detach().clone().requires_grad_(True)
    w1r, b1r = (t.detach().clone().requires_grad_(True) for t in (w1, b1))
    w2r, b2r = (t.detach().clone().requires_grad_(True) for t in (w2, b2))
    yr = F.linear(F.gelu(F.linear(xr, w1r, b1r)), w2r, b2r)
    yr.pow(2).sum().backward()
    assert torch.allclose(y, yr, atol=1e-6)
    assert torch.allclose(x.grad, xr.grad, atol=1e-5)
    assert torch.allclose(w1.grad, w1r.grad, atol=1e-5)
    assert torch.allclose(b1.grad, b1r.grad, atol=1e-5)
    assert torch.allclose(w2.grad, w2r.grad, atol=1e-5)
    assert torch.allclose(b2.grad, b2r.grad, atol=1e-5)


def test_unfused_gelu_untouched(fused_env):
    """A gelu whose input we did not produce must run the stock op."""
    _, handler = fused_env
    x = torch.randn(TOK, HID)
    with _mode(handler):
        y = F.gelu(x)
    assert torch.allclose(y, F.gelu(x), atol=1e-6)
