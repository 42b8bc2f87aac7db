"""Fused-MLP-GELU dispatch routing (ops/linear.py TunedGemmMode):
CPU harness with a mocked extension implementing the hipblaslt-ext
epilogue semantics (tanh GELU) in plain torch, so the intricate
caching/weakref/early-stop logic is covered without a GPU.  The GPU
counterpart (real _C kernels) lives in tests/test_gpu_kernels.py.
"""

import types

import pytest
import torch
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from vit_10b_fsdp_example_amd.ops import linear as linmod

TOK, D, HID = 8, 4, 16


def _tanh_gelu(x):
    return F.gelu(x.float(), approximate="tanh").to(x.dtype)


class FakeExt(types.SimpleNamespace):
    """hipblaslt-ext epilogue semantics in torch."""

    def __init__(self):
        super().__init__()
        self.calls = {"gelu": 0, "dgelu": 0}

    def lt_gemm_gelu(self, a, b, bias, algo_index):
        pre = a @ b + (bias if bias is not None else 0)
        self.calls["gelu"] += 1
        return _tanh_gelu(pre), pre

    def lt_gemm_dgelu_bgrad(self, dy, w, aux, algo_index):
        d_gelu_out = dy @ w
        dpre = linmod._dgelu_tanh(d_gelu_out, aux)
        self.calls["dgelu"] += 1
        return dpre, dpre.sum(0)


@pytest.fixture()
def fused_env(monkeypatch):
    fake = FakeExt()
    monkeypatch.setattr(linmod, "ext", lambda: fake)
    monkeypatch.setattr(linmod.TunedGemmMode, "_gpu_ok", lambda self, t: True)
    monkeypatch.setenv("VITFSDP_FUSED_GELU", "1")
    monkeypatch.setitem(linmod._GELU_CFG, "d", D)
    monkeypatch.setitem(linmod._GELU_CFG, "hid", HID)
    return fake


def _mlp_params(seed=0):
    g = torch.Generator().manual_seed(seed)
    w1 = torch.randn(HID, D, generator=g, requires_grad=True)
    b1 = torch.randn(HID, generator=g, requires_grad=True)
    w2 = torch.randn(D, HID, generator=g, requires_grad=True)
    b2 = torch.randn(D, generator=g, requires_grad=True)
    return w1, b1, w2, b2


def _reference(x, w1, b1, w2, b2):
    """Eager tanh-GELU MLP (what the fused path should reproduce)."""
    h = F.linear(x, w1, b1)
    return F.linear(_tanh_gelu(h), w2, b2)


def test_fused_mlp_matches_tanh_reference(fused_env):
    w1, b1, w2, b2 = _mlp_params()
    x = torch.randn(TOK, D, requires_grad=True)

    # fused forward+backward
    xf = x.detach().clone().requires_grad_(True)
    w1f, b1f = (t.detach().clone().requires_grad_(True) for t in (w1, b1))
    w2f, b2f = (t.detach().clone().requires_grad_(True) for t in (w2, b2))
    with linmod.TunedGemmMode() as m:
        yf = F.linear(F.gelu(F.linear(xf, w1f, b1f)), w2f, b2f)
        yf.pow(2).sum().backward()
    assert m.gelu_hits == 2, "fc1 fwd and fc2 dgrad must both fuse"
    assert fused_env.calls == {"gelu": 1, "dgelu": 1}

    # reference: same math eagerly with tanh GELU
    xr = x.detach().clone().requires_grad_(True)
    w1r, b1r = (t.detach().clone().requires_grad_(True) for t in (w1, b1))
    w2r, b2r = (t.detach().clone().requires_grad_(True) for t in (w2, b2))
    yr = _reference(xr, w1r, b1r, w2r, b2r)
    yr.pow(2).sum().backward()

    assert torch.allclose(yf, yr, atol=1e-5)
    for got, ref in [(xf, xr), (w1f, w1r), (b1f, b1r), (w2f, w2r), (b2f, b2r)]:
        assert torch.allclose(got.grad, ref.grad, atol=1e-4), got.shape


def test_fused_mlp_under_checkpoint(fused_env):
    """Non-reentrant checkpointing: the recompute re-runs the fused fc1
    (gelu included, free via the epilogue), early-stop still skips fc2's
    forward, and the backward fusions consume the RECOMPUTED aux."""
    w1, b1, w2, b2 = _mlp_params(seed=3)
    x = torch.randn(TOK, D, requires_grad=True)

    def block(t):
        return F.linear(F.gelu(F.linear(t, w1, b1)), w2, b2)

    with linmod.TunedGemmMode() as m:
        y = checkpoint(block, x, use_reentrant=False)
        y.pow(2).sum().backward()
    # fc1 fused twice (forward + recompute), fc2 dgrad fused once
    assert fused_env.calls == {"gelu": 2, "dgelu": 1}
    assert m.gelu_hits == 3

    xr = x.detach().clone().requires_grad_(True)
    w1r, b1r = (t.detach().clone().requires_grad_(True) for t in (w1, b1))
    w2r, b2r = (t.detach().clone().requires_grad_(True) for t in (w2, b2))
    yr = _reference(xr, w1r, b1r, w2r, b2r)
    yr.pow(2).sum().backward()
    assert torch.allclose(y, yr, atol=1e-5)
    assert torch.allclose(x.grad, xr.grad, atol=1e-4)
    assert torch.allclose(w1.grad, w1r.grad, atol=1e-4)
    assert torch.allclose(b1.grad, b1r.grad, atol=1e-4)
    assert torch.allclose(w2.grad, w2r.grad, atol=1e-4)
    assert torch.allclose(b2.grad, b2r.grad, atol=1e-4)
    w1.grad = b1.grad = w2.grad = b2.grad = None


def test_unfused_gelu_untouched(fused_env):
    """A gelu whose input we did not produce must run the stock op."""
    x = torch.randn(TOK, HID)
    with linmod.TunedGemmMode():
        y = F.gelu(x)
    assert torch.allclose(y, F.gelu(x), atol=1e-6)
