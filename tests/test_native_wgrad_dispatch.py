"""NativeWgradMode (ops/linear.py, VITFSDP_NATIVE_WGRAD=2): dispatcher-
level dW rerouting that, unlike the Function path, preserves
non-reentrant checkpoint early-stop (tests/test_checkpoint_earlystop.py
pins why that matters)."""

import torch
import torch.nn.functional as F

from tests.utils_mp import CountMM
from torch.utils.checkpoint import checkpoint

from vit_10b_fsdp_example_amd.ops import NativeWgradMode
from vit_10b_fsdp_example_amd.ops.linear import _is_wgrad_mm

# gate-passing dims: k % 64 == 0, m % 256 == 0, n % 256 == 0
K, IN, OUT = 64, 256, 512


def _handler(a, b):
    # reference implementation of what wgrad_gemm computes on GPU:
    # a [K, M] contiguous, b [K, N] contiguous -> A^T B [M, N]
    return (a.t().double() @ b.double()).to(b.dtype)


def _no_min_tiles(monkeypatch):
    import vit_10b_fsdp_example_amd.ops.linear as linear_mod
    monkeypatch.setattr(linear_mod, "_MIN_TILES", 0)


def test_pattern_matcher(monkeypatch):
    _no_min_tiles(monkeypatch)
    x = torch.randn(K, IN)
    dy = torch.randn(K, OUT)
    assert _is_wgrad_mm(x.t(), dy)          # the AddmmBackward dW shape
    assert not _is_wgrad_mm(dy, x)          # dgrad: contiguous first arg
    assert not _is_wgrad_mm(x.t(), dy.t())  # non-contiguous second arg
    assert not _is_wgrad_mm(torch.randn(K, 100).t(), dy)  # gate: 100%256


def test_intercepts_linear_wgrad_and_matches_stock(monkeypatch):
    _no_min_tiles(monkeypatch)
    torch.manual_seed(0)
    lin = torch.nn.Linear(IN, OUT)
    x = torch.randn(K, IN, requires_grad=True)
    # a non-degenerate upstream grad: .sum().backward() would feed the
    # dW mm an EXPANDED dy (strides (0,0)) which the strict stride
    # matcher correctly declines (real training grads are contiguous)
    g = torch.randn(K, OUT)

    # stock baseline
    lin(x).backward(g)
    ref_w, ref_x = lin.weight.grad.clone(), x.grad.clone()
    lin.weight.grad = lin.bias.grad = x.grad = None

    mode = NativeWgradMode(handler=_handler)
    with mode:
        lin(x).backward(g)
    assert mode.hits == 1  # exactly the dW GEMM, not dgrad/forward
    torch.testing.assert_close(lin.weight.grad, ref_w, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(x.grad, ref_x, rtol=1e-6, atol=1e-6)


def test_mode_preserves_checkpoint_early_stop(monkeypatch):
    """The whole point of mode "2": under the dispatch mode the stock
    addmm nodes remain, so the last recompute GEMM is still skipped —
    total backward GEMM count stays 5 (vs 6 for the Function path)."""
    _no_min_tiles(monkeypatch)
    torch.manual_seed(0)
    w1 = torch.randn(256, 256, requires_grad=True)
    b1 = torch.randn(256, requires_grad=True)
    w2 = torch.randn(256, 256, requires_grad=True)
    b2 = torch.randn(256, requires_grad=True)
    x = torch.randn(K, 256, requires_grad=True)

    def block(t):
        return F.linear(F.gelu(F.linear(t, w1, b1)), w2, b2)

    out = checkpoint(block, x, use_reentrant=False)
    counter = CountMM()
    wmode = NativeWgradMode(handler=_handler)
    # counter innermost: it sees each ORIGINAL dispatch (then redispatches
    # into wmode), so its count is comparable to the stock-vs-Function
    # 5-vs-6 measurement in test_checkpoint_earlystop.py
    with wmode, counter:
        out.backward(torch.randn_like(out))  # contiguous upstream grad
    assert counter.n == 5  # early-stop intact: same count as stock
    assert wmode.hits == 2  # ...and both wgrads rerouted
