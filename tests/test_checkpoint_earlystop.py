"""Pin the non-reentrant-checkpoint early-stop property that decides
how NativeLinear's wgrad must be wired (ops/linear.py, ROADMAP item 4).

Mechanism (verified against MI355X ktrace, profiles/PROFILES.md
"ktrace diff"): torch's codegen'd ops (addmm/mm) create their grad_fn
and pack input SavedVariables BEFORE dispatching the kernel, so under
``checkpoint(use_reentrant=False)`` the recompute's early-stop exception
fires before the LAST GEMM of the region ever runs — the fc2 forward is
never recomputed.  A custom ``torch.autograd.Function`` packs its saved
tensors only after ``forward`` returns, so wrapping a linear in one
forces that GEMM back into every recompute: +1 forward GEMM per block
per step (~164 ms/step at ViT-10B).  Any future wgrad integration must
keep the stock autograd node (or restructure the region) rather than
substitute a Function.
"""

import torch
import torch.nn.functional as F

from tests.utils_mp import CountMM
from torch.utils.checkpoint import checkpoint


class _LinearFn(torch.autograd.Function):
    """Minimal stand-in with _NativeLinearFn's save/compute structure."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return F.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        return torch.matmul(dy, w), torch.matmul(dy.t(), x), dy.sum(0)


def _backward_gemm_count(use_function):
    torch.manual_seed(0)
    w1 = torch.randn(8, 8, requires_grad=True)
    b1 = torch.randn(8, requires_grad=True)
    w2 = torch.randn(8, 8, requires_grad=True)
    b2 = torch.randn(8, requires_grad=True)
    x = torch.randn(4, 8, requires_grad=True)
    lin = _LinearFn.apply if use_function else F.linear

    def block(t):
        return lin(F.gelu(lin(t, w1, b1)), w2, b2)

    out = checkpoint(block, x, use_reentrant=False)
    counter = CountMM()
    with counter:
        out.sum().backward()
    return counter.n


def test_early_stop_skips_last_recompute_gemm():
    # recompute fc1 only (fc2 early-stopped) + 2 dgrads + 2 wgrads
    assert _backward_gemm_count(use_function=False) == 5


def test_custom_function_defeats_early_stop():
    # the Function's post-forward packing forces fc2's forward GEMM
    # back into the recompute: exactly one extra GEMM per region
    assert _backward_gemm_count(use_function=True) == 6
