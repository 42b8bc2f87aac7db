"""Linear layer with a native weight-gradient kernel.

Forward and the input-gradient GEMM go through hipBLASLt (the sanctioned
library path for plain GEMMs); the weight gradient dW = dY^T X can route
through csrc/wgemm.hip (gfx950 ds_read_b64_tr_b16 hardware transpose
reads, fused dbias column sums).

Measured ISOLATED (benchmarks/bench_wgemm.py, K=32768):
    shape (M=out, N=in)             ours    hipBLASLt
    qkv   15360 x 5120               931        660  TF/s
    proj   5120 x 5120               884        825
    fc1   20480 x 5120               898       1076
    fc2    5120 x 20480              918       1080

Measurement history (all on MI355X, receipts in profiles/PROFILES.md):
  * first in-step number ~520 TF/s was caused by the FUSED dbias column
    sums (per-fragment unpack+accumulate VALU, 945 -> 535 TF on the qkv
    shape); dbias is now a separate memory-bound reduction;
  * a cold-cache bench (4 rotating operand sets past the 256 MB L3)
    confirmed the GEMM core itself beats hipBLASLt isolated;
  * but a within-box end-to-end A/B still reads 54.3 vs 58.0 img/s with
    the custom autograd Function active;
  * per-layer isolation (benchmarks/bench_linear_bwd.py) shows stock,
    Function-with-library-wgrad, and Function-with-native-wgrad all
    EQUAL (qkv 12.9/12.6/12.9 ms fwd+bwd) — i.e. (a) torch's fused
    linear backward already picks a dW kernel as fast as ours (the
    standalone torch.matmul(a.t(), b) used as the bench baseline is the
    SLOW formulation, not what training actually runs), and (b) the
    end-to-end Function cost is an interaction with the grad-checkpoint
    context, not a per-layer kernel effect;
  * ROOT-CAUSED via benchmarks/ktrace_diff.py on the ON/OFF traces
    (profiles/PROFILES.md "ktrace diff") + a CPU dispatch-count repro
    (tests/test_checkpoint_earlystop.py): torch's codegen'd addmm packs
    its input SavedVariables BEFORE dispatching the kernel, so
    checkpoint(use_reentrant=False) early-stops the recompute before
    the region's LAST GEMM (fc2 forward is never recomputed).  A custom
    autograd.Function packs only after forward returns, so wrapping the
    linears in one forces that GEMM back into every block recompute:
    +128 forward GEMMs (+657 ms) per 4 profiled steps, ~164 ms/step —
    which is the previously "unexplained" part of the A/B gap.  The
    rest of the gap was the since-removed bias-fused wgemm variant.
The corrected integration is VITFSDP_NATIVE_WGRAD=2 (NativeWgradMode
below): dispatcher-level interception that keeps the stock addmm node.
Measured on MI355X (ViT-Large bs=128, within-box where noted):
  * mode 2 ungated: 1040 -> 702 img/s — the kernel has no split-K, so
    ViT-Large dW grids (16-64 workgroups) cannot fill 256 CUs;
  * mode 2 with the _MIN_TILES>=256 gate (nothing routed at Large):
    117.4 ms/step vs 123.1 baseline (different boxes, within the ±5%
    box variance) — the TorchDispatchMode python overhead itself is
    NOT measurable: backward is GPU-bound and dispatch overlaps with
    queued kernels.
Both modes stay OPT-IN pending a ViT-10B within-box A/B (qkv/proj pass
the gate there at 1200/400 workgroups); per-layer parity says expect
neutral-to-small effect (ROADMAP item 4).
"""

import contextlib
import os

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils._python_dispatch import TorchDispatchMode

from ._extension import ext, use_hip
from ..tuning import lt_algo_table


# "1": custom autograd.Function (kept for A/B history; defeats checkpoint
#      early-stop, see tests/test_checkpoint_earlystop.py)
# "2": dispatcher interception below autograd (NativeWgradMode) — the
#      stock addmm node stays, early-stop is preserved
_WGRAD_MODE = os.environ.get("VITFSDP_NATIVE_WGRAD", "0")
_NATIVE_WGRAD = _WGRAD_MODE == "1"
_NATIVE_WGRAD_DISPATCH = _WGRAD_MODE == "2"


def _use_native_wgrad(dy2, x2, w):
    if not _NATIVE_WGRAD:
        return False
    if not (dy2.is_cuda and dy2.dtype == torch.bfloat16):
        return False
    m, n = w.shape[0], w.shape[1]
    k = dy2.shape[0]
    if not (k % 64 == 0 and m % 256 == 0 and n % 256 == 0):
        return False
    # measured crossover: hipBLASLt wins on the very wide/tall shapes
    if m > 16384 or n > 8192:
        return False
    return use_hip(dy2) and hasattr(ext(), "wgrad_gemm")


class _NativeLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return F.linear(x, weight, bias)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = torch.matmul(dy, weight)  # hipBLASLt dgrad

        if _use_native_wgrad(dy2, x2, weight):
            # dbias via the fused in-kernel column sums measured ~1.7x
            # slower end-to-end (per-fragment unpack+accumulate VALU);
            # the plain reduction is one cheap memory-bound pass
            (dw,) = ext().wgrad_gemm(dy2.contiguous(), x2.contiguous(), False)
            db = dy2.sum(0) if ctx.has_bias else None
        else:
            dw = torch.matmul(dy2.t(), x2)
            db = dy2.sum(0) if ctx.has_bias else None
        return dx, dw, db


# The kernel has no split-K: its grid is (m/256)*(n/256) workgroups, so
# small dW matrices cannot fill 256 CUs (ViT-Large proj dW is 16
# workgroups and measured 1040 -> 702 img/s when routed native).  10B
# qkv/proj are 1200/400 workgroups.  Env override for experiments.
_MIN_TILES = int(os.environ.get("VITFSDP_WGRAD_MIN_TILES", "256"))


def _wgrad_mm_shapes_ok(m, n, k):
    """Shape gate for the native wgrad kernel: the 256x256x64 tile must
    divide evenly, the grid must be large enough to fill the chip
    (_MIN_TILES), and the measured ViT-10B crossover keeps the very
    wide/tall MLP shapes (fc1/fc2, 20480-wide) on hipBLASLt while qkv
    and proj go native.  Symmetric bound because the dispatcher sees dW
    in whichever orientation AddmmBackward chose."""
    return (k % 64 == 0 and m % 256 == 0 and n % 256 == 0
            and m <= 16384 and n <= 16384
            and (m // 256) * (n // 256) >= _MIN_TILES)


def _is_wgrad_mm(at, b):
    """Match the weight-gradient GEMM AddmmBackward emits:
    mm(saved_input.t(), grad_out) — first operand a transposed view of a
    contiguous [K, M] base, second operand contiguous [K, N]."""
    if at.dim() != 2 or b.dim() != 2 or at.shape[1] != b.shape[0]:
        return False
    m, k = at.shape
    if m < 2 or k < 2:
        return False  # degenerate strides are ambiguous
    if not (at.stride(0) == 1 and at.stride(1) == m):
        return False
    if not (b.stride(1) == 1 and b.stride(0) == b.shape[1]):
        return False
    return _wgrad_mm_shapes_ok(m, b.shape[1], k)


class NativeWgradMode(TorchDispatchMode):
    """Reroute ONLY the dW GEMMs to csrc/wgemm.hip, below autograd.

    This is the corrected integration after the Function-path root cause
    (see module docstring): pushing a mode around ``loss.backward()``
    leaves every forward an ordinary codegen'd addmm — non-reentrant
    checkpoint early-stop keeps skipping the last recompute GEMM — and
    the mode swaps the kernel at dispatch time when AddmmBackward runs
    its mm(x^T, dy).  The engine propagates the mode into its worker
    threads via ThreadLocalState, which
    tests/test_native_wgrad_dispatch.py verifies on CPU.

    ``handler(a, b)`` (a: contiguous [K, M] base, b: contiguous [K, N],
    returns [M, N]) is a test seam; the default requires the HIP
    extension and bf16 CUDA tensors.
    """

    def __init__(self, handler=None):
        super().__init__()
        self.hits = 0
        self._handler = handler

    def _route(self, at, b):
        if self._handler is not None:
            return self._handler(at.t(), b)
        (dw,) = ext().wgrad_gemm(at.t().contiguous(), b.contiguous(), False)
        return dw

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        if (
            func is torch.ops.aten.mm.default
            and _is_wgrad_mm(*args)
            and (
                self._handler is not None
                or (
                    args[0].is_cuda
                    and args[0].dtype == torch.bfloat16
                    and args[1].dtype == torch.bfloat16
                    and use_hip(args[0])
                    and hasattr(ext(), "wgrad_gemm")
                )
            )
        ):
            self.hits += 1
            return self._route(*args)
        return func(*args, **kwargs)


def _op_layout(t):
    """'N' for a row-contiguous 2-D tensor, 'T' for a transposed view of
    one, None otherwise (mirrors csrc/ltgemm.cpp classify())."""
    if t.dim() != 2:
        return None
    r, c = t.shape
    s0, s1 = t.stride()
    if s1 == 1 and (s0 == c or r == 1):
        return "N"
    if s0 == 1 and (s1 == r or c == 1):
        return "T"
    return None


def _dual_key(a, b):
    """Column-major-dual shape key for row-major a @ b, matching the
    offline search rows (csrc/tools/hipblaslt_search.cpp): the A-slot
    takes b's memory, the B-slot a's."""
    la, lb = _op_layout(a), _op_layout(b)
    if la is None or lb is None:
        return None
    return (lb, la, b.shape[1], a.shape[0], a.shape[1])


# --------------------------------------------------------------------------
# Fused MLP GELU (ROADMAP item 6 / VERDICT 3).
#
# Enabled by configure_gelu_fusion(embed_dim, hidden_dim) + the
# VITFSDP_FUSED_GELU env knob.  Mechanism (below autograd, so the stock
# addmm/gelu graph nodes — and checkpoint early-stop — are untouched):
# fc1's forward addmm runs as _C.fwd_gemm_gelu (csrc/fgemm.hip, the
# hand-written CDNA4 GEMM with an EXACT-erf GELU+bias epilogue); the
# dispatch returns the PRE-activation as the addmm result (so
# GeluBackward saves exactly what the stock graph expects) and answers
# the immediately following aten.gelu with the cached gelu output.  The
# backward is entirely stock — erf forward + erf gelu_backward, i.e.
# reference numerics up to GEMM rounding.
#
# (hipblaslt-ext's GELU_AUX/DGELU epilogues ship no algorithms for bf16
# on this ROCm 7.2 / gfx950 stack — probed on-box, r2 GPU call 2 — and
# are tanh-approx anyway; the lt_gemm_gelu / lt_gemm_dgelu_bgrad
# entries remain in csrc/ltgemm.cpp as dormant alternatives.)
# --------------------------------------------------------------------------

_GELU_CFG = {"d": 0, "hid": 0}

# minimum (M/256)*(N/256) workgroups before fc1 routes to fgemm (grid
# fill, same rationale as the wgrad gate)
_FGEMM_MIN_TILES = int(os.environ.get("VITFSDP_FGEMM_MIN_TILES", "256"))


def configure_gelu_fusion(embed_dim, hidden_dim):
    """Register the MLP dims so the dispatch mode can recognize fc1's
    forward addmm by shape.  No-op when embed == hidden (the shape key
    would be ambiguous)."""
    if embed_dim != hidden_dim:
        _GELU_CFG["d"] = int(embed_dim)
        _GELU_CFG["hid"] = int(hidden_dim)


def _gelu_fusion_on():
    return (
        os.environ.get("VITFSDP_FUSED_GELU", "0") == "1"
        and _GELU_CFG["hid"] > 0
    )




class TunedGemmMode(TorchDispatchMode):
    """Reroute training GEMMs whose shape has an offline-searched
    hipBLASLt algorithm (tuned/lt_algos_gfx950.json) through
    _C.lt_gemm with the explicit index — same library, better kernel
    pick than the heuristic (ROADMAP item 5).  Dispatcher-level like
    NativeWgradMode, so the autograd graph keeps its stock addmm/mm
    nodes and non-reentrant checkpoint early-stop is preserved.

    Also hosts the native-wgrad reroute when VITFSDP_NATIVE_WGRAD=2 and
    the fused MLP GELU epilogues when configured, so one mode covers
    the whole step (push around forward AND backward).
    """

    def __init__(self, table=None, native_wgrad=None, handler=None):
        super().__init__()
        self.table = lt_algo_table() if table is None else table
        self.native_wgrad = (
            _NATIVE_WGRAD_DISPATCH if native_wgrad is None else native_wgrad
        )
        # test seam: handler(a, b, idx, bias) replaces _C.lt_gemm (and
        # lifts the CUDA/bf16 gate) so the routing logic runs on CPU
        self._handler = handler
        self.hits = 0
        self.wgrad_hits = 0
        # fused-GELU state (see module comment above):
        self.fused_gelu = _gelu_fusion_on()
        self._pending_gelu = {}  # pre-act data_ptr -> gelu_out (transient)
        self.gelu_hits = 0

    def _tuned_index(self, a, b):
        if not self.table:
            return None
        key = _dual_key(a, b)
        if key is None:
            return None
        return self.table.get(key)

    def _gpu_ok(self, t):
        if self._handler is not None:
            return True
        return t.is_cuda and t.dtype == torch.bfloat16 and use_hip(t)

    def _route(self, a, b, idx, bias=None):
        if self._handler is not None:
            return self._handler(a, b, idx, bias)
        return ext().lt_gemm(a, b, idx, bias)

    # -- fused-GELU helpers -------------------------------------------------

    def _is_fc1_fwd(self, key):
        if not (
            key is not None
            and key[0] == "T" and key[1] == "N"
            and key[2] == _GELU_CFG["hid"] and key[4] == _GELU_CFG["d"]
        ):
            return False
        if self._handler is not None:
            return True  # CPU test seam has no kernel shape limits
        hid, tok = key[2], key[3]
        return (
            tok % 256 == 0 and hid % 256 == 0 and key[4] % 64 == 0
            and (tok // 256) * (hid // 256) >= _FGEMM_MIN_TILES
        )

    def _fused_fc1_fwd(self, a, b, bias):
        if self._handler is not None:  # CPU test seam
            out, pre = self._handler(a, b, "gelu", bias)
        else:
            # b is the transposed view of fc1's weight [hid, d]: b.t()
            # IS the weight, row-major contiguous — fgemm reads it
            # directly (W [N, K], K contiguous)
            out, pre = ext().fwd_gemm_gelu(
                a.contiguous(), b.t().contiguous(), bias
            )
        self.gelu_hits += 1
        if len(self._pending_gelu) > 4:  # unconsumed strays (shouldn't happen)
            self._pending_gelu.clear()
        self._pending_gelu[pre.data_ptr()] = out
        return pre

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        if func is torch.ops.aten.mm.default and self._gpu_ok(args[0]):
            a, b = args
            if (
                self.native_wgrad
                and self._handler is None
                and _is_wgrad_mm(a, b)
                and hasattr(ext(), "wgrad_gemm")
            ):
                self.wgrad_hits += 1
                (dw,) = ext().wgrad_gemm(a.t().contiguous(), b.contiguous(),
                                         False)
                return dw
            idx = self._tuned_index(a, b)
            if idx is not None:
                self.hits += 1
                return self._route(a, b, idx)
        elif (
            func is torch.ops.aten.addmm.default
            and len(args) == 3
            and kwargs.get("beta", 1) == 1
            and kwargs.get("alpha", 1) == 1
            and self._gpu_ok(args[1])
            and args[0].dim() == 1
        ):
            bias, a, b = args
            key = _dual_key(a, b)
            if (
                self.fused_gelu
                and self._is_fc1_fwd(key)
                and (self._handler is not None
                     or hasattr(ext(), "fwd_gemm_gelu"))
            ):
                return self._fused_fc1_fwd(a, b, bias)
            idx = self.table.get(key) if (self.table and key) else None
            if idx is not None:
                self.hits += 1
                return self._route(a, b, idx, bias)
        elif self.fused_gelu and func is torch.ops.aten.gelu.default:
            out = self._pending_gelu.pop(args[0].data_ptr(), None)
            if out is not None and out.shape == args[0].shape:
                return out
        return func(*args, **kwargs)


def gemm_dispatch_context():
    """Context manager for the training step (forward and backward):
    activates the dispatcher-level GEMM rerouting when the tuned
    algorithm table is present, fused MLP GELU is configured, or
    VITFSDP_NATIVE_WGRAD=2; otherwise a no-op."""
    if lt_algo_table() or _gelu_fusion_on():
        return TunedGemmMode()
    if _NATIVE_WGRAD_DISPATCH:
        return NativeWgradMode()  # backward-compatible, wgrad only
    return contextlib.nullcontext()


def wgrad_backward_context():
    """Backward-compatible alias of gemm_dispatch_context (historical
    name from when only the dW GEMMs were rerouted)."""
    return gemm_dispatch_context()


class NativeLinear(nn.Linear):
    """Drop-in nn.Linear that routes the weight gradient through the
    native wgrad kernel when VITFSDP_NATIVE_WGRAD=1 and the shape
    qualifies (identical state_dict keys and initialization).

    When the native path is disabled it falls back to stock autograd
    entirely: a within-box A/B showed the custom Function with the
    LIBRARY wgrad formulation costs ~126 ms/step at ViT-10B — torch's
    own matmul backward picks a faster dW GEMM layout than the explicit
    dy^T @ x call — so there is no reason to take the Function without
    the native kernel."""

    def forward(self, x):
        if _NATIVE_WGRAD:
            return _NativeLinearFn.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)
