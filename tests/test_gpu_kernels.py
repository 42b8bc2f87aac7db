"""GPU kernel numerics: every HIP kernel vs a plain PyTorch fp32
reference of the same op.  All tests require an MI355X."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from vit_10b_fsdp_example_amd.ops import _extension

    EXT = _extension.ext()
else:
    EXT = None


def _dev():
    return torch.device("cuda", 0)


def test_extension_loaded():
    """The native extension must be present on a GPU box (no silent
    eager fallback)."""
    assert EXT is not None, "_C extension missing on GPU box"
    t = torch.zeros(4, device=_dev())
    from vit_10b_fsdp_example_amd.ops import use_hip

    assert use_hip(t) is True


def test_mfma_probe_layout():
    """Pin the 16x16x32 bf16 MFMA fragment layout: probe kernel output
    must equal the row-major matmul.  Asymmetric operands so a transposed
    C-write cannot pass (guide §3)."""
    torch.manual_seed(0)
    a = torch.randn(16, 32, device=_dev())
    b = torch.arange(32 * 16, device=_dev(), dtype=torch.float32).reshape(32, 16)
    b = (b % 7) - 3.0 + 0.1 * torch.randn(32, 16, device=_dev())
    c = EXT.mfma_probe(a, b)
    ref = a.to(torch.bfloat16).float() @ b.to(torch.bfloat16).float()
    np.testing.assert_allclose(
        c.cpu().numpy(), ref.cpu().numpy(), rtol=1e-2, atol=1e-2
    )


@pytest.mark.parametrize("shape", [(128, 256, 5120), (4, 7, 64), (2, 3, 1024)])
def test_layernorm_fwd_bwd(shape):
    from vit_10b_fsdp_example_amd.ops import layer_norm

    torch.manual_seed(0)
    x = torch.randn(*shape, device=_dev(), dtype=torch.bfloat16)
    w = torch.randn(shape[-1], device=_dev(), dtype=torch.bfloat16)
    b = torch.randn(shape[-1], device=_dev(), dtype=torch.bfloat16)
    x.requires_grad_(True)
    w.requires_grad_(True)
    b.requires_grad_(True)

    y = layer_norm(x, w, b, 1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yf = torch.nn.functional.layer_norm(xf, (shape[-1],), wf, bf, 1e-6)
    yf.backward(dy.float())

    np.testing.assert_allclose(
        y.detach().float().cpu(), yf.detach().cpu(), rtol=0.05, atol=0.03
    )
    np.testing.assert_allclose(
        x.grad.float().cpu(), xf.grad.cpu(), rtol=0.1, atol=0.05
    )
    # column reductions accumulate over many rows: compare with a scale-
    # aware tolerance
    n_rows = int(np.prod(shape[:-1]))
    np.testing.assert_allclose(
        w.grad.float().cpu(), wf.grad.cpu(),
        rtol=0.05, atol=0.02 * max(1.0, n_rows ** 0.5),
    )
    np.testing.assert_allclose(
        b.grad.float().cpu(), bf.grad.cpu(),
        rtol=0.05, atol=0.02 * max(1.0, n_rows ** 0.5),
    )


def test_cross_entropy_fwd_bwd():
    from vit_10b_fsdp_example_amd.ops import cross_entropy

    torch.manual_seed(0)
    logits = torch.randn(128, 1000, device=_dev(), dtype=torch.bfloat16)
    target = torch.randint(0, 1000, (128,), device=_dev())
    logits.requires_grad_(True)
    loss = cross_entropy(logits, target)
    loss.backward()

    lf = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lf, target)
    ref.backward()

    assert abs(loss.item() - ref.item()) < 0.02
    np.testing.assert_allclose(
        logits.grad.float().cpu(), lf.grad.cpu(), rtol=0.05, atol=1e-4
    )


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    n = 100003  # not a multiple of 4: exercises the scalar tail
    p1 = torch.randn(n, device=_dev())
    g = torch.randn(n, device=_dev())
    p2 = p1.clone()

    from vit_10b_fsdp_example_amd.ops import FusedAdamW

    a = torch.nn.Parameter(p1)
    a.grad = g.clone()
    opt1 = FusedAdamW([a], lr=1e-3, weight_decay=0.1)
    b = torch.nn.Parameter(p2)
    b.grad = g.clone()
    opt2 = torch.optim.AdamW([b], lr=1e-3, weight_decay=0.1)
    for _ in range(5):
        opt1.step()
        opt2.step()
    np.testing.assert_allclose(
        a.detach().cpu().numpy(), b.detach().cpu().numpy(), rtol=1e-5, atol=1e-6
    )


def test_multi_tensor_sqnorm_scale():
    from vit_10b_fsdp_example_amd.ops import local_sqnorm, scale_

    torch.manual_seed(0)
    ts = [torch.randn(1000 + i * 7, device=_dev()) for i in range(5)]
    ref = sum(float(t.pow(2).sum()) for t in ts)
    got = float(local_sqnorm(ts))
    assert abs(got - ref) / ref < 1e-5
    refs = [t.clone() * 0.5 for t in ts]
    scale_(ts, torch.tensor(0.5, device=_dev()))
    for t, r in zip(ts, refs):
        np.testing.assert_allclose(t.cpu().numpy(), r.cpu().numpy(), rtol=1e-6)


@pytest.mark.parametrize(
    "B,H,T,D",
    [
        (2, 4, 256, 160),  # ViT-10B head shape
        (2, 2, 128, 64),
        (1, 2, 64, 16),
        (1, 3, 80, 32),    # T not a multiple of the q/k tiles
        (1, 1, 1024, 160), # long sequence (image_size 448 class)
    ],
)
def test_fmha_fwd_vs_fp32_reference(B, H, T, D):
    torch.manual_seed(0)
    q = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    k = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    v = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    scale = D ** -0.5
    o, lse = EXT.fmha_fwd(q, k, v, scale)

    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    p = torch.softmax(s, dim=-1)
    ref_o = p @ v.float()
    ref_lse = torch.logsumexp(s, dim=-1)

    np.testing.assert_allclose(
        o.float().cpu().numpy(), ref_o.cpu().numpy(), rtol=0.05, atol=0.03
    )
    np.testing.assert_allclose(
        lse.cpu().numpy(), ref_lse.cpu().numpy(), rtol=1e-3, atol=1e-2
    )


@pytest.mark.parametrize("B,H,T,D", [(2, 4, 256, 160), (1, 2, 80, 64)])
def test_fmha_bwd_vs_fp32_reference(B, H, T, D):
    torch.manual_seed(1)
    q = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    k = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    v = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    do = torch.randn(B, H, T, D, device=_dev(), dtype=torch.bfloat16)
    scale = D ** -0.5
    o, lse = EXT.fmha_fwd(q, k, v, scale)
    dq, dk, dv = EXT.fmha_bwd(do, q, k, v, o, lse, scale)

    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    s = (qf @ kf.transpose(-2, -1)) * scale
    ref_o = torch.softmax(s, dim=-1) @ vf
    ref_o.backward(do.float())

    np.testing.assert_allclose(
        dv.float().cpu().numpy(), vf.grad.cpu().numpy(), rtol=0.1, atol=0.06
    )
    np.testing.assert_allclose(
        dq.float().cpu().numpy(), qf.grad.cpu().numpy(), rtol=0.1, atol=0.06
    )
    np.testing.assert_allclose(
        dk.float().cpu().numpy(), kf.grad.cpu().numpy(), rtol=0.1, atol=0.06
    )


def test_attention_autograd_path():
    """ops.attention dispatches to the HIP kernel on GPU and its autograd
    matches the fp32 math composition."""
    from vit_10b_fsdp_example_amd.ops import attention

    torch.manual_seed(2)
    q = torch.randn(2, 4, 128, 160, device=_dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    o = attention(q, k, v)
    o.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad.float()).all()

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    s = (qf @ kf.transpose(-2, -1)) * (160 ** -0.5)
    ref = torch.softmax(s, dim=-1) @ vf
    np.testing.assert_allclose(
        o.detach().float().cpu().numpy(), ref.detach().cpu().numpy(),
        rtol=0.05, atol=0.03,
    )


def test_fmha_qkv_fused_path():
    """The zero-copy qkv path (strided kernels) matches the fp32 math
    reference for outputs and the fused dqkv gradient."""
    from vit_10b_fsdp_example_amd.ops import attention_qkv

    torch.manual_seed(3)
    B, T, H, D = 2, 256, 4, 160
    qkv = torch.randn(B, T, 3, H, D, device=_dev(), dtype=torch.bfloat16,
                      requires_grad=True)
    o = attention_qkv(qkv, H)
    assert o.shape == (B, T, H * D)
    do = torch.randn_like(o)
    o.backward(do)

    qf = qkv.detach().float().requires_grad_(True)
    q, k, v = qf.permute(2, 0, 3, 1, 4).unbind(0)
    s = (q @ k.transpose(-2, -1)) * (D ** -0.5)
    ref = (torch.softmax(s, dim=-1) @ v).transpose(1, 2).reshape(B, T, H * D)
    ref.backward(do.float())

    np.testing.assert_allclose(
        o.detach().float().cpu().numpy(), ref.detach().cpu().numpy(),
        rtol=0.05, atol=0.03,
    )
    np.testing.assert_allclose(
        qkv.grad.float().cpu().numpy(), qf.grad.cpu().numpy(),
        rtol=0.1, atol=0.06,
    )


def test_fused_add_layernorm():
    """Fused residual-add+LN matches the fp32 composition, forward and
    backward (both inputs, gamma/beta), including the downstream use of
    the summed output."""
    from vit_10b_fsdp_example_amd.ops import fused_add_layer_norm

    torch.manual_seed(4)
    N, D = 512, 1024
    x = torch.randn(4, N // 4, D, device=_dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    r = torch.randn_like(x, requires_grad=True)
    w = torch.randn(D, device=_dev(), dtype=torch.bfloat16, requires_grad=True)
    b = torch.randn(D, device=_dev(), dtype=torch.bfloat16, requires_grad=True)

    s, y = fused_add_layer_norm(x, r, w, b, 1e-6)
    out = s * 0.5 + y * 2.0  # use both outputs downstream
    dy = torch.randn_like(out)
    out.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    rf = r.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    sf = xf + rf
    yf = torch.nn.functional.layer_norm(sf, (D,), wf, bf, 1e-6)
    (sf * 0.5 + yf * 2.0).backward(dy.float())

    np.testing.assert_allclose(s.detach().float().cpu(), sf.detach().cpu(),
                               rtol=0.05, atol=0.03)
    np.testing.assert_allclose(y.detach().float().cpu(), yf.detach().cpu(),
                               rtol=0.05, atol=0.05)
    np.testing.assert_allclose(x.grad.float().cpu(), xf.grad.cpu(),
                               rtol=0.1, atol=0.1)
    np.testing.assert_allclose(r.grad.float().cpu(), rf.grad.cpu(),
                               rtol=0.1, atol=0.1)
    np.testing.assert_allclose(w.grad.float().cpu(), wf.grad.cpu(),
                               rtol=0.05, atol=0.02 * N ** 0.5)
    np.testing.assert_allclose(b.grad.float().cpu(), bf.grad.cpu(),
                               rtol=0.05, atol=0.02 * N ** 0.5)


def test_wgrad_gemm_vs_reference():
    """Native wgrad (A^T B via hardware transpose reads) vs fp32 matmul,
    including the fused bias column sums."""
    torch.manual_seed(5)
    K, M, N = 2048, 512, 256
    a = torch.randn(K, M, device=_dev(), dtype=torch.bfloat16)
    b = torch.randn(K, N, device=_dev(), dtype=torch.bfloat16)
    c, db = EXT.wgrad_gemm(a, b, True)
    ref = a.float().t() @ b.float()
    ref_db = a.float().sum(0)
    scale = ref.abs().max().item()
    np.testing.assert_allclose(c.float().cpu().numpy(), ref.cpu().numpy(),
                               atol=0.02 * scale, rtol=0.05)
    np.testing.assert_allclose(db.cpu().numpy(), ref_db.cpu().numpy(),
                               atol=0.02 * ref_db.abs().max().item() + 0.5,
                               rtol=0.05)


def test_native_linear_grads(monkeypatch):
    """NativeLinear's backward (native wgrad + fused dbias) matches
    nn.Linear's autograd (native path forced on for the test)."""
    from vit_10b_fsdp_example_amd.ops import NativeLinear
    from vit_10b_fsdp_example_amd.ops import linear as linear_mod

    monkeypatch.setattr(linear_mod, "_NATIVE_WGRAD", True)

    torch.manual_seed(6)
    K, N, M = 1024, 256, 512  # qualifies for the native path
    lin = NativeLinear(N, M).to(_dev(), torch.bfloat16)
    x = torch.randn(4, K // 4, N, device=_dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    y = lin(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    import torch.nn as nn
    ref = nn.Linear(N, M).to(_dev(), torch.bfloat16)
    with torch.no_grad():
        ref.weight.copy_(lin.weight)
        ref.bias.copy_(lin.bias)
    xr = x.detach().clone().requires_grad_(True)
    ref(xr).backward(dy)

    for got, want, name in [
        (x.grad, xr.grad, "dx"),
        (lin.weight.grad, ref.weight.grad, "dw"),
        (lin.bias.grad, ref.bias.grad, "db"),
    ]:
        g, w = got.float().cpu().numpy(), want.float().cpu().numpy()
        scale = max(abs(w).max(), 1.0)
        np.testing.assert_allclose(g, w, atol=0.03 * scale, rtol=0.1,
                                   err_msg=name)


def test_wgrad_dispatch_mode_real_kernel(monkeypatch):
    """NativeWgradMode (VITFSDP_NATIVE_WGRAD=2 path) on GPU: the
    dispatcher interception routes the AddmmBackward dW mm to
    csrc/wgemm.hip and matches stock autograd, with the stock addmm
    node left in place (see tests/test_native_wgrad_dispatch.py for
    the early-stop rationale)."""
    from vit_10b_fsdp_example_amd.ops import NativeWgradMode
    import vit_10b_fsdp_example_amd.ops.linear as linear_mod
    import torch.nn as nn

    monkeypatch.setattr(linear_mod, "_MIN_TILES", 0)  # small test shape
    torch.manual_seed(7)
    K, IN, OUT = 2048, 512, 768  # passes the 256x256x64 gate
    lin = nn.Linear(IN, OUT).to(_dev(), torch.bfloat16)
    x = torch.randn(K, IN, device=_dev(), dtype=torch.bfloat16,
                    requires_grad=True)
    dy = torch.randn(K, OUT, device=_dev(), dtype=torch.bfloat16)

    # stock baseline
    lin(x).backward(dy)
    ref_w = lin.weight.grad.float().cpu().numpy()
    ref_x = x.grad.float().cpu().numpy()
    lin.weight.grad = lin.bias.grad = x.grad = None

    mode = NativeWgradMode()
    with mode:
        lin(x).backward(dy)
    assert mode.hits == 1, "dW mm was not intercepted on the GPU path"

    got_w = lin.weight.grad.float().cpu().numpy()
    got_x = x.grad.float().cpu().numpy()
    scale = max(abs(ref_w).max(), 1.0)
    np.testing.assert_allclose(got_w, ref_w, atol=0.03 * scale, rtol=0.1,
                               err_msg="dw")
    np.testing.assert_allclose(got_x, ref_x, atol=1e-3, rtol=1e-3,
                               err_msg="dx")


def test_lt_gemm_matches_matmul():
    """_C.lt_gemm (hipblaslt-ext apply path for offline-searched
    algorithm indices, csrc/ltgemm.cpp) computes the same row-major
    bf16 GEMM as torch.matmul when using the library heuristic (-1).
    Validates the column-major duality before any tuned index is wired
    in (ROADMAP item 5)."""
    torch.manual_seed(8)
    M, K, N = 512, 640, 384  # deliberately all-distinct dims
    a = torch.randn(M, K, device=_dev(), dtype=torch.bfloat16)
    b = torch.randn(K, N, device=_dev(), dtype=torch.bfloat16)
    try:
        d = EXT.lt_gemm(a, b, -1)
    except RuntimeError as exc:  # library/setup quirk, not a mapping bug
        pytest.skip(f"lt_gemm unavailable: {exc}")
    ref = (a.float() @ b.float())
    scale = ref.abs().max().item()
    np.testing.assert_allclose(d.float().cpu().numpy(), ref.cpu().numpy(),
                               atol=0.02 * scale, rtol=0.05)
    # fused bias epilogue: per-output-feature vector of length N
    bias = torch.randn(N, device=_dev(), dtype=torch.bfloat16)
    db = EXT.lt_gemm(a, b, -1, bias)
    refb = ref + bias.float()
    np.testing.assert_allclose(db.float().cpu().numpy(), refb.cpu().numpy(),
                               atol=0.02 * scale, rtol=0.05)


def test_fused_adamw_bf16_comm_grads_prescale():
    """The engine's comm-dtype handoff: bf16 _comm_grad + _grad_prescale
    (+ deferred clip scale) must match torch.optim.AdamW fed the
    explicitly scaled fp32 gradient (fsdp.py _finalize_unit contract)."""
    torch.manual_seed(1)
    n = 65539  # scalar-tail exercise
    from vit_10b_fsdp_example_amd.ops import FusedAdamW

    base = torch.randn(n, device=_dev())
    g_bf16 = torch.randn(n, device=_dev()).to(torch.bfloat16)
    prescale = 1.0 / 8.0
    clip = torch.tensor(0.37, device=_dev())

    a = torch.nn.Parameter(base.clone())
    a._comm_grad = g_bf16.clone()
    a._grad_prescale = prescale
    a._deferred_grad_scale = clip
    opt1 = FusedAdamW([a], lr=1e-3, weight_decay=0.1)
    opt1.step()
    assert a._comm_grad is None  # consumed

    b = torch.nn.Parameter(base.clone())
    b.grad = g_bf16.to(torch.float32) * (prescale * float(clip))
    opt2 = torch.optim.AdamW([b], lr=1e-3, weight_decay=0.1)
    opt2.step()
    np.testing.assert_allclose(
        a.detach().cpu().numpy(), b.detach().cpu().numpy(), rtol=1e-5, atol=1e-6
    )


def test_multi_tensor_sqnorm_scale_bf16():
    from vit_10b_fsdp_example_amd.ops import local_sqnorm, scale_

    torch.manual_seed(2)
    ts = [
        torch.randn(1000 + i * 9, device=_dev()).to(torch.bfloat16)
        for i in range(4)
    ] + [torch.randn(777, device=_dev())]  # mixed fp32+bf16 list
    ref = sum(float(t.float().pow(2).sum()) for t in ts)
    got = float(local_sqnorm(ts))
    assert abs(got - ref) / ref < 1e-4
    refs = [(t.float() * 0.25).to(t.dtype).clone() for t in ts]
    scale_(ts, torch.tensor(0.25, device=_dev()))
    for t, r in zip(ts, refs):
        np.testing.assert_allclose(
            t.float().cpu().numpy(), r.float().cpu().numpy(), rtol=1e-2,
            atol=1e-3,
        )


def test_lt_gemm_strided_views_match_matmul():
    """lt_gemm must accept transposed views copy-free: fwd (a, W^T),
    dgrad (dy, W), wgrad (dy^T, x) all vs torch.matmul."""
    torch.manual_seed(5)
    tok, din, dout = 512, 256, 384
    x = torch.randn(tok, din, device=_dev()).to(torch.bfloat16)
    w = torch.randn(dout, din, device=_dev()).to(torch.bfloat16)
    dy = torch.randn(tok, dout, device=_dev()).to(torch.bfloat16)
    for a, b in [(x, w.t()), (dy, w), (dy.t(), x)]:
        got = EXT.lt_gemm(a, b, -1)
        ref = torch.matmul(a.float(), b.float())
        err = (got.float() - ref).abs().max() / (ref.abs().max() + 1e-6)
        assert float(err) < 3e-2, (a.shape, b.shape, float(err))


def test_lt_gemm_gelu_matches_tanh_reference():
    """hipblaslt-ext GELU_AUX epilogue — SKIPPED when the library ships
    no algorithms for it on this stack (probed: ROCm 7.2 / gfx950 has
    none; the live fusion path is fwd_gemm_gelu instead)."""
    torch.manual_seed(6)
    tok, din, dout = 512, 256, 384
    x = torch.randn(tok, din, device=_dev()).to(torch.bfloat16)
    w = torch.randn(dout, din, device=_dev()).to(torch.bfloat16) * 0.05
    bias = torch.randn(dout, device=_dev()).to(torch.bfloat16)
    try:
        out, pre = EXT.lt_gemm_gelu(x, w.t(), bias, -1)
    except RuntimeError as exc:
        assert "no heuristic algorithm" in str(exc)
        pytest.skip("hipBLASLt ships no GELU_AUX algorithms on this stack")
    pre_ref = torch.matmul(x.float(), w.t().float()) + bias.float()
    out_ref = torch.nn.functional.gelu(pre_ref, approximate="tanh")
    assert float((pre.float() - pre_ref).abs().max()) < 0.05
    assert float((out.float() - out_ref).abs().max()) < 0.05


def test_lt_gemm_dgelu_bgrad_matches_reference():
    torch.manual_seed(7)
    tok, dout, hid = 512, 256, 1024
    dy = torch.randn(tok, dout, device=_dev()).to(torch.bfloat16)
    w2 = torch.randn(dout, hid, device=_dev()).to(torch.bfloat16) * 0.05
    pre = torch.randn(tok, hid, device=_dev()).to(torch.bfloat16)
    try:
        dpre, dbias = EXT.lt_gemm_dgelu_bgrad(dy, w2, pre, -1)
    except RuntimeError as exc:
        assert "no heuristic algorithm" in str(exc)
        pytest.skip("hipBLASLt ships no DGELU algorithms on this stack")

    dgelu_out_ref = torch.matmul(dy.float(), w2.float())
    p = pre.float().detach().requires_grad_(True)
    torch.nn.functional.gelu(p, approximate="tanh").backward(dgelu_out_ref)
    dpre_ref = p.grad
    rel = (dpre.float() - dpre_ref).abs().max() / (dpre_ref.abs().max() + 1e-6)
    assert float(rel) < 3e-2
    dbias_ref = dpre_ref.sum(0)
    relb = (dbias.float() - dbias_ref).abs().max() / (
        dbias_ref.abs().max() + 1e-6
    )
    assert float(relb) < 3e-2


def test_fused_gelu_dispatch_gpu(monkeypatch):
    """End-to-end dispatch fusion with the real fwd_gemm_gelu kernel on
    a small MLP: fused-on must match the STOCK eager erf composition
    (same numerics up to GEMM rounding), under checkpointing, gradients
    included."""
    import torch.nn.functional as F
    from torch.utils.checkpoint import checkpoint
    from vit_10b_fsdp_example_amd.ops import linear as linmod

    tok, d, hid = 256, 128, 512
    monkeypatch.setenv("VITFSDP_FUSED_GELU", "1")
    monkeypatch.setattr(linmod, "_FGEMM_MIN_TILES", 0)
    monkeypatch.setitem(linmod._GELU_CFG, "d", d)
    monkeypatch.setitem(linmod._GELU_CFG, "hid", hid)

    torch.manual_seed(8)
    dt = torch.bfloat16
    w1 = (torch.randn(hid, d, device=_dev()) * 0.05).to(dt).requires_grad_(True)
    b1 = torch.zeros(hid, device=_dev(), dtype=dt).requires_grad_(True)
    w2 = (torch.randn(d, hid, device=_dev()) * 0.05).to(dt).requires_grad_(True)
    b2 = torch.zeros(d, device=_dev(), dtype=dt).requires_grad_(True)
    x = torch.randn(tok, d, device=_dev()).to(dt).requires_grad_(True)

    def block(t):
        return F.linear(F.gelu(F.linear(t, w1, b1)), w2, b2)

    with linmod.TunedGemmMode(table={}, native_wgrad=False) as m:
        y = checkpoint(block, x, use_reentrant=False)
        y.float().pow(2).sum().backward()
    assert m.gelu_hits == 2, m.gelu_hits  # forward + recompute

    grads = [t.grad.clone() for t in (x, w1, b1, w2, b2)]
    for t in (x, w1, b1, w2, b2):
        t.grad = None

    yr = block(x)  # stock erf path
    yr.float().pow(2).sum().backward()
    assert float((y.float() - yr.float()).abs().max()) < 0.02
    for got, (t, name) in zip(
        grads, [(x, "x"), (w1, "w1"), (b1, "b1"), (w2, "w2"), (b2, "b2")]
    ):
        ref = t.grad.float()
        rel = (got.float() - ref).abs().max() / (ref.abs().max() + 1e-6)
        assert float(rel) < 3e-2, (name, float(rel))



def test_fwd_gemm_vs_reference():
    """csrc/fgemm.hip (hand-written CDNA4 forward Linear GEMM) vs fp32
    torch reference, with and without the fused bias."""
    torch.manual_seed(9)
    M, N, K = 512, 256, 320
    x = torch.randn(M, K, device=_dev()).to(torch.bfloat16)
    w = torch.randn(N, K, device=_dev()).to(torch.bfloat16)
    bias = torch.randn(N, device=_dev()).to(torch.bfloat16)
    ref = x.float() @ w.float().t()
    got = EXT.fwd_gemm(x, w)
    rel = (got.float() - ref).abs().max() / ref.abs().max()
    assert float(rel) < 2e-2, float(rel)
    got_b = EXT.fwd_gemm(x, w, bias)
    ref_b = ref + bias.float()
    rel_b = (got_b.float() - ref_b).abs().max() / ref_b.abs().max()
    assert float(rel_b) < 2e-2, float(rel_b)


def test_fwd_gemm_gelu_erf_vs_reference():
    """fgemm's fused GELU epilogue must match torch's ERF-exact
    F.gelu(linear(x)) — the reference numerics, unlike hipBLASLt's
    tanh-approx epilogue."""
    torch.manual_seed(10)
    M, N, K = 512, 256, 320
    x = torch.randn(M, K, device=_dev()).to(torch.bfloat16)
    w = (torch.randn(N, K, device=_dev()) * 0.05).to(torch.bfloat16)
    bias = torch.randn(N, device=_dev()).to(torch.bfloat16)
    out, pre = EXT.fwd_gemm_gelu(x, w, bias)
    pre_ref = x.float() @ w.float().t() + bias.float()
    out_ref = torch.nn.functional.gelu(pre_ref)  # erf-exact
    assert float((pre.float() - pre_ref).abs().max()) < 0.05
    assert float((out.float() - out_ref).abs().max()) < 0.05


@pytest.mark.parametrize(
    "B,H,T,D",
    [
        (2, 3, 128, 160),
        (1, 1, 1024, 160),  # long context (448px class)
        (1, 2, 200, 160),   # ragged: T not a tile multiple
    ],
)
def test_fmha_dropout_in_kernel_fwd_bwd(B, H, T, D):
    """In-kernel attention dropout: the kernel must match the math
    composition evaluated with the EXACT mask (bit-reproduced from the
    counter hash), forward and backward — including long-context and
    ragged tile shapes."""
    from vit_10b_fsdp_example_amd.ops.attention import (
        dropout_mask_reference,
    )

    torch.manual_seed(0)
    p, seed, scale = 0.3, 98765, D ** -0.5
    mk = lambda: (torch.randn(B, H, T, D, device=_dev()) * 0.5).to(
        torch.bfloat16
    )
    q, k, v = mk(), mk(), mk()
    for t in (q, k, v):
        t.requires_grad_(True)

    o, lse = EXT.fmha_fwd(q, k, v, scale, p, seed)

    # exact-mask reference in fp32
    mask = torch.empty(B, H, T, T, device=_dev())
    for b in range(B):
        for h in range(H):
            rows = torch.arange(T) + (b * H + h) * T
            mask[b, h] = dropout_mask_reference(
                seed, rows, torch.arange(T), p
            ).float().to(_dev())
    qf, kf, vf = (t.detach().float().requires_grad_(True) for t in (q, k, v))
    probs = torch.softmax(qf @ kf.transpose(-2, -1) * scale, dim=-1)
    o_ref = (probs * mask / (1 - p)) @ vf
    err = (o.float() - o_ref).abs().max()
    assert float(err) < 0.05, float(err)

    # same seed -> identical output; different seed -> different
    o2, _ = EXT.fmha_fwd(q, k, v, scale, p, seed)
    assert torch.equal(o, o2)
    o3, _ = EXT.fmha_fwd(q, k, v, scale, p, seed + 1)
    assert not torch.equal(o, o3)

    # measured drop rate ~ p (the output of a row with all-ones V would
    # be the masked row-mean; check via probs-mask stats instead)
    keep_frac = mask.mean().item()
    assert abs(keep_frac - (1 - p)) < 0.02

    # backward vs the exact-mask reference
    do = torch.randn_like(o)
    dq, dk, dv = EXT.fmha_bwd(do, q, k, v, o, lse, scale, p, seed)
    o_ref.backward(do.float())
    for got, ref, name in [(dq, qf.grad, "dq"), (dk, kf.grad, "dk"),
                           (dv, vf.grad, "dv")]:
        rel = (got.float() - ref).abs().max() / (ref.abs().max() + 1e-6)
        assert float(rel) < 6e-2, (name, float(rel))


def test_attention_dropout_autograd_path():
    """attention_qkv with dropout_p > 0 must use the flash kernel (no
    O(T^2) fallback) and train: grads finite, loss decreasing on a tiny
    overfit."""
    from vit_10b_fsdp_example_amd.ops import attention_qkv

    torch.manual_seed(1)
    B, T, H, D = 2, 64, 2, 64
    qkv = (torch.randn(B, T, 3, H, D, device=_dev()) * 0.2).to(
        torch.bfloat16
    ).requires_grad_(True)
    torch.manual_seed(7)
    out = attention_qkv(qkv, H, dropout_p=0.25, training=True)
    assert out.shape == (B, T, H * D)
    out.float().pow(2).sum().backward()
    assert torch.isfinite(qkv.grad.float()).all()
    # reproducibility through torch.manual_seed (seed drawn from CPU RNG)
    g1 = qkv.grad.clone()
    qkv.grad = None
    torch.manual_seed(7)
    out2 = attention_qkv(qkv, H, dropout_p=0.25, training=True)
    out2.float().pow(2).sum().backward()
    assert torch.equal(out, out2) and torch.equal(g1, qkv.grad)
    # eval mode: no dropout -> deterministic equality with p=0 path
    oe = attention_qkv(qkv, H, dropout_p=0.25, training=False)
    o0 = attention_qkv(qkv, H, dropout_p=0.0, training=False)
    assert torch.equal(oe, o0)


def test_mfma32_probe_layout():
    """Pin the 32x32x16 bf16 MFMA fragment layout (csrc/fgemm.hip's
    wide-MFMA variant relies on it)."""
    torch.manual_seed(0)
    a = torch.randn(32, 16, device=_dev())
    b = (torch.arange(16 * 32, device=_dev(), dtype=torch.float32)
         .reshape(16, 32) % 7) - 3.0 + 0.1 * torch.randn(16, 32, device=_dev())
    c = EXT.mfma32_probe(a, b)
    ref = a.to(torch.bfloat16).float() @ b.to(torch.bfloat16).float()
    err = (c - ref).abs().max() / ref.abs().max()
    assert float(err) < 2e-2, float(err)
