"""TunedGemmMode routing logic (ops/linear.py): the dual-shape keys
must match csrc/tools/hipblaslt_search.cpp's problem rows, the mode must
reroute exactly the tabled training GEMMs (forward addmm with a
transposed weight view, dgrad mm, wgrad mm), and checkpoint early-stop
must survive the mode being active through forward + backward.
"""

import json
import subprocess
import sys

import torch
import torch.nn.functional as F

from vit_10b_fsdp_example_amd.ops.linear import (
    TunedGemmMode, _dual_key, _op_layout,
)


def test_op_layout_classification():
    a = torch.randn(6, 8)
    assert _op_layout(a) == "N"
    assert _op_layout(a.t()) == "T"
    assert _op_layout(a[:, ::2]) is None  # neither row- nor col-contiguous
    assert _op_layout(torch.randn(6)) is None


def test_dual_keys_match_search_rows():
    """fwd / dgrad / wgrad of a Linear(in=8, out=12) on 16 rows must key
    exactly as the search tool writes them: (opA, opB, m=out|in, n=tok,
    k) in the column-major dual."""
    tok, din, dout = 16, 8, 12
    x = torch.randn(tok, din)
    w = torch.randn(dout, din)
    dy = torch.randn(tok, dout)
    # forward x @ W^T -> TN, m=out, n=tok, k=in
    assert _dual_key(x, w.t()) == ("T", "N", dout, tok, din)
    # dgrad dy @ W -> NN, m=in, n=tok, k=out
    assert _dual_key(dy, w) == ("N", "N", din, tok, dout)
    # wgrad: AddmmBackward emits mm(dy.t(), x) -> NT, m=in, n=out, k=tok
    assert _dual_key(dy.t(), x) == ("N", "T", din, dout, tok)


def test_mode_routes_only_tabled_shapes():
    tok, din, dout = 16, 8, 12
    table = {
        ("T", "N", dout, tok, din): 111,  # fwd
        ("N", "T", din, dout, tok): 333,  # wgrad
    }
    calls = []

    def handler(a, b, idx, bias):
        calls.append((idx, bias is not None))
        out = a @ b
        return out + bias if bias is not None else out

    x = torch.randn(tok, din, requires_grad=True)
    w = torch.randn(dout, din, requires_grad=True)
    bias = torch.randn(dout, requires_grad=True)
    with TunedGemmMode(table=table, native_wgrad=False, handler=handler) as m:
        y = F.linear(x, w, bias)
        y.pow(2).sum().backward()  # sum() alone gives a degenerate
        # zero-stride grad_output that is (correctly) not routed
    # fwd routed with bias; wgrad routed; dgrad (not tabled) untouched
    assert (111, True) in calls
    assert (333, False) in calls
    assert m.hits == 2
    # numerics equal the unrouted computation
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = bias.detach().clone().requires_grad_(True)
    y2 = F.linear(x2, w2, b2)
    y2.pow(2).sum().backward()
    assert torch.allclose(y, y2, atol=1e-6)
    assert torch.allclose(w.grad, w2.grad, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(bias.grad, b2.grad, atol=1e-5)


def test_mode_checkpoint_early_stop_preserved():
    """With the mode active around forward AND backward, non-reentrant
    checkpointing must still skip the last recompute GEMM (the round-1
    root cause, tests/test_checkpoint_earlystop.py)."""
    from torch.utils.checkpoint import checkpoint

    tok, d = 8, 4
    w1 = torch.randn(d, d, requires_grad=True)
    w2 = torch.randn(d, d, requires_grad=True)

    routed = []

    def handler(a, b, idx, bias):
        routed.append(idx)
        return a @ b

    def block(x):
        return F.linear(F.gelu(F.linear(x, w1)), w2)

    x = torch.randn(tok, d, requires_grad=True)
    table = {("T", "N", d, tok, d): 7}  # both fwd GEMMs share this shape
    with TunedGemmMode(table=table, native_wgrad=False, handler=handler):
        y = checkpoint(block, x, use_reentrant=False)
        y.pow(2).sum().backward()
    # forward: 2 routed fwd GEMMs; recompute: ONLY w1's forward re-runs
    # (early-stop skips the last GEMM of the region) -> 3 total
    assert routed.count(7) == 3, routed


def test_make_lt_table_script(tmp_path):
    csv = tmp_path / "search.csv"
    csv.write_text(
        "problem,opA,opB,m,n,k,algo_index,ms,tflops,note\n"
        "qkv_fwd,T,N,15360,32768,5120,100,4.00,1288.0,heuristic\n"
        "qkv_fwd,T,N,15360,32768,5120,200,3.60,1431.0,top1\n"  # 11% gain
        "proj_fwd,T,N,5120,32768,5120,300,1.40,1226.0,heuristic\n"
        "proj_fwd,T,N,5120,32768,5120,301,1.39,1235.0,top1\n"  # <3%: keep
    )
    out = tmp_path / "table.json"
    res = subprocess.run(
        [sys.executable, "scripts/make_lt_table.py", str(csv), "-o", str(out)],
        capture_output=True, text=True,
        cwd=__import__("os").path.dirname(__import__("os").path.dirname(
            __import__("os").path.abspath(__file__))),
    )
    assert res.returncode == 0, res.stderr
    table = json.load(open(out))["entries"]
    assert table == {
        "T,N,15360,32768,5120": {
            "index": 200, "name": "qkv_fwd", "heuristic_index": 100,
            "heuristic_ms": 4.0, "best_ms": 3.6, "gain": 1.1111,
        }
    }
