#!/usr/bin/env python3
"""On-box diagnostic for the tuned-GEMM dispatch route: is the table
loaded, do the training keys hit, and what do the individual algorithm
choices actually cost inside torch vs through _C.lt_gemm?"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from vit_10b_fsdp_example_amd.ops import linear as linmod  # noqa: E402
from vit_10b_fsdp_example_amd.ops import ext  # noqa: E402
from vit_10b_fsdp_example_amd.tuning import lt_algo_table  # noqa: E402

dev = torch.device("cuda", 0)
table = lt_algo_table()
print(f"table entries: {len(table)}", flush=True)
for k, v in sorted(table.items())[:4]:
    print("  ", k, "->", v)


def timeit(fn, iters=10):
    # cold-ish: two operand sets
    fn(0)
    torch.cuda.synchronize()
    t0 = time.time()
    for i in range(iters):
        fn(i % 2)
    torch.cuda.synchronize()
    return (time.time() - t0) / iters * 1e3


torch.manual_seed(0)
tok, din, dout = 32768, 5120, 15360
xs = [torch.randn(tok, din, device=dev, dtype=torch.bfloat16) for _ in range(2)]
ws = [torch.randn(dout, din, device=dev, dtype=torch.bfloat16) for _ in range(2)]
bs = [torch.randn(dout, device=dev, dtype=torch.bfloat16) for _ in range(2)]
dys = [torch.randn(tok, dout, device=dev, dtype=torch.bfloat16) for _ in range(2)]

fl = 2.0 * tok * din * dout / 1e12
key_fwd = ("T", "N", dout, tok, din)
idx_fwd = table.get(key_fwd, -1)
key_dg = ("N", "N", din, tok, dout)
idx_dg = table.get(key_dg, -1)
print(f"qkv fwd idx {idx_fwd}, dgrad idx {idx_dg}")

t = timeit(lambda i: F.linear(xs[i], ws[i], bs[i]))
print(f"torch F.linear(qkv fwd):        {t:7.3f} ms {fl/t*1e3:7.0f} TF")
t = timeit(lambda i: torch.matmul(xs[i], ws[i].t()))
print(f"torch matmul no-bias:           {t:7.3f} ms {fl/t*1e3:7.0f} TF")
t = timeit(lambda i: ext().lt_gemm(xs[i], ws[i].t(), -1))
print(f"lt_gemm heuristic no-bias:      {t:7.3f} ms {fl/t*1e3:7.0f} TF")
t = timeit(lambda i: ext().lt_gemm(xs[i], ws[i].t(), -1, bs[i]))
print(f"lt_gemm heuristic +bias:        {t:7.3f} ms {fl/t*1e3:7.0f} TF")
if idx_fwd >= 0:
    t = timeit(lambda i: ext().lt_gemm(xs[i], ws[i].t(), idx_fwd))
    print(f"lt_gemm tuned({idx_fwd}) no-bias: {t:7.3f} ms {fl/t*1e3:7.0f} TF")
    t = timeit(lambda i: ext().lt_gemm(xs[i], ws[i].t(), idx_fwd, bs[i]))
    print(f"lt_gemm tuned({idx_fwd}) +bias:   {t:7.3f} ms {fl/t*1e3:7.0f} TF")
t = timeit(lambda i: torch.matmul(dys[i], ws[i]))
print(f"torch dgrad matmul:             {t:7.3f} ms {fl/t*1e3:7.0f} TF")
if idx_dg >= 0:
    t = timeit(lambda i: ext().lt_gemm(dys[i], ws[i], idx_dg))
    print(f"lt_gemm tuned dgrad({idx_dg}):  {t:7.3f} ms {fl/t*1e3:7.0f} TF")

# dispatch-hit check on the real ops
with linmod.TunedGemmMode() as m:
    x = xs[0].requires_grad_(True)
    w = ws[0].requires_grad_(True)
    b = bs[0].requires_grad_(True)
    y = F.linear(x, w, b)
    y.float().pow(2).sum().backward()
print(f"dispatch hits on one qkv-shaped linear fwd+bwd: {m.hits}", flush=True)

# timed A/B of the same fwd+bwd with and without the mode
def step(mode):
    def run(i):
        x = xs[i].detach().requires_grad_(True)
        w = ws[i].detach().requires_grad_(True)
        b = bs[i].detach().requires_grad_(True)
        if mode:
            with linmod.TunedGemmMode():
                y = F.linear(x, w, b)
                y.backward(dys[i])
        else:
            y = F.linear(x, w, b)
            y.backward(dys[i])
    return run

t_off = timeit(step(False), iters=6)
t_on = timeit(step(True), iters=6)
print(f"linear fwd+bwd: mode OFF {t_off:7.3f} ms | mode ON {t_on:7.3f} ms")
