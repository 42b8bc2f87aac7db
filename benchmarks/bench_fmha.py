#!/usr/bin/env python3
"""FMHA kernel micro-benchmark: TF/s on ViT shapes, forward and backward,
plus a numerics check against the fp32 math reference.  Run on an MI355X:

    python benchmarks/bench_fmha.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_shape(B, H, T, D, iters=20):
    import vit_10b_fsdp_example_amd._C as C

    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, H, T, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    scale = D ** -0.5

    o, lse = C.fmha_fwd(q, k, v, scale)
    # numerics check
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    ref = torch.softmax(s, dim=-1) @ v.float()
    err = (o.float() - ref).abs().max().item()

    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        o, lse = C.fmha_fwd(q, k, v, scale)
    torch.cuda.synchronize()
    fwd_t = (time.time() - t0) / iters

    t0 = time.time()
    for _ in range(iters):
        C.fmha_bwd(do, q, k, v, o, lse, scale)
    torch.cuda.synchronize()
    bwd_t = (time.time() - t0) / iters

    fwd_flops = 4.0 * B * H * T * T * D
    bwd_flops = 2.5 * fwd_flops  # dQ,dK,dV,dP,S-recompute
    print(
        f"B{B} H{H} T{T} D{D}: fwd {fwd_t * 1e3:7.2f} ms "
        f"{fwd_flops / fwd_t / 1e12:7.1f} TF/s | "
        f"bwd {bwd_t * 1e3:7.2f} ms {bwd_flops / bwd_t / 1e12:7.1f} TF/s | "
        f"max_err {err:.4f}"
    )


if __name__ == "__main__":
    assert torch.cuda.is_available()
    bench_shape(128, 32, 256, 160)   # ViT-10B per-GPU shape
    bench_shape(128, 16, 256, 64)    # ViT-Large
    bench_shape(8, 32, 1024, 160)    # long-sequence (448px) class
    bench_shape(128, 64, 256, 128)   # ViT-60B-class head shape
