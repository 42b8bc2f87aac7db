#!/usr/bin/env python3
"""Per-GEMM comparison of nn.Linear's fused autograd backward vs our
_NativeLinearFn (ROADMAP #4): identifies which gradient's kernel
selection makes the custom Function slower end-to-end even when the
native dW kernel wins in isolation.  Run on an MI355X:

    python benchmarks/bench_linear_bwd.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn


def bench_case(B, T, n_in, n_out, iters=10, label=""):
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    x = torch.randn(B, T, n_in, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    dy = torch.randn(B, T, n_out, device=dev, dtype=torch.bfloat16)

    lin = nn.Linear(n_in, n_out).to(dev, torch.bfloat16)

    def run_stock():
        y = torch.nn.functional.linear(x, lin.weight, lin.bias)
        y.backward(dy)
        x.grad = None
        lin.weight.grad = None
        lin.bias.grad = None

    from vit_10b_fsdp_example_amd.ops.linear import _NativeLinearFn
    from vit_10b_fsdp_example_amd.ops import linear as lmod

    def run_fn(native):
        lmod._NATIVE_WGRAD = native
        y = _NativeLinearFn.apply(x, lin.weight, lin.bias)
        y.backward(dy)
        x.grad = None
        lin.weight.grad = None
        lin.bias.grad = None

    for fn, name in [(run_stock, "stock"),
                     (lambda: run_fn(False), "Fn/lib-wgrad"),
                     (lambda: run_fn(True), "Fn/native-wgrad")]:
        fn()  # warmup
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        dt = (time.time() - t0) / iters
        print(f"  {label} {name:16s}: {dt * 1e3:7.2f} ms")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    print("qkv shape (5120 -> 15360):")
    bench_case(128, 256, 5120, 15360, label="qkv")
    print("proj shape (5120 -> 5120):")
    bench_case(128, 256, 5120, 5120, label="proj")
    print("fc1 shape (5120 -> 20480):")
    bench_case(128, 256, 5120, 20480, label="fc1")
