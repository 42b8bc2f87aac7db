"""Minimal fmha fwd loop for PMC counter collection."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import vit_10b_fsdp_example_amd._C as C

dev = torch.device("cuda", 0)
torch.manual_seed(0)
q = torch.randn(128, 32, 256, 160, device=dev, dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
for _ in range(10):
    o, lse = C.fmha_fwd(q, k, v, 160 ** -0.5)
torch.cuda.synchronize()
print("done")
