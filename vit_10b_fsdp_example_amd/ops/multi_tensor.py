"""Multi-tensor gradient-norm and scaling (kernel K11 in SURVEY.md §2D).

Backs FSDP.clip_grad_norm_ (reference run_vit_training.py:270): the
local sum of squares over every grad shard in one chunked HIP launch
(fp32 accumulation), a single scalar RCCL all-reduce done by the
caller, then one fused scale launch.  CPU path uses torch._foreach.
"""

import torch

from ._extension import ext


def local_sqnorm(tensors):
    """Sum of squares over a list of tensors -> fp32 scalar tensor (on the
    tensors' device)."""
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return torch.zeros((), dtype=torch.float32)
    if tensors[0].is_cuda and ext() is not None:
        return ext().multi_tensor_sqnorm(tensors)
    acc = torch.zeros((), dtype=torch.float32, device=tensors[0].device)
    for t in tensors:
        acc += t.detach().float().pow(2).sum()
    return acc


def scale_(tensors, factor):
    """In-place multiply every tensor by a scalar (python float or 0-dim
    tensor)."""
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return
    if tensors[0].is_cuda and ext() is not None:
        if torch.is_tensor(factor):
            ext().multi_tensor_scale_tensor(tensors, factor)
        else:
            ext().multi_tensor_scale(tensors, float(factor))
        return
    if torch.is_tensor(factor):
        factor = float(factor)
    torch._foreach_mul_(tensors, factor)
