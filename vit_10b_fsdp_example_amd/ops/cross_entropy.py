"""Fused softmax cross-entropy (kernel K9 in SURVEY.md §2D).

The reference uses torch.nn.CrossEntropyLoss over 1000 classes
(run_vit_training.py:229,262).  On GPU the fused HIP kernel computes
log-softmax + NLL in one pass (fp32 accumulation from bf16 logits) and
the backward produces dLogits = (softmax - onehot) * dLoss / N in one
kernel.  Mean reduction, matching the reference's default.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._extension import ext, use_hip


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        loss, lse = ext().cross_entropy_fwd(logits, target)
        ctx.save_for_backward(logits, target, lse)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, target, lse = ctx.saved_tensors
        dlogits = ext().cross_entropy_bwd(dloss, logits, target, lse)
        return dlogits, None


def cross_entropy(logits, target):
    """Mean-reduced cross entropy on [N, C] logits, [N] int64 targets."""
    if logits.dtype == torch.bfloat16 and use_hip(logits):
        return _CrossEntropyFn.apply(logits.contiguous(), target.contiguous())
    return F.cross_entropy(logits.float(), target)


class CrossEntropyLoss(nn.Module):
    """Drop-in replacement for torch.nn.CrossEntropyLoss() as the
    reference constructs it (mean reduction, no label smoothing)."""

    def forward(self, logits, target):
        return cross_entropy(logits, target)
