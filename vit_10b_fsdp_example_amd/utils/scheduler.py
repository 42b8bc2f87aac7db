"""Linear-warmup + cosine-decay LR schedule (capability parity with
/root/reference/utils.py:11-21; schedule length computed by the trainer as
len(dataset) // batch_size * num_epochs, reference run_vit_training.py:238-240)."""

import math

import torch


def get_warmup_cosine_scheduler(optimizer, warmup_iteration, max_iteration):
    """LR ratio: step/warmup during warmup, then a half cosine from 1 to 0
    across the remaining steps."""

    def _ratio(step):
        if step < warmup_iteration:
            return step / float(warmup_iteration)
        denom = max(max_iteration - warmup_iteration, 1)
        progress = (step - warmup_iteration) / float(denom)
        return 0.5 * (1.0 + math.cos(math.pi * progress))

    return torch.optim.lr_scheduler.LambdaLR(optimizer, _ratio)
