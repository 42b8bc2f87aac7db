"""SmoothedValue and LR schedule behavior (reference utils.py:11-21,60-102)."""

import math

import pytest
import torch

from vit_10b_fsdp_example_amd.utils import (
    SmoothedValue, get_warmup_cosine_scheduler,
)


def test_smoothed_value_window():
    sv = SmoothedValue(window_size=3)
    for v in [1.0, 2.0, 3.0, 4.0]:
        sv.update(v, batch_size=1)
    assert sv.avg == pytest.approx((2 + 3 + 4) / 3)
    assert sv.median == pytest.approx(3.0)
    assert sv.global_avg == pytest.approx((1 + 2 + 3 + 4) / 4)
    assert sv.get_latest() == 4.0


def test_smoothed_value_weighted():
    sv = SmoothedValue(window_size=5)
    sv.update(1.0, batch_size=3)
    sv.update(2.0, batch_size=1)
    assert sv.avg == pytest.approx((1.0 * 3 + 2.0 * 1) / 4)


def test_warmup_cosine_schedule():
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=1.0)
    sched = get_warmup_cosine_scheduler(opt, warmup_iteration=10, max_iteration=110)
    lrs = []
    for _ in range(110):
        lrs.append(opt.param_groups[0]["lr"])
        opt.step()
        sched.step()
    # linear warmup
    assert lrs[0] == pytest.approx(0.0)
    assert lrs[5] == pytest.approx(0.5)
    # peak at end of warmup
    assert lrs[10] == pytest.approx(1.0)
    # cosine midpoint
    assert lrs[60] == pytest.approx(0.5 * (1 + math.cos(math.pi * 0.5)), abs=1e-6)
    # near-zero at the end
    assert lrs[-1] < 0.01
