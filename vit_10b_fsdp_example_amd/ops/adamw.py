"""Fused multi-tensor AdamW (kernel K8 in SURVEY.md §2D).

The reference runs torch.optim.AdamW over the *sharded* params
(run_vit_training.py:237: lr 1e-3, weight_decay 0.1), so the optimizer
state is sharded by construction (ZeRO-3).  Here the step over the
fp32 master shards runs as one chunked HIP launch (multi-tensor: a
chunk table of pointers, decoupled weight decay, fp32 math) instead of
a python loop of elementwise ops.  CPU path uses torch._foreach with
identical math, which is what the FSDP-vs-DDP CPU parity tests rely on.
"""

import math

import torch

from ._extension import ext


class FusedAdamW(torch.optim.Optimizer):
    """Decoupled-weight-decay Adam, identical math to torch.optim.AdamW:

        m = b1*m + (1-b1)*g        v = b2*v + (1-b2)*g^2
        p -= lr * wd * p           (decoupled decay, applied first)
        p -= lr * (m/(1-b1^t)) / (sqrt(v/(1-b2^t)) + eps)
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8, weight_decay=1e-2):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p)
                    state["exp_avg_sq"] = torch.zeros_like(p)
                state["step"] += 1
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state["exp_avg"])
                exp_avg_sqs.append(state["exp_avg_sq"])
            if not params:
                continue

            # all params in a group share the step count in practice
            # (they are stepped together every iteration)
            step_t = self.state[params[0]]["step"]
            beta1, beta2 = group["betas"]
            bias_c1 = 1.0 - beta1 ** step_t
            bias_c2 = 1.0 - beta2 ** step_t

            # deferred gradient clipping (clip_grad_norm_(defer_scale=True)
            # leaves a device-scalar coefficient on the params instead of
            # running a separate full scale pass over the grads)
            grad_scale = None
            for p in params:
                s = getattr(p, "_deferred_grad_scale", None)
                if s is not None:
                    grad_scale = s if grad_scale is None else grad_scale
                    p._deferred_grad_scale = None

            if params[0].is_cuda and ext() is not None:
                # bf16 comm mirrors (FSDP attaches _bf16_mirror so the
                # gather path never re-casts the fp32 master): the step
                # kernel refreshes them in its epilogue
                mirrors = [getattr(p, "_bf16_mirror", None) for p in params]
                if not any(m is not None for m in mirrors):
                    mirrors = []
                ext().fused_adamw(
                    params,
                    grads,
                    exp_avgs,
                    exp_avg_sqs,
                    mirrors,
                    group["lr"],
                    beta1,
                    beta2,
                    group["eps"],
                    group["weight_decay"],
                    bias_c1,
                    bias_c2,
                    grad_scale,
                )
                for p, m in zip(params, mirrors or []):
                    if m is not None:
                        p._mirror_fresh = True
            else:
                if grad_scale is not None:
                    grads = torch._foreach_mul(grads, float(grad_scale))
                self._foreach_step(
                    params, grads, exp_avgs, exp_avg_sqs,
                    group["lr"], beta1, beta2, group["eps"],
                    group["weight_decay"], bias_c1, bias_c2,
                )
                for p in params:
                    m = getattr(p, "_bf16_mirror", None)
                    if m is not None:
                        m.copy_(p.detach().to(m.dtype))
                        p._mirror_fresh = True
        return loss

    @staticmethod
    def _foreach_step(params, grads, exp_avgs, exp_avg_sqs, lr, beta1, beta2,
                      eps, weight_decay, bias_c1, bias_c2):
        if weight_decay != 0.0:
            torch._foreach_mul_(params, 1.0 - lr * weight_decay)
        torch._foreach_lerp_(exp_avgs, grads, 1.0 - beta1)
        torch._foreach_mul_(exp_avg_sqs, beta2)
        torch._foreach_addcmul_(exp_avg_sqs, grads, grads, 1.0 - beta2)
        step_size = lr / bias_c1
        denom = torch._foreach_sqrt(exp_avg_sqs)
        torch._foreach_div_(denom, math.sqrt(bias_c2))
        torch._foreach_add_(denom, eps)
        torch._foreach_addcdiv_(params, exp_avgs, denom, -step_size)
