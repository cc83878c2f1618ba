"""Fused AdamW over flat fp32 master shards.

Replaces DeepSpeed FusedAdam (reference: models/model_utils.py:62-72).
The ZeRO optimizer hands us ONE flat fp32 master tensor + flat grad per
bucket, so the fused kernel is a single elementwise pass (7 eager passes →
1 HBM round-trip; memory-bound, target ~6.3 TB/s).  HIP kernel:
csrc/adamw.hip; this file holds the dispatch + the eager oracle.
"""
from __future__ import annotations


import torch

from fengshen_amd.ops import use_hip, get_ext


@torch.no_grad()
def _eager_adamw_flat_(master, grad, exp_avg, exp_avg_sq, out_param,
                       lr, beta1, beta2, eps, weight_decay, step):
    grad32 = grad.float() if grad.dtype != torch.float32 else grad
    if weight_decay != 0.0:
        master.mul_(1.0 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(grad32, alpha=1.0 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad32, grad32, value=1.0 - beta2)
    bias_correction1 = 1.0 - beta1 ** step
    bias_correction2 = 1.0 - beta2 ** step
    denom = (exp_avg_sq / bias_correction2).sqrt_().add_(eps)
    master.addcdiv_(exp_avg, denom, value=-lr / bias_correction1)
    if out_param is not None and out_param.data_ptr() != master.data_ptr():
        out_param.copy_(master)


@torch.no_grad()
def fused_adamw_flat_(master: torch.Tensor, grad: torch.Tensor,
                      exp_avg: torch.Tensor, exp_avg_sq: torch.Tensor,
                      out_param: torch.Tensor, *, lr: float, beta1: float,
                      beta2: float, eps: float, weight_decay: float, step: int):
    """In-place AdamW on a flat fp32 master; writes bf16/fp16 copy to out_param.

    master/exp_avg/exp_avg_sq: fp32 1-D, same numel.
    grad: fp32/bf16/fp16 1-D, same numel.
    out_param: model-dtype 1-D view to refresh (may alias master for fp32 runs).
    """
    if use_hip(master):
        get_ext().fused_adamw(
            master, grad, exp_avg, exp_avg_sq, out_param,
            float(lr), float(beta1), float(beta2), float(eps),
            float(weight_decay), int(step))
        return
    _eager_adamw_flat_(master, grad, exp_avg, exp_avg_sq, out_param,
                       lr, beta1, beta2, eps, weight_decay, step)


class FusedAdamW(torch.optim.Optimizer):
    """Drop-in AdamW with fp32 master weights for bf16 params.

    Used by configure_optimizers when no ZeRO strategy is active (the
    reference picks deepspeed FusedAdam / torch AdamW by strategy,
    model_utils.py:62-83).
    """

    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    master = (p.detach().float().clone()
                              if p.dtype != torch.float32 else p)
                    state["master"] = master
                    state["exp_avg"] = torch.zeros_like(master)
                    state["exp_avg_sq"] = torch.zeros_like(master)
                state["step"] += 1
                master = state["master"] if p.dtype != torch.float32 else p
                fused_adamw_flat_(
                    master.view(-1), p.grad.reshape(-1),
                    state["exp_avg"].view(-1), state["exp_avg_sq"].view(-1),
                    p.data.view(-1),
                    lr=group["lr"], beta1=beta1, beta2=beta2,
                    eps=group["eps"], weight_decay=group["weight_decay"],
                    step=state["step"])
        return loss
