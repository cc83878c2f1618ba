"""Optimizer/scheduler factory shared by every training app.

Behavioral parity: reference fengshen/models/model_utils.py —
add_module_args (:13), configure_optimizers (:50-98: decay/no-decay split,
optimizer pick by strategy), scheduler registry incl. polynomial w/ lr_end
(:212-254), inverse_square_root (:101-191), get_total_steps (:194-209).
MI355X redesign: DeepSpeedCPUAdam/FusedAdam are replaced by our ZeRO
optimizer (fused AdamW HIP kernel inside) and FusedAdamW.
"""
from __future__ import annotations

import argparse
import math
import torch
from torch.optim.lr_scheduler import LambdaLR


def add_module_args(parent_args: argparse.ArgumentParser):
    parser = parent_args.add_argument_group("Basic Module")
    parser.add_argument("--learning_rate", default=1e-4, type=float)
    parser.add_argument("--min_learning_rate", default=1e-7, type=float)
    parser.add_argument("--lr_decay_steps", default=0, type=int)
    parser.add_argument("--lr_decay_ratio", default=1.0, type=float)
    parser.add_argument("--warmup_steps", default=0, type=int)
    parser.add_argument("--warmup_ratio", default=0.1, type=float)
    parser.add_argument("--weight_decay", default=1e-1, type=float)
    parser.add_argument("--adam_beta1", default=0.9, type=float)
    parser.add_argument("--adam_beta2", default=0.999, type=float)
    parser.add_argument("--adam_epsilon", default=1e-8, type=float)
    parser.add_argument("--model_path", default=None, type=str)
    parser.add_argument(
        "--scheduler_type", default="polynomial", type=str,
        choices=["polynomial", "linear", "cosine", "constant",
                 "constant_with_warmup", "inverse_sqrt", "direct"])
    return parent_args


def get_default_update_params(model: torch.nn.Module, weight_decay: float):
    """decay / no-decay param split (reference model_utils.py:53-60)."""
    no_decay = ["bias", "LayerNorm.weight", "layer_norm.weight", "layernorm.weight",
                "norm.weight", "ln_f.weight", "ln_1.weight", "ln_2.weight"]
    decay_params = []
    no_decay_params = []
    for n, p in model.named_parameters():
        if not p.requires_grad:
            continue
        if p.dim() < 2 or any(nd in n for nd in no_decay):
            no_decay_params.append(p)
        else:
            decay_params.append(p)
    return [
        {"params": decay_params, "weight_decay": weight_decay},
        {"params": no_decay_params, "weight_decay": 0.0},
    ]


def configure_optimizers(pl_model, model_params=None):
    """Build (optimizer, scheduler) from module hparams + trainer strategy.

    Mirrors the reference call pattern: every LightningModule's
    configure_optimizers delegates here (model_utils.py:50).
    """
    args = pl_model.hparams
    optim_groups = model_params if model_params is not None else \
        get_default_update_params(pl_model, args.weight_decay)

    trainer = pl_model.trainer
    strategy = trainer.strategy if trainer is not None else None
    betas = (getattr(args, "adam_beta1", 0.9), getattr(args, "adam_beta2", 0.999))
    eps = getattr(args, "adam_epsilon", 1e-8)

    if strategy is not None and strategy.kind == "zero" and strategy.stage == 3:
        from fengshen_amd.parallel.zero3 import Zero3Engine
        from fengshen_amd.parallel import groups as pgroups
        optimizer = Zero3Engine(
            pl_model, lr=args.learning_rate, betas=betas, eps=eps,
            weight_decay=args.weight_decay,
            process_group=pgroups.get_data_parallel_group())
    elif strategy is not None and strategy.kind == "zero" and strategy.stage >= 1:
        from fengshen_amd.parallel.zero import ZeroOptimizer
        from fengshen_amd.parallel import groups as pgroups
        optimizer = ZeroOptimizer(
            optim_groups, stage=strategy.stage, lr=args.learning_rate,
            betas=betas, eps=eps, weight_decay=args.weight_decay,
            process_group=pgroups.get_data_parallel_group(),
            bucket_numel=strategy.bucket_numel,
            overlap_comm=strategy.overlap_comm,
            cpu_offload=getattr(strategy, "cpu_offload", False))
    else:
        from fengshen_amd.ops.adamw import FusedAdamW
        optimizer = FusedAdamW(optim_groups, lr=args.learning_rate,
                               betas=betas, eps=eps,
                               weight_decay=args.weight_decay)

    total_steps = get_total_steps(trainer, args)
    warmup_steps = int(getattr(args, "warmup_steps", 0) or
                       getattr(args, "warmup_ratio", 0.0) * total_steps)
    lr_decay_steps = getattr(args, "lr_decay_steps", 0) or int(
        total_steps * getattr(args, "lr_decay_ratio", 1.0))
    scheduler = get_scheduler(
        getattr(args, "scheduler_type", "polynomial"), optimizer,
        num_warmup_steps=warmup_steps, num_training_steps=lr_decay_steps,
        lr_end=getattr(args, "min_learning_rate", 1e-7),
        lr_init=args.learning_rate)
    return {
        "optimizer": optimizer,
        "lr_scheduler": {"scheduler": scheduler, "interval": "step"},
    }


def get_total_steps(trainer, args) -> int:
    """Total optimizer steps (reference model_utils.py:194-209, DP-aware)."""
    if trainer is None:
        return getattr(args, "max_steps", 0) or 100000
    if trainer.max_steps and trainer.max_steps > 0:
        return trainer.max_steps
    return trainer.estimated_stepping_batches


# ---------------------------------------------------------------------------
# scheduler registry (reference model_utils.py:101-254)
# ---------------------------------------------------------------------------
def get_scheduler(name: str, optimizer, num_warmup_steps: int = 0,
                  num_training_steps: int = 100000, lr_end: float = 1e-7,
                  lr_init: float = 1e-4, power: float = 1.0):
    name = name.lower()
    if name == "constant":
        return LambdaLR(optimizer, lambda step: 1.0)
    if name == "constant_with_warmup":
        def fn(step):
            return min(1.0, step / max(1, num_warmup_steps))
        return LambdaLR(optimizer, fn)
    if name == "linear":
        def fn(step):
            if step < num_warmup_steps:
                return step / max(1, num_warmup_steps)
            return max(0.0, (num_training_steps - step) /
                       max(1, num_training_steps - num_warmup_steps))
        return LambdaLR(optimizer, fn)
    if name == "cosine":
        def fn(step):
            if step < num_warmup_steps:
                return step / max(1, num_warmup_steps)
            progress = (step - num_warmup_steps) / max(
                1, num_training_steps - num_warmup_steps)
            return max(0.0, 0.5 * (1.0 + math.cos(math.pi * min(progress, 1.0))))
        return LambdaLR(optimizer, fn)
    if name == "polynomial":
        # polynomial decay w/ floor lr_end (reference :212-254)
        def fn(step):
            if step < num_warmup_steps:
                return step / max(1, num_warmup_steps)
            if step > num_training_steps:
                return lr_end / lr_init
            lr_range = lr_init - lr_end
            decay_steps = max(1, num_training_steps - num_warmup_steps)
            pct_remaining = 1 - (step - num_warmup_steps) / decay_steps
            decay = lr_range * pct_remaining ** power + lr_end
            return decay / lr_init
        return LambdaLR(optimizer, fn)
    if name == "inverse_sqrt":
        # reference inverse_square_root_schedule (:101-139)
        def fn(step):
            if step < num_warmup_steps:
                return step / max(1, num_warmup_steps)
            return (max(1, num_warmup_steps) ** 0.5) / (step ** 0.5)
        return LambdaLR(optimizer, fn)
    if name == "direct":
        # reference Direct_LR (:141-191): constant after warmup
        def fn(step):
            if step < num_warmup_steps:
                return step / max(1, num_warmup_steps)
            return 1.0
        return LambdaLR(optimizer, fn)
    raise ValueError(f"unknown scheduler {name!r}")
