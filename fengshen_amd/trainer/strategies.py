"""Training strategies: single-device / DDP / native ZeRO-1/2/3, × TP.

Replaces the reference's PL strategy zoo + fengshen DeepSpeedStrategy
(strategies/megatron_deepspeed.py:51-399).  A strategy owns:
  * distributed + model-parallel group setup (ref: setup_mpu :339-369)
  * precision conversion (bf16 params, fp32 masters in the optimizer)
  * optimizer construction/wrapping (ref: deepspeed.initialize :302-320)
  * backward + grad sync + clip + step
"""
from __future__ import annotations

import logging
import os
from typing import Optional

import torch
import torch.distributed as dist

from fengshen_amd.parallel import groups as pgroups
from fengshen_amd.parallel.ddp import GradReducer
from fengshen_amd.parallel.random import model_parallel_manual_seed
from fengshen_amd.parallel.zero import ZeroOptimizer
from fengshen_amd.parallel.zero3 import Zero3Engine
from fengshen_amd.ops.adamw import FusedAdamW

logger = logging.getLogger(__name__)

_ADAM_FAMILY = (torch.optim.AdamW, torch.optim.Adam, FusedAdamW)


def parse_strategy(name: str) -> dict:
    """'auto' | 'single' | 'ddp' | 'zero1' | 'zero2' | 'zero3'."""
    name = (name or "auto").lower()
    if name in ("auto", "ddp", "single", "ddp_sharded"):
        return {"kind": name if name != "ddp_sharded" else "zero2", "stage": 0}
    if name.startswith("zero"):
        # 'zero2_offload' => ZeRO-offload (optimizer states in host RAM,
        # reference parity: deepspeed offload_optimizer config)
        offload = name.endswith("_offload")
        stage = int(name[4:].split("_")[0])
        return {"kind": "zero", "stage": stage, "cpu_offload": offload}
    if name.startswith("deepspeed_stage_"):  # reference-compat alias
        return {"kind": "zero", "stage": int(name.rsplit("_", 1)[1])}
    raise ValueError(f"unknown strategy {name!r}")


class Strategy:
    """One strategy instance per Trainer; world topology from torchrun env."""

    def __init__(self, kind: str = "auto", stage: int = 0,
                 tensor_model_parallel_size: int = 1,
                 pipe_model_parallel_size: int = 1,
                 mpu_seed: int = 42,
                 bucket_numel: int = 128 * 1024 * 1024,
                 overlap_comm: bool = True,
                 cpu_offload: bool = False):
        self.kind = kind
        self.stage = stage
        self.cpu_offload = cpu_offload
        self.tp_size = tensor_model_parallel_size
        self.pp_size = pipe_model_parallel_size
        self.mpu_seed = mpu_seed
        self.bucket_numel = bucket_numel
        self.overlap_comm = overlap_comm
        self.zero_optimizer: Optional[ZeroOptimizer] = None
        self.grad_reducer: Optional[GradReducer] = None
        self.trainer = None

    # ------------------------------------------------------------------
    def setup_environment(self, trainer) -> None:
        self.trainer = trainer
        world = int(os.environ.get("WORLD_SIZE", "1"))
        need_dist = world > 1 or self.tp_size > 1 or self.pp_size > 1
        if need_dist:
            pgroups.init_distributed()
        if self.kind == "auto":
            self.kind = "ddp" if world > 1 else "single"
        if (self.tp_size > 1 or self.pp_size > 1) and not pgroups.model_parallel_is_initialized():
            pgroups.initialize_model_parallel(self.tp_size, self.pp_size)
        model_parallel_manual_seed(self.mpu_seed + trainer.seed_offset)

    @property
    def data_parallel_world_size(self) -> int:
        return pgroups.get_data_parallel_world_size()

    @property
    def data_parallel_rank(self) -> int:
        return pgroups.get_data_parallel_rank()

    # ------------------------------------------------------------------
    def setup_model(self, model: torch.nn.Module, device, precision: str):
        if precision in ("bf16", "bf16-mixed", "bf16-true"):
            model = model.to(dtype=torch.bfloat16)
        elif precision in ("16", "fp16", "16-mixed"):
            model = model.to(dtype=torch.float16)
        model = model.to(device)
        return model

    def setup_optimizers(self, model, optimizer, scheduler_cfg):
        """Wrap/convert the module's optimizer for this strategy."""
        dp_world = self.data_parallel_world_size
        if isinstance(optimizer, (ZeroOptimizer, Zero3Engine)):
            self.zero_optimizer = optimizer
            return optimizer, scheduler_cfg
        if self.kind == "zero" and self.stage == 3 \
                and isinstance(optimizer, _ADAM_FAMILY):
            d = optimizer.defaults
            eng = Zero3Engine(
                model, lr=d.get("lr", 1e-3), betas=d.get("betas", (0.9, 0.999)),
                eps=d.get("eps", 1e-8), weight_decay=d.get("weight_decay", 0.0),
                process_group=pgroups.get_data_parallel_group())
            self.zero_optimizer = eng
            if scheduler_cfg is not None:
                scheduler_cfg["scheduler"].optimizer = eng
            return eng, scheduler_cfg
        use_zero = self.kind == "zero" or (self.kind == "ddp" and dp_world > 1)
        if use_zero and isinstance(optimizer, _ADAM_FAMILY):
            d = optimizer.defaults
            zopt = ZeroOptimizer(
                optimizer.param_groups,
                stage=self.stage,
                lr=d.get("lr", 1e-3), betas=d.get("betas", (0.9, 0.999)),
                eps=d.get("eps", 1e-8), weight_decay=d.get("weight_decay", 0.0),
                process_group=pgroups.get_data_parallel_group(),
                bucket_numel=self.bucket_numel,
                overlap_comm=self.overlap_comm,
                cpu_offload=self.cpu_offload,
            )
            self.zero_optimizer = zopt
            if scheduler_cfg is not None:
                scheduler_cfg["scheduler"].optimizer = zopt
            return zopt, scheduler_cfg
        if self.kind == "zero" and 1 <= self.stage <= 2:
            raise ValueError(
                f"ZeRO stage {self.stage} requires an Adam-family optimizer, "
                f"got {type(optimizer).__name__}")
        if dp_world > 1:
            # generic optimizer under DDP: external bucketed reducer
            self.grad_reducer = GradReducer(
                model, process_group=pgroups.get_data_parallel_group(),
                bucket_numel=self.bucket_numel)
        if isinstance(optimizer, _ADAM_FAMILY) and not isinstance(optimizer, FusedAdamW) \
                and next(model.parameters()).dtype != torch.float32:
            # low-precision params need master weights: swap in FusedAdamW
            fopt = FusedAdamW(optimizer.param_groups, **{
                k: optimizer.defaults[k] for k in ("lr", "betas", "eps", "weight_decay")
                if k in optimizer.defaults})
            if scheduler_cfg is not None:
                scheduler_cfg["scheduler"].optimizer = fopt
            optimizer = fopt
        return optimizer, scheduler_cfg

    # ------------------------------------------------------------------
    def set_sync(self, flag: bool):
        if self.zero_optimizer is not None:
            self.zero_optimizer.set_sync(flag)
        if self.grad_reducer is not None:
            self.grad_reducer.set_sync(flag)

    def backward(self, loss: torch.Tensor):
        loss.backward()

    def pre_step(self):
        if self.grad_reducer is not None:
            self.grad_reducer.finalize()

    def clip_gradients(self, optimizer, model, clip_val: float):
        if not clip_val or clip_val <= 0:
            return None
        if isinstance(optimizer, ZeroOptimizer):
            optimizer.clip_grad = 0.0  # we call explicitly for the norm value
            return optimizer.clip_grad_norm_(clip_val)
        return torch.nn.utils.clip_grad_norm_(model.parameters(), clip_val)

    def reduce_metric(self, value: torch.Tensor) -> torch.Tensor:
        """Average a scalar across all ranks (sync_dist=True)."""
        if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
            value = value.detach().clone()
            dist.all_reduce(value)
            value /= dist.get_world_size()
        return value

    def barrier(self):
        if dist.is_available() and dist.is_initialized():
            dist.barrier()
