"""Native ZeRO-0/1/2 optimizer — replaces DeepSpeed's engine.

The reference outsources ZeRO to deepspeed.initialize(..., mpu=mpu)
(strategies/megatron_deepspeed.py:302-320); we own it.  MI355X-first design:

  * One flat bf16 parameter buffer per bucket; model params are *views* into
    it, so the all-gather after the step updates weights with zero copies.
  * Gradients are views into a flat per-bucket grad buffer — autograd
    accumulates in place, no reducer copy.
  * Backward hooks launch the bucket's reduce-scatter (stage 2) or
    all-reduce (stage 0/1) as soon as the bucket's grads are complete;
    RCCL runs on its own stream so communication overlaps the rest of
    backward (reference parity: overlap_comm=True, megatron_deepspeed.py:81).
  * Buckets default to 128 Mi elements (256 MiB bf16) — sized for xGMI ring
    collectives (per-link ~153 GB/s) and 288 GB HBM, vs the reference's
    2e8-element DeepSpeed default.
  * Optimizer states (fp32 master + Adam moments) are sharded across the
    *data-parallel* group only — TP ranks each keep their own shard space
    (reference parity: DeepSpeed given mpu=).
  * The update itself is a fused multi-tensor AdamW HIP kernel over the flat
    fp32 shards (ops/adamw.py), replacing DeepSpeed FusedAdam.

Stage semantics:
  0 = DDP + mixed precision (full fp32 master + moments on every rank)
  1 = optimizer-state sharding (all-reduce grads, update own shard, all-gather)
  2 = + gradient sharding via reduce-scatter (half the grad traffic of AR)
  3 = + parameter sharding (see zero3.py)
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from fengshen_amd.ops.adamw import fused_adamw_flat_

_ALIGN = 128  # element alignment for shards (512B for fp32)


def _pad_to(n: int, mult: int) -> int:
    return ((n + mult - 1) // mult) * mult


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], group_idx: int,
                 world_size: int, device, param_dtype, shard_states: bool):
        self.params = params
        self.group_idx = group_idx
        numel = sum(p.numel() for p in params)
        self.numel = numel
        self.numel_padded = _pad_to(numel, _ALIGN * world_size)
        self.shard_numel = self.numel_padded // world_size if shard_states else self.numel_padded
        self.flat_param = torch.zeros(self.numel_padded, dtype=param_dtype, device=device)
        self.flat_grad: Optional[torch.Tensor] = None
        self.grad_shard: Optional[torch.Tensor] = None  # stage-2 RS output
        self.master_shard: Optional[torch.Tensor] = None
        self.exp_avg: Optional[torch.Tensor] = None
        self.exp_avg_sq: Optional[torch.Tensor] = None
        self.param_offsets: List[int] = []
        self.ready = 0
        self.work = None  # in-flight comm handle
        self.launched = False

        # move params into the flat buffer
        offset = 0
        for p in params:
            n = p.numel()
            self.flat_param[offset:offset + n].copy_(p.data.reshape(-1).to(param_dtype))
            p.data = self.flat_param[offset:offset + n].view_as(p.data)
            self.param_offsets.append(offset)
            offset += n

    def alloc_grads(self, grad_dtype):
        self.flat_grad = torch.zeros(self.numel_padded, dtype=grad_dtype,
                                     device=self.flat_param.device)
        for p, off in zip(self.params, self.param_offsets):
            p.grad = self.flat_grad[off:off + p.numel()].view_as(p.data)


class ZeroOptimizer(torch.optim.Optimizer):
    """ZeRO stage 0/1/2 optimizer with fused AdamW.

    Accepts standard param_groups (lr / weight_decay / betas / eps per group,
    so model_utils-style decay/no-decay groups work) and behaves like a torch
    optimizer for LR schedulers (mutate param_groups[i]['lr']).
    """

    def __init__(self, params, stage: int = 2, lr: float = 1e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8, weight_decay: float = 0.01,
                 process_group=None, bucket_numel: int = 128 * 1024 * 1024,
                 overlap_comm: bool = True, grad_dtype: Optional[torch.dtype] = None,
                 clip_grad: float = 0.0, cpu_offload: bool = False):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        assert stage in (0, 1, 2), "use Zero3Model for stage 3"
        self.stage = stage
        # ZeRO-offload parity (reference: DeepSpeedCPUAdam + offload_optimizer,
        # strategies/megatron_deepspeed.py:124-167): fp32 master + Adam moments
        # live in pinned host memory; the step streams the grad shard to the
        # host, updates there, and streams the new weights back.  Saves
        # 12 bytes/param of HBM per shard at the cost of 2 PCIe transits.
        self.cpu_offload = cpu_offload
        self.group = process_group
        if self.group is None and dist.is_available() and dist.is_initialized():
            from fengshen_amd.parallel import groups as pgroups
            self.group = pgroups.get_data_parallel_group()
        self.world_size = dist.get_world_size(self.group) if (
            dist.is_available() and dist.is_initialized()) else 1
        self.rank = dist.get_rank(self.group) if (
            dist.is_available() and dist.is_initialized()) else 0
        self.overlap_comm = overlap_comm and self.world_size > 1
        self.clip_grad = clip_grad
        self.bucket_numel = bucket_numel
        self._step_count = 0
        self._sync_grads = True  # False during grad-accumulation micro-steps
        self._hook_handles = []
        self.buckets: List[_Bucket] = []

        shard = stage >= 1 and self.world_size > 1
        # build buckets per param group, in reverse registration order
        # (approximates backward completion order for overlap)
        seen = set()
        for gi, g in enumerate(self.param_groups):
            plist = [p for p in g["params"] if p.requires_grad and id(p) not in seen]
            for p in plist:
                seen.add(id(p))
            device = plist[0].device if plist else torch.device("cpu")
            param_dtype = plist[0].dtype if plist else torch.float32
            cur: List[torch.nn.Parameter] = []
            cur_n = 0
            for p in reversed(plist):
                cur.append(p)
                cur_n += p.numel()
                if cur_n >= self.bucket_numel:
                    self.buckets.append(_Bucket(cur, gi, self.world_size, device,
                                                param_dtype, shard))
                    cur, cur_n = [], 0
            if cur:
                self.buckets.append(_Bucket(cur, gi, self.world_size, device,
                                            param_dtype, shard))

        grad_dtype = grad_dtype or (self.buckets[0].flat_param.dtype
                                    if self.buckets else torch.float32)
        self._grad_dtype = grad_dtype
        for b in self.buckets:
            b.alloc_grads(grad_dtype)
            self._init_states(b, shard)

        # param -> bucket for hooks
        self._param_bucket: Dict[int, _Bucket] = {}
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[id(p)] = b
        if self.world_size > 1:
            self._register_hooks()

    # ------------------------------------------------------------------
    def _init_states(self, b: _Bucket, shard: bool):
        if shard:
            start = self.rank * b.shard_numel
            src = b.flat_param[start:start + b.shard_numel]
        else:
            src = b.flat_param
        if self.cpu_offload:
            pin = src.is_cuda
            b.master_shard = src.detach().to(
                torch.float32).cpu().clone()
            if pin:
                b.master_shard = b.master_shard.pin_memory()
            b.exp_avg = torch.zeros_like(b.master_shard)
            b.exp_avg_sq = torch.zeros_like(b.master_shard)
            b.cpu_grad = torch.zeros(b.master_shard.numel(),
                                     dtype=b.flat_grad.dtype)
            if pin:
                b.cpu_grad = b.cpu_grad.pin_memory()
        else:
            b.master_shard = src.detach().to(torch.float32).clone()
            b.exp_avg = torch.zeros_like(b.master_shard)
            b.exp_avg_sq = torch.zeros_like(b.master_shard)

    def _register_hooks(self):
        for b in self.buckets:
            for p in b.params:
                h = p.register_post_accumulate_grad_hook(self._make_hook(b))
                self._hook_handles.append(h)

    def _make_hook(self, bucket: _Bucket):
        def hook(_param):
            if not self._sync_grads:
                return
            bucket.ready += 1
            if bucket.ready == len(bucket.params) and self.overlap_comm:
                self._launch_comm(bucket)
        return hook

    def _launch_comm(self, b: _Bucket):
        if b.launched or self.world_size == 1:
            return
        b.launched = True
        b.flat_grad.div_(self.world_size)
        if self.stage >= 2 and self._rs_supported():
            if b.grad_shard is None:
                b.grad_shard = torch.empty(b.shard_numel, dtype=self._grad_dtype,
                                           device=b.flat_grad.device)
            b.work = dist.reduce_scatter_tensor(b.grad_shard, b.flat_grad,
                                                group=self.group, async_op=True)
        else:
            b.work = dist.all_reduce(b.flat_grad, group=self.group, async_op=True)

    def _rs_supported(self) -> bool:
        if not dist.is_initialized():
            return False
        return dist.get_backend(self.group) == "nccl"

    def _finish_comm(self):
        for b in self.buckets:
            if not b.launched:
                self._launch_comm(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None

    def _grad_shard_view(self, b: _Bucket) -> torch.Tensor:
        """The gradient slice this rank's adamw consumes."""
        if self.stage == 0 or self.world_size == 1:
            return b.flat_grad
        if self.stage >= 2 and self._rs_supported():
            return b.grad_shard
        start = self.rank * b.shard_numel
        return b.flat_grad[start:start + b.shard_numel]

    # ------------------------------------------------------------------
    def set_sync(self, flag: bool):
        """False during gradient-accumulation micro-steps (no comm).

        This is a hard contract, not an optimization: with sync on and
        overlap_comm, the backward hook launches the bucket's async
        reduction as soon as every param has accumulated ONCE — a later
        micro-backward would then race the in-flight collective on the
        same flat grad buffer.  step() launches any pending reductions
        itself (trailing flush), so the pattern is:
        set_sync(False) -> N micro-backwards -> set_sync(True) -> step().
        """
        self._sync_grads = flag

    @torch.no_grad()
    def _global_grad_norm(self) -> torch.Tensor:
        """L2 norm of the full (DP-averaged) gradient.

        Works on the post-comm grad shards; sums squares over the DP group
        when sharded, and over the TP group for TP-parallel params
        (duplicated params counted only on TP rank 0).
        """
        from fengshen_amd.parallel import groups as pgroups

        device = self.buckets[0].flat_param.device if self.buckets else "cpu"
        sharded = self.stage >= 1 and self.world_size > 1
        sq = torch.zeros((), dtype=torch.float32, device=device)
        tp = pgroups.get_tensor_model_parallel_world_size()
        tp_rank = pgroups.get_tensor_model_parallel_rank()
        if sharded:
            for b in self.buckets:
                g = self._grad_shard_view(b).float()
                sq += (g * g).sum()
            if dist.is_initialized():
                dist.all_reduce(sq, group=self.group)
        else:
            for b in self.buckets:
                g = b.flat_grad.float()
                sq += (g * g).sum()
        if tp > 1 and dist.is_initialized():
            # sum parallel shards across the TP group.  Duplicated params
            # (norms/biases, identical grads on every TP rank) get counted
            # tp times — a <0.1%-of-elements overcount that only makes
            # clipping marginally more conservative; exact dedup needs
            # per-param bookkeeping the flat buckets do not keep.
            dist.all_reduce(sq, group=pgroups.get_tensor_model_parallel_group())
        return sq.sqrt()

    @torch.no_grad()
    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        total_norm = self._global_grad_norm()
        clip_coef = max_norm / (total_norm + 1e-6)
        if clip_coef < 1.0:
            for b in self.buckets:
                self._grad_shard_view(b).mul_(clip_coef)
                if self.stage < 2 or self.world_size == 1 or not self._rs_supported():
                    pass  # shard is a view of flat_grad; already scaled
                else:
                    b.flat_grad.mul_(clip_coef)  # keep full grads consistent too
        return total_norm

    # ------------------------------------------------------------------
    @torch.no_grad()
    def step(self, closure=None):  # noqa: C901
        loss = closure() if closure is not None else None
        self._finish_comm()
        self._step_count += 1
        grad_norm = None
        if self.clip_grad and self.clip_grad > 0:
            grad_norm = self.clip_grad_norm_(self.clip_grad)

        ag_works = []
        for b in self.buckets:
            g = self.param_groups[b.group_idx]
            beta1, beta2 = g["betas"]
            grad = self._grad_shard_view(b)
            sharded = self.stage >= 1 and self.world_size > 1
            if sharded:
                start = self.rank * b.shard_numel
                out_param = b.flat_param[start:start + b.shard_numel]
            else:
                out_param = b.flat_param
            if self.cpu_offload:
                b.cpu_grad.copy_(grad)
                fused_adamw_flat_(
                    b.master_shard, b.cpu_grad, b.exp_avg, b.exp_avg_sq,
                    None,
                    lr=g["lr"], beta1=beta1, beta2=beta2, eps=g["eps"],
                    weight_decay=g["weight_decay"], step=self._step_count,
                )
                out_param.copy_(b.master_shard, non_blocking=True)
            else:
                fused_adamw_flat_(
                    b.master_shard, grad, b.exp_avg, b.exp_avg_sq,
                    out_param,
                    lr=g["lr"], beta1=beta1, beta2=beta2, eps=g["eps"],
                    weight_decay=g["weight_decay"], step=self._step_count,
                )
            if sharded:
                shard = out_param.contiguous()
                if dist.get_backend(self.group) == "nccl":
                    w = dist.all_gather_into_tensor(b.flat_param, shard,
                                                    group=self.group, async_op=True)
                else:
                    chunks = list(b.flat_param.chunk(self.world_size))
                    w = dist.all_gather(chunks, shard, group=self.group,
                                        async_op=True)
                ag_works.append(w)
        for w in ag_works:
            w.wait()
        self._last_grad_norm = grad_norm
        return loss

    @torch.no_grad()
    def zero_grad(self, set_to_none: bool = False):
        for b in self.buckets:
            b.flat_grad.zero_()
            b.ready = 0
            b.launched = False
            b.work = None

    # ------------------------------------------------------------------
    def state_dict(self) -> dict:
        return {
            "stage": self.stage,
            "step": self._step_count,
            "world_size": self.world_size,
            "rank": self.rank,
            "param_groups": [
                {k: v for k, v in g.items() if k != "params"}
                for g in self.param_groups
            ],
            "buckets": [
                {
                    "master_shard": b.master_shard,
                    "exp_avg": b.exp_avg,
                    "exp_avg_sq": b.exp_avg_sq,
                }
                for b in self.buckets
            ],
        }

    def load_state_dict(self, sd: dict):
        assert sd["world_size"] == self.world_size, (
            "ZeRO resume requires the same DP world size "
            f"(ckpt {sd['world_size']} != current {self.world_size})")
        self._step_count = sd["step"]
        for g, gs in zip(self.param_groups, sd["param_groups"]):
            g.update(gs)
        for b, bs in zip(self.buckets, sd["buckets"]):
            b.master_shard.copy_(bs["master_shard"])
            b.exp_avg.copy_(bs["exp_avg"])
            b.exp_avg_sq.copy_(bs["exp_avg_sq"])
        # refresh low-precision params from masters
        sharded = self.stage >= 1 and self.world_size > 1
        for b in self.buckets:
            if sharded:
                start = self.rank * b.shard_numel
                b.flat_param[start:start + b.shard_numel].copy_(b.master_shard)
                if dist.get_backend(self.group) == "nccl":
                    dist.all_gather_into_tensor(
                        b.flat_param,
                        b.flat_param[start:start + b.shard_numel].contiguous(),
                        group=self.group)
                else:
                    chunks = list(b.flat_param.chunk(self.world_size))
                    dist.all_gather(
                        chunks,
                        b.flat_param[start:start + b.shard_numel].contiguous(),
                        group=self.group)
            else:
                b.flat_param.copy_(b.master_shard)
