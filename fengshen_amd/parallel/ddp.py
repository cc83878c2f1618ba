"""Bucketed gradient all-reduce for plain DDP with arbitrary optimizers.

Used when the user's optimizer is not Adam-family (otherwise the trainer
routes through ZeroOptimizer stage 0, which fuses the reducer + master
weights).  Same design as our ZeRO reducer: post-accumulate-grad hooks fill
flat buckets, async RCCL all-reduce overlaps the rest of backward.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist


class GradReducer:
    def __init__(self, module: torch.nn.Module, process_group=None,
                 bucket_numel: int = 64 * 1024 * 1024):
        self.group = process_group
        self.world_size = dist.get_world_size(process_group) if dist.is_initialized() else 1
        self.enabled = self.world_size > 1
        self._sync = True
        self.buckets: List[dict] = []
        if not self.enabled:
            return
        params = [p for p in module.parameters() if p.requires_grad]
        cur, cur_n = [], 0
        for p in reversed(params):
            cur.append(p)
            cur_n += p.numel()
            if cur_n >= bucket_numel:
                self._add_bucket(cur)
                cur, cur_n = [], 0
        if cur:
            self._add_bucket(cur)
        for b in self.buckets:
            for p in b["params"]:
                p.register_post_accumulate_grad_hook(self._make_hook(b))

    def _add_bucket(self, params):
        numel = sum(p.numel() for p in params)
        device = params[0].device
        dtype = params[0].dtype
        self.buckets.append(dict(params=params, numel=numel, device=device,
                                 dtype=dtype, flat=None, ready=0, work=None))

    def set_sync(self, flag: bool):
        self._sync = flag

    def _make_hook(self, b):
        def hook(_p):
            if not self._sync or not self.enabled:
                return
            b["ready"] += 1
            if b["ready"] == len(b["params"]):
                self._launch(b)
        return hook

    def _launch(self, b):
        if b["flat"] is None:
            b["flat"] = torch.empty(b["numel"], dtype=b["dtype"], device=b["device"])
        off = 0
        for p in b["params"]:
            n = p.numel()
            if p.grad is None:
                b["flat"][off:off + n].zero_()
            else:
                b["flat"][off:off + n].copy_(p.grad.reshape(-1))
            off += n
        b["flat"].div_(self.world_size)
        b["work"] = dist.all_reduce(b["flat"], group=self.group, async_op=True)

    def finalize(self):
        """Wait for comm and scatter averaged grads back into param.grad."""
        if not self.enabled or not self._sync:
            return
        for b in self.buckets:
            # ready==0 with grads present happens when every micro-batch ran
            # under set_sync(False) (trailing grad-accum flush): hooks never
            # fired with sync on, so launch the reduction here.
            if b["work"] is None and (
                    b["ready"] > 0
                    or any(p.grad is not None for p in b["params"])):
                self._launch(b)
        for b in self.buckets:
            if b["work"] is None:
                continue
            b["work"].wait()
            off = 0
            for p in b["params"]:
                n = p.numel()
                if p.grad is not None:
                    p.grad.reshape(-1).copy_(b["flat"][off:off + n])
                off += n
            b["work"] = None
            b["ready"] = 0
