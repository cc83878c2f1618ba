"""TP checkpoint resharding — shard a full (TP=1) state_dict for a TP=N
model and merge TP-sharded state_dicts back to full.

Behavioral parity: reference utils/llama_convert/convert_fs_llama_tp.py
(offline reshard into part_{rank} dirs, dim-aware split :85) and
fs_merge_weight.py — generalized: instead of a hand-written per-model table,
the rules are derived from the target model's parallel modules
(partition_dim attrs + MergedColumnParallelLinear segment sizes).
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn


def _module_param_rules(model: nn.Module) -> Dict[str, dict]:
    """name -> {dim, segments} for every TP-partitioned parameter."""
    from fengshen_amd.models.layers import MergedColumnParallelLinear

    rules: Dict[str, dict] = {}
    for mod_name, mod in model.named_modules():
        segs: List[int] = getattr(mod, "segment_sizes", None) \
            if isinstance(mod, MergedColumnParallelLinear) else None
        for p_name, p in mod.named_parameters(recurse=False):
            if not getattr(p, "tensor_model_parallel", False):
                continue
            full = f"{mod_name}.{p_name}" if mod_name else p_name
            rules[full] = {
                "dim": getattr(p, "partition_dim", 0),
                "segments": segs if p_name in ("weight", "bias") else None,
            }
    return rules


def shard_state_dict(model: nn.Module, full_sd: Dict[str, torch.Tensor],
                     tp_size: int, tp_rank: int) -> Dict[str, torch.Tensor]:
    """Slice a full state_dict for one TP rank of `model`'s topology."""
    rules = _module_param_rules(model)
    out = {}
    for k, v in full_sd.items():
        rule = rules.get(k)
        if rule is None:
            out[k] = v
            continue
        dim = rule["dim"]
        segs = rule["segments"]
        if segs is None or dim != 0:
            per = v.shape[dim] // tp_size
            out[k] = v.narrow(dim, tp_rank * per, per).clone()
        else:
            # merged segments: take this rank's slice of EACH segment
            pieces = []
            off = 0
            for seg in segs:
                seg_per = seg // tp_size
                pieces.append(v.narrow(0, off + tp_rank * seg_per, seg_per))
                off += seg
            out[k] = torch.cat(pieces, dim=0).clone()
    return out


def merge_state_dicts(model: nn.Module,
                      shards: List[Dict[str, torch.Tensor]]) -> Dict[str, torch.Tensor]:
    """Inverse of shard_state_dict: merge per-TP-rank shards to a full sd."""
    rules = _module_param_rules(model)
    tp = len(shards)
    out = {}
    for k, v0 in shards[0].items():
        rule = rules.get(k)
        if rule is None:
            out[k] = v0
            continue
        dim = rule["dim"]
        segs = rule["segments"]
        parts = [s[k] for s in shards]
        if segs is None or dim != 0:
            out[k] = torch.cat(parts, dim=dim)
        else:
            seg_chunks = [[] for _ in segs]
            for part in parts:
                off = 0
                for i, seg in enumerate(segs):
                    seg_per = seg // tp
                    seg_chunks[i].append(part.narrow(0, off, seg_per))
                    off += seg_per
            out[k] = torch.cat([torch.cat(c, dim=0) for c in seg_chunks], dim=0)
    return out
