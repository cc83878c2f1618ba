"""broadcast_data — feed identical batches to all TP ranks.

Behavioral parity: reference mpu/data.py:79 (sizes broadcast then flattened
payload broadcast from TP rank 0).
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist

from fengshen_amd.parallel import groups


def _check_data_types(keys, data, target_dtype):
    for key in keys:
        assert data[key].dtype == target_dtype, (
            f"{key} has dtype {data[key].dtype} != {target_dtype}")


def _build_key_size_numel_dictionaries(keys: List[str], data):
    max_dim = 8
    sizes = [0 for _ in range(max_dim * len(keys))]
    if groups.get_tensor_model_parallel_rank() == 0:
        offset = 0
        for key in keys:
            assert data[key].dim() < max_dim
            size = data[key].size()
            for i, s in enumerate(size):
                sizes[i + offset] = s
            offset += max_dim
    device = "cuda" if torch.cuda.is_available() else "cpu"
    sizes_cuda = torch.tensor(sizes, dtype=torch.long, device=device)
    if groups.get_tensor_model_parallel_world_size() > 1:
        dist.broadcast(sizes_cuda, groups.get_tensor_model_parallel_src_rank(),
                       group=groups.get_tensor_model_parallel_group())
    sizes_cpu = sizes_cuda.cpu().tolist()
    key_size, key_numel, total_numel = {}, {}, 0
    offset = 0
    for key in keys:
        shape = []
        for i in range(max_dim):
            s = sizes_cpu[i + offset]
            if s == 0:
                break
            shape.append(s)
        numel = 1
        for s in shape:
            numel *= s
        key_size[key] = shape
        key_numel[key] = numel
        total_numel += numel
        offset += max_dim
    return key_size, key_numel, total_numel


def broadcast_data(keys: List[str], data: Dict[str, torch.Tensor],
                   datatype: torch.dtype) -> Dict[str, torch.Tensor]:
    """Broadcast data[keys] from TP rank 0 to all TP ranks."""
    key_size, key_numel, total_numel = _build_key_size_numel_dictionaries(keys, data)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if groups.get_tensor_model_parallel_rank() == 0:
        _check_data_types(keys, data, datatype)
        flatten = torch.cat(
            [data[key].to(device).contiguous().view(-1) for key in keys], dim=0)
    else:
        flatten = torch.empty(total_numel, device=device, dtype=datatype)
    if groups.get_tensor_model_parallel_world_size() > 1:
        dist.broadcast(flatten, groups.get_tensor_model_parallel_src_rank(),
                       group=groups.get_tensor_model_parallel_group())
    output, offset = {}, 0
    for key in keys:
        numel = key_numel[key]
        output[key] = flatten.narrow(0, offset, numel).view(key_size[key])
        offset += numel
    return output
