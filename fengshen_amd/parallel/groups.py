"""Process-group topology for DP × TP × PP over RCCL/xGMI.

Behavioral parity: reference fengshen/models/megatron/mpu/initialize.py:61-228
(initialize_model_parallel, group getters) — redesigned, not translated.

MI355X-first design notes:
  * One process per GPU; ``torch.distributed`` backend "nccl" IS RCCL on ROCm.
  * TP ranks are *innermost* (contiguous global ranks) so the per-layer TP
    all-reduce — the latency-critical collective (2/layer fwd + 2 bwd,
    SURVEY.md §2.3) — stays inside one node and rides direct xGMI
    point-to-point links (7 × ~153 GB/s per GPU).
  * Topology order is pipe-outer → data → tensor-inner, matching the
    reference's PipeModelDataParallelTopology (megatron_deepspeed.py:349-354)
    so per-TP-rank checkpoint layouts (part_{rank} dirs) are compatible.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TENSOR_MODEL_PARALLEL_GROUP = None
_PIPELINE_MODEL_PARALLEL_GROUP = None
_DATA_PARALLEL_GROUP = None
_MODEL_PARALLEL_GLOBAL_RANKS = None
_TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = None

# cached sizes/ranks so getters work without re-deriving
_TP_WORLD_SIZE: Optional[int] = None
_TP_RANK: Optional[int] = None
_PP_WORLD_SIZE: Optional[int] = None
_PP_RANK: Optional[int] = None

# fp32 all-reduce toggle for bf16 TP activations
# (reference: mpu/initialize.py:165-167 + mappings.py:37-46)
_FP32_ALLREDUCE = False


def init_distributed(backend: Optional[str] = None, timeout_minutes: int = 30) -> None:
    """Initialize torch.distributed from torchrun env vars (idempotent).

    backend defaults to nccl (=RCCL) when a GPU is visible, else gloo.
    """
    if dist.is_initialized():
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    import datetime

    kwargs = dict(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(minutes=timeout_minutes),
    )
    if backend == "nccl" and torch.cuda.is_available():
        local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
        torch.cuda.set_device(local_rank)
        kwargs["device_id"] = torch.device("cuda", local_rank)
    try:
        dist.init_process_group(**kwargs)
    except TypeError:  # older torch without device_id
        kwargs.pop("device_id", None)
        dist.init_process_group(**kwargs)


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
) -> None:
    """Build TP/PP/DP groups.

    With world size W = pp * dp * tp, global rank r decomposes as
      tp_rank = r % tp
      dp_rank = (r // tp) % dp
      pp_rank = r // (tp * dp)
    """
    global _TENSOR_MODEL_PARALLEL_GROUP, _PIPELINE_MODEL_PARALLEL_GROUP
    global _DATA_PARALLEL_GROUP, _TP_WORLD_SIZE, _TP_RANK, _PP_WORLD_SIZE, _PP_RANK
    global _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS

    assert dist.is_initialized(), "call init_distributed() first"
    world_size = dist.get_world_size()
    tp = tensor_model_parallel_size
    pp = pipeline_model_parallel_size
    assert world_size % (tp * pp) == 0, (
        f"world size {world_size} not divisible by tp({tp}) * pp({pp})"
    )
    dp = world_size // (tp * pp)
    rank = dist.get_rank()

    # data-parallel groups: ranks with same (tp_rank, pp_rank)
    for p in range(pp):
        for t in range(tp):
            ranks = [p * dp * tp + d * tp + t for d in range(dp)]
            group = dist.new_group(ranks)
            if rank in ranks:
                _DATA_PARALLEL_GROUP = group

    # tensor-parallel groups: contiguous ranks
    for i in range(world_size // tp):
        ranks = list(range(i * tp, (i + 1) * tp))
        group = dist.new_group(ranks)
        if rank in ranks:
            _TENSOR_MODEL_PARALLEL_GROUP = group
            _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = ranks

    # pipeline groups: same (dp_rank, tp_rank) across pp stages
    for d in range(dp):
        for t in range(tp):
            ranks = [p * dp * tp + d * tp + t for p in range(pp)]
            group = dist.new_group(ranks)
            if rank in ranks:
                _PIPELINE_MODEL_PARALLEL_GROUP = group

    _TP_WORLD_SIZE = tp
    _TP_RANK = rank % tp
    _PP_WORLD_SIZE = pp
    _PP_RANK = rank // (tp * dp)


def model_parallel_is_initialized() -> bool:
    return _TENSOR_MODEL_PARALLEL_GROUP is not None


def destroy_model_parallel() -> None:
    global _TENSOR_MODEL_PARALLEL_GROUP, _PIPELINE_MODEL_PARALLEL_GROUP
    global _DATA_PARALLEL_GROUP, _TP_WORLD_SIZE, _TP_RANK, _PP_WORLD_SIZE, _PP_RANK
    global _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS
    _TENSOR_MODEL_PARALLEL_GROUP = None
    _PIPELINE_MODEL_PARALLEL_GROUP = None
    _DATA_PARALLEL_GROUP = None
    _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS = None
    _TP_WORLD_SIZE = _TP_RANK = _PP_WORLD_SIZE = _PP_RANK = None


# ---------------------------------------------------------------------------
# getters — safe to call without init (degrade to single-process semantics)
# ---------------------------------------------------------------------------
def get_tensor_model_parallel_group():
    return _TENSOR_MODEL_PARALLEL_GROUP


def get_tensor_model_parallel_world_size() -> int:
    if _TP_WORLD_SIZE is not None:
        return _TP_WORLD_SIZE
    return 1


def get_tensor_model_parallel_rank() -> int:
    if _TP_RANK is not None:
        return _TP_RANK
    return 0


def get_tensor_model_parallel_src_rank() -> int:
    """Global rank of local TP rank 0 (reference: mpu/initialize.py:223-228)."""
    if _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS is not None:
        return _TENSOR_MODEL_PARALLEL_GLOBAL_RANKS[0]
    return 0


def get_data_parallel_group():
    if _DATA_PARALLEL_GROUP is not None:
        return _DATA_PARALLEL_GROUP
    if dist.is_available() and dist.is_initialized():
        return dist.group.WORLD
    return None


def get_data_parallel_world_size() -> int:
    if _DATA_PARALLEL_GROUP is not None:
        return dist.get_world_size(group=_DATA_PARALLEL_GROUP)
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return 1


def get_data_parallel_rank() -> int:
    if _DATA_PARALLEL_GROUP is not None:
        return dist.get_rank(group=_DATA_PARALLEL_GROUP)
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank()
    return 0


def get_pipeline_model_parallel_group():
    return _PIPELINE_MODEL_PARALLEL_GROUP


def get_pipeline_model_parallel_world_size() -> int:
    return _PP_WORLD_SIZE if _PP_WORLD_SIZE is not None else 1


def get_pipeline_model_parallel_rank() -> int:
    return _PP_RANK if _PP_RANK is not None else 0


def set_fp32_allreduce(flag: bool) -> None:
    """Upcast bf16 TP activation all-reduces to fp32
    (reference: mpu/mappings.py:37-46 toggle)."""
    global _FP32_ALLREDUCE
    _FP32_ALLREDUCE = flag


def get_fp32_allreduce() -> bool:
    return _FP32_ALLREDUCE
