"""Flash attention wrapper: fused HIP forward + fused FA2-style backward.

csrc/flash_attn.hip (no SxS materialization, LSE saved).  Round-2
generality (ref flash_attention.py:174 call semantics):
- causal self-attention (LLaMA/GPT): sq == sk, sq % 64 == 0;
- bidirectional with optional per-batch key lengths (BERT suffix padding
  masks become klens int32 [b]);
- cross-attention (sq != sk, ragged sk) for the SD UNet;
- head_dim in {40, 64, 80, 96, 128, 160} (BERT 64, GPT2-3.5B 96,
  LLaMA 128, SD 40/80/160).
"""
from __future__ import annotations

from typing import Optional

import torch

from fengshen_amd.ops import get_ext

_FLASH_DIMS = (40, 64, 80, 96, 128, 160)


def _v3_eligible(ext, q, k, causal, klens, dropout_p):
    """v3 (swapped-QK^T 32x32 schedule, ~1.8x): self-attention at
    d in {64, 128} with s % 64 == 0; causal or klens-masked bidirectional.
    Dropout routes to the general kernels by default: a same-box A/B on
    the BERT step (b128 s512 d64 p=0.1) measured them identical (177.2
    vs 177.5/178.8 samples/s — the step is no longer attention-bound),
    so the simpler routing stands; FENGSHEN_FLASH_V3_DROP=1 flips
    dropout onto v3 at d 64/96 for experiments."""
    if not (hasattr(ext, "flash_attn_fwd_v3")
            and q.shape[-1] in (64, 96, 128)
            and q.shape[-2] == k.shape[-2]
            and q.shape[-2] % 64 == 0 and q.shape[-2] >= 64):
        return False
    if dropout_p > 0:
        import os
        return os.environ.get("FENGSHEN_FLASH_V3_DROP") == "1" \
            and q.shape[-1] in (64, 96)
    return True


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, causal, klens, dropout_p, seed):
        ext = get_ext()
        if _v3_eligible(ext, q, k, causal, klens, dropout_p):
            o, lse = ext.flash_attn_fwd_v3(q, k, v, scale, causal, klens,
                                           dropout_p, seed)
        else:
            o, lse = ext.flash_attn_fwd(q, k, v, scale, causal, klens,
                                        dropout_p, seed)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        ctx.causal = causal
        ctx.klens = klens
        ctx.dropout_p = dropout_p
        ctx.seed = seed
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = get_ext()
        if _v3_eligible(ext, q, k, ctx.causal, ctx.klens, ctx.dropout_p):
            dq, dk, dv = ext.flash_attn_bwd_v3(
                q, k, v, o, do.contiguous(), lse, ctx.scale, ctx.causal,
                ctx.klens, ctx.dropout_p, ctx.seed)
        else:
            dq, dk, dv = ext.flash_attn_bwd(
                q, k, v, o, do.contiguous(), lse, ctx.scale, ctx.causal,
                ctx.klens, ctx.dropout_p, ctx.seed)
        return dq, dk, dv, None, None, None, None, None


def _draw_seed(device) -> int:
    """Per-call dropout seed: CPU torch RNG (deterministic under
    torch.manual_seed) mixed with the TP rank so TP-local heads get
    decorrelated masks (eager path forks the mpu RNG tracker for the
    same reason)."""
    base = int(torch.randint(0, 2 ** 62, (1,)).item())
    try:
        from fengshen_amd.parallel import groups as pg
        base += pg.get_tensor_model_parallel_rank() * 0x9E3779B9
    except Exception:
        pass
    return base


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: float, causal: bool = True,
                    klens: Optional[torch.Tensor] = None,
                    dropout_p: float = 0.0) -> torch.Tensor:
    """q [b,h,sq,d], k/v [b,h,sk,d] bf16 contiguous.  dropout_p > 0
    applies attention dropout with an in-kernel counter hash (no mask
    tensor); the normalizer uses the undropped row sum, matching eager
    dropout(softmax(S)) @ V."""
    seed = _draw_seed(q.device) if dropout_p > 0 else 0
    return _FlashAttention.apply(q.contiguous(), k.contiguous(),
                                 v.contiguous(), scale, causal, klens,
                                 float(dropout_p), seed)


def mask_to_klens(mask: torch.Tensor, sk: int) -> Optional[torch.Tensor]:
    """Convert a True=masked pad mask into per-batch key lengths when it is
    a pure suffix-padding pattern ([b,1,1,sk] or [b,sk]); else None."""
    if mask.dtype != torch.bool:
        mask = mask != 0
    if mask.dim() == 4:
        if mask.shape[2] != 1 or mask.shape[1] != 1 or mask.shape[3] != sk:
            return None
        m2 = mask[:, 0, 0, :]
    elif mask.dim() == 2 and mask.shape[1] == sk:
        m2 = mask
    else:
        return None
    klens = (~m2).sum(dim=1, dtype=torch.int32)
    ar = torch.arange(sk, device=mask.device, dtype=torch.int32)
    if not torch.equal(m2, ar.unsqueeze(0) >= klens.unsqueeze(1)):
        return None  # not suffix padding
    return klens.contiguous()


def flash_attn_supported(q, k, v, causal, mask, dropout_p) -> bool:
    if q.dtype != torch.bfloat16:
        return False
    if q.shape[-1] not in _FLASH_DIMS:
        return False
    sq, sk = q.shape[-2], k.shape[-2]
    if causal:
        return mask is None and sq == sk and sq % 64 == 0
    # bidirectional: sq tiles at 16; sk arbitrary; mask must convert to
    # suffix-pad klens (checked by the caller via mask_to_klens)
    return sq % 16 == 0 and sq >= 16
