"""Flash attention wrapper: fused HIP forward, composed backward.

Forward: csrc/flash_attn.hip (no SxS materialization, LSE saved).
Backward (v1): recompute-from-QKV composition on hipBLASLt bmm + the fused
softmax-backward identity — dV = P^T dO; dP = dO V^T;
dS = P*(dP - rowsum(dP*P)); dQ = dS K * scale; dK = dS^T Q * scale.
P is rebuilt row-block-exactly from the saved LSE, so forward and backward
agree bitwise on the softmax normalizer.  A fully fused backward kernel is
the planned next step.
"""
from __future__ import annotations

import torch

from fengshen_amd.ops import get_ext


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        o, lse = get_ext().flash_attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        scale = ctx.scale
        if hasattr(get_ext(), "flash_attn_bwd"):
            dq, dk, dv = get_ext().flash_attn_bwd(q, k, v, o, do.contiguous(),
                                                  lse, scale)
            return dq, dk, dv, None
        b, h, s, d = q.shape
        qf = q.reshape(b * h, s, d)
        kf = k.reshape(b * h, s, d)
        vf = v.reshape(b * h, s, d)
        dof = do.contiguous().reshape(b * h, s, d)
        # rebuild P from QK^T and the saved LSE (fp32 softmax, bf16 P)
        scores = torch.baddbmm(
            torch.empty(b * h, s, s, dtype=q.dtype, device=q.device),
            qf, kf.transpose(1, 2), beta=0.0, alpha=scale).float()
        causal = torch.ones(s, s, dtype=torch.bool, device=q.device).triu(1)
        scores.masked_fill_(causal, float("-inf"))
        p = torch.exp(scores - lse.reshape(b * h, s, 1))
        pb = p.to(q.dtype)
        dv = torch.bmm(pb.transpose(1, 2), dof)
        dp = torch.bmm(dof, vf.transpose(1, 2)).float()
        delta = (dp * p).sum(-1, keepdim=True)
        ds = (p * (dp - delta)).to(q.dtype)
        dq = torch.bmm(ds, kf) * scale
        dk = torch.bmm(ds.transpose(1, 2), qf) * scale
        return (dq.view(b, h, s, d), dk.view(b, h, s, d),
                dv.view(b, h, s, d), None)


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: float) -> torch.Tensor:
    """q,k,v [b, h, s, 128] bf16 contiguous; causal."""
    return _FlashAttention.apply(q.contiguous(), k.contiguous(),
                                 v.contiguous(), scale)


def flash_attn_supported(q, k, v, causal, mask, dropout_p) -> bool:
    return (causal and mask is None and dropout_p == 0.0
            and q.dtype == torch.bfloat16 and q.shape[-1] == 128
            and q.shape[-2] == k.shape[-2] and q.shape[-2] % 64 == 0)
