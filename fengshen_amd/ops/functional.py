"""Fused-op functional API: HIP kernel dispatch + eager oracles.

Each op mirrors a hot-path op the reference fuses (or left unfused):
  scaled_masked_softmax   — ref fused_kernels/scaled_masked_softmax.h (wave64 redesign)
  scaled_causal_softmax   — ref scaled_upper_triang_masked_softmax.h
  layer_norm              — ref layer_norm_cuda.cpp (vestigial apex LN, now real)
  rms_norm                — ref layers/norms.py:35-52 (fp32 variance)
  bias_gelu               — ref layers/activations.py:60-94 (jit fused)
  bias_dropout_add        — ref layers/fused_bias_dropout.py:44-55
  rotary (apply_rotary)   — ref layers/positional_embeddings.py:54-87
  swiglu                  — ref LLaMAParallelMLP silu(w1)*w3 (transformer.py:571-623)
  fused attention         — ref flash_attention.py (flash_attn v1 import)

The eager implementations double as numerics oracles in tests
(SURVEY.md §4 "fallback-path pattern").
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from fengshen_amd.ops import use_hip, get_ext


# ---------------------------------------------------------------------------
# softmax family
# ---------------------------------------------------------------------------
def eager_scaled_masked_softmax(x: torch.Tensor, mask: Optional[torch.Tensor],
                                scale: float) -> torch.Tensor:
    """x [b, np, sq, sk]; mask [b or 1, 1, sq, sk] (True = MASKED, like the
    reference's uint8 pad mask); fp32 accumulation."""
    y = x.float() * scale
    if mask is not None:
        y = y.masked_fill(mask.bool(), -10000.0)
    return torch.softmax(y, dim=-1).to(x.dtype)


def eager_scaled_causal_softmax(x: torch.Tensor, scale: float) -> torch.Tensor:
    """x [attn_batches, sq, sk] with sq == sk; causal mask by index compare."""
    sq, sk = x.shape[-2], x.shape[-1]
    y = x.float() * scale
    causal = torch.ones(sq, sk, dtype=torch.bool, device=x.device).triu(1)
    y = y.masked_fill(causal, -10000.0)
    return torch.softmax(y, dim=-1).to(x.dtype)


class _ScaledMaskedSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, mask, scale):
        if use_hip(x):
            out = get_ext().scaled_masked_softmax_fwd(x.contiguous(), mask,
                                                      scale)
        else:
            out = eager_scaled_masked_softmax(x, mask, scale)
        ctx.save_for_backward(out)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (out,) = ctx.saved_tensors
        if use_hip(out):
            grad_in = get_ext().scaled_softmax_bwd(grad_out.contiguous(), out,
                                                   ctx.scale)
        else:
            g = grad_out.float() * out.float()
            grad_in = ((g - out.float() * g.sum(dim=-1, keepdim=True))
                       * ctx.scale).to(out.dtype)
        return grad_in, None, None


class _ScaledCausalSoftmax(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, scale):
        if use_hip(x):
            out = get_ext().scaled_causal_softmax_fwd(x.contiguous(), scale)
        else:
            out = eager_scaled_causal_softmax(x, scale)
        ctx.save_for_backward(out)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (out,) = ctx.saved_tensors
        if use_hip(out):
            grad_in = get_ext().scaled_softmax_bwd(grad_out.contiguous(), out,
                                                   ctx.scale)
        else:
            g = grad_out.float() * out.float()
            grad_in = ((g - out.float() * g.sum(dim=-1, keepdim=True))
                       * ctx.scale).to(out.dtype)
        return grad_in, None


def scaled_masked_softmax(x, mask, scale: float = 1.0):
    return _ScaledMaskedSoftmax.apply(x, mask, scale)


def scaled_causal_softmax(x, scale: float = 1.0):
    return _ScaledCausalSoftmax.apply(x, scale)


# ---------------------------------------------------------------------------
# norms
# ---------------------------------------------------------------------------
def eager_rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float):
    """fp32 variance like the reference RMSNorm (norms.py:35-52)."""
    x32 = x.float()
    var = x32.pow(2).mean(-1, keepdim=True)
    y = x32 * torch.rsqrt(var + eps)
    return (weight.float() * y).to(x.dtype)


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if use_hip(x):
            x = x.contiguous()  # kernel reads rows as flat vectors
            out, invrms = get_ext().rms_norm_fwd(x, weight, eps)
        else:
            x32 = x.float()
            var = x32.pow(2).mean(-1, keepdim=True)
            invrms = torch.rsqrt(var + eps)
            out = (x32 * invrms * weight.float()).to(x.dtype)
        ctx.save_for_backward(x, weight, invrms)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x, weight, invrms = ctx.saved_tensors
        if use_hip(x):
            gx, gw = get_ext().rms_norm_bwd(grad_out.contiguous(), x, weight,
                                            invrms)
            return gx, gw, None
        x32 = x.float()
        g32 = grad_out.float()
        w32 = weight.float()
        xhat = x32 * invrms
        gy_w = g32 * w32
        # d/dx: invrms * (gy_w - xhat * mean(gy_w * xhat))
        mean_term = (gy_w * xhat).mean(-1, keepdim=True)
        gx = (invrms * (gy_w - xhat * mean_term)).to(x.dtype)
        gw = (g32 * xhat).sum(dim=tuple(range(x.dim() - 1))).to(weight.dtype)
        return gx, gw, None


def rms_norm(x, weight, eps: float = 1e-6):
    return _RMSNorm.apply(x, weight, eps)


def eager_layer_norm(x, weight, bias, eps):
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), weight.float(),
        bias.float() if bias is not None else None, eps).to(x.dtype)


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        if use_hip(x):
            x = x.contiguous()  # kernel reads rows as flat vectors
            out, mean, invstd = get_ext().layer_norm_fwd(x, weight, bias, eps)
            ctx.save_for_backward(x, weight, mean, invstd)
            ctx.has_bias = bias is not None
            return out
        ctx.eps = eps
        ctx.fallback = True
        x32 = x.float()
        mean = x32.mean(-1, keepdim=True)
        var = x32.var(-1, unbiased=False, keepdim=True)
        invstd = torch.rsqrt(var + eps)
        out = (x32 - mean) * invstd * weight.float()
        if bias is not None:
            out = out + bias.float()
        ctx.save_for_backward(x, weight, mean, invstd)
        ctx.has_bias = bias is not None
        return out.to(x.dtype)

    @staticmethod
    def backward(ctx, grad_out):
        x, weight, mean, invstd = ctx.saved_tensors
        if use_hip(x):
            gx, gw, gb = get_ext().layer_norm_bwd(
                grad_out.contiguous(), x, weight, mean, invstd)
            return gx, gw, (gb if ctx.has_bias else None), None
        x32 = x.float()
        g32 = grad_out.float()
        w32 = weight.float()
        xhat = (x32 - mean) * invstd
        gy_w = g32 * w32
        n = x.shape[-1]
        gx = (invstd * (gy_w - gy_w.mean(-1, keepdim=True)
                        - xhat * (gy_w * xhat).mean(-1, keepdim=True))).to(x.dtype)
        dims = tuple(range(x.dim() - 1))
        gw = (g32 * xhat).sum(dims).to(weight.dtype)
        gb = g32.sum(dims).to(weight.dtype) if ctx.has_bias else None
        return gx, gw, gb, None


def layer_norm(x, weight, bias=None, eps: float = 1e-5):
    return _LayerNorm.apply(x, weight, bias, eps)


# ---------------------------------------------------------------------------
# activations
# ---------------------------------------------------------------------------
def eager_gelu(x):
    """tanh-approx GELU (reference bias_gelu, activations.py:60-77)."""
    return x * 0.5 * (1.0 + torch.tanh(0.79788456 * x * (1 + 0.044715 * x * x)))


class _BiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        if use_hip(x):
            return get_ext().bias_gelu_fwd(x, bias)
        y = x.float() + bias.float()
        return eager_gelu(y).to(x.dtype)

    @staticmethod
    def backward(ctx, grad_out):
        x, bias = ctx.saved_tensors
        if use_hip(x):
            gx = get_ext().bias_gelu_bwd(grad_out.contiguous(), x, bias)
        else:
            y = (x.float() + bias.float())
            t = torch.tanh(0.79788456 * y * (1 + 0.044715 * y * y))
            ff = 0.5 * y * ((1 - t * t) * (0.79788456 + 0.1070322243 * y * y)) \
                + 0.5 * (1 + t)
            gx = (ff * grad_out.float()).to(x.dtype)
        gb = gx.sum(dim=tuple(range(gx.dim() - 1))).to(bias.dtype)
        return gx, gb


def bias_gelu(x, bias):
    return _BiasGelu.apply(x, bias)


def eager_swiglu(gate, up):
    return torch.nn.functional.silu(gate.float()).to(gate.dtype) * up


class _SwiGLU(torch.autograd.Function):
    """fused silu(gate) * up over a [..., 2h] packed tensor (gate|up)."""

    @staticmethod
    def forward(ctx, packed):
        ctx.save_for_backward(packed)
        if use_hip(packed):
            return get_ext().swiglu_fwd(packed)
        gate, up = packed.chunk(2, dim=-1)
        return eager_swiglu(gate, up)

    @staticmethod
    def backward(ctx, grad_out):
        (packed,) = ctx.saved_tensors
        if use_hip(packed):
            return get_ext().swiglu_bwd(grad_out.contiguous(), packed)
        gate, up = packed.chunk(2, dim=-1)
        g32, u32, go = gate.float(), up.float(), grad_out.float()
        sig = torch.sigmoid(g32)
        silu = g32 * sig
        dgate = go * u32 * sig * (1 + g32 * (1 - sig))
        dup = go * silu
        return torch.cat([dgate, dup], dim=-1).to(packed.dtype)


def swiglu(packed):
    return _SwiGLU.apply(packed)


class _BiasDropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias, residual, p, training):
        if not training or p == 0.0:
            ctx.p = 0.0
            ctx.save_for_backward(torch.tensor([]))
            y = x if bias is None else x + bias
            ctx.has_bias = bias is not None
            return y + residual
        keep = 1.0 - p
        mask = (torch.rand_like(x, dtype=torch.float32) < keep)
        y = x if bias is None else x + bias
        out = y * mask.to(x.dtype) / keep + residual
        ctx.p = p
        ctx.has_bias = bias is not None
        ctx.save_for_backward(mask)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        if ctx.p == 0.0:
            gb = (grad_out.sum(dim=tuple(range(grad_out.dim() - 1)))
                  if ctx.has_bias else None)
            return grad_out, gb, grad_out, None, None
        (mask,) = ctx.saved_tensors
        keep = 1.0 - ctx.p
        gx = grad_out * mask.to(grad_out.dtype) / keep
        gb = gx.sum(dim=tuple(range(gx.dim() - 1))) if ctx.has_bias else None
        return gx, gb, grad_out, None, None


def bias_dropout_add(x, bias, residual, p: float, training: bool):
    return _BiasDropoutAdd.apply(x, bias, residual, p, training)


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear with a decode fast path: for inference-shaped inputs
    (<=8 rows, bf16, no bias) the hand-written bf16 GEMV kernel streams
    the weight matrix at HBM rate instead of hipBLASLt's skinny-M GEMM
    tiles.  Falls back to F.linear everywhere else (training, CPU,
    other dtypes) so autograd semantics are unchanged."""
    if (bias is None and x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and not torch.is_grad_enabled()
            and weight.shape[1] % 16 == 0
            and x.shape[-1] == weight.shape[1]):
        rows = x.numel() // x.shape[-1]
        # rows == 1 only: the B>1 GEMV re-reads x per weight vector and
        # measured SLOWER than hipBLASLt's M=8 GEMM (17.2 vs 11.0 ms/tok
        # at batch 8) — multi-row decode stays on the library path.
        if rows == 1 and use_hip(x):
            y = get_ext().bf16_gemv(weight.contiguous(), x.contiguous())
            return y.view(*x.shape[:-1], weight.shape[0])
    return torch.nn.functional.linear(x, weight, bias)


# ---------------------------------------------------------------------------
# rotary embedding
# ---------------------------------------------------------------------------
def build_rope_cache(seq_len: int, dim: int, base: float = 10000.0,
                     device=None, dtype=torch.float32):
    """Host-precomputed cos/sin tables (guide Appendix B: no on-device trig)."""
    inv_freq = 1.0 / (base ** (torch.arange(0, dim, 2, device=device,
                                            dtype=torch.float32) / dim))
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return emb.cos().to(dtype), emb.sin().to(dtype)


def _rotate_half(x):
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def eager_apply_rotary(q, k, cos, sin, offset: int = 0):
    """q,k [b, np, s, hn]; cos/sin [max_s, hn]."""
    s = q.shape[-2]
    c = cos[offset:offset + s].to(q.dtype)
    si = sin[offset:offset + s].to(q.dtype)
    q_out = q * c + _rotate_half(q) * si
    k_out = k * c + _rotate_half(k) * si
    return q_out, k_out


class _ApplyRotary(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin, offset):
        ctx.save_for_backward(cos, sin)
        ctx.offset = offset
        if use_hip(q):
            return tuple(get_ext().rope_fwd(q.contiguous(), k.contiguous(),
                                            cos, sin, offset))
        return eager_apply_rotary(q, k, cos, sin, offset)

    @staticmethod
    def backward(ctx, gq, gk):
        cos, sin = ctx.saved_tensors
        offset = ctx.offset
        # rotation is orthogonal: grad = rotate by -theta
        if use_hip(gq):
            gq_out, gk_out = get_ext().rope_bwd(gq.contiguous(), gk.contiguous(),
                                                cos, sin, offset)
            return gq_out, gk_out, None, None, None
        s = gq.shape[-2]
        c = cos[offset:offset + s].to(gq.dtype)
        si = sin[offset:offset + s].to(gq.dtype)
        gq_out = gq * c - _rotate_half(gq * si)
        gk_out = gk * c - _rotate_half(gk * si)
        return gq_out, gk_out, None, None, None


def apply_rotary(q, k, cos, sin, offset: int = 0):
    return _ApplyRotary.apply(q, k, cos, sin, offset)


# ---------------------------------------------------------------------------
# attention (composite; flash HIP kernel replaces this on GPU)
# ---------------------------------------------------------------------------
def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              causal: bool = False, mask: Optional[torch.Tensor] = None,
              dropout_p: float = 0.0, training: bool = False,
              scale: Optional[float] = None) -> torch.Tensor:
    """q,k,v: [b, np, s, hn] -> context [b, np, sq, hn].

    GPU path: flash-style fused HIP kernel when available (csrc/flash_attn.hip)
    else hipBLASLt bmm + fused-softmax kernel.  CPU path: eager oracle.
    """
    b, np_, sq, hn = q.shape
    sk = k.shape[-2]
    if scale is None:
        scale = 1.0 / math.sqrt(hn)
    ext = get_ext() if use_hip(q) else None
    if ext is not None and hasattr(ext, "flash_attn_fwd"):
        from fengshen_amd.ops.flash import (
            flash_attention, flash_attn_supported, mask_to_klens)
        drop = dropout_p if training else 0.0
        if flash_attn_supported(q, k, v, causal, mask, drop):
            if causal and mask is None:
                return flash_attention(q, k, v, scale, causal=True,
                                       dropout_p=drop)
            if not causal:
                klens = None
                if mask is not None:
                    klens = mask_to_klens(mask, sk)
                if mask is None or klens is not None:
                    return flash_attention(q, k, v, scale, causal=False,
                                           klens=klens, dropout_p=drop)
    # bmm path: scores in [b*np, sq, sk]
    q2 = q.reshape(b * np_, sq, hn)
    k2 = k.reshape(b * np_, sk, hn)
    scores = torch.bmm(q2, k2.transpose(1, 2)).view(b, np_, sq, sk)
    if causal and sq == sk and mask is None:
        probs = scaled_causal_softmax(scores.view(b * np_, sq, sk), scale)
        probs = probs.view(b, np_, sq, sk)
    else:
        m = mask
        if causal:
            # Rectangular causal for chunked prefill (sq < sk with a KV
            # cache): query i sits at global position (sk - sq + i), so key
            # j is visible iff j <= sk - sq + i.
            cm = torch.ones(sq, sk, dtype=torch.bool,
                            device=q.device).triu(1 + sk - sq)
            m = cm if mask is None else (mask.bool() | cm)
        probs = scaled_masked_softmax(scores, m, scale)
    if dropout_p > 0.0 and training:
        from fengshen_amd.parallel.random import get_rng_tracker
        with get_rng_tracker().fork():
            probs = torch.nn.functional.dropout(probs, p=dropout_p)
    ctx = torch.bmm(probs.view(b * np_, sq, sk), v.reshape(b * np_, sk, hn))
    return ctx.view(b, np_, sq, hn)
