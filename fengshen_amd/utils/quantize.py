"""Weight-only int8 quantization for inference.

Behavioral parity: reference examples/ziya_inference (bitsandbytes
Linear8bitLt conversion, hf_quantizatin_inference.py:9-10; llama.cpp q5
path) — MI355X equivalent: W8A16 per-channel symmetric quantization; the
dequant+GEMM runs in bf16 (HBM traffic halves, which is what matters for
memory-bound decode).
"""
from __future__ import annotations

import torch
import torch.nn as nn


class W8Linear(nn.Module):
    """int8 weight + per-output-channel fp scale; bf16 activations."""

    def __init__(self, weight: torch.Tensor, bias=None):
        super().__init__()
        w = weight.detach().float()
        scale = w.abs().amax(dim=1, keepdim=True) / 127.0
        scale = scale.clamp(min=1e-8)
        qw = torch.round(w / scale).clamp(-127, 127).to(torch.int8)
        self.register_buffer("qweight", qw)
        self.register_buffer("scale", scale.to(torch.float32))
        self.bias = None
        if bias is not None:
            self.register_buffer("bias_buf", bias.detach().clone())
            self.bias = True
        self.out_features, self.in_features = weight.shape

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from fengshen_amd.ops import use_hip, get_ext
        rows = x.numel() // x.shape[-1]
        if (use_hip(x) and x.dtype == torch.bfloat16 and rows <= 16
                and x.shape[-1] % 16 == 0):
            # decode path: fused int8 GEMV (weights stay int8 in HBM)
            y = get_ext().w8_gemv(self.qweight, self.scale.view(-1),
                                  x.reshape(rows, -1).contiguous())
            y = y.view(*x.shape[:-1], self.out_features)
        else:
            w = (self.qweight.to(x.dtype) * self.scale.to(x.dtype))
            y = torch.nn.functional.linear(x, w)
        if self.bias:
            y = y + self.bias_buf.to(x.dtype)
        return y

    def extra_repr(self):
        return f"in={self.in_features}, out={self.out_features}, w8"


_QUANT_TARGETS = ("qkv_proj", "out_proj", "gate_up_proj", "down_proj",
                  "fc_in", "fc_out", "lm_head")


def quantize_model_int8(model: nn.Module, targets=_QUANT_TARGETS) -> nn.Module:
    """Replace target Linear-like modules with W8Linear (in place)."""
    from fengshen_amd.parallel.layers import (
        ColumnParallelLinear, RowParallelLinear)

    for name, module in model.named_children():
        if isinstance(module, (nn.Linear, ColumnParallelLinear,
                               RowParallelLinear)) and \
                any(t in name for t in targets):
            setattr(model, name, W8Linear(module.weight,
                                          getattr(module, "bias", None)))
        else:
            quantize_model_int8(module, targets)
    return model


def quantized_bytes(model: nn.Module) -> int:
    total = 0
    for p in model.parameters():
        total += p.numel() * p.element_size()
    for b in model.buffers():
        total += b.numel() * b.element_size()
    return total
