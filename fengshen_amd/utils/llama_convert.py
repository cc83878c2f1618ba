"""HF-LLaMA <-> fengshen_amd LLaMA weight conversion + delta weights.

Behavioral parity: reference utils/llama_convert/ (hf_to_fs.py incl. rotary
permute + vocab rounding :56-110, fs_to_hf.py, convert_fs_llama_tp.py) and
utils/apply_delta.py / make_delta.py (Ziya delta-weight release flow).

Our LLaMA differs from HF's in two ways: fused qkv_proj (q;k;v rows) and
fused gate_up_proj (gate;up rows).  RoPE convention matches HF's
half-rotation layout, so no rotary permute is needed.
"""
from __future__ import annotations

from typing import Dict

import torch


def hf_to_fs_llama(hf_sd: Dict[str, torch.Tensor],
                   num_layers: int) -> Dict[str, torch.Tensor]:
    """Map HF LlamaForCausalLM names -> our fused layout."""
    out = {}
    out["model.embed_tokens.weight"] = hf_sd["model.embed_tokens.weight"]
    out["model.norm.weight"] = hf_sd["model.norm.weight"]
    out["lm_head.weight"] = hf_sd["lm_head.weight"]
    for i in range(num_layers):
        p = f"model.layers.{i}"
        out[f"{p}.input_norm.weight"] = hf_sd[f"{p}.input_layernorm.weight"]
        out[f"{p}.post_attention_norm.weight"] = \
            hf_sd[f"{p}.post_attention_layernorm.weight"]
        out[f"{p}.attention.qkv_proj.weight"] = torch.cat([
            hf_sd[f"{p}.self_attn.q_proj.weight"],
            hf_sd[f"{p}.self_attn.k_proj.weight"],
            hf_sd[f"{p}.self_attn.v_proj.weight"]], dim=0)
        out[f"{p}.attention.out_proj.weight"] = \
            hf_sd[f"{p}.self_attn.o_proj.weight"]
        out[f"{p}.mlp.gate_up_proj.weight"] = torch.cat([
            hf_sd[f"{p}.mlp.gate_proj.weight"],
            hf_sd[f"{p}.mlp.up_proj.weight"]], dim=0)
        out[f"{p}.mlp.down_proj.weight"] = hf_sd[f"{p}.mlp.down_proj.weight"]
    return out


def fs_to_hf_llama(fs_sd: Dict[str, torch.Tensor],
                   num_layers: int) -> Dict[str, torch.Tensor]:
    out = {}
    out["model.embed_tokens.weight"] = fs_sd["model.embed_tokens.weight"]
    out["model.norm.weight"] = fs_sd["model.norm.weight"]
    out["lm_head.weight"] = fs_sd["lm_head.weight"]
    for i in range(num_layers):
        p = f"model.layers.{i}"
        out[f"{p}.input_layernorm.weight"] = fs_sd[f"{p}.input_norm.weight"]
        out[f"{p}.post_attention_layernorm.weight"] = \
            fs_sd[f"{p}.post_attention_norm.weight"]
        qkv = fs_sd[f"{p}.attention.qkv_proj.weight"]
        h = qkv.shape[0] // 3
        out[f"{p}.self_attn.q_proj.weight"] = qkv[:h]
        out[f"{p}.self_attn.k_proj.weight"] = qkv[h:2 * h]
        out[f"{p}.self_attn.v_proj.weight"] = qkv[2 * h:]
        out[f"{p}.self_attn.o_proj.weight"] = \
            fs_sd[f"{p}.attention.out_proj.weight"]
        gu = fs_sd[f"{p}.mlp.gate_up_proj.weight"]
        ff = gu.shape[0] // 2
        out[f"{p}.mlp.gate_proj.weight"] = gu[:ff]
        out[f"{p}.mlp.up_proj.weight"] = gu[ff:]
        out[f"{p}.mlp.down_proj.weight"] = fs_sd[f"{p}.mlp.down_proj.weight"]
    return out


def pad_vocab(weight: torch.Tensor, multiple: int = 128) -> torch.Tensor:
    """Round the vocab dim up (reference hf_to_fs.py:56 vocab rounding)."""
    v, h = weight.shape
    target = ((v + multiple - 1) // multiple) * multiple
    if target == v:
        return weight
    out = torch.zeros(target, h, dtype=weight.dtype)
    out[:v] = weight
    return out


# ---------------------------------------------------------------------------
# delta weights (reference utils/make_delta.py / apply_delta.py)
# ---------------------------------------------------------------------------
def make_delta(base_sd: Dict[str, torch.Tensor],
               target_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """delta = target - base (released instead of the full finetune)."""
    delta = {}
    for k, v in target_sd.items():
        delta[k] = (v.float() - base_sd[k].float()).to(v.dtype) \
            if k in base_sd else v
    return delta


def apply_delta(base_sd: Dict[str, torch.Tensor],
                delta_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    out = {}
    for k, v in delta_sd.items():
        out[k] = (base_sd[k].float() + v.float()).to(v.dtype) \
            if k in base_sd else v
    return out
