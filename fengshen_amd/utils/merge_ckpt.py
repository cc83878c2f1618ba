"""Merge a trainer checkpoint directory into an HF save_pretrained dir.

Behavioral parity: reference utils/llama_convert/merge_lt_mp_to_hf.py +
utils/fs_merge_weight.py — take the per-TP-rank `model_part_{i}.pt`
shards that Trainer.save_checkpoint writes, merge them along each
parallel layer's shard dim (tp_convert rules), and export a plain HF
checkpoint (config.json + safetensors) loadable with from_pretrained on
any topology.
"""
from __future__ import annotations

import glob
import os
import re
from typing import Callable, Optional

import torch


def load_trainer_model_state(ckpt_dir: str, reference_model=None) -> dict:
    """Read model_part_*.pt from a Trainer checkpoint dir; merge TP shards
    (needs `reference_model` built at TP=1 for the shard rules when more
    than one part exists)."""
    parts = sorted(
        glob.glob(os.path.join(ckpt_dir, "model_part_*.pt")),
        key=lambda p: int(re.search(r"model_part_(\d+)", p).group(1)))
    if not parts:
        raise FileNotFoundError(f"no model_part_*.pt under {ckpt_dir}")
    shards = [torch.load(p, map_location="cpu", weights_only=True)
              for p in parts]
    if len(shards) == 1:
        return shards[0]
    if reference_model is None:
        raise ValueError("TP>1 checkpoint: pass reference_model (built at "
                         "TP=1) so shard dims are known")
    from fengshen_amd.utils.tp_convert import merge_state_dicts
    return merge_state_dicts(reference_model, shards)


def trainer_ckpt_to_hf(ckpt_dir: str, model_ctor: Callable[[], torch.nn.Module],
                       out_dir: str, model_attr: Optional[str] = "model") -> str:
    """ckpt_dir: Trainer.save_checkpoint output.  model_ctor: builds the
    HF model at TP=1 (e.g. `lambda: LlamaForCausalLM(cfg)`).  model_attr:
    the FengshenModule attribute the app stored the HF model under
    ("model" in every shipped example) — its prefix is stripped from the
    checkpoint keys; None if the module itself was saved."""
    hf_model = model_ctor()
    sd = load_trainer_model_state(ckpt_dir, reference_model=hf_model)
    if model_attr:
        prefix = model_attr + "."
        sd = {k[len(prefix):]: v for k, v in sd.items()
              if k.startswith(prefix)}
    missing, unexpected = hf_model.load_state_dict(sd, strict=False)
    if missing:
        raise RuntimeError(f"missing keys after merge: {missing[:5]}...")
    hf_model.save_pretrained(out_dir)
    return out_dir
