"""Paraphrase generation with the fixed pretrain prompt
(ref models/transfo_xl_paraphrase/generate.py:16-59)."""
from __future__ import annotations

import torch

from fengshen_amd.utils.transfo_xl_utils import sample_sequence


@torch.no_grad()
def paraphrase_generate(model, tokenizer, input_text: str,
                        device=None, temperature: float = 1.0,
                        top_p: float = 0.9, max_out_seq: int = 100,
                        eod_token: int = 50000) -> str:
    """Generate a paraphrase with the fixed prompt
    “<input>”的相似句是“ ... ” (ref :39-41 prompt construction)."""
    prompt = f"“{input_text}”的相似句是“"
    prompt_tokens = tokenizer.encode(prompt)
    if prompt_tokens and prompt_tokens[-1] in (
            getattr(tokenizer, "sep_token_id", None),
            getattr(tokenizer, "eos_token_id", None)):
        prompt_tokens = prompt_tokens[:-1]
    dev = device if device is not None else next(model.parameters()).device
    tokens = torch.tensor([prompt_tokens], dtype=torch.long, device=dev)
    model = model.eval().to(dev)
    out_tokens, _mems = sample_sequence(
        model, tokens, do_sampling=True, temperature=temperature,
        top_p=top_p, max_out_seq=len(prompt_tokens) + max_out_seq,
        end_token_id=eod_token)
    res = tokenizer.decode(out_tokens[len(prompt_tokens):])
    return res.split("”")[0]
