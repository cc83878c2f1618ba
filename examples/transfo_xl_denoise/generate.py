"""Bigan-Transfo-XL denoise generation demo.

Behavioral parity: reference examples/{transfo_xl_denoise usage} +
utils/transfo_xl_utils.py sampling helpers — segment-recurrent greedy /
top-k decoding with cached memories.

Run:  python generate.py [--seq_len 48]
Uses the tiny config with random weights when no --model_path is given.
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch

from fengshen_amd.models.transfo_xl_denoise.modeling_transfo_xl_denoise import (
    TransfoXLDenoiseModel,
    transfo_xl_tiny_config,
)


@torch.no_grad()
def generate(model, input_ids, max_new_tokens=32, top_k=1, segment=16):
    """Segment-recurrent decode: feed segments, carry mems."""
    ids = input_ids
    mems = None
    # prime memories with the prompt in segment chunks
    for s in range(0, ids.shape[1], segment):
        out = model(ids[:, s:s + segment], mems=mems)
        mems = out.mems
    cur = ids[:, -1:]
    for _ in range(max_new_tokens):
        out = model(cur, mems=mems)
        mems = out.mems
        logits = out.logits[:, -1]
        if top_k > 1:
            v, ix = logits.topk(top_k)
            probs = torch.softmax(v.float(), -1)
            nxt = ix.gather(-1, torch.multinomial(probs, 1))
        else:
            nxt = logits.argmax(-1, keepdim=True)
        ids = torch.cat([ids, nxt], dim=1)
        cur = nxt
    return ids


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_path", default=None)
    parser.add_argument("--seq_len", default=48, type=int)
    parser.add_argument("--top_k", default=1, type=int)
    args = parser.parse_args()

    if args.model_path:
        model = TransfoXLDenoiseModel.from_pretrained(args.model_path)
    else:
        model = TransfoXLDenoiseModel(
            transfo_xl_tiny_config(torch_dtype="float32")).float()
    model.eval()
    torch.manual_seed(0)
    prompt = torch.randint(5, 200, (1, 24))
    out = generate(model, prompt, max_new_tokens=args.seq_len - 24,
                   top_k=args.top_k)
    print("generated ids:", out[0].tolist())


if __name__ == "__main__":
    main()
