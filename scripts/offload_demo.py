"""Low-VRAM finetune demo: Erlangshen-1.3B with ZeRO-offload on one GPU.

Reference parity: fengshen/README.md:88 — "7 GB VRAM finetune of 1.3B".
Here: bf16 weights+grads on device, fp32 master + Adam moments in host
RAM (zero2_offload), activation checkpointing on.  Prints peak VRAM.

Run on a GPU box:  python scripts/offload_demo.py [--steps 5]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..")))

import torch

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    erlangshen_1b3_config,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForPreTraining,
)
from fengshen_amd.parallel.zero import ZeroOptimizer


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--steps", default=5, type=int)
    parser.add_argument("--batch", default=4, type=int)
    parser.add_argument("--seq_len", default=512, type=int)
    args = parser.parse_args()
    assert torch.cuda.is_available()
    device = torch.device("cuda:0")

    torch.manual_seed(0)
    torch.set_default_dtype(torch.bfloat16)
    with device:
        model = MegatronBertForPreTraining(erlangshen_1b3_config())
    torch.set_default_dtype(torch.float32)
    model = model.to(torch.bfloat16).to(device)
    model.gradient_checkpointing_enable()
    model.train()

    opt = ZeroOptimizer(model.parameters(), stage=2, lr=1e-5,
                        weight_decay=0.01, cpu_offload=True)
    vocab = model.config.vocab_size
    b, s = args.batch, args.seq_len
    ids = torch.randint(3, vocab, (b, s), device=device)
    labels = ids.clone()
    labels[torch.rand(b, s, device=device) > 0.15] = -100
    batch = dict(input_ids=ids, labels=labels,
                 next_sentence_label=torch.randint(0, 2, (b,), device=device),
                 attention_mask=torch.ones_like(ids))

    torch.cuda.reset_peak_memory_stats()
    for i in range(args.steps):
        out = model(**batch)
        opt.zero_grad()
        out.loss.backward()
        opt.step()
        print(f"step {i}: loss {out.loss.item():.4f}")
    peak = torch.cuda.max_memory_allocated() / 2**30
    host = sum(bk.master_shard.numel() * 12 for bk in opt.buckets) / 2**30
    print(f"peak VRAM: {peak:.2f} GiB (optimizer states offloaded: "
          f"{host:.2f} GiB in host RAM)")


if __name__ == "__main__":
    main()
