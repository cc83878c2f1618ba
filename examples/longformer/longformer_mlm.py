"""Longformer MLM demo: band + global attention at long context
(ref examples/longformer README usage)."""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

from fengshen_amd.models.longformer.modeling_longformer import (  # noqa: E402
    LongformerForMaskedLM,
    longformer_tiny_config,
)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model_path", default=None)
    p.add_argument("--seq_len", type=int, default=1024)
    p.add_argument("--batch", type=int, default=2)
    p.add_argument("--device", default="cuda" if torch.cuda.is_available()
                   else "cpu")
    args = p.parse_args()
    torch.manual_seed(0)
    if args.model_path:
        model = LongformerForMaskedLM.from_pretrained(args.model_path)
    else:
        model = LongformerForMaskedLM(longformer_tiny_config(
            max_position_embeddings=max(args.seq_len, 512)))
    model = model.to(args.device).train()
    vocab = model.config.vocab_size
    ids = torch.randint(3, vocab, (args.batch, args.seq_len),
                        device=args.device)
    labels = ids.clone()
    labels[torch.rand_like(ids, dtype=torch.float) > 0.15] = -100
    gmask = torch.zeros_like(ids)
    gmask[:, 0] = 1  # CLS attends globally
    out = model(ids, attention_mask=torch.ones_like(ids),
                global_attention_mask=gmask, labels=labels)
    out.loss.backward()
    print(f"seq_len={args.seq_len} mlm_loss={out.loss.item():.4f}")


if __name__ == "__main__":
    main()
