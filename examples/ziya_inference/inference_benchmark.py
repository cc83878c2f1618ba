"""Ziya-LLaMA inference benchmark: ms/token for bf16 and int8-weight paths.

Behavioral parity: reference examples/ziya_inference/readme.md:17-24 (the
repo's only hard throughput table: fp16 49.89 ms/token @2x3090, int8 147.06
@1x3090).  Ours: one MI355X, bf16 vs W8 weight-only.
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse
import json
import time

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model_size", default="13b", choices=["tiny", "13b"])
    p.add_argument("--model_path", default=None)
    p.add_argument("--prompt_len", type=int, default=64)
    p.add_argument("--new_tokens", type=int, default=64)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--quant", choices=["bf16", "int8"], default="bf16")
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-captured decode loop (serving fast path)")
    p.add_argument("--sample", action="store_true",
                   help="top-k/top-p sampling inside the graph "
                        "(gumbel-max; greedy otherwise)")
    p.add_argument("--top_k", type=int, default=50)
    p.add_argument("--top_p", type=float, default=0.95)
    p.add_argument("--temperature", type=float, default=0.8)
    args = p.parse_args()
    assert torch.cuda.is_available()

    from fengshen_amd.models.llama.configuration_llama import (
        llama_tiny_config, ziya_llama_13b_config)
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM

    torch.manual_seed(0)
    if args.model_path:
        model = LlamaForCausalLM.from_pretrained(args.model_path)
    else:
        cfg = (ziya_llama_13b_config() if args.model_size == "13b"
               else llama_tiny_config())
        model = LlamaForCausalLM(cfg)
    model = model.to(torch.bfloat16).to("cuda").eval()

    if args.quant == "int8":
        from fengshen_amd.utils.quantize import (
            quantize_model_int8, quantized_bytes)
        quantize_model_int8(model)
        print(f"int8 model bytes: {quantized_bytes(model) / 2**30:.1f} GiB")

    vocab = model.config.vocab_size
    ids = torch.randint(3, vocab, (args.batch, args.prompt_len),
                        device="cuda")
    if args.graph:
        from fengshen_amd.serving.graphed_decode import GraphedDecoder
        dec = GraphedDecoder(model, batch=args.batch,
                             max_len=args.prompt_len + args.new_tokens + 8,
                             max_new_tokens=args.new_tokens,
                             do_sample=args.sample, top_k=args.top_k,
                             top_p=args.top_p, temperature=args.temperature)
        dec.generate(ids, max_new_tokens=4)  # warmup + capture
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = dec.generate(ids, max_new_tokens=args.new_tokens)
        torch.cuda.synchronize()
    else:
        model.generate(ids, max_new_tokens=4, do_sample=False)  # warmup
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = model.generate(ids, max_new_tokens=args.new_tokens,
                             do_sample=False)
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    n_new = out.shape[1] - args.prompt_len
    print(json.dumps({
        "metric": "ms/token Ziya-LLaMA-13B generate",
        "value": round(dt / n_new * 1000, 2),
        "unit": "ms/token",
        "quant": args.quant,
        "graph": bool(args.graph),
        "batch": args.batch,
        "prompt_len": args.prompt_len,
        "new_tokens": n_new,
        "hbm_gib": round(torch.cuda.max_memory_allocated() / 2**30, 1),
        "reference_anchor": "fp16 49.89 ms/tok @2x3090; int8 147.06 @1x3090",
    }))


if __name__ == "__main__":
    main()
