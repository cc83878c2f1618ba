"""TCBert prompt topic-classification demo.

Behavioral parity: reference examples/tcbert/example.py — prompt
"这是一条关于{label}的新闻：" + text, label read from the MLM head at the
masked label positions, via TCBertPipeline.

Run:  python example_tcbert.py [--model_path IDEA-style checkpoint]
With no --model_path a tiny random-weight model runs (smoke mode).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

from fengshen_amd.models.tcbert.modeling_tcbert import TCBertConfig
from fengshen_amd.pipelines.tcbert import TCBertPipeline
from fengshen_amd.tokenizer import SimpleCharTokenizer


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_path", default=None)
    args = parser.parse_args()

    labels = ["体育", "财经", "科技"]
    if args.model_path:
        from transformers import AutoTokenizer
        pipe = TCBertPipeline(model=args.model_path,
                              tokenizer=AutoTokenizer.from_pretrained(
                                  args.model_path),
                              labels=labels)
    else:
        cfg = TCBertConfig(vocab_size=300, hidden_size=64,
                           num_hidden_layers=2, num_attention_heads=4,
                           intermediate_size=128, torch_dtype="float32")
        pipe = TCBertPipeline(tokenizer=SimpleCharTokenizer(),
                              config=cfg, labels=labels)

    samples = ["昨晚的比赛中主队以三比一获胜。",
               "股市今日大幅上涨，成交量创新高。",
               "新一代芯片的算力再次翻倍。"]
    for text, res in zip(samples, pipe(samples)):
        print(f"{res['label_name']}: {text}")


if __name__ == "__main__":
    main()
