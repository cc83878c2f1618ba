"""Text-classification finetune app (the reference's most-used finetune
family: examples/classification with 14 sbatch configs).

Uses the TextClassificationPipeline training path over any two-sentence or
single-sentence dataset (AFQMC/TNEWS-style json: {"sentence": ..., "label": N}).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

from fengshen_amd.pipelines.text_classification import (
    TextClassificationPipeline,
)


def synthetic_clue(n=128):
    pos = "这家餐厅的菜品味道非常好值得推荐"
    neg = "等了一个小时都没有上菜体验很差"
    return [{"sentence": (pos if i % 2 == 0 else neg) + f"第{i}条",
             "label": i % 2} for i in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", type=str, default=None)
    parser.add_argument("--num_labels", type=int, default=2)
    parser.add_argument("--tokenizer", type=str, default=None)
    TextClassificationPipeline.add_pipeline_specific_args(parser)
    args = parser.parse_args()

    if args.tokenizer:
        from transformers import AutoTokenizer
        tokenizer = AutoTokenizer.from_pretrained(args.tokenizer)
    else:
        from fengshen_amd.tokenizer import SimpleCharTokenizer
        tokenizer = SimpleCharTokenizer()

    config = None
    if args.model is None:
        from fengshen_amd.models.megatron_bert.configuration_megatron_bert \
            import bert_tiny_config
        config = bert_tiny_config()
        config.num_labels = args.num_labels
    pipe = TextClassificationPipeline(args=args, model=args.model,
                                      tokenizer=tokenizer, config=config)
    datasets = {"train": synthetic_clue()} if not args.train_file else None
    pipe.train(datasets)
    print(pipe("这家店的服务特别棒"))


if __name__ == "__main__":
    main()
