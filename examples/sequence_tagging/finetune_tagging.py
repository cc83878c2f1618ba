"""NER finetune app (reference examples/sequence_tagging)."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse

from fengshen_amd.pipelines.sequence_tagging import SequenceTaggingPipeline


def synthetic_ner(n=64):
    # 李明 = PER, 北京 = LOC
    return [{"text": "李明住在北京",
             "labels": ["B-PER", "I-PER", "O", "O", "B-LOC", "I-LOC"]}
            for _ in range(n)]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", type=str, default=None)
    parser.add_argument("--head", type=str, default="crf",
                        choices=["linear", "crf"])
    SequenceTaggingPipeline.add_pipeline_specific_args(parser)
    args = parser.parse_args()

    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    id2label = {0: "O", 1: "B-PER", 2: "I-PER", 3: "B-LOC", 4: "I-LOC"}
    config = None
    if args.model is None:
        from fengshen_amd.models.megatron_bert.configuration_megatron_bert \
            import bert_tiny_config
        config = bert_tiny_config()
    pipe = SequenceTaggingPipeline(args=args, model=args.model,
                                   tokenizer=tokenizer, id2label=id2label,
                                   config=config, head=args.head)
    datasets = {"train": synthetic_ner()} if not args.train_file else None
    pipe.train(datasets)
    print(pipe("李明去了北京"))


if __name__ == "__main__":
    main()
