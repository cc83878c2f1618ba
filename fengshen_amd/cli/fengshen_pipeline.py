"""fengshen-pipeline CLI.

Behavioral parity: reference cli/fengshen_pipeline.py:7-34 —
`fengshen-pipeline <task> <train|predict> --model=... --datasets=... text`.
"""
from __future__ import annotations

import argparse
import sys
from importlib import import_module


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    if len(argv) < 2:
        print("usage: fengshen-pipeline <task> <train|predict> [args...]")
        return 1
    task, verb = argv[0], argv[1]
    try:
        mod = import_module(f"fengshen_amd.pipelines.{task}")
    except ImportError as e:
        print(f"unknown task {task!r}: {e}")
        return 1
    Pipeline = mod.Pipeline

    parser = argparse.ArgumentParser(prog=f"fengshen-pipeline {task} {verb}")
    parser.add_argument("--model", type=str, default=None)
    parser.add_argument("--datasets", type=str, default=None)
    parser.add_argument("--text", type=str, default=None)
    Pipeline.add_pipeline_specific_args(parser)
    args, _unknown = parser.parse_known_args(argv[2:])

    from transformers import AutoTokenizer
    tokenizer = AutoTokenizer.from_pretrained(args.model) \
        if args.model else None
    pipe = Pipeline(args=args, model=args.model, tokenizer=tokenizer)

    if verb == "train":
        from datasets import load_dataset
        datasets = load_dataset(args.datasets)
        pipe.train(datasets)
    elif verb == "predict":
        result = pipe(args.text)
        print(result)
    else:
        print(f"unknown verb {verb!r} (use train|predict)")
        return 1
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
