"""Predictions -> CLUE submission files
(reference examples/clue1.1/predict2submit/*_submit.py): map unified
predictions back to each task's submission schema."""
from __future__ import annotations

import argparse
import json

from clue2unidata import TNEWS_LABELS

DESC2LABEL = {v: k for k, v in TNEWS_LABELS.items()}


def tnews_submit(pred: dict) -> dict:
    return {"id": pred["id"], "label_desc": DESC2LABEL.get(pred["label"])}


def afqmc_submit(pred: dict) -> dict:
    return {"id": pred["id"], "label": str(pred["label_index"])}


def ocnli_submit(pred: dict) -> dict:
    m = {"蕴含": "entailment", "矛盾": "contradiction", "中立": "neutral"}
    return {"id": pred["id"], "label": m.get(pred["label"])}


SUBMITTERS = {"tnews": tnews_submit, "afqmc": afqmc_submit,
              "ocnli": ocnli_submit, "cmnli": ocnli_submit,
              "csl": afqmc_submit, "wsc": afqmc_submit}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--task", required=True, choices=sorted(SUBMITTERS))
    p.add_argument("--input", required=True)
    p.add_argument("--output", required=True)
    args = p.parse_args()
    sub = SUBMITTERS[args.task]
    with open(args.input, encoding="utf8") as f, \
            open(args.output, "w", encoding="utf8") as out:
        for line in f:
            if line.strip():
                out.write(json.dumps(sub(json.loads(line)),
                                     ensure_ascii=False) + "\n")


if __name__ == "__main__":
    main()
