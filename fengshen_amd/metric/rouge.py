"""ROUGE for Chinese summarization (ref: torchmetrics ROUGEScore usage in
examples/summary/seq2seq_summary.py:38 — torchmetrics is not in this image,
so this is a native implementation).

Operates on pre-tokenized strings (use utils.chinese_char_tokenize to get
char-level tokens, matching the reference's chinese_char_tokenize +
normalizer=identity recipe).  Reports precision/recall/fmeasure for
rouge-1, rouge-2 and rouge-L (LCS).
"""
from __future__ import annotations

from collections import Counter
from typing import Dict, Iterable, List, Sequence


def _ngrams(tokens: Sequence[str], n: int) -> Counter:
    return Counter(tuple(tokens[i:i + n])
                   for i in range(len(tokens) - n + 1))


def _prf(match: int, pred: int, ref: int) -> Dict[str, float]:
    p = match / pred if pred else 0.0
    r = match / ref if ref else 0.0
    f = 2 * p * r / (p + r) if p + r else 0.0
    return {"precision": p, "recall": r, "fmeasure": f}


def _lcs_len(a: Sequence[str], b: Sequence[str]) -> int:
    if not a or not b:
        return 0
    prev = [0] * (len(b) + 1)
    for i in range(1, len(a) + 1):
        cur = [0] * (len(b) + 1)
        ai = a[i - 1]
        for j in range(1, len(b) + 1):
            if ai == b[j - 1]:
                cur[j] = prev[j - 1] + 1
            else:
                cur[j] = max(prev[j], cur[j - 1])
        prev = cur
    return prev[-1]


def rouge_n(pred: Sequence[str], ref: Sequence[str],
            n: int) -> Dict[str, float]:
    pg, rg = _ngrams(pred, n), _ngrams(ref, n)
    match = sum((pg & rg).values())
    return _prf(match, max(sum(pg.values()), 0), max(sum(rg.values()), 0))


def rouge_l(pred: Sequence[str], ref: Sequence[str]) -> Dict[str, float]:
    lcs = _lcs_len(pred, ref)
    return _prf(lcs, len(pred), len(ref))


class RougeScore:
    """Streaming ROUGE accumulator (mirrors torchmetrics' interface shape:
    update(preds, targets) then compute() -> {rouge1_fmeasure, ...})."""

    def __init__(self, rouge_keys=("rouge1", "rouge2", "rougeL")):
        self.rouge_keys = tuple(rouge_keys)
        self.reset()

    def reset(self):
        self._scores: Dict[str, List[float]] = {
            f"{k}_{m}": [] for k in self.rouge_keys
            for m in ("precision", "recall", "fmeasure")}

    def _one(self, key: str, pred: Sequence[str], ref: Sequence[str]):
        if key == "rouge1":
            return rouge_n(pred, ref, 1)
        if key == "rouge2":
            return rouge_n(pred, ref, 2)
        if key in ("rougeL", "rougeLsum"):
            return rouge_l(pred, ref)
        raise ValueError(key)

    def update(self, preds: Iterable[str], targets: Iterable[str]):
        """preds/targets: whitespace-pre-tokenized strings (one per
        sample), as produced by chinese_char_tokenize."""
        if isinstance(preds, str):
            preds, targets = [preds], [targets]  # type: ignore[list-item]
        for p, t in zip(preds, targets):
            ptok = p.split() if isinstance(p, str) else list(p)
            ttok = t.split() if isinstance(t, str) else list(t)
            for key in self.rouge_keys:
                sc = self._one(key, ptok, ttok)
                for m, v in sc.items():
                    self._scores[f"{key}_{m}"].append(v)

    def compute(self) -> Dict[str, float]:
        return {k: (sum(v) / len(v) if v else 0.0)
                for k, v in self._scores.items()}
