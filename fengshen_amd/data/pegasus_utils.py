"""Pegasus gap-sentence pretraining utilities.

Behavioral parity: reference examples/pegasus/data_utils.py:99-127 — select
the top-k "principal" sentences by ROUGE-1-like overlap with the rest of the
document, replace them with a mask sentinel in the input, use them as the
summarization target.
"""
from __future__ import annotations

from typing import List, Tuple

from fengshen_amd.data.data_utils import ChineseSentenceSplitter


def _unigram_overlap(a: str, b: str) -> float:
    sa, sb = set(a), set(b)
    if not sa or not sb:
        return 0.0
    inter = len(sa & sb)
    prec = inter / len(sa)
    rec = inter / len(sb)
    return 0.0 if prec + rec == 0 else 2 * prec * rec / (prec + rec)


def select_gap_sentences(sentences: List[str],
                         gap_ratio: float = 0.3) -> List[int]:
    """indices of principal sentences (highest overlap with the rest)."""
    n = len(sentences)
    k = max(1, int(round(n * gap_ratio)))
    scores = []
    for i, s in enumerate(sentences):
        rest = "".join(sentences[:i] + sentences[i + 1:])
        scores.append((_unigram_overlap(s, rest), i))
    scores.sort(reverse=True)
    return sorted(i for _, i in scores[:k])


def build_gap_sentence_sample(text: str, mask_token: str = "[MASK]",
                              gap_ratio: float = 0.3) -> Tuple[str, str]:
    """-> (masked input text, target text of the gap sentences)."""
    sentences = ChineseSentenceSplitter().tokenize(text)
    if len(sentences) < 2:
        return text, text
    gaps = set(select_gap_sentences(sentences, gap_ratio))
    inp = "".join(mask_token if i in gaps else s
                  for i, s in enumerate(sentences))
    tgt = "".join(sentences[i] for i in sorted(gaps))
    return inp, tgt
