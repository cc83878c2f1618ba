"""MLM sample construction utilities.

Behavioral parity: reference data/data_utils/ —
create_masked_lm_predictions (mask_utils.py:18: whole-word-mask + n-gram,
bert/t5 styles), get_a_and_b_segments (sop_utils.py:3), truncate_segments
(truncate_utils.py:2), create_tokens_and_tokentypes (token_type_utils.py:1),
ChineseSentenceSplitter (sentence_split.py:4).
"""
from __future__ import annotations

import re
from typing import List, Sequence, Tuple

import numpy as np


class ChineseSentenceSplitter:
    """Split Chinese text into sentences on 。！？；… and newline."""

    _PAT = re.compile(r"([。！？；…\!\?;]+[”’\"']?|\n+)")

    def tokenize(self, text: str) -> List[str]:
        parts = self._PAT.split(text)
        sentences = []
        cur = ""
        for p in parts:
            if p is None:
                continue
            if self._PAT.fullmatch(p):
                cur += p.strip("\n")
                if cur.strip():
                    sentences.append(cur.strip())
                cur = ""
            else:
                cur += p
        if cur.strip():
            sentences.append(cur.strip())
        return sentences


def get_a_and_b_segments(sample: Sequence[Sequence[int]], np_rng) -> Tuple[
        List[int], List[int], bool]:
    """Split a doc's sentences into A/B; 50% swapped => SOP label
    (reference sop_utils.py:3)."""
    n_sentences = len(sample)
    assert n_sentences > 1
    a_end = 1 if n_sentences == 2 else np_rng.randint(1, n_sentences)
    tokens_a: List[int] = []
    for j in range(a_end):
        tokens_a.extend(sample[j])
    tokens_b: List[int] = []
    for j in range(a_end, n_sentences):
        tokens_b.extend(sample[j])
    is_next_random = False
    if np_rng.random() < 0.5:
        is_next_random = True
        tokens_a, tokens_b = tokens_b, tokens_a
    return tokens_a, tokens_b, is_next_random


def truncate_segments(tokens_a: List[int], tokens_b: List[int], len_a: int,
                      len_b: int, max_num_tokens: int, np_rng) -> bool:
    """Trim front/back until total fits (reference truncate_utils.py:2)."""
    assert len_a > 0
    if len_a + len_b <= max_num_tokens:
        return False
    while len_a + len_b > max_num_tokens:
        if len_a > len_b:
            len_a -= 1
            tokens = tokens_a
        else:
            len_b -= 1
            tokens = tokens_b
        if np_rng.random() < 0.5:
            del tokens[0]
        else:
            tokens.pop()
    return True


def create_tokens_and_tokentypes(tokens_a, tokens_b, cls_id: int, sep_id: int):
    """[CLS] A [SEP] B [SEP] + token types (reference token_type_utils.py:1)."""
    tokens = [cls_id] + list(tokens_a) + [sep_id]
    tokentypes = [0] * len(tokens)
    if tokens_b:
        tokens += list(tokens_b) + [sep_id]
        tokentypes += [1] * (len(tokens_b) + 1)
    return tokens, tokentypes


def is_start_piece(piece: str) -> bool:
    return not piece.startswith("##")


MaskedLmInstance = Tuple[int, int]  # (position, original label)


def create_masked_lm_predictions(
        tokens: List[int], vocab_id_list: List[int], vocab_id_to_token_dict,
        masked_lm_prob: float, cls_id: int, sep_id: int, mask_id: int,
        max_predictions_per_seq: int, np_rng,
        max_ngrams: int = 3, do_whole_word_mask: bool = True,
        favor_longer_ngram: bool = False, geometric_dist: bool = False,
        masking_style: str = "bert"):
    """Whole-word / n-gram masking (reference mask_utils.py:18).

    Returns (output_tokens, masked_positions, masked_labels).
    masking_style 'bert': 80% [MASK] / 10% random / 10% keep.
    masking_style 't5':   always replace (caller builds sentinels).
    """
    cand_indexes: List[List[int]] = []
    token_boundary = [0] * len(tokens)
    for (i, token) in enumerate(tokens):
        if token == cls_id or token == sep_id:
            token_boundary[i] = 1
            continue
        piece = vocab_id_to_token_dict.get(token, "") \
            if hasattr(vocab_id_to_token_dict, "get") \
            else vocab_id_to_token_dict[token]
        if (do_whole_word_mask and len(cand_indexes) >= 1
                and not is_start_piece(piece)):
            cand_indexes[-1].append(i)
        else:
            cand_indexes.append([i])
            if is_start_piece(piece):
                token_boundary[i] = 1

    output_tokens = list(tokens)
    num_to_predict = min(max_predictions_per_seq,
                         max(1, int(round(len(tokens) * masked_lm_prob))))

    ngrams = np.arange(1, max_ngrams + 1, dtype=np.int64)
    if not geometric_dist:
        pvals = 1.0 / np.arange(1, max_ngrams + 1)
        pvals /= pvals.sum(keepdims=True)
        if favor_longer_ngram:
            pvals = pvals[::-1]

    ngram_indexes = []
    for idx in range(len(cand_indexes)):
        ngram_index = []
        for n in ngrams:
            ngram_index.append(cand_indexes[idx:idx + n])
        ngram_indexes.append(ngram_index)
    np_rng.shuffle(ngram_indexes)

    masked_lms: List[MaskedLmInstance] = []
    covered = set()
    for cand_index_set in ngram_indexes:
        if len(masked_lms) >= num_to_predict:
            break
        if not cand_index_set:
            continue
        if any(i in covered
               for index_set in cand_index_set[0] for i in index_set):
            continue
        if not geometric_dist:
            n = np_rng.choice(
                ngrams[:len(cand_index_set)],
                p=pvals[:len(cand_index_set)]
                / pvals[:len(cand_index_set)].sum(keepdims=True))
        else:
            n = min(np_rng.geometric(0.2), max_ngrams)
        index_set = sum(cand_index_set[n - 1], [])
        while len(masked_lms) + len(index_set) > num_to_predict and n > 1:
            n -= 1
            index_set = sum(cand_index_set[n - 1], [])
        if len(masked_lms) + len(index_set) > num_to_predict:
            continue
        for index in index_set:
            covered.add(index)
            label = tokens[index]
            if masking_style == "bert":
                r = np_rng.random()
                if r < 0.8:
                    output_tokens[index] = mask_id
                elif r < 0.9:
                    output_tokens[index] = vocab_id_list[
                        np_rng.randint(0, len(vocab_id_list))]
                # else keep original
            else:  # t5-style: always mask
                output_tokens[index] = mask_id
            masked_lms.append((index, label))
    masked_lms.sort(key=lambda t: t[0])
    positions = [p for p, _ in masked_lms]
    labels = [l for _, l in masked_lms]
    return output_tokens, positions, labels
