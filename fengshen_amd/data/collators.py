"""Task collators (the reference keeps one per workload family).

Behavioral parity:
  MlmSopCollator   — pretrain_erlangshen.py:36-123 (sentence split -> SOP
                     pair -> wwm/n-gram MLM masking)
  SftCollator      — examples/ziya_llama/finetune_ziya_llama.py:35-85
                     ("<human>:...\n<bot>:" prompt masked to -100)
  T5SpanCollator   — data/t5_dataloader/t5_datasets.py (span corruption
                     with sentinel ids, compute_input_and_target_lengths)
  GptFimCollator   — plain causal LM packing
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List

import numpy as np
import torch

from fengshen_amd.data.data_utils import (
    ChineseSentenceSplitter,
    create_masked_lm_predictions,
    create_tokens_and_tokentypes,
    get_a_and_b_segments,
    truncate_segments,
)


def _pad(seq: List[int], length: int, pad_id: int) -> List[int]:
    return list(seq) + [pad_id] * (length - len(seq))


@dataclass
class MlmSopCollator:
    """MLM + sentence-order-prediction batches from raw text."""

    tokenizer: Any
    max_seq_length: int = 512
    masked_lm_prob: float = 0.15
    max_ngrams: int = 3
    content_key: str = "text"
    seed: int = 1234

    def __post_init__(self):
        self.splitter = ChineseSentenceSplitter()
        tk = self.tokenizer
        self.cls_id = tk.cls_token_id
        self.sep_id = tk.sep_token_id
        self.mask_id = tk.mask_token_id
        self.pad_id = tk.pad_token_id
        self.vocab_id_list = list(tk.get_vocab().values())
        self.vocab_id_to_token = {v: k for k, v in tk.get_vocab().items()}
        self._rng = np.random.RandomState(self.seed)

    def __call__(self, samples: List[Dict]) -> Dict[str, torch.Tensor]:
        batch = {"input_ids": [], "attention_mask": [], "token_type_ids": [],
                 "labels": [], "next_sentence_label": []}
        max_pred = int(self.max_seq_length * self.masked_lm_prob) + 1
        for s in samples:
            text = s[self.content_key] if isinstance(s, dict) else s
            sentences = self.splitter.tokenize(text)
            sent_ids = [self.tokenizer.encode(x, add_special_tokens=False)
                        for x in sentences if x.strip()]
            sent_ids = [x for x in sent_ids if x]
            if len(sent_ids) < 2:
                ids = (sent_ids[0] if sent_ids else
                       self.tokenizer.encode(text, add_special_tokens=False))
                half = max(1, len(ids) // 2)
                sent_ids = [ids[:half], ids[half:] or [self.sep_id]]
            tokens_a, tokens_b, is_random = get_a_and_b_segments(
                sent_ids, self._rng)
            truncate_segments(tokens_a, tokens_b, len(tokens_a), len(tokens_b),
                              self.max_seq_length - 3, self._rng)
            tokens, tokentypes = create_tokens_and_tokentypes(
                tokens_a, tokens_b, self.cls_id, self.sep_id)
            tokens, positions, labels_ = create_masked_lm_predictions(
                tokens, self.vocab_id_list, self.vocab_id_to_token,
                self.masked_lm_prob, self.cls_id, self.sep_id, self.mask_id,
                max_pred, self._rng, max_ngrams=self.max_ngrams)
            labels = [-100] * len(tokens)
            for p, l in zip(positions, labels_):
                labels[p] = l
            attn = [1] * len(tokens)
            L = self.max_seq_length
            batch["input_ids"].append(_pad(tokens[:L], L, self.pad_id))
            batch["attention_mask"].append(_pad(attn[:L], L, 0))
            batch["token_type_ids"].append(_pad(tokentypes[:L], L, 0))
            batch["labels"].append(_pad(labels[:L], L, -100))
            batch["next_sentence_label"].append(int(is_random))
        return {k: torch.tensor(v, dtype=torch.long) for k, v in batch.items()}


@dataclass
class SftCollator:
    """Causal SFT with prompt masking (ref LlamaSFTCollator :35-85)."""

    tokenizer: Any
    max_seq_length: int = 2048
    human_prefix: str = "<human>:"
    bot_prefix: str = "<bot>:"
    query_key: str = "query"
    answer_key: str = "answer"
    pad_to_max: bool = False

    def __call__(self, samples: List[Dict]) -> Dict[str, torch.Tensor]:
        input_ids, labels = [], []
        for s in samples:
            queries = s[self.query_key]
            answers = s[self.answer_key]
            if isinstance(queries, str):
                queries, answers = [queries], [answers]
            ids: List[int] = []
            lbl: List[int] = []
            if self.tokenizer.bos_token_id is not None:
                ids.append(self.tokenizer.bos_token_id)
                lbl.append(-100)
            for q, a in zip(queries, answers):
                prompt = f"{self.human_prefix}{q}\n{self.bot_prefix}"
                p_ids = self.tokenizer.encode(prompt, add_special_tokens=False)
                a_ids = self.tokenizer.encode(a, add_special_tokens=False)
                if self.tokenizer.eos_token_id is not None:
                    a_ids = a_ids + [self.tokenizer.eos_token_id]
                ids += p_ids + a_ids
                lbl += [-100] * len(p_ids) + a_ids
            ids = ids[:self.max_seq_length]
            lbl = lbl[:self.max_seq_length]
            input_ids.append(ids)
            labels.append(lbl)
        L = self.max_seq_length if self.pad_to_max else \
            max(len(x) for x in input_ids)
        pad_id = self.tokenizer.pad_token_id or 0
        attn = [[1] * len(x) + [0] * (L - len(x)) for x in input_ids]
        input_ids = [_pad(x, L, pad_id) for x in input_ids]
        labels = [_pad(x, L, -100) for x in labels]
        return {
            "input_ids": torch.tensor(input_ids, dtype=torch.long),
            "attention_mask": torch.tensor(attn, dtype=torch.long),
            "labels": torch.tensor(labels, dtype=torch.long),
        }


def compute_input_and_target_lengths(inputs_length: int, noise_density: float,
                                     mean_noise_span_length: float):
    """T5 span-corruption length math (ref t5_datasets.py:14)."""
    def _lengths(tokens_length):
        num_noise_tokens = int(round(tokens_length * noise_density))
        num_nonnoise_tokens = tokens_length - num_noise_tokens
        num_noise_spans = int(round(num_noise_tokens / mean_noise_span_length))
        num_noise_spans = max(num_noise_spans, 1)
        return (num_nonnoise_tokens + num_noise_spans + 1,
                num_noise_tokens + num_noise_spans + 1)

    tokens_length = inputs_length
    while _lengths(tokens_length + 1)[0] <= inputs_length:
        tokens_length += 1
    inputs_len, targets_len = _lengths(tokens_length)
    return tokens_length, targets_len


@dataclass
class T5SpanCollator:
    """UL2/T5 span corruption with sentinel tokens (ref t5_datasets.py)."""

    tokenizer: Any
    max_seq_length: int = 512
    noise_density: float = 0.15
    mean_noise_span_length: float = 3.0
    content_key: str = "text"
    seed: int = 1234

    def __post_init__(self):
        self._rng = np.random.RandomState(self.seed)
        self.pad_id = self.tokenizer.pad_token_id or 0
        self.eos_id = self.tokenizer.eos_token_id
        # sentinels: highest vocab ids (<extra_id_k>) if present, else tail ids
        self.sentinel_base = None
        for k in range(3):
            tok = f"<extra_id_{k}>"
            if tok in self.tokenizer.get_vocab():
                if k == 0:
                    self.sentinel_base = self.tokenizer.get_vocab()[tok]
                break
        if self.sentinel_base is None:
            self.sentinel_base = len(self.tokenizer.get_vocab()) - 1
        self.expanded_length, self.target_length = \
            compute_input_and_target_lengths(
                self.max_seq_length, self.noise_density,
                self.mean_noise_span_length)

    def _random_spans_noise_mask(self, length: int) -> np.ndarray:
        num_noise = int(round(length * self.noise_density))
        num_noise = min(max(num_noise, 1), length - 1)
        num_spans = max(int(round(num_noise / self.mean_noise_span_length)), 1)
        num_nonnoise = length - num_noise

        def segment(total, n):
            cut = np.arange(total - 1) < (n - 1)
            self._rng.shuffle(cut)
            first_in_seg = np.pad(cut, [[1, 0]])
            segment_id = np.cumsum(first_in_seg)
            return np.asarray(np.bincount(segment_id, minlength=n))

        noise_spans = segment(num_noise, num_spans)
        nonnoise_spans = segment(num_nonnoise, num_spans)
        interleaved = np.empty(num_spans * 2, dtype=np.int64)
        interleaved[0::2] = nonnoise_spans
        interleaved[1::2] = noise_spans
        span_starts = np.cumsum(interleaved)[:-1]
        mask = np.zeros(length, dtype=bool)
        start_indicator = np.zeros(length, dtype=np.int64)
        start_indicator[span_starts] = 1
        span_num = np.cumsum(start_indicator)
        mask = (span_num % 2) == 1
        return mask

    def _sentinel_ids(self, mask: np.ndarray) -> np.ndarray:
        start = mask & ~np.roll(mask, 1)
        start[0] = mask[0]
        ids = np.where(start, np.cumsum(start), 0)
        out = np.where(ids != 0, self.sentinel_base - (ids - 1), 0)
        out[mask & ~start] = -1
        return out

    def _filter(self, ids: np.ndarray, sentinel: np.ndarray) -> List[int]:
        combined = np.where(sentinel != 0, sentinel, ids)
        return [int(x) for x in combined if x >= 0]

    def __call__(self, samples: List[Dict]) -> Dict[str, torch.Tensor]:
        enc_in, dec_lab = [], []
        for s in samples:
            text = s[self.content_key] if isinstance(s, dict) else s
            ids = self.tokenizer.encode(text, add_special_tokens=False)
            ids = ids[:self.expanded_length]
            if len(ids) < 8:
                ids = (ids * 8)[:8]
            arr = np.array(ids)
            mask = self._random_spans_noise_mask(len(arr))
            in_ids = self._filter(arr, self._sentinel_ids(mask))
            lab_ids = self._filter(arr, self._sentinel_ids(~mask))
            if self.eos_id is not None:
                in_ids.append(self.eos_id)
                lab_ids.append(self.eos_id)
            enc_in.append(in_ids[:self.max_seq_length])
            dec_lab.append(lab_ids[:self.target_length])
        Li = max(len(x) for x in enc_in)
        Lt = max(len(x) for x in dec_lab)
        attn = [[1] * len(x) + [0] * (Li - len(x)) for x in enc_in]
        return {
            "input_ids": torch.tensor(
                [_pad(x, Li, self.pad_id) for x in enc_in], dtype=torch.long),
            "attention_mask": torch.tensor(attn, dtype=torch.long),
            "labels": torch.tensor(
                [_pad(x, Lt, -100) for x in dec_lab], dtype=torch.long),
        }


@dataclass
class CausalCollator:
    """Plain causal-LM collator over pre-tokenized or raw text samples."""

    tokenizer: Any
    max_seq_length: int = 1024
    content_key: str = "text"

    def __call__(self, samples: List[Dict]) -> Dict[str, torch.Tensor]:
        seqs = []
        for s in samples:
            if isinstance(s, dict) and "input_ids" in s:
                ids = list(s["input_ids"])
            else:
                text = s[self.content_key] if isinstance(s, dict) else s
                ids = self.tokenizer.encode(text)
            seqs.append(ids[:self.max_seq_length])
        L = max(len(x) for x in seqs)
        pad_id = self.tokenizer.pad_token_id or 0
        attn = [[1] * len(x) + [0] * (L - len(x)) for x in seqs]
        labels = [x + [-100] * (L - len(x)) for x in seqs]
        return {
            "input_ids": torch.tensor(
                [_pad(x, L, pad_id) for x in seqs], dtype=torch.long),
            "attention_mask": torch.tensor(attn, dtype=torch.long),
            "labels": torch.tensor(labels, dtype=torch.long),
        }
