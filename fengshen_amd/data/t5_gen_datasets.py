"""Generation-task (dialog) dataset classes for Randeng-T5/mT5
(ref fengshen/data/t5_dataloader/t5_gen_datasets.py).

DialogDataset tokenizes {context: [turns...], knowledge, target} samples
into grounded-dialog encoder/decoder tensors:
  input  = [CTSTART] context-tail [CTEND] [KNSTART] knowledge [KNEND]
  token_types = speaker alternation (0/1) over context, 2 over knowledge
  labels = target + </s>; decoder_input_ids = shift-right(labels)
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch
from torch.nn.utils.rnn import pad_sequence
from torch.utils.data import Dataset

SPECIAL_TOKENS = ["[CTSTART]", "[CTEND]", "[SEP]", "[KNSTART]", "[KNEND]"]


def add_dialog_special_tokens(tokenizer):
    """Register the dialog span markers (ref special_token_dict :27-36)."""
    if hasattr(tokenizer, "add_special_tokens"):
        tokenizer.add_special_tokens(
            {"additional_special_tokens": SPECIAL_TOKENS})
    return tokenizer


class DialogDataset(Dataset):
    """Knowledge-grounded dialog dataset (ref DialogDataset :38-210).

    data: list of {"context": [utterance, ...], "knowledge": str,
                   "target": str}.
    """

    def __init__(self, data: List[dict], tokenizer,
                 max_seq_length: int = 512,
                 max_knowledge_length: int = 128,
                 max_target_length: int = 128,
                 eos_token_id: Optional[int] = None):
        super().__init__()
        self.tokenizer = tokenizer
        self.max_seq_length = max_seq_length
        self.max_knowledge_length = max_knowledge_length
        self.max_target_length = max_target_length
        self.eos_token_id = (eos_token_id if eos_token_id is not None
                             else getattr(tokenizer, "eos_token_id", 1) or 1)
        self.data = [self.regular_tokenize(s) for s in data]

    def __len__(self):
        return len(self.data)

    def __getitem__(self, index):
        return self.data[index]

    # ------------------------------------------------------------------
    def _tok_id(self, token: str) -> int:
        if hasattr(self.tokenizer, "convert_tokens_to_ids"):
            return self.tokenizer.convert_tokens_to_ids(token)
        return self.tokenizer.get_vocab().get(
            token, getattr(self.tokenizer, "unk_token_id", 0))

    def get_token_type(self, context: List[str],
                       tokentypes=None) -> List[int]:
        """Speaker alternation 0/1 per utterance (ref :193-210)."""
        context_token_types: List[int] = []
        for i, line in enumerate(context):
            if tokentypes:
                n = len(tokentypes[i])
            else:
                n = 1 + len(line)
            context_token_types.extend([i % 2] * n)
        return context_token_types

    def regular_tokenize(self, sample: dict) -> Dict[str, np.ndarray]:
        tk = self.tokenizer
        per_turn_ids = [tk.encode(line) for line in sample["context"]]
        context_types = self.get_token_type(
            sample["context"], per_turn_ids)

        knowledge_ids = tk.encode(sample["knowledge"],
                                  add_special_tokens=False)
        if isinstance(knowledge_ids, int):
            knowledge_ids = [knowledge_ids]
        target_ids = tk.encode(sample["target"],
                               add_special_tokens=False)
        target_ids = target_ids[:self.max_target_length - 1]

        knowledge_ids = ([self._tok_id("[KNSTART]")]
                         + knowledge_ids[:self.max_knowledge_length - 2]
                         + [self._tok_id("[KNEND]")])
        l_kn = len(knowledge_ids)
        knowledge_types = [2] * l_kn

        flatten_context: List[int] = []
        for line in per_turn_ids:
            flatten_context.extend(line)
        l_ct = min(len(flatten_context), self.max_seq_length - l_kn - 2)
        context_ids = ([self._tok_id("[CTSTART]")]
                       + flatten_context[-l_ct:]
                       + [self._tok_id("[CTEND]")])
        context_types = context_types[-l_ct:] + [0]
        context_types.insert(0, context_types[0])

        target_ids = target_ids + [self.eos_token_id]
        return {
            "input_ids": np.array(context_ids + knowledge_ids,
                                  dtype=np.int32),
            "token_types": np.array(context_types + knowledge_types,
                                    dtype=np.int32),
            "attention_mask": np.ones(len(context_types) + l_kn,
                                      dtype=np.int8),
            "labels": np.array(target_ids, dtype=np.int32),
        }


def shift_tokens_right(input_ids: np.ndarray, pad_token_id: int,
                       decoder_start_token_id: int) -> np.ndarray:
    """Shift right for decoder inputs; -100 -> pad (ref :288-301)."""
    input_ids = np.asarray(input_ids)
    shifted = np.zeros_like(input_ids)
    shifted[:, 1:] = input_ids[:, :-1]
    shifted[:, 0] = decoder_start_token_id
    return np.where(shifted == -100, pad_token_id, shifted)


class DialogCollator:
    """Batch collation for DialogDataset samples (ref collate_fn :263-286)."""

    def __init__(self, pad_token_id: int = 0,
                 decoder_start_token_id: int = 0):
        self.pad_token_id = pad_token_id
        self.decoder_start_token_id = decoder_start_token_id

    def __call__(self, samples: List[dict]) -> Dict[str, torch.Tensor]:
        batch = {
            k: [torch.tensor(np.asarray(s[k]), dtype=torch.int64)
                for s in samples]
            for k in ["input_ids", "token_types", "attention_mask", "labels"]
        }
        for k, v in batch.items():
            pad = -100 if k == "labels" else self.pad_token_id
            batch[k] = pad_sequence(v, batch_first=True, padding_value=pad)
        batch["decoder_input_ids"] = torch.tensor(
            shift_tokens_right(batch["labels"].numpy(), self.pad_token_id,
                               self.decoder_start_token_id),
            dtype=torch.long)
        return batch
