"""Minimal character-level tokenizer (HF-compatible surface) for smoke runs
and tests.  Production flows use HF AutoTokenizer / sentencepiece models
(reference tokenizer/sentencepiece: see sentencepiece_trainer.py)."""
from __future__ import annotations

from typing import List


class SimpleCharTokenizer:
    def __init__(self):
        chars = [chr(c) for c in range(0x4E00, 0x4E00 + 200)] + \
            list("abcdefghijklmnopqrstuvwxyz0123456789.,!? :\n<>_")
        self._vocab = {"[PAD]": 0, "[CLS]": 1, "[SEP]": 2, "[MASK]": 3,
                       "[UNK]": 4, "<s>": 5, "</s>": 6}
        for c in chars:
            self._vocab.setdefault(c, len(self._vocab))
        self._inv = {v: k for k, v in self._vocab.items()}
        self.pad_token_id = 0
        self.cls_token_id = 1
        self.sep_token_id = 2
        self.mask_token_id = 3
        self.unk_token_id = 4
        self.bos_token_id = 5
        self.eos_token_id = 6

    def get_vocab(self):
        return self._vocab

    def __len__(self):
        return len(self._vocab)

    def encode(self, text: str, add_special_tokens: bool = True) -> List[int]:
        ids = [self._vocab.get(c, self.unk_token_id) for c in text]
        if add_special_tokens:
            return [self.cls_token_id] + ids + [self.sep_token_id]
        return ids

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        out = []
        for i in ids:
            tok = self._inv.get(int(i), "[UNK]")
            if skip_special_tokens and tok.startswith("["):
                continue
            out.append(tok)
        return "".join(out)
