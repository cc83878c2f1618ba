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

        self.pad_token = "[PAD]"
        self.cls_token = "[CLS]"
        self.sep_token = "[SEP]"
        self.mask_token = "[MASK]"
        self.unk_token = "[UNK]"

    def get_vocab(self):
        return self._vocab

    def __len__(self):
        return len(self._vocab)

    @property
    def vocab_size(self):
        return len(self._vocab)

    def _tokenize_ids(self, text: str) -> List[int]:
        """Char-level with special-token substrings kept whole."""
        ids: List[int] = []
        i = 0
        specials = ("[PAD]", "[CLS]", "[SEP]", "[MASK]", "[UNK]")
        while i < len(text):
            matched = False
            if text[i] == "[":
                for sp in specials:
                    if text.startswith(sp, i):
                        ids.append(self._vocab[sp])
                        i += len(sp)
                        matched = True
                        break
            if not matched:
                ids.append(self._vocab.get(text[i], self.unk_token_id))
                i += 1
        return ids

    def encode(self, text: str, add_special_tokens: bool = True) -> List[int]:
        ids = self._tokenize_ids(text)
        if add_special_tokens:
            return [self.cls_token_id] + ids + [self.sep_token_id]
        return ids

    def encode_plus(self, text: str, max_length: int = 512,
                    padding: str = "max_length",
                    truncation="longest_first", **_kw):
        ids = self.encode(text)
        if truncation and len(ids) > max_length:
            ids = ids[:max_length - 1] + [self.sep_token_id]
        attn = [1] * len(ids)
        tok_type = [0] * len(ids)
        if padding == "max_length" and len(ids) < max_length:
            pad = max_length - len(ids)
            ids = ids + [self.pad_token_id] * pad
            attn = attn + [0] * pad
            tok_type = tok_type + [0] * pad
        return {"input_ids": ids, "attention_mask": attn,
                "token_type_ids": tok_type}

    def __call__(self, text, padding=True, truncation=True,
                 max_length: int = 512, return_tensors=None, **_kw):
        """HF-style batch encode: str or list of str; padding to the
        longest sequence (or max_length when padding='max_length')."""
        texts = [text] if isinstance(text, str) else list(text)
        encs = []
        for t in texts:
            ids = self.encode(t)
            if truncation and len(ids) > max_length:
                ids = ids[:max_length - 1] + [self.sep_token_id]
            encs.append(ids)
        tgt = (max_length if padding == "max_length"
               else max(len(e) for e in encs))
        input_ids, attn = [], []
        for e in encs:
            pad = tgt - len(e) if padding else 0
            input_ids.append(e + [self.pad_token_id] * pad)
            attn.append([1] * len(e) + [0] * pad)
        out = {"input_ids": input_ids, "attention_mask": attn}
        if return_tensors == "pt":
            import torch
            out = {k: torch.tensor(v) for k, v in out.items()}
        return out

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        out = []
        for i in ids:
            tok = self._inv.get(int(i), "[UNK]")
            if skip_special_tokens and tok.startswith("["):
                continue
            out.append(tok)
        return "".join(out)
