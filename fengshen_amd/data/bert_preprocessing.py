"""Offline corpus preprocessing for BERT-family pretraining.

Behavioral parity: reference data/bert_dataloader/ (auto_split.sh shards
>1GB files into ~300MB pieces; preprocessing.py pre-splits documents into
sentences so DataLoader workers don't pay for it online; load.py caches via
HF datasets so "180GB loads in seconds").  Ours adds a direct
jsonl -> mmap (.bin/.idx) tokenized path.
"""
from __future__ import annotations

import json
import os
from typing import Optional

from fengshen_amd.data.data_utils import ChineseSentenceSplitter


def split_shards(input_path: str, out_dir: str,
                 shard_bytes: int = 300 * 1024 * 1024) -> list:
    """Split a large jsonl into ~shard_bytes pieces (auto_split.sh parity)."""
    os.makedirs(out_dir, exist_ok=True)
    base = os.path.splitext(os.path.basename(input_path))[0]
    shards = []
    idx = 0
    out = None
    written = 0
    with open(input_path, encoding="utf-8") as f:
        for line in f:
            if out is None or written >= shard_bytes:
                if out:
                    out.close()
                path = os.path.join(out_dir, f"{base}_{idx:04d}.jsonl")
                shards.append(path)
                out = open(path, "w", encoding="utf-8")
                written = 0
                idx += 1
            out.write(line)
            written += len(line.encode("utf-8"))
    if out:
        out.close()
    return shards


def presplit_sentences(input_path: str, output_path: str,
                       content_key: str = "text") -> int:
    """Rewrite jsonl so each doc carries its sentence list
    (preprocessing.py parity)."""
    splitter = ChineseSentenceSplitter()
    n = 0
    with open(input_path, encoding="utf-8") as fin, \
            open(output_path, "w", encoding="utf-8") as fout:
        for line in fin:
            if not line.strip():
                continue
            doc = json.loads(line)
            doc["sentences"] = splitter.tokenize(doc.get(content_key, ""))
            fout.write(json.dumps(doc, ensure_ascii=False) + "\n")
            n += 1
    return n


def jsonl_to_mmap(input_path: str, out_prefix: str, tokenizer,
                  content_key: str = "text",
                  sentence_level: bool = True,
                  vocab_size: Optional[int] = None) -> int:
    """Tokenize a jsonl corpus into the .bin/.idx mmap format (one item per
    sentence, doc boundaries preserved) — feeds BertMmapDataset/GPTDataset."""
    from fengshen_amd.data.indexed_dataset import make_builder

    splitter = ChineseSentenceSplitter()
    builder = make_builder(out_prefix + ".bin",
                           vocab_size or len(tokenizer.get_vocab()))
    docs = 0
    with open(input_path, encoding="utf-8") as f:
        for line in f:
            if not line.strip():
                continue
            doc = json.loads(line)
            text = doc.get(content_key, "")
            if not text:
                continue
            pieces = splitter.tokenize(text) if sentence_level else [text]
            wrote = False
            for sent in pieces:
                ids = tokenizer.encode(sent, add_special_tokens=False)
                if ids:
                    builder.add_item(ids)
                    wrote = True
            if wrote:
                builder.end_document()
                docs += 1
    builder.finalize(out_prefix + ".idx")
    return docs
