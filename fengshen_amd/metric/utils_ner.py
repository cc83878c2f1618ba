"""BIO/BIOS entity decoding (reference fengshen/metric/utils_ner.py)."""
from __future__ import annotations

from typing import List, Tuple


def get_entity_bios(seq: List[str]) -> List[Tuple]:
    """decode BIOS: B-X / I-X / S-X / O."""
    chunks = []
    chunk = [-1, -1, -1]
    for i, tag in enumerate(seq):
        if tag.startswith("S-"):
            if chunk[2] != -1:
                chunks.append(tuple(chunk))
            chunks.append((tag.split("-", 1)[1], i, i))
            chunk = [-1, -1, -1]
        elif tag.startswith("B-"):
            if chunk[2] != -1:
                chunks.append(tuple(chunk))
            chunk = [tag.split("-", 1)[1], i, i]
        elif tag.startswith("I-") and chunk[1] != -1:
            if tag.split("-", 1)[1] == chunk[0]:
                chunk[2] = i
            if i == len(seq) - 1:
                chunks.append(tuple(chunk))
                chunk = [-1, -1, -1]
        else:
            if chunk[2] != -1:
                chunks.append(tuple(chunk))
            chunk = [-1, -1, -1]
    if chunk[2] != -1:
        chunks.append(tuple(chunk))
    return chunks


def get_entity_bio(seq: List[str]) -> List[Tuple]:
    """decode BIO: B-X / I-X / O."""
    chunks = []
    chunk = [-1, -1, -1]
    for i, tag in enumerate(seq):
        if tag.startswith("B-"):
            if chunk[2] != -1:
                chunks.append(tuple(chunk))
            chunk = [tag.split("-", 1)[1], i, i]
            if i == len(seq) - 1:
                chunks.append(tuple(chunk))
                chunk = [-1, -1, -1]
        elif tag.startswith("I-") and chunk[1] != -1:
            if tag.split("-", 1)[1] == chunk[0]:
                chunk[2] = i
            if i == len(seq) - 1:
                chunks.append(tuple(chunk))
                chunk = [-1, -1, -1]
        else:
            if chunk[2] != -1:
                chunks.append(tuple(chunk))
            chunk = [-1, -1, -1]
    return chunks


def get_entities(seq: List[str], markup: str = "bios") -> List[Tuple]:
    assert markup in ("bio", "bios")
    return get_entity_bios(seq) if markup == "bios" else get_entity_bio(seq)
