"""Misc utilities.

Behavioral parity: reference fengshen/utils/utils.py (report_memory :62-74,
chinese_char_tokenize :44) — reimplemented for ROCm (torch.cuda on ROCm
reports HBM via the hip runtime).
"""
import logging
import torch

logger = logging.getLogger(__name__)


def report_memory(name: str = "") -> str:
    """Report current/peak HBM usage on this rank's GPU (MB)."""
    if not torch.cuda.is_available():
        return f"{name} memory (MB): no GPU"
    mega = 1024.0 * 1024.0
    s = (
        f"{name} memory (MB) | allocated: {torch.cuda.memory_allocated() / mega:.1f}"
        f" | max allocated: {torch.cuda.max_memory_allocated() / mega:.1f}"
        f" | reserved: {torch.cuda.memory_reserved() / mega:.1f}"
        f" | max reserved: {torch.cuda.max_memory_reserved() / mega:.1f}"
    )
    logger.info(s)
    return s


def _is_chinese_char(cp: int) -> bool:
    return (
        (0x4E00 <= cp <= 0x9FFF)
        or (0x3400 <= cp <= 0x4DBF)
        or (0x20000 <= cp <= 0x2A6DF)
        or (0x2A700 <= cp <= 0x2B73F)
        or (0x2B740 <= cp <= 0x2B81F)
        or (0x2B820 <= cp <= 0x2CEAF)
        or (0xF900 <= cp <= 0xFAFF)
        or (0x2F800 <= cp <= 0x2FA1F)
    )


def chinese_char_tokenize(line: str) -> str:
    """Insert spaces around CJK chars (used before CE/metric comparisons)."""
    line = line.strip()
    out = []
    for ch in line:
        if _is_chinese_char(ord(ch)):
            out.append(" ")
            out.append(ch)
            out.append(" ")
        else:
            out.append(ch)
    return "".join(out).strip()
