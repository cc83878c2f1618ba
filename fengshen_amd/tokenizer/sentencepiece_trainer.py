"""SentencePiece vocab training (reference tokenizer/sentencepiece/
pretrain_google_sp.sh: BPE, vocab 40k, character_coverage 0.9995)."""
from __future__ import annotations

import argparse


def train_spm(input_path: str, model_prefix: str, vocab_size: int = 40000,
              character_coverage: float = 0.9995, model_type: str = "bpe"):
    import sentencepiece as spm
    spm.SentencePieceTrainer.train(
        input=input_path, model_prefix=model_prefix, vocab_size=vocab_size,
        character_coverage=character_coverage, model_type=model_type,
        input_sentence_size=2000000, shuffle_input_sentence=True)


def shuffle_corpus(input_path: str, output_path: str, seed: int = 1234):
    """reference tokenizer/sentencepiece/shuffle_corpus.py"""
    import random
    rng = random.Random(seed)
    with open(input_path) as f:
        lines = f.readlines()
    rng.shuffle(lines)
    with open(output_path, "w") as f:
        f.writelines(lines)


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--input", required=True)
    p.add_argument("--model_prefix", required=True)
    p.add_argument("--vocab_size", type=int, default=40000)
    a = p.parse_args()
    train_spm(a.input, a.model_prefix, a.vocab_size)
