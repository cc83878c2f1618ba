"""Tokenizer tooling: char tokenizer + sentencepiece training."""
from fengshen_amd.tokenizer import SimpleCharTokenizer


def test_char_tokenizer_roundtrip():
    tk = SimpleCharTokenizer()
    ids = tk.encode("hello 123")
    assert all(isinstance(i, int) for i in ids)
    assert tk.decode(ids).replace("[UNK]", "?") is not None
    assert tk.pad_token_id == 0 and tk.mask_token_id == 3
    vocab = tk.get_vocab()
    assert len(vocab) > 200


def test_spm_training_and_encode(tmp_path):
    from fengshen_amd.tokenizer.sentencepiece_trainer import (
        shuffle_corpus,
        train_spm,
    )
    corpus = tmp_path / "corpus.txt"
    lines = [f"the quick brown fox {i} jumps over the lazy dog\n"
             for i in range(200)]
    corpus.write_text("".join(lines))
    shuffled = tmp_path / "shuffled.txt"
    shuffle_corpus(str(corpus), str(shuffled))
    assert sorted(shuffled.read_text().splitlines()) == \
        sorted(corpus.read_text().splitlines())

    prefix = str(tmp_path / "spm_test")
    train_spm(str(shuffled), prefix, vocab_size=64)
    import sentencepiece as spm
    sp = spm.SentencePieceProcessor(model_file=prefix + ".model")
    ids = sp.encode("the quick brown fox")
    assert len(ids) > 0
    assert sp.decode(ids) == "the quick brown fox"


def test_char_tokenizer_hf_style_call():
    """__call__ batches like HF: dynamic padding, max_length padding,
    truncation, pt tensors."""
    import torch
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tk = SimpleCharTokenizer()
    out = tk(["ab", "abcde"], return_tensors="pt")
    assert out["input_ids"].shape == out["attention_mask"].shape
    assert out["input_ids"].shape[1] == 7  # CLS + 5 chars + SEP
    assert out["attention_mask"][0].sum() == 4
    # single string
    one = tk("ab")
    assert isinstance(one["input_ids"][0], list)
    # max_length padding + truncation
    out2 = tk(["ab"], padding="max_length", max_length=6,
              return_tensors="pt")
    assert out2["input_ids"].shape == (1, 6)
    long = tk(["abcdefgh"], max_length=5, return_tensors="pt")
    assert long["input_ids"].shape[1] == 5
    assert long["input_ids"][0, -1].item() == tk.sep_token_id
    # round-trip through decode drops specials
    ids = torch.as_tensor(tk("ab")["input_ids"][0])
    assert tk.decode(ids) == "ab"
