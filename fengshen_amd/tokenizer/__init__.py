from fengshen_amd.tokenizer.char_tokenizer import SimpleCharTokenizer  # noqa: F401
