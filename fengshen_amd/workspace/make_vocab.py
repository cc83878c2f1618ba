"""Generate from-scratch pretraining vocabularies (ref workspace/*/pretrain
ships config.json + vocab.txt per model).

The vocab is built programmatically — specials, ASCII, CJK punctuation,
then the BMP unified-CJK range in codepoint order — sized to each
config's vocab_size.  For training against released IDEA-CCNL
checkpoints use the hub tokenizer instead; these files are for
from-scratch runs only.
"""
import json
import os


def build_vocab(size: int):
    toks = ["[PAD]"] + [f"[unused{i}]" for i in range(1, 100)] + \
        ["[UNK]", "[CLS]", "[SEP]", "[MASK]", "<S>", "<T>"]
    toks += [chr(c) for c in range(0x21, 0x7F)]              # ASCII printable
    toks += [chr(c) for c in range(0x3001, 0x3018)]          # CJK punct
    toks += [chr(c) for c in range(0xFF01, 0xFF5F)]          # fullwidth
    c = 0x4E00
    while len(toks) < size and c <= 0x9FFF:                  # unified CJK
        ch = chr(c)
        if ch not in toks[:106]:
            toks.append(ch)
        c += 1
    i = 0
    while len(toks) < size:                                  # pad the tail
        toks.append(f"[unused{100 + i}]")
        i += 1
    return toks[:size]


def main():
    root = os.path.dirname(os.path.abspath(__file__))
    for d in sorted(os.listdir(root)):
        cfg_path = os.path.join(root, d, "config.json")
        if not os.path.isfile(cfg_path):
            continue
        size = json.load(open(cfg_path)).get("vocab_size", 21128)
        out = os.path.join(root, d, "vocab.txt")
        with open(out, "w", encoding="utf8") as f:
            f.write("\n".join(build_vocab(size)) + "\n")
        print(f"{out}: {size} tokens")


if __name__ == "__main__":
    main()
