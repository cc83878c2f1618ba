"""Misc utils: memory probe string, Chinese char tokenization."""
from fengshen_amd.utils.utils import chinese_char_tokenize, report_memory


def test_report_memory_cpu_safe():
    s = report_memory("probe")
    assert "probe" in s and "memory" in s


def test_chinese_char_tokenize():
    out = chinese_char_tokenize("abc中文def")
    # CJK chars get space-separated; latin runs preserved
    assert "中" in out.split() and "文" in out.split()
    assert "abc" in out
    # idempotent-ish: already separated text keeps tokens
    again = chinese_char_tokenize(out)
    assert "中" in again.split()


# ---------------------------------------------------------------------------
# TF checkpoint importer (ref convert_tf_checkpoint_to_pytorch.py; no TF in
# the image — native TensorBundle reader).  The test writes a
# spec-conformant LevelDB table + data shard by hand and reads it back.
# ---------------------------------------------------------------------------
def _varint(n):
    out = b""
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _pb_tag(field, wire):
    return _varint((field << 3) | wire)


def _bundle_entry(dtype, shape, offset, size):
    shape_msg = b""
    for d in shape:
        dim = _pb_tag(1, 0) + _varint(d)
        shape_msg += _pb_tag(2, 2) + _varint(len(dim)) + dim
    msg = _pb_tag(1, 0) + _varint(dtype)
    msg += _pb_tag(2, 2) + _varint(len(shape_msg)) + shape_msg
    msg += _pb_tag(4, 0) + _varint(offset)
    msg += _pb_tag(5, 0) + _varint(size)
    return msg


def _leveldb_block(entries):
    """One block, no prefix sharing, single restart at 0."""
    import struct
    body = b""
    for k, v in entries:
        body += _varint(0) + _varint(len(k)) + _varint(len(v)) + k + v
    body += struct.pack("<I", 0) + struct.pack("<I", 1)
    return body


def _write_tf_checkpoint(tmp_path, tensors):
    """tensors: {name: np.ndarray (float32)}"""
    import struct
    import numpy as np
    data = b""
    entries = []
    for name in sorted(tensors):
        arr = np.ascontiguousarray(tensors[name], dtype=np.float32)
        entries.append((name.encode(), _bundle_entry(
            1, arr.shape, len(data), arr.nbytes)))
        data += arr.tobytes()
    (tmp_path / "model.ckpt.data-00000-of-00001").write_bytes(data)

    blk = _leveldb_block(entries)
    out = blk + b"\x00" + struct.pack("<I", 0)  # type + crc (crc unchecked)
    data_handle = _varint(0) + _varint(len(blk))
    # index block: one entry pointing at the data block
    idx = _leveldb_block([(b"\xff", data_handle)])
    idx_off = len(out)
    out += idx + b"\x00" + struct.pack("<I", 0)
    meta_off = len(out)
    meta = _leveldb_block([])
    out += meta + b"\x00" + struct.pack("<I", 0)
    footer = (_varint(meta_off) + _varint(len(meta))
              + _varint(idx_off) + _varint(len(idx)))
    footer += b"\x00" * (40 - len(footer))
    out += footer + struct.pack("<Q", 0xDB4775248B80FB57)
    (tmp_path / "model.ckpt.index").write_bytes(out)
    return str(tmp_path / "model.ckpt")


def test_tf_checkpoint_reader(tmp_path):
    import numpy as np
    from fengshen_amd.utils.tf_checkpoint import TFCheckpointReader
    rng = np.random.RandomState(0)
    tensors = {
        "bert/encoder/layer_0/attention/self/query/kernel":
            rng.randn(8, 8).astype(np.float32),
        "bert/embeddings/word_embeddings":
            rng.randn(16, 8).astype(np.float32),
        "global_step": np.array([7.0], dtype=np.float32),
    }
    prefix = _write_tf_checkpoint(tmp_path, tensors)
    r = TFCheckpointReader(prefix)
    assert set(r.variable_names()) == set(tensors)
    for name, arr in tensors.items():
        got = r.load_variable(name)
        assert got.shape == arr.shape
        assert np.allclose(got, arr)


def test_tf_bert_name_mapping(tmp_path):
    import numpy as np
    from fengshen_amd.utils.tf_checkpoint import (
        convert_tf_bert_to_state_dict)
    rng = np.random.RandomState(1)
    kern = rng.randn(4, 6).astype(np.float32)
    tensors = {
        "bert/encoder/layer_0/attention/self/query/kernel": kern,
        "bert/encoder/layer_0/attention/output/LayerNorm/gamma":
            rng.randn(4).astype(np.float32),
        "bert/embeddings/word_embeddings":
            rng.randn(16, 4).astype(np.float32),
        "cls/predictions/output_bias": rng.randn(16).astype(np.float32),
        "bert/adam_m/skip_me": rng.randn(2).astype(np.float32),
    }
    prefix = _write_tf_checkpoint(tmp_path, tensors)
    sd = convert_tf_bert_to_state_dict(prefix)
    assert "bert.encoder.layer.0.attention.self.query.weight" in sd
    # TF kernels are [in, out]; torch wants [out, in]
    assert np.allclose(
        sd["bert.encoder.layer.0.attention.self.query.weight"], kern.T)
    assert "bert.encoder.layer.0.attention.output.LayerNorm.weight" in sd
    assert "bert.embeddings.word_embeddings.weight" in sd
    assert "cls.predictions.bias" in sd
    assert not any("adam_m" in k for k in sd)


def test_ner_bio_bios_decode():
    """BIO/BIOS entity decode (ref metric/utils_ner.py behavior)."""
    from fengshen_amd.metric.utils_ner import get_entities

    bios = ["O", "B-PER", "I-PER", "O", "S-LOC", "B-ORG", "I-ORG"]
    assert get_entities(bios, "bios") == [("PER", 1, 2), ("LOC", 4, 4),
                                          ("ORG", 5, 6)]
    bio = ["B-PER", "I-PER", "O", "B-LOC", "B-ORG", "I-ORG"]
    assert get_entities(bio, "bio") == [("PER", 0, 1), ("LOC", 3, 3),
                                        ("ORG", 4, 5)]
    # I- with mismatched type does not extend the chunk
    assert get_entities(["B-PER", "I-LOC", "O"], "bio") == [("PER", 0, 0)]
    # trailing entity at sequence end is flushed
    assert get_entities(["O", "B-LOC"], "bio") == [("LOC", 1, 1)]
    assert get_entities([], "bios") == []


def test_scheduler_registry_formulas():
    """Scheduler registry (ref model_utils.py:101-254): warmup ramp,
    polynomial floor at lr_end, inverse-sqrt decay, direct constant."""
    import math
    import torch
    from fengshen_amd.models.model_utils import get_scheduler

    def lrs(name, n, **kw):
        p = torch.nn.Parameter(torch.zeros(1))
        opt = torch.optim.SGD([p], lr=1e-4)
        sch = get_scheduler(name, opt, **kw)
        out = []
        for _ in range(n):
            out.append(opt.param_groups[0]["lr"])
            opt.step()
            sch.step()
        return out

    poly = lrs("polynomial", 30, num_warmup_steps=5, num_training_steps=20,
               lr_end=1e-6, lr_init=1e-4)
    assert poly[0] == 0.0 and abs(poly[5] - 1e-4) < 1e-9   # ramp to peak
    assert abs(poly[-1] - 1e-6) < 1e-9                      # floor at lr_end
    assert all(a >= b - 1e-12 for a, b in zip(poly[5:], poly[6:]))

    inv = lrs("inverse_sqrt", 25, num_warmup_steps=4)
    assert abs(inv[4] - 1e-4) < 1e-9
    assert abs(inv[16] - 1e-4 * math.sqrt(4 / 16)) < 1e-9   # ~ 1/sqrt(t)

    direct = lrs("direct", 10, num_warmup_steps=4)
    assert abs(direct[2] - 1e-4 * 2 / 4) < 1e-9
    assert all(abs(v - 1e-4) < 1e-9 for v in direct[4:])

    cos = lrs("cosine", 21, num_warmup_steps=0, num_training_steps=20)
    assert abs(cos[10] - 5e-5) < 1e-7                       # half way
    assert cos[-1] < 1e-5


def test_rouge_score_hand_values():
    """RougeScore against hand-computed rouge-1/2/L on token strings."""
    from fengshen_amd.metric.rouge import RougeScore

    rs = RougeScore()
    rs.update(["a b c d"], ["a b e d"])
    out = rs.compute()
    # unigrams: overlap {a,b,d} = 3 of 4 -> P=R=F=0.75
    assert abs(out["rouge1_fmeasure"] - 0.75) < 1e-6
    # bigrams: pred {ab,bc,cd}, ref {ab,be,ed} -> overlap {ab} = 1/3
    assert abs(out["rouge2_precision"] - 1 / 3) < 1e-6
    # LCS("abcd","abed") = "abd" (3) -> F = 0.75
    assert abs(out["rougeL_fmeasure"] - 0.75) < 1e-6
    # perfect match accumulates with the first sample
    rs.update(["x y"], ["x y"])
    out2 = rs.compute()
    assert abs(out2["rouge1_fmeasure"] - (0.75 + 1.0) / 2) < 1e-6
    rs.reset()
    assert rs.compute()["rouge1_fmeasure"] == 0.0
