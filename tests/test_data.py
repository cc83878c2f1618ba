"""Data layer tests: mmap dataset roundtrip, GPT sample windows, blending,
samplers, masking utils, collators."""
import numpy as np
import torch

from fengshen_amd.data.indexed_dataset import (
    MMapIndexedDataset,
    MMapIndexedDatasetBuilder,
    best_fitting_dtype,
)


from fengshen_amd.tokenizer import SimpleCharTokenizer as FakeTokenizer  # noqa


def test_mmap_dataset_roundtrip(tmp_path):
    prefix = str(tmp_path / "corpus")
    builder = MMapIndexedDatasetBuilder(prefix + ".bin",
                                        dtype=best_fitting_dtype(30000))
    docs = [[list(range(10, 20)), list(range(30, 35))],
            [list(range(100, 140))]]
    for doc in docs:
        for sent in doc:
            builder.add_item(np.array(sent))
        builder.end_document()
    builder.finalize(prefix + ".idx")

    ds = MMapIndexedDataset(prefix)
    assert len(ds) == 3
    assert ds.get(0).tolist() == list(range(10, 20))
    assert ds.get(1).tolist() == list(range(30, 35))
    assert ds.get(2).tolist() == list(range(100, 140))
    assert ds.get(2, offset=5, length=3).tolist() == [105, 106, 107]
    assert ds.doc_idx.tolist() == [0, 2, 3]
    assert ds.dtype == np.uint16


def test_gpt_dataset_windows(tmp_path):
    from fengshen_amd.data.gpt_dataset import GPTDataset
    prefix = str(tmp_path / "gpt")
    builder = MMapIndexedDatasetBuilder(prefix + ".bin", dtype=np.int32)
    rng = np.random.RandomState(0)
    total = 0
    for _ in range(7):
        n = rng.randint(5, 40)
        builder.add_item(rng.randint(0, 1000, size=n))
        builder.end_document()
        total += n
    builder.finalize(prefix + ".idx")
    ds = MMapIndexedDataset(prefix)
    g = GPTDataset(ds, seq_length=16, num_epochs=2, seed=1)
    assert len(g) == (2 * total - 1) // 16
    for i in range(len(g)):
        item = g[i]
        assert item["input_ids"].shape == (16,)
        assert item["labels"].shape == (16,)
    # shifted-by-one relationship
    it = g[0]
    full = torch.cat([it["input_ids"][:1], it["labels"]])
    assert torch.equal(it["input_ids"], full[:-1])


def test_blendable_dataset_weights():
    from fengshen_amd.data.gpt_dataset import BlendableDataset

    class Const(torch.utils.data.Dataset):
        def __init__(self, v, n=100):
            self.v, self.n = v, n

        def __len__(self):
            return self.n

        def __getitem__(self, i):
            return self.v

    b = BlendableDataset([Const(0), Const(1)], [0.75, 0.25], size=400)
    vals = [b[i] for i in range(400)]
    assert abs(sum(vals) / 400 - 0.25) < 0.02


def test_mlm_sop_collator():
    from fengshen_amd.data.collators import MlmSopCollator
    tk = FakeTokenizer()
    coll = MlmSopCollator(tk, max_seq_length=64)
    texts = [{"text": "今天天气不错。我们出去玩吧！明天继续工作。"},
             {"text": "他说：好的。然后就走了。再也没有回来。"}]
    out = coll(texts)
    assert out["input_ids"].shape == (2, 64)
    assert out["labels"].shape == (2, 64)
    assert out["next_sentence_label"].shape == (2,)
    # some positions masked, labels set only there
    masked = (out["labels"] != -100)
    assert masked.any()
    # every masked label is a valid token id
    assert (out["labels"][masked] >= 0).all()
    # CLS at position 0, unmasked rows start with cls
    assert (out["input_ids"][:, 0] == tk.cls_token_id).all()


def test_sft_collator_prompt_masking():
    from fengshen_amd.data.collators import SftCollator
    tk = FakeTokenizer()
    coll = SftCollator(tk, max_seq_length=128)
    out = coll([{"query": "1+1?", "answer": "2"},
                {"query": ["hi", "more"], "answer": ["yo", "ok"]}])
    assert out["input_ids"].shape == out["labels"].shape
    labels = out["labels"]
    # prompt region must be -100; answers must appear in labels
    assert (labels[0] == -100).sum() > 2
    assert (labels[0] != -100).sum() >= 2  # answer + eos


def test_t5_span_collator():
    from fengshen_amd.data.collators import T5SpanCollator
    tk = FakeTokenizer()
    coll = T5SpanCollator(tk, max_seq_length=64)
    out = coll([{"text": "abcdefghij" * 10}, {"text": "0123456789" * 8}])
    assert out["input_ids"].dim() == 2
    assert out["labels"].dim() == 2
    # corrupted input shorter than raw
    assert out["input_ids"].shape[1] <= 64


def test_causal_collator():
    from fengshen_amd.data.collators import CausalCollator
    tk = FakeTokenizer()
    coll = CausalCollator(tk, max_seq_length=32)
    out = coll([{"text": "hello world"}, {"text": "a"}])
    assert out["input_ids"].shape == out["labels"].shape
    assert (out["labels"][1] == -100).sum() > 0  # padding masked


def test_sampler_exact_resume():
    from fengshen_amd.data.universal_sampler import PretrainingRandomSampler
    full = PretrainingRandomSampler(
        total_samples=100, consumed_samples=0, micro_batch_size=4,
        data_parallel_rank=0, data_parallel_size=2, epoch=0, seed=7)
    batches = list(full)
    resumed = PretrainingRandomSampler(
        total_samples=100, consumed_samples=24, micro_batch_size=4,
        data_parallel_rank=0, data_parallel_size=2, epoch=0, seed=7)
    resumed_batches = list(resumed)
    assert batches[3:] == resumed_batches


def test_masking_utils():
    from fengshen_amd.data.data_utils import (
        ChineseSentenceSplitter, create_masked_lm_predictions)
    sp = ChineseSentenceSplitter()
    s = sp.tokenize("你好。今天怎么样？很好！")
    assert len(s) == 3
    rng = np.random.RandomState(0)
    tokens = list(range(10, 40))
    vocab = list(range(5, 100))
    id2tok = {i: f"t{i}" for i in vocab + tokens}
    out, pos, lab = create_masked_lm_predictions(
        tokens, vocab, id2tok, 0.3, cls_id=1, sep_id=2, mask_id=3,
        max_predictions_per_seq=10, np_rng=rng)
    assert len(pos) == len(lab) > 0
    for p, l in zip(pos, lab):
        assert tokens[p] == l


def test_bert_mmap_dataset(tmp_path):
    from fengshen_amd.data.bert_dataset import BertMmapDataset
    from fengshen_amd.data.indexed_dataset import MMapIndexedDatasetBuilder, MMapIndexedDataset
    rng = np.random.RandomState(0)
    prefix = str(tmp_path / "bertcorp")
    builder = MMapIndexedDatasetBuilder(prefix + ".bin", dtype=np.int32)
    for _doc in range(6):
        for _sent in range(rng.randint(2, 5)):
            builder.add_item(rng.randint(10, 200, size=rng.randint(5, 20)))
        builder.end_document()
    builder.finalize(prefix + ".idx")
    ds = MMapIndexedDataset(prefix)
    vocab = list(range(10, 200))
    id2tok = {i: f"t{i}" for i in range(210)}
    bert_ds = BertMmapDataset(ds, vocab, id2tok, cls_id=1, sep_id=2,
                              mask_id=3, pad_id=0, max_seq_length=64,
                              num_epochs=2)
    assert len(bert_ds) > 0
    item = bert_ds[0]
    assert item["input_ids"].shape == (64,)
    assert item["input_ids"][0] == 1  # CLS
    assert (item["labels"] != -100).sum() > 0
    assert item["next_sentence_label"].item() in (0, 1)
    # deterministic per-sample
    item2 = bert_ds[0]
    assert torch.equal(item["input_ids"], item2["input_ids"])


def test_bert_preprocessing_pipeline(tmp_path):
    import json
    from fengshen_amd.data.bert_preprocessing import (
        split_shards, presplit_sentences, jsonl_to_mmap)
    from fengshen_amd.data.indexed_dataset import MMapIndexedDataset
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    src = tmp_path / "corpus.jsonl"
    with open(src, "w", encoding="utf-8") as f:
        for i in range(20):
            f.write(json.dumps(
                {"text": f"第{i}篇文章的第一句。第二句内容在这里。最后一句。"},
                ensure_ascii=False) + "\n")
    shards = split_shards(str(src), str(tmp_path / "shards"), shard_bytes=500)
    assert len(shards) > 1
    n = presplit_sentences(str(src), str(tmp_path / "presplit.jsonl"))
    assert n == 20
    with open(tmp_path / "presplit.jsonl", encoding="utf-8") as f:
        doc = json.loads(f.readline())
        assert len(doc["sentences"]) == 3
    tk = SimpleCharTokenizer()
    docs = jsonl_to_mmap(str(src), str(tmp_path / "corpus"), tk)
    assert docs == 20
    ds = MMapIndexedDataset(str(tmp_path / "corpus"))
    assert len(ds.doc_idx) == 21  # 20 docs + leading 0
    assert len(ds) == 60  # 3 sentences each


def test_bart_mmap_dataset(tmp_path):
    from fengshen_amd.data.bart_dataset import BartMmapDataset
    from fengshen_amd.data.indexed_dataset import (
        MMapIndexedDataset,
        MMapIndexedDatasetBuilder,
    )
    rng = np.random.RandomState(0)
    prefix = str(tmp_path / "bartcorp")
    builder = MMapIndexedDatasetBuilder(prefix + ".bin", dtype=np.int32)
    for _doc in range(6):
        for _sent in range(rng.randint(2, 5)):
            builder.add_item(rng.randint(10, 200, size=rng.randint(5, 20)))
        builder.end_document()
    builder.finalize(prefix + ".idx")
    ds = MMapIndexedDataset(prefix)
    id2tok = {i: f"t{i}" for i in range(210)}
    bart_ds = BartMmapDataset(ds, id2tok, cls_id=1, sep_id=2, mask_id=3,
                              pad_id=0, vocab_size=210, max_seq_length=64,
                              num_epochs=2)
    assert len(bart_ds) > 0
    item = bart_ds[0]
    assert item["input_ids"].shape == (64,)
    assert item["input_ids"][0] == 1                      # CLS kept
    assert (item["input_ids"] == 3).sum() > 0             # masking happened
    assert (item["labels"] != -100).sum() > 0
    # labels are the clean shifted stream, never containing [MASK]
    assert (item["labels"][item["labels"] != -100] != 3).all()
    # deterministic per-sample
    assert torch.equal(item["input_ids"], bart_ds[0]["input_ids"])
    # different samples get different noise
    assert not torch.equal(bart_ds[0]["input_ids"], bart_ds[1]["input_ids"])


def test_legacy_indexed_dataset_roundtrip(tmp_path):
    from fengshen_amd.data.indexed_dataset import (
        IndexedCachedDataset,
        IndexedDataset,
        IndexedDatasetBuilder,
        infer_dataset_impl,
    )
    p = str(tmp_path / "legacy")
    b = IndexedDatasetBuilder(p + ".bin", dtype=np.int32)
    arrs = [np.arange(5), np.arange(3) + 100, np.arange(7) + 200]
    for a in arrs:
        b.add_item(a)
    b.end_document()
    b.finalize(p + ".idx")
    assert infer_dataset_impl(p) == "cached"
    assert IndexedDataset.exists(p)
    ds = IndexedCachedDataset(p)
    ds.prefetch([0, 2])
    for i, a in enumerate(arrs):
        assert (ds[i] == a).all()
    assert len(ds) == 3


def test_get_samples_mapping_cache(tmp_path):
    from fengshen_amd.data.helpers_py import get_samples_mapping
    from fengshen_amd.data.indexed_dataset import (
        MMapIndexedDataset,
        MMapIndexedDatasetBuilder,
    )
    rng = np.random.RandomState(0)
    p = str(tmp_path / "corp")
    b = MMapIndexedDatasetBuilder(p + ".bin", dtype=np.int32)
    for _ in range(5):
        for _ in range(3):
            b.add_item(rng.randint(10, 200, size=12))
        b.end_document()
    b.finalize(p + ".idx")
    ds = MMapIndexedDataset(p)
    m1 = get_samples_mapping(ds, p, 2, 2 ** 62, 32, 0.1, 1234, "t")
    cache = [f for f in tmp_path.iterdir() if "indexmap" in f.name]
    assert len(cache) == 1
    m2 = get_samples_mapping(ds, p, 2, 2 ** 62, 32, 0.1, 1234, "t")
    assert (np.asarray(m1) == np.asarray(m2)).all()


def test_tagging_collators_feed_heads():
    """span/biaffine collators produce exactly what BertSpan/BertBiaffine
    consume (ref sequence_tagging_collator.py:9-206)."""
    from fengshen_amd.data.tagging_collators import (
        CollatorForBiaffine,
        CollatorForCrf,
        CollatorForLinear,
        CollatorForSpan,
    )
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config,
    )
    from fengshen_amd.models.tagging_models.bert_for_tagging import (
        BertBiaffine,
        BertSpan,
    )
    from fengshen_amd.tokenizer import SimpleCharTokenizer

    tk = SimpleCharTokenizer()
    samples = [{"text": "abcdef", "entities": [(0, 1, 2), (3, 5, 1)],
                "labels": ["B-a", "I-a", "O", "B-b", "I-b", "I-b"]},
               {"text": "abc", "entities": [(1, 2, 3)],
                "labels": ["O", "B-c", "I-c"]}]

    lin = CollatorForLinear(tk, {"O": 0, "B-a": 1, "I-a": 2, "B-b": 3,
                                 "I-b": 4, "B-c": 5, "I-c": 6})
    b = lin(samples)
    assert b["labels"].shape == b["input_ids"].shape
    assert CollatorForCrf is CollatorForLinear

    cfg = bert_tiny_config(vocab_size=300, torch_dtype="float32")
    span_batch = CollatorForSpan(tk)(samples)
    m = BertSpan(cfg, num_labels=5).float()
    out = m(**span_batch)
    assert out.loss.isfinite()
    # boundary labels landed (+1 for CLS)
    assert span_batch["start_positions"][0, 1] == 2
    assert span_batch["end_positions"][0, 2] == 2

    bia_batch = CollatorForBiaffine(tk)(samples)
    m2 = BertBiaffine(cfg, num_labels=5).float()
    out2 = m2(**bia_batch)
    assert out2.loss.isfinite()
    assert bia_batch["span_labels"][0, 1, 2] == 2       # entity cell
    assert bia_batch["span_labels"][0, 2, 1] == -100    # below diagonal


# ---------------------------------------------------------------------------
# t5_gen_datasets: knowledge-grounded dialog dataset
# (ref data/t5_dataloader/t5_gen_datasets.py)
# ---------------------------------------------------------------------------
def test_dialog_dataset_layout():
    from fengshen_amd.data.t5_gen_datasets import (
        DialogCollator, DialogDataset, add_dialog_special_tokens)
    from fengshen_amd.tokenizer import SimpleCharTokenizer

    class Tok(SimpleCharTokenizer):
        def add_special_tokens(self, d):
            for t in d["additional_special_tokens"]:
                self._vocab.setdefault(t, len(self._vocab))
            self._inv = {v: k for k, v in self._vocab.items()}

        def convert_tokens_to_ids(self, t):
            return self._vocab.get(t, self.unk_token_id)

    tk = add_dialog_special_tokens(Tok())
    data = [{"context": ["你好", "你好啊", "天气如何"],
             "knowledge": "今天晴",
             "target": "天气很好"}]
    ds = DialogDataset(data, tk, max_seq_length=64,
                       max_knowledge_length=16, max_target_length=8,
                       eos_token_id=tk.eos_token_id)
    s = ds[0]
    kn_start = tk.convert_tokens_to_ids("[KNSTART]")
    kn_end = tk.convert_tokens_to_ids("[KNEND]")
    ct_start = tk.convert_tokens_to_ids("[CTSTART]")
    ids = s["input_ids"].tolist()
    assert ids[0] == ct_start and kn_start in ids and ids[-1] == kn_end
    # knowledge region typed 2, context alternates 0/1
    tt = s["token_types"].tolist()
    kn_i = ids.index(kn_start)
    assert all(t == 2 for t in tt[kn_i:])
    assert set(tt[:kn_i]) <= {0, 1}
    assert len(ids) == len(tt) == len(s["attention_mask"])
    assert s["labels"][-1] == tk.eos_token_id


def test_dialog_collator_shift_right():
    import numpy as np
    from fengshen_amd.data.t5_gen_datasets import (
        DialogCollator, shift_tokens_right)
    labels = np.array([[7, 8, 9, -100], [5, 6, -100, -100]])
    shifted = shift_tokens_right(labels, pad_token_id=0,
                                 decoder_start_token_id=2)
    assert shifted.tolist() == [[2, 7, 8, 9], [2, 5, 6, 0]]
    samples = [{"input_ids": [3, 4], "token_types": [0, 0],
                "attention_mask": [1, 1], "labels": [7, 8]},
               {"input_ids": [3, 4, 5], "token_types": [0, 0, 1],
                "attention_mask": [1, 1, 1], "labels": [9]}]
    batch = DialogCollator(pad_token_id=0, decoder_start_token_id=2)(samples)
    assert batch["input_ids"].shape == (2, 3)
    assert batch["labels"][1].tolist() == [9, -100]
    assert batch["decoder_input_ids"][0].tolist() == [2, 7]
