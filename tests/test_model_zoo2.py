"""Model-zoo batch 3: Longformer, DeBERTa-v2, ZEN, ALBERT, Transfo-XL,
DAVAE."""
import torch


def test_longformer_band_and_global():
    from fengshen_amd.models.longformer.modeling_longformer import (
        LongformerForMaskedLM, longformer_tiny_config)
    torch.manual_seed(0)
    m = LongformerForMaskedLM(longformer_tiny_config())
    ids = torch.randint(3, 256, (2, 64))
    labels = ids.clone()
    labels[:, ::2] = -100
    gmask = torch.zeros_like(ids)
    gmask[:, 0] = 1  # CLS global
    out = m(ids, attention_mask=torch.ones_like(ids),
            global_attention_mask=gmask, labels=labels)
    assert out.loss.isfinite()
    out.loss.backward()


def test_longformer_window_locality():
    """tokens beyond the window must not influence a local token's output."""
    from fengshen_amd.models.longformer.modeling_longformer import (
        LongformerModel, longformer_tiny_config)
    torch.manual_seed(0)
    m = LongformerModel(longformer_tiny_config()).eval()
    ids = torch.randint(3, 256, (1, 64))
    ids2 = ids.clone()
    ids2[0, -1] = (ids2[0, -1] + 1) % 253 + 3  # perturb far-away token
    with torch.no_grad():
        h1 = m(ids).last_hidden_state
        h2 = m(ids2).last_hidden_state
    # token 0 (distance 63 > window 16) unaffected
    assert torch.allclose(h1[0, 0], h2[0, 0], atol=1e-5)
    # last token obviously affected
    assert not torch.allclose(h1[0, -1], h2[0, -1], atol=1e-3)


def test_deberta_disentangled_attention():
    from fengshen_amd.models.deberta_v2.modeling_deberta_v2 import (
        DebertaV2ForMaskedLM, deberta_tiny_config)
    torch.manual_seed(0)
    m = DebertaV2ForMaskedLM(deberta_tiny_config())
    ids = torch.randint(3, 256, (2, 20))
    labels = ids.clone()
    labels[:, ::3] = -100
    out = m(ids, attention_mask=torch.ones_like(ids), labels=labels)
    assert out.loss.isfinite()
    out.loss.backward()
    # position-sensitivity: with non-trivial relative embeddings, reversing
    # the sequence must change per-token outputs (deberta has NO absolute
    # positions — all position signal flows through c2p/p2c)
    from fengshen_amd.models.deberta_v2.modeling_deberta_v2 import DebertaV2Model
    enc = DebertaV2Model(deberta_tiny_config()).eval()
    with torch.no_grad():
        enc.rel_embeddings.weight.normal_(0, 0.5)
        for layer in enc.layers:
            layer.attn.pos_key.weight.normal_(0, 0.5)
            layer.attn.pos_query.weight.normal_(0, 0.5)
        a = enc(ids).last_hidden_state
        b = enc(ids.flip(dims=[1])).last_hidden_state.flip(dims=[1])
    assert not torch.allclose(a, b, atol=1e-3)


def test_zen_ngram_fusion():
    from fengshen_amd.models.zen.modeling_zen import (
        ZenForSequenceClassification, ZenNgramDict, zen_tiny_config)
    torch.manual_seed(0)
    cfg = zen_tiny_config()
    cfg.num_labels = 2
    m = ZenForSequenceClassification(cfg)
    b, s, ng = 2, 24, 8
    ids = torch.randint(3, 256, (b, s))
    ngram_ids = torch.randint(1, 512, (b, ng))
    pos_mat = torch.zeros(b, s, ng)
    pos_mat[:, 2, 0] = 1
    pos_mat[:, 3, 0] = 1  # ngram 0 covers tokens 2-3
    labels = torch.tensor([0, 1])
    out = m(ids, ngram_ids=ngram_ids, ngram_position_matrix=pos_mat,
            attention_mask=torch.ones_like(ids), labels=labels)
    assert out.loss.isfinite()
    out.loss.backward()
    # the ngram dict matcher
    d = ZenNgramDict(["你好", "世界", "你好世"])
    matches = d.match(list("你好世界"))
    assert ("你好" in [d.id_to_ngram[g] for g, _, _ in matches])


def test_albert_shared_layers():
    from fengshen_amd.models.albert.modeling_albert import (
        AlbertForMaskedLM, albert_tiny_config)
    torch.manual_seed(0)
    m = AlbertForMaskedLM(albert_tiny_config())
    # parameter sharing: model is much smaller than an unshared 3-layer
    n_params = sum(p.numel() for p in m.albert.parameters())
    ids = torch.randint(3, 256, (2, 16))
    labels = ids.clone()
    labels[:, ::2] = -100
    out = m(ids, labels=labels)
    assert out.loss.isfinite()
    out.loss.backward()
    assert n_params < 200_000


def test_transfo_xl_memory_recurrence():
    from fengshen_amd.models.transfo_xl_denoise.modeling_transfo_xl_denoise import (
        TransfoXLDenoiseModel, transfo_xl_tiny_config)
    torch.manual_seed(0)
    m = TransfoXLDenoiseModel(transfo_xl_tiny_config()).eval()
    a = torch.randint(3, 256, (2, 16))
    bseg = torch.randint(3, 256, (2, 16))
    with torch.no_grad():
        out1 = m(a)
        out2_with_mem = m(bseg, mems=out1.mems)
        out2_no_mem = m(bseg)
    # memory must change the second segment's outputs
    assert not torch.allclose(out2_with_mem.logits, out2_no_mem.logits,
                              atol=1e-3)
    # training step
    m.train()
    out = m(a, labels=a)
    assert out.loss.isfinite()
    out.loss.backward()


def test_davae_elbo_and_sample():
    from fengshen_amd.models.davae.modeling_davae import (
        DAVAEModel, davae_tiny_config)
    torch.manual_seed(0)
    m = DAVAEModel(davae_tiny_config())
    ids = torch.randint(3, 256, (2, 20))
    out = m(ids, labels=ids)
    assert out.loss.isfinite() and out.kl_loss.isfinite()
    out.loss.backward()
    m.eval()
    gen = m.sample(2, 8)
    assert gen.shape == (2, 8)


def test_tcbert_prompt_classification():
    from fengshen_amd.models.tcbert.modeling_tcbert import TCBertModel
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    torch.manual_seed(0)
    m = TCBertModel(bert_tiny_config())
    ids = torch.randint(3, 256, (2, 20))
    mask_pos = torch.tensor([[1, 2], [1, 2]])
    label_tok = torch.randint(3, 256, (4, 2))  # 4 labels x 2 verbalizer toks
    labels = torch.tensor([0, 3])
    out = m(ids, mask_positions=mask_pos, label_token_ids=label_tok,
            labels=labels)
    assert out.loss.isfinite() and out.label_logits.shape == (2, 4)
    out.loss.backward()


def test_uniex_triaffine_training_and_extract():
    """Reference UniEXBertModel semantics (:885-1025): triaffine span
    scoring with an index head (label 0) over the full sequence and type
    heads over gathered text tokens; full + fast extract modes."""
    from fengshen_amd.models.uniex.modeling_uniex import (
        UniEXModel, span_gather)
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    torch.manual_seed(0)
    m = UniEXModel(bert_tiny_config(), triaffine_hidden_size=32)
    b, s, n_lab = 2, 16, 3  # label 0 = index head, 1..2 = types
    ids = torch.randint(3, 256, (b, s))
    label_token_idx = torch.tensor([[1, 3, 5]] * b)
    text_token_idx = torch.tensor([[7 + i for i in range(8)]] * b)
    span_labels = torch.zeros(b, s, s, n_lab)
    span_labels[:, 8, 10, 0] = 1   # index head hit
    span_labels[:, 8, 10, 1] = 1   # type 1
    span_mask = torch.zeros(b, s, s, n_lab) - 10000.0
    span_mask[:, 7:15, 7:15, :] = 0.0
    out = m(ids, span_labels=span_labels, span_labels_mask=span_mask,
            label_token_idx=label_token_idx, text_token_idx=text_token_idx)
    assert out.loss.isfinite()
    assert out.span_logits.shape == (b, 8, 8, n_lab - 1)
    out.loss.backward()
    # span_gather subsets the grid correctly
    sub = span_gather(span_labels[:, :, :, 1:], text_token_idx)
    assert sub.shape == (b, 8, 8, n_lab - 1)
    assert sub[0, 1, 3, 0] == 1  # (8,10) -> text-relative (1,3)
    # full extract
    full = m(ids, label_token_idx=label_token_idx,
             text_token_idx=text_token_idx, fast_ex_mode=False)
    assert full.span_logits.shape == (b, 8, 8, n_lab)
    # fast extract returns span/type dicts
    fast = m(ids, label_token_idx=label_token_idx,
             text_token_idx=text_token_idx, fast_ex_mode=True,
             threshold=0.0)
    assert len(fast) == b
    if fast[0]:
        assert {"span", "type", "score"} <= set(fast[0][0])


def test_tcbert_pipeline():
    from fengshen_amd.pipelines.tcbert import TCBertPipeline
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    torch.manual_seed(0)
    pipe = TCBertPipeline(tokenizer=SimpleCharTokenizer(),
                          config=bert_tiny_config(),
                          labels=["体育", "科技", "财经"])
    out = pipe("这是一段测试文本")
    assert out["label_name"] in ["体育", "科技", "财经"]


def test_llama_convert_roundtrip():
    import torch as t
    from fengshen_amd.utils.llama_convert import (
        hf_to_fs_llama, fs_to_hf_llama, make_delta, apply_delta)
    from fengshen_amd.models.llama.configuration_llama import llama_tiny_config
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    t.manual_seed(0)
    m = LlamaForCausalLM(llama_tiny_config())
    fs_sd = m.state_dict()
    hf_sd = fs_to_hf_llama(fs_sd, num_layers=2)
    back = hf_to_fs_llama(hf_sd, num_layers=2)
    for k, v in fs_sd.items():
        assert t.equal(v, back[k]), k
    # delta roundtrip
    target = {k: v + 0.5 for k, v in hf_sd.items()}
    delta = make_delta(hf_sd, target)
    rec = apply_delta(hf_sd, delta)
    for k in hf_sd:
        assert t.allclose(rec[k], target[k], atol=1e-6)


def test_mmap_index_dataset(tmp_path):
    from fengshen_amd.data.mmap_dataloader import (
        MMapIndexDataset, convert_py_to_npy)
    samples = [{"input_ids": list(range(5))},
               {"input_ids": list(range(10, 17))}]
    prefix = str(tmp_path / "simple")
    convert_py_to_npy(samples, prefix)
    ds = MMapIndexDataset(prefix)
    assert len(ds) == 2
    assert ds[1]["input_ids"].tolist() == list(range(10, 17))


def test_int8_quantization_close():
    from fengshen_amd.utils.quantize import quantize_model_int8
    from fengshen_amd.models.llama.configuration_llama import llama_tiny_config
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    torch.manual_seed(0)
    m = LlamaForCausalLM(llama_tiny_config()).eval()
    ids = torch.randint(3, 256, (2, 16))
    with torch.no_grad():
        ref = m(ids).logits
    quantize_model_int8(m)
    with torch.no_grad():
        q = m(ids).logits
    rel = (q - ref).abs().max() / ref.abs().max()
    assert rel < 0.1, rel.item()


def test_uniex_metrics():
    """get_entity_f1 / get_rel_f1 (ref modeling_uniex.py:44-162)."""
    from fengshen_amd.models.uniex.modeling_uniex import (
        get_entity_f1, get_rel_f1)
    test = [{"entity_list": [
        {"entity_type": "人名", "entity_index": [[0, 1]]},
        {"entity_type": "地名", "entity_index": [[4, 5]]}]}]
    pred = [{"entity_list": [
        {"entity_type": "人名", "entity_index": [[0, 1]]},
        {"entity_type": "地名", "entity_index": [[3, 5]]}]}]
    f1, r, p = get_entity_f1(test, pred)
    assert abs(p - 0.5) < 1e-9 and abs(r - 0.5) < 1e-9
    assert abs(f1 - 0.5) < 1e-9
    # spo fallback when entity_list is empty
    t2 = [{"spo_list": [{"predicate": "位于",
                         "subject": {"entity_type": "机构",
                                     "entity_index": [[0, 1]]},
                         "object": {"entity_type": "地名",
                                    "entity_index": [[4, 5]]}}]}]
    f1e, _, _ = get_entity_f1(t2, t2)
    assert f1e == 1.0
    f1r, _, _ = get_rel_f1(t2, t2)
    assert f1r == 1.0


def test_uniex_data_encoder_trains():
    """UniEXDataEncoder feeds the triaffine training path end-to-end and
    the index-head grid marks the char-aligned entity span."""
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    from fengshen_amd.models.uniex.modeling_uniex import (
        UniEXDataEncoder, UniEXModel)
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    torch.manual_seed(0)
    tk = SimpleCharTokenizer()
    enc = UniEXDataEncoder(tk, max_length=48)
    types = ["人名", "地名"]
    items = [{"text": "ab cd ef",
              "entity_list": [{"entity_type": "人名",
                               "entity_index": [[0, 1]]}]},
             {"text": "xy zw", "entity_list": []}]
    batch = enc.collate([enc.encode(it, types) for it in items])
    assert batch["span_labels"].shape[-1] == 1 + len(types)
    # index + type heads hit at the encoded span
    s0 = enc.encode(items[0], types)
    ts = s0["text_start"]
    assert s0["span_labels"][ts, ts + 1, 0] == 1
    assert s0["span_labels"][ts, ts + 1, 1] == 1
    m = UniEXModel(bert_tiny_config(), triaffine_hidden_size=32)
    out = m(**batch)
    assert out.loss.isfinite()
    out.loss.backward()
