"""Zoo batch 4: PPVAE, deepVAE, GAVAE, pegasus utils, hubert dataset."""
import numpy as np
import torch


def test_ppvae():
    from fengshen_amd.models.ppvae.modeling_ppvae import (
        PPVAEModel, PPVAEConfig)
    torch.manual_seed(0)
    m = PPVAEModel(PPVAEConfig(latent_dim=32))
    lat = torch.randn(8, 32)
    out = m(lat)
    assert out.loss.isfinite()
    out.loss.backward()
    z = m.sample_latent(4)
    assert z.shape == (4, 32)


def test_deep_vae_layerwise_latents():
    from fengshen_amd.models.deep_vae.modeling_deep_vae import (
        DeepVAEModel, deep_vae_tiny_config)
    torch.manual_seed(0)
    m = DeepVAEModel(deep_vae_tiny_config())
    ids = torch.randint(3, 256, (2, 16))
    out = m(ids, labels=ids)
    assert out.loss.isfinite() and out.kl_loss.isfinite()
    out.loss.backward()


def test_gavae_gan_losses():
    from fengshen_amd.models.gavae.modeling_gavae import (
        GAVAEModel, GAVAEConfig)
    torch.manual_seed(0)
    m = GAVAEModel(GAVAEConfig(latent_dim=16, n_labels=3))
    real = torch.randn(6, 16)
    labels = torch.randint(0, 3, (6,))
    out = m(real, labels)
    assert out.g_loss.isfinite() and out.d_loss.isfinite()
    (out.g_loss + out.d_loss).backward()
    z = m.generate_latent(labels)
    assert z.shape == (6, 16)


def test_pegasus_gap_sentences():
    from fengshen_amd.data.pegasus_utils import build_gap_sentence_sample
    text = "今天天气很好。我们去公园玩。公园里有很多人。天气好人就多。回家吃饭。"
    inp, tgt = build_gap_sentence_sample(text, gap_ratio=0.4)
    assert "[MASK]" in inp
    assert len(tgt) > 0
    # target sentences are removed from input
    for sent in tgt.split("。"):
        if sent:
            assert sent + "。" not in inp


def test_hubert_dataset():
    from fengshen_amd.data.hubert_dataset import HubertDataset
    rng = np.random.RandomState(0)
    wavs = [rng.randn(32000).astype("float32"),
            rng.randn(48000).astype("float32")]
    labs = [rng.randint(0, 100, 100), rng.randint(0, 100, 150)]
    ds = HubertDataset(wavs, labs, max_sample_size=16000)
    a = ds[0]
    assert a["source"].shape[0] == 16000
    assert a["label"].numel() <= 51
    batch = ds.collater([ds[0], ds[1]])
    assert batch["source"].shape[0] == 2
    assert batch["padding_mask"].dtype == torch.bool


def test_auto_registry_roundtrip(tmp_path):
    from fengshen_amd.models.auto import register_fengshen_auto_classes
    register_fengshen_auto_classes()
    from transformers import AutoConfig, AutoModelForCausalLM
    from fengshen_amd.models.llama.configuration_llama import llama_tiny_config
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    torch.manual_seed(0)
    m = LlamaForCausalLM(llama_tiny_config()).eval()
    m.save_pretrained(tmp_path)
    cfg = AutoConfig.from_pretrained(tmp_path)
    assert cfg.model_type == "fengshen_llama"
    m2 = AutoModelForCausalLM.from_pretrained(tmp_path).eval()
    ids = torch.randint(3, 256, (1, 8))
    with torch.no_grad():
        assert torch.allclose(m(ids).logits, m2(ids).logits, atol=1e-5)


def test_deltalm_interleaved_decoder():
    from fengshen_amd.models.deltalm.modeling_deltalm import (
        DeltaLMForConditionalGeneration, deltalm_tiny_config)
    torch.manual_seed(0)
    m = DeltaLMForConditionalGeneration(deltalm_tiny_config())
    src = torch.randint(3, 256, (2, 14))
    lab = torch.randint(3, 256, (2, 9))
    out = m(input_ids=src, labels=lab)
    assert out.loss.isfinite()
    out.loss.backward()
    # signature structure: decoder layer has two FFNs
    assert hasattr(m.dec_layers[0], "ffn1") and hasattr(m.dec_layers[0], "ffn2")
    m.eval()
    gen = m.generate(src, max_new_tokens=4, do_sample=False)
    assert gen.shape[0] == 2
