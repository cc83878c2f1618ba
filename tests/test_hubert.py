"""HuBERT model + dataset integration."""
import numpy as np
import torch

from fengshen_amd.data.hubert_dataset import HubertDataset
from fengshen_amd.models.hubert import (
    HubertForPreTraining,
    HubertModel,
    hubert_tiny_config,
)
from fengshen_amd.models.hubert.modeling_hubert import compute_mask_indices


def test_frame_lengths_match_conv_output():
    cfg = hubert_tiny_config(torch_dtype="float32")
    m = HubertModel(cfg).float()
    for n in [2000, 4000, 5555]:
        src = torch.randn(1, n)
        T = m.feature_extractor(src).shape[1]
        assert m.frame_lengths(torch.tensor([n]))[0].item() == T


def test_mask_indices_respect_padding():
    torch.manual_seed(0)
    pad = torch.zeros(2, 50, dtype=torch.bool)
    pad[1, 10:] = True
    mi = compute_mask_indices((2, 50), 0.5, 4, pad)
    assert mi[0].any()
    # spans start inside the valid region for the padded row
    assert mi[1, :14].any()


def test_pretraining_loss_and_grads():
    torch.manual_seed(0)
    cfg = hubert_tiny_config(torch_dtype="float32")
    m = HubertForPreTraining(cfg).float()
    src = torch.randn(2, 4000)
    pad = torch.zeros(2, 4000, dtype=torch.bool)
    pad[1, 3000:] = True
    T = m.hubert.frame_lengths(torch.tensor([4000]))[0].item()
    lab = torch.randint(0, cfg.vocab_size, (2, T))
    out = m(src, padding_mask=pad, labels=lab)
    assert out.loss.isfinite()
    out.loss.backward()
    assert m.label_embs.grad is not None
    assert m.hubert.feature_extractor.conv_layers[0].weight.grad is not None
    # padded frames never contribute to the loss selection
    assert not (out.mask_time_indices & torch.zeros(1, dtype=torch.bool)).any()


def test_dataset_collater_feeds_model():
    torch.manual_seed(0)
    rng = np.random.RandomState(0)
    waves = [rng.randn(3000 + 500 * i).astype(np.float32) for i in range(3)]
    labels = [rng.randint(0, 16, size=(len(w) // 320,)) for w in waves]
    ds = HubertDataset(waves, labels, max_sample_size=3200)
    batch = ds.collater([ds[i] for i in range(3)])
    cfg = hubert_tiny_config(torch_dtype="float32")
    m = HubertForPreTraining(cfg).float()
    out = m(batch["source"], padding_mask=batch["padding_mask"],
            labels=batch["labels"])
    assert out.loss.isfinite()


def test_hubert_save_load_roundtrip(tmp_path):
    torch.manual_seed(0)
    cfg = hubert_tiny_config(torch_dtype="float32")
    m = HubertForPreTraining(cfg).float()
    m.save_pretrained(tmp_path / "hubert")
    m2 = HubertForPreTraining.from_pretrained(tmp_path / "hubert").float()
    m.eval()
    m2.eval()  # dropout off for a deterministic comparison
    src = torch.randn(1, 3200)
    mi = torch.zeros(1, m.hubert.frame_lengths(torch.tensor([3200]))[0],
                     dtype=torch.bool)
    mi[:, :4] = True
    with torch.no_grad():
        a = m.hubert(src, apply_mask=True, mask_time_indices=mi)
        b = m2.hubert(src, apply_mask=True, mask_time_indices=mi)
    assert torch.allclose(a.last_hidden_state, b.last_hidden_state,
                          atol=1e-6)
