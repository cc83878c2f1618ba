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


# ---------------------------------------------------------------------------
# Full-parity manifest dataset (ref data/hubert/hubert_dataset.py)
# ---------------------------------------------------------------------------
def _write_manifest(tmp_path, n=4, sr=16000, label_rate=50.0):
    import numpy as np
    root = tmp_path / "audio"
    root.mkdir()
    sizes = [3200, 4800, 1600, 6400]
    lines = [str(root)]
    label_lines = []
    for i in range(n):
        wav = np.random.RandomState(i).randn(sizes[i]).astype(np.float32)
        np.save(root / f"a{i}.npy", wav)
        lines.append(f"a{i}.npy\t{sizes[i]}")
        n_frames = int(sizes[i] / sr * label_rate)
        label_lines.append(" ".join(
            str(j % 5) for j in range(n_frames)))
    man = tmp_path / "train.tsv"
    man.write_text("\n".join(lines) + "\n")
    lab = tmp_path / "train.km"
    lab.write_text("\n".join(label_lines) + "\n")
    return str(man), str(lab)


def test_label_dictionary_fairseq_semantics():
    from fengshen_amd.data.hubert_dataset import LabelDictionary
    d = LabelDictionary([str(i) for i in range(5)])
    assert d.pad() == 1 and d.eos() == 2 and d.unk() == 3
    assert d.index("0") == 4  # first symbol after the 4 specials
    enc = d.encode_line("0 3 zzz")
    assert enc.tolist() == [4, 7, d.unk()]
    assert d.string([4, 5]) == "0 1"


def test_hubert_manifest_dataset_frame_alignment(tmp_path):
    from fengshen_amd.data.hubert_dataset import (
        HubertManifestDataset, LabelDictionary)
    man, lab = _write_manifest(tmp_path)
    d = LabelDictionary([str(i) for i in range(5)])
    ds = HubertManifestDataset(
        man, 16000, [lab], label_rates=50.0, pad_list=[d.pad()],
        label_processors=[lambda s: d.encode_line(s)],
        max_sample_size=3200, random_crop=True, store_labels=True)
    assert len(ds) == 4
    item = ds[1]
    assert item["source"].shape[0] == 4800  # crop happens in collater
    batch = ds.collater([ds[0], ds[1]])
    src = batch["net_input"]["source"]
    assert src.shape == (2, 3200)  # min(min sizes, max_sample_size)
    t = batch["target_list"][0]
    # 3200 samples @50Hz/16k = 10 frames
    assert t.shape[1] == 10
    assert batch["ntokens_list"][0] == 20


def test_hubert_manifest_offsets_and_filtering(tmp_path):
    from fengshen_amd.data.hubert_dataset import (
        HubertManifestDataset, LabelDictionary, load_label_offset)
    man, lab = _write_manifest(tmp_path)
    d = LabelDictionary([str(i) for i in range(5)])
    # min_keep filters the 1600-sample clip
    ds = HubertManifestDataset(
        man, 16000, [lab], label_rates=50.0, pad_list=[d.pad()],
        label_processors=[lambda s: d.encode_line(s)],
        min_keep_sample_size=3000, store_labels=False,
        pad_audio=True, max_sample_size=8000, single_target=True)
    assert len(ds) == 3
    # streamed label equals stored label for a filtered index set
    assert ds.get_label(2, 0).tolist()[:3] == [4, 5, 6]
    batch = ds.collater([ds[0], ds[2]])
    assert batch["net_input"]["source"].shape == (2, 6400)
    assert batch["net_input"]["padding_mask"][0, -1]  # padded tail
    assert "target" in batch and "ntokens" in batch


def test_hubert_ordered_indices_size_sorted(tmp_path):
    from fengshen_amd.data.hubert_dataset import (
        HubertManifestDataset, LabelDictionary)
    man, lab = _write_manifest(tmp_path)
    d = LabelDictionary([str(i) for i in range(5)])
    ds = HubertManifestDataset(
        man, 16000, [lab], label_rates=50.0, pad_list=[d.pad()],
        label_processors=[lambda s: d.encode_line(s)], shuffle=False)
    idx = ds.ordered_indices()
    sizes = [ds.sizes[i] for i in idx]
    assert sizes == sorted(sizes, reverse=True)
