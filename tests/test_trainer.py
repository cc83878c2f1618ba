"""End-to-end CPU tests of the trainer stack (module -> trainer -> optimizer
-> checkpoint/resume) on a toy regression problem."""
import argparse
import math
import os

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, Dataset

from fengshen_amd import FengshenModule, Trainer
from fengshen_amd.models.model_utils import (
    add_module_args,
    configure_optimizers,
    get_scheduler,
)
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class ToyDataset(Dataset):
    def __init__(self, n=256, d=8, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(n, d, generator=g)
        w = torch.arange(1, d + 1, dtype=torch.float32)
        self.y = self.x @ w

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return {"x": self.x[i], "y": self.y[i]}


class ToyModule(FengshenModule):
    def __init__(self, args, d=8):
        super().__init__()
        self.save_hyperparameters(args)
        self.net = nn.Sequential(nn.Linear(d, 32), nn.Tanh(), nn.Linear(32, 1))
        self.losses = []

    def training_step(self, batch, batch_idx):
        pred = self.net(batch["x"]).squeeze(-1)
        loss = torch.nn.functional.mse_loss(pred, batch["y"])
        self.log("train_loss", loss)
        self.losses.append(loss.item())
        return loss

    def validation_step(self, batch, batch_idx):
        pred = self.net(batch["x"]).squeeze(-1)
        loss = torch.nn.functional.mse_loss(pred, batch["y"])
        self.log("val_loss", loss, sync_dist=True)
        return loss

    def configure_optimizers(self):
        return configure_optimizers(self)


def _make_args(**over):
    parser = argparse.ArgumentParser()
    add_module_args(parser)
    args = parser.parse_args([])
    args.learning_rate = 1e-2
    args.warmup_steps = 2
    for k, v in over.items():
        setattr(args, k, v)
    return args


def test_fit_loss_decreases(tmp_path):
    args = _make_args()
    torch.manual_seed(0)
    model = ToyModule(args)
    loader = DataLoader(ToyDataset(), batch_size=32, shuffle=True)
    trainer = Trainer(max_steps=60, precision="fp32",
                      default_root_dir=str(tmp_path), log_every_n_steps=20)
    trainer.fit(model, train_dataloaders=loader)
    assert trainer.global_step == 60
    first = sum(model.losses[:5]) / 5
    last = sum(model.losses[-5:]) / 5
    assert last < first * 0.7, f"loss did not decrease: {first} -> {last}"


def test_grad_accumulation_counts(tmp_path):
    args = _make_args()
    model = ToyModule(args)
    loader = DataLoader(ToyDataset(n=64), batch_size=8)
    trainer = Trainer(max_steps=4, precision="fp32", accumulate_grad_batches=2,
                      default_root_dir=str(tmp_path))
    trainer.fit(model, train_dataloaders=loader)
    assert trainer.global_step == 4
    # 4 optimizer steps * 2 micro * 8 samples
    assert trainer.global_samples == 64


def _toy_datamodule(args, n=512):
    from fengshen_amd import UniversalDataModule
    args.train_batchsize = 16
    args.val_batchsize = 16
    args.sampler_type = "single"  # sequential => deterministic order
    args.num_workers = 0
    args.train_datasets_field = "train"
    args.val_datasets_field = "validation"
    return UniversalDataModule(
        tokenizer=None, collate_fn=None, args=args,
        datasets={"train": ToyDataset(n=n)})


def test_checkpoint_resume_exact(tmp_path):
    # run 1: 10 steps uninterrupted; run 2: 5 steps -> ckpt -> resume 5 more.
    # exact data-order resume comes from consumed_samples + PretrainingSampler.
    args = _make_args()
    args.lr_decay_steps = 10  # fix schedule horizon across the interrupted run
    torch.manual_seed(7)
    model_a = ToyModule(args)
    tr_a = Trainer(max_steps=10, precision="fp32",
                   default_root_dir=str(tmp_path / "a"))
    tr_a.fit(model_a, datamodule=_toy_datamodule(args))

    torch.manual_seed(7)
    model_b = ToyModule(args)
    tr_b = Trainer(max_steps=5, precision="fp32",
                   default_root_dir=str(tmp_path / "b"))
    tr_b.fit(model_b, datamodule=_toy_datamodule(args))
    ckpt = str(tmp_path / "ckpt5")
    tr_b.save_checkpoint(ckpt)

    model_c = ToyModule(args)
    tr_c = Trainer(max_steps=10, precision="fp32",
                   default_root_dir=str(tmp_path / "c"))
    tr_c.fit(model_c, datamodule=_toy_datamodule(args), ckpt_path=ckpt)
    assert tr_c.global_step == 10

    # same final weights as the uninterrupted run (same data order: sequential
    # 512-sample dataset consumed deterministically per fit call)
    for pa, pc in zip(model_a.parameters(), model_c.parameters()):
        assert torch.allclose(pa, pc, atol=1e-5), "resume diverged"


def test_universal_checkpoint_callback(tmp_path):
    args = _make_args()
    args.save_ckpt_path = str(tmp_path / "ckpts")
    args.every_n_train_steps = 5
    args.save_top_k = 2
    args.save_last = False
    args.monitor = "step"
    args.mode = "max"
    args.filename = "model-{step:02d}"
    args.save_weights_only = False
    args.every_n_epochs = None
    cb = UniversalCheckpoint(args)
    model = ToyModule(args)
    loader = DataLoader(ToyDataset(n=640), batch_size=16)
    trainer = Trainer(max_steps=20, precision="fp32", callbacks=[cb],
                      default_root_dir=str(tmp_path))
    trainer.fit(model, train_dataloaders=loader)
    saved = sorted(os.listdir(args.save_ckpt_path))
    assert len(saved) == 2  # top-k pruning
    assert "model-15.ckpt" in saved and "model-20.ckpt" in saved


def test_schedulers_shapes():
    m = nn.Linear(2, 2)
    opt = torch.optim.AdamW(m.parameters(), lr=1.0)
    for name in ["constant", "constant_with_warmup", "linear", "cosine",
                 "polynomial", "inverse_sqrt", "direct"]:
        sch = get_scheduler(name, opt, num_warmup_steps=5,
                            num_training_steps=20, lr_init=1.0, lr_end=0.1)
        lrs = []
        for _ in range(25):
            lrs.append(opt.param_groups[0]["lr"])
            opt.step()
            sch.step()
        assert all(not math.isnan(x) for x in lrs), name
        if name != "constant":
            assert lrs[1] < lrs[5], f"{name}: warmup missing"


def test_validation_loop(tmp_path):
    args = _make_args()
    model = ToyModule(args)
    train = DataLoader(ToyDataset(n=128), batch_size=16)
    val = DataLoader(ToyDataset(n=64, seed=1), batch_size=16)
    trainer = Trainer(max_steps=8, precision="fp32", val_check_interval=4,
                      default_root_dir=str(tmp_path))
    trainer.fit(model, train_dataloaders=train, val_dataloaders=val)
    assert "val_loss" in trainer._metrics


def test_step_profiler_callback(tmp_path):
    from fengshen_amd.utils.profiling import StepProfiler
    args = _make_args()
    torch.manual_seed(0)
    model = ToyModule(args)
    loader = DataLoader(ToyDataset(n=128), batch_size=16)
    prof = StepProfiler(start_step=2, num_steps=1, out_dir=str(tmp_path / "p"))
    trainer = Trainer(max_steps=5, precision="fp32", callbacks=[prof],
                      default_root_dir=str(tmp_path))
    trainer.fit(model, train_dataloaders=loader)
    assert (tmp_path / "p" / "step_profile.txt").exists()


def test_roctx_marker_callback(tmp_path):
    from fengshen_amd.utils.profiling import RocTXMarker
    args = _make_args()
    torch.manual_seed(0)
    model = ToyModule(args)
    loader = DataLoader(ToyDataset(n=64), batch_size=16)
    marker = RocTXMarker()
    trainer = Trainer(max_steps=3, precision="fp32", callbacks=[marker],
                      default_root_dir=str(tmp_path))
    trainer.fit(model, train_dataloaders=loader)
    assert not marker._open  # every pushed range was popped


def test_predict_loop(tmp_path):
    class PredModule(ToyModule):
        def predict_step(self, batch, batch_idx):
            return self.net(batch["x"]).squeeze(-1)

    args = _make_args()
    torch.manual_seed(0)
    model = PredModule(args)
    loader = DataLoader(ToyDataset(n=32), batch_size=8)
    trainer = Trainer(precision="fp32", default_root_dir=str(tmp_path))
    outs = trainer.predict(model, loader)
    assert len(outs) == 4
    assert all(o.shape == (8,) for o in outs)


# ---------------------------------------------------------------------------
# experiment loggers (ref finetune_ziya_llama.py:218 WandbLogger, PL TB)
# ---------------------------------------------------------------------------
def test_tensorboard_logger_event_file(tmp_path):
    """The native writer must produce valid TFRecord framing (CRC-checked)
    and decodable simple_value summaries."""
    import struct
    from fengshen_amd.trainer.loggers import (
        TensorBoardLogger, _masked_crc)
    lg = TensorBoardLogger(str(tmp_path), name="exp", version="0")
    lg.log_metrics({"train_loss": 1.5, "lr": 0.001}, step=7)
    lg.log_metrics({"train_loss": 1.25}, step=8)
    lg.finalize()
    files = [f for f in os.listdir(tmp_path / "exp" / "0")
             if f.startswith("events.out.tfevents")]
    assert len(files) == 1
    raw = (tmp_path / "exp" / "0" / files[0]).read_bytes()
    # walk the TFRecord stream verifying CRCs
    off, events = 0, []
    while off < len(raw):
        (ln,) = struct.unpack_from("<Q", raw, off)
        (hcrc,) = struct.unpack_from("<I", raw, off + 8)
        assert hcrc == _masked_crc(raw[off:off + 8])
        payload = raw[off + 12:off + 12 + ln]
        (pcrc,) = struct.unpack_from("<I", raw, off + 12 + ln)
        assert pcrc == _masked_crc(payload)
        events.append(payload)
        off += 12 + ln + 4
    assert len(events) == 4  # file_version + 3 scalars
    # the 2nd event carries tag "train_loss" and float 1.5
    assert b"train_loss" in events[1]
    assert struct.pack("<f", 1.5) in events[1]


def test_wandb_logger_offline(tmp_path):
    import json
    from fengshen_amd.trainer.loggers import WandbLogger
    lg = WandbLogger(project="p", name="r1", save_dir=str(tmp_path),
                     config={"lr": 0.1})
    lg.log_metrics({"loss": 2.0}, step=1)
    lg.log_metrics({"loss": 1.0}, step=2)
    lg.finalize()
    hist = [json.loads(x) for x in
            (tmp_path / "r1" / "history.jsonl").read_text().splitlines()]
    assert hist[0]["_step"] == 1 and hist[0]["loss"] == 2.0
    cfg = json.loads((tmp_path / "r1" / "config.json").read_text())
    assert cfg["lr"] == 0.1


def test_trainer_with_loggers(tmp_path):
    """Trainer forwards flushed metrics to attached loggers."""
    from fengshen_amd.trainer.loggers import Logger

    class Capture(Logger):
        def __init__(self):
            self.rows = []
            self.hparams = None

        def log_hyperparams(self, params):
            self.hparams = params

        def log_metrics(self, metrics, step):
            self.rows.append((step, dict(metrics)))

    cap = Capture()
    model = ToyModule(_make_args())
    tr = Trainer(max_steps=3, precision="fp32",
                 default_root_dir=str(tmp_path),
                 log_every_n_steps=1, logger=cap)
    tr.fit(model, train_dataloaders=DataLoader(ToyDataset(), batch_size=16))
    assert len(cap.rows) >= 2
    assert any("train_loss" in m for _s, m in cap.rows)
