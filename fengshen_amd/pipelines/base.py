"""Shared pipeline plumbing.

Behavioral parity: reference pipelines/text_classification.py:134-231 —
ctor(model=path-or-config, args), add_pipeline_specific_args nests
DataModule/Trainer/checkpoint/module args, train() spins a Trainer,
__call__ preprocess->forward->postprocess.
"""
from __future__ import annotations

import argparse

import torch

from fengshen_amd.data.universal_datamodule import UniversalDataModule
from fengshen_amd.models.model_utils import add_module_args, configure_optimizers
from fengshen_amd.trainer.module import FengshenModule
from fengshen_amd.trainer.trainer import Trainer
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


def add_common_pipeline_args(parser: argparse.ArgumentParser):
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    return parser


class _TaskModule(FengshenModule):
    """LightningModule wrapper around an HF-style model
    (ref text_classification.py:38-91)."""

    def __init__(self, args, model):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = model

    def forward(self, **batch):
        return self.model(**batch)

    def training_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def validation_step(self, batch, batch_idx):
        out = self.model(**batch)
        self.log("val_loss", out.loss, sync_dist=True)
        return out.loss

    def configure_optimizers(self):
        return configure_optimizers(self)


class BasePipeline:
    """Shared pipeline base: ctor(model=hub-id-or-instance, tokenizer),
    `__call__(inputs)` for inference, `.train(datasets)` spins a Trainer
    (reference pipelines/*.Pipeline interface)."""

    task_name = "base"

    def __init__(self, args=None, model=None, tokenizer=None):
        self.args = args
        self.tokenizer = tokenizer
        self.model = model
        self.device = torch.device(
            "cuda" if torch.cuda.is_available() else "cpu")

    @classmethod
    def add_pipeline_specific_args(cls, parser: argparse.ArgumentParser):
        return add_common_pipeline_args(parser)

    # -- training ----------------------------------------------------------
    def train(self, datasets, collate_fn=None):
        module = _TaskModule(self.args, self.model)
        dm = UniversalDataModule(
            tokenizer=self.tokenizer, collate_fn=collate_fn or self.collator(),
            args=self.args, datasets=datasets)
        ckpt_cb = UniversalCheckpoint(self.args)
        trainer = Trainer.from_argparse_args(self.args, callbacks=[ckpt_cb])
        trainer.fit(module, datamodule=dm)
        self.model = module.model
        return trainer

    def collator(self):
        raise NotImplementedError

    def __call__(self, inputs):
        raise NotImplementedError
