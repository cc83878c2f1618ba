"""UniversalDataModule — the one data module every training app uses.

Behavioral parity: reference data/universal_datamodule/universal_datamodule.py
(:20-189): dataset dict {'train','validation','test'}, custom megatron
samplers when sampler_type is set (exact resume via consumed_samples),
DistributedSampler for val/test, DP-rank-aware when TP is active
(use_mpu path :84-85), datasets loadable from HF `datasets` json files
or passed in directly.
"""
from __future__ import annotations

import argparse
from typing import Optional

from torch.utils.data import DataLoader, DistributedSampler

from fengshen_amd.data.universal_sampler import (
    PretrainingRandomSampler,
    PretrainingSampler,
)


def get_consume_samples(data_model: "UniversalDataModule") -> int:
    """reference universal_datamodule.py:8-17."""
    trainer = data_model.trainer
    if trainer is None:
        return 0
    return trainer.consumed_samples


class UniversalDataModule:
    @staticmethod
    def add_data_specific_args(parent_args: argparse.ArgumentParser):
        parser = parent_args.add_argument_group("Universal DataModule")
        parser.add_argument("--num_workers", default=2, type=int)
        parser.add_argument("--dataloader_workers", default=2, type=int)
        parser.add_argument("--train_batchsize", default=16, type=int)
        parser.add_argument("--val_batchsize", default=16, type=int)
        parser.add_argument("--test_batchsize", default=16, type=int)
        parser.add_argument("--datasets_name", type=str, default=None)
        parser.add_argument("--train_datasets_field", type=str, default="train")
        parser.add_argument("--val_datasets_field", type=str, default="validation")
        parser.add_argument("--test_datasets_field", type=str, default="test")
        parser.add_argument("--train_file", type=str, default=None)
        parser.add_argument("--val_file", type=str, default=None)
        parser.add_argument("--test_file", type=str, default=None)
        parser.add_argument("--raw_file_type", type=str, default="json")
        parser.add_argument("--sampler_type", type=str,
                            choices=["single", "random"], default="random")
        parser.add_argument("--use_mpu", action="store_true", default=False)
        return parent_args

    def __init__(self, tokenizer, collate_fn, args, datasets=None, **kwargs):
        self.tokenizer = tokenizer
        self.collate_fn = collate_fn
        self.hparams = args
        self.trainer = None
        self._datasets = datasets

    # ------------------------------------------------------------------
    def setup(self, stage: Optional[str] = None):
        if self._datasets is not None:
            return
        args = self.hparams
        if getattr(args, "datasets_name", None) is not None:
            try:
                from datasets import load_dataset
                self._datasets = load_dataset(args.datasets_name)
                return
            except Exception as e:
                raise RuntimeError(
                    f"could not load dataset {args.datasets_name}: {e}") from e
        # raw json/csv files via HF datasets (reference :66-71)
        data_files = {}
        if getattr(args, "train_file", None):
            data_files["train"] = args.train_file
        if getattr(args, "val_file", None):
            data_files["validation"] = args.val_file
        if getattr(args, "test_file", None):
            data_files["test"] = args.test_file
        if data_files:
            from datasets import load_dataset
            self._datasets = load_dataset(
                getattr(args, "raw_file_type", "json"), data_files=data_files)
            return
        raise ValueError("no datasets provided: pass datasets=, or set "
                         "--datasets_name / --train_file")

    @property
    def datasets(self):
        return self._datasets

    @datasets.setter
    def datasets(self, v):
        self._datasets = v

    # ------------------------------------------------------------------
    def _dp_rank_size(self):
        from fengshen_amd.parallel import groups as pg
        if getattr(self.hparams, "use_mpu", False) or \
                pg.get_tensor_model_parallel_world_size() > 1:
            return pg.get_data_parallel_rank(), pg.get_data_parallel_world_size()
        if self.trainer is not None:
            return (self.trainer.strategy.data_parallel_rank,
                    self.trainer.strategy.data_parallel_world_size)
        return 0, 1

    def train_dataloader(self):
        ds = self._datasets[self.hparams.train_datasets_field]
        rank, world = self._dp_rank_size()
        consumed = get_consume_samples(self)
        sampler_type = getattr(self.hparams, "sampler_type", "random")
        if sampler_type == "random":
            batch_sampler = PretrainingRandomSampler(
                total_samples=len(ds), consumed_samples=consumed,
                micro_batch_size=self.hparams.train_batchsize,
                data_parallel_rank=rank, data_parallel_size=world,
                epoch=self.trainer.current_epoch if self.trainer else 0,
                seed=getattr(self.hparams, "seed", 1234))
        else:
            batch_sampler = PretrainingSampler(
                total_samples=len(ds), consumed_samples=consumed,
                micro_batch_size=self.hparams.train_batchsize,
                data_parallel_rank=rank, data_parallel_size=world)
        return DataLoader(
            ds, batch_sampler=batch_sampler,
            num_workers=getattr(self.hparams, "num_workers", 2),
            collate_fn=self.collate_fn, pin_memory=True)

    def _eval_dataloader(self, field: str, batch_size: int):
        if self._datasets is None or field not in self._datasets:
            return None
        ds = self._datasets[field]
        rank, world = self._dp_rank_size()
        sampler = DistributedSampler(
            ds, num_replicas=world, rank=rank, shuffle=False) if world > 1 else None
        return DataLoader(
            ds, batch_size=batch_size, sampler=sampler, shuffle=False,
            num_workers=getattr(self.hparams, "num_workers", 2),
            collate_fn=self.collate_fn, pin_memory=True)

    def val_dataloader(self):
        return self._eval_dataloader(
            self.hparams.val_datasets_field,
            getattr(self.hparams, "val_batchsize", 16))

    def test_dataloader(self):
        return self._eval_dataloader(
            self.hparams.test_datasets_field,
            getattr(self.hparams, "test_batchsize", 16))
