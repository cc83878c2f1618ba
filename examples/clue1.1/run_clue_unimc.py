"""CLUE-1.1 UniMC solution runner
(reference examples/clue1.1/solution/clue_unimc.py + run_clue_unimc.sh):
train UniMC on converted unified data, predict the eval split, and emit
per-id labels for predict2submit."""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))

import argparse
import json

import torch

from fengshen_amd import FengshenModule, Trainer, UniversalDataModule
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    bert_tiny_config)
from fengshen_amd.models.model_utils import (
    add_module_args, configure_optimizers)
from fengshen_amd.models.unimc.modeling_unimc import (
    UniMCEncoder, UniMCModel, unimc_collate)
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint


class CLUECollator:
    def __init__(self, tokenizer, max_len=128):
        self.encoder = UniMCEncoder(tokenizer, yes_token=5, no_token=6,
                                    max_length=max_len)

    def __call__(self, samples):
        batch = unimc_collate([self.encoder.encode(s) for s in samples])
        batch.pop("mlmlabels_mask", None)
        return batch


class CLUEUniMC(FengshenModule):
    def __init__(self, args):
        super().__init__()
        self.save_hyperparameters(args)
        self.model = UniMCModel(bert_tiny_config(), yes_token_id=5)

    def training_step(self, batch, batch_idx):
        batch.pop("option_positions", None)
        out = self.model(**batch)
        self.log("train_loss", out.loss)
        return out.loss

    def predict_step(self, batch, batch_idx):
        opts = batch.pop("option_positions")
        out = self.model(**{k: v for k, v in batch.items()
                            if k not in ("mlmlabels", "clslabels")})
        picked = out.cls_logits.argmax(-1)  # anchor position
        # map anchor position back to option index
        return (opts == picked.unsqueeze(1)).float().argmax(-1)

    def configure_optimizers(self):
        return configure_optimizers(self)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--data", default=None,
                        help="unified-format jsonl from clue2unidata.py")
    parser.add_argument("--predict_output", default="clue_predict.jsonl")
    add_module_args(parser)
    UniversalDataModule.add_data_specific_args(parser)
    Trainer.add_argparse_args(parser)
    UniversalCheckpoint.add_argparse_args(parser)
    args = parser.parse_args()

    if args.data:
        with open(args.data, encoding="utf8") as f:
            data = [json.loads(x) for x in f if x.strip()]
    else:  # synthetic smoke
        data = [{"texta": "球队赢了比赛" if i % 2 == 0 else "新芯片发布",
                 "textb": "", "question": "下面新闻属于哪一个类别？",
                 "choice": ["体育", "科技"], "label": i % 2, "id": i}
                for i in range(32)]

    from fengshen_amd.tokenizer import SimpleCharTokenizer
    tokenizer = SimpleCharTokenizer()
    dm = UniversalDataModule(tokenizer, CLUECollator(tokenizer), args,
                             datasets={"train": data, "test": data})
    trainer = Trainer.from_argparse_args(
        args, callbacks=[UniversalCheckpoint(args)])
    task = CLUEUniMC(args)
    trainer.fit(task, datamodule=dm)
    preds = trainer.predict(task, datamodule=dm)
    with open(args.predict_output, "w", encoding="utf8") as f:
        i = 0
        for batch_pred in preds:
            for p in batch_pred.tolist():
                f.write(json.dumps(
                    {"id": data[i]["id"],
                     "label_index": int(p),
                     "label": data[i]["choice"][int(p)]},
                    ensure_ascii=False) + "\n")
                i += 1
    print(f"predictions -> {args.predict_output}")


if __name__ == "__main__":
    main()
