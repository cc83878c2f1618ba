"""Zero-shot multiple choice (UniMC) pipeline —
reference pipelines/multiplechoice.py."""
from __future__ import annotations

import torch

from fengshen_amd.pipelines.base import BasePipeline


class MultipleChoicePipeline(BasePipeline):
    """UniMC zero-shot label-as-option choice."""

    task_name = "multiplechoice"

    def __init__(self, args=None, model=None, tokenizer=None, config=None,
                 yes_token_id: int = 1):
        super().__init__(args, model, tokenizer)
        if self.model is None:
            from fengshen_amd.models.unimc.modeling_unimc import UniMCModel
            if isinstance(model, str):
                self.model = UniMCModel.from_pretrained(model)
            else:
                self.model = UniMCModel(config, yes_token_id=yes_token_id)
        self.model.eval()

    @torch.no_grad()
    def __call__(self, samples):
        """samples: [{'texta': ..., 'choices': [...], 'question': ...}]

        Uses the reference-parity UniMCEncoder (option-isolation attention
        mask, per-option restarted position ids, yes/no anchors)."""
        from fengshen_amd.models.unimc.modeling_unimc import (
            UniMCEncoder, unimc_collate)
        single = isinstance(samples, dict)
        if single:
            samples = [samples]
        enc = UniMCEncoder(self.tokenizer,
                           yes_token=self.model.yes_token_id,
                           no_token=getattr(self.model, "no_token_id", 6))
        dev = next(self.model.parameters()).device
        batch = unimc_collate([enc.encode(
            {"texta": s["texta"], "choice": s["choices"],
             "question": s.get("question", "")}) for s in samples])
        opts = batch["option_positions"].to(dev)
        out = self.model(
            batch["input_ids"].to(dev),
            attention_mask=batch["attention_mask"].to(dev),
            token_type_ids=batch["token_type_ids"].to(dev),
            position_ids=batch["position_ids"].to(dev),
            clslabels_mask=batch["clslabels_mask"].to(dev))
        picked = out.cls_logits.argmax(-1)  # anchor position
        pred = (opts == picked.unsqueeze(1)).float().argmax(-1)
        results = [{"label": int(p), "choice": s["choices"][int(p)]}
                   for p, s in zip(pred, samples)]
        return results[0] if single else results


Pipeline = MultipleChoicePipeline
