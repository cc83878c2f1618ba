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
        """samples: [{'texta': ..., 'choices': [...], 'question': ...}]"""
        single = isinstance(samples, dict)
        if single:
            samples = [samples]
        vocab = self.tokenizer.get_vocab()
        results = []
        for s in samples:
            ids = [self.tokenizer.cls_token_id]
            opt_pos = []
            for choice in s["choices"]:
                opt_pos.append(len(ids))
                ids += [vocab.get(c, 4) for c in choice]
                ids.append(self.tokenizer.sep_token_id)
            ids += [vocab.get(c, 4) for c in s.get("question", "")]
            ids += [vocab.get(c, 4) for c in s["texta"]]
            ids.append(self.tokenizer.sep_token_id)
            dev = next(self.model.parameters()).device
            pred = self.model.predict(
                torch.tensor([ids], dtype=torch.long, device=dev), None, None,
                torch.tensor([opt_pos], dtype=torch.long, device=dev))
            results.append({"label": int(pred[0]),
                            "choice": s["choices"][int(pred[0])]})
        return results[0] if single else results


Pipeline = MultipleChoicePipeline
