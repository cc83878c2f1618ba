"""Topic-classification prompt pipeline (TCBert) —
reference pipelines/tcbert.py."""
from __future__ import annotations

import torch

from fengshen_amd.pipelines.base import BasePipeline


class TCBertPipeline(BasePipeline):
    """Prompt-based topic classification (label read at [MASK])."""

    task_name = "tcbert"

    def __init__(self, args=None, model=None, tokenizer=None, config=None,
                 labels=None, prompt: str = "这是一条关于{}的新闻："):
        super().__init__(args, model, tokenizer)
        self.labels = labels or []
        self.prompt = prompt
        if self.model is None:
            from fengshen_amd.models.tcbert.modeling_tcbert import TCBertModel
            if isinstance(model, str):
                self.model = TCBertModel.from_pretrained(model)
            else:
                self.model = TCBertModel(config)
        self.model.eval()
        vocab = self.tokenizer.get_vocab()
        n_mask = max(len(l) for l in self.labels) if self.labels else 2
        self.n_mask = n_mask
        rows = []
        for lab in self.labels:
            toks = [vocab.get(c, self.tokenizer.unk_token_id)
                    for c in lab[:n_mask]]
            toks += [self.tokenizer.pad_token_id] * (n_mask - len(toks))
            rows.append(toks)
        self.label_token_ids = torch.tensor(rows, dtype=torch.long)

    @torch.no_grad()
    def __call__(self, texts):
        single = isinstance(texts, str)
        if single:
            texts = [texts]
        vocab = self.tokenizer.get_vocab()
        batch_ids, mask_pos = [], []
        for t in texts:
            cut = self.prompt.index("{}")  # masks inserted at the "{}" site
            ids = [self.tokenizer.cls_token_id]
            ids += [vocab.get(c, 4) for c in self.prompt[:cut]]
            mp = list(range(len(ids), len(ids) + self.n_mask))
            ids += [self.tokenizer.mask_token_id] * self.n_mask
            ids += [vocab.get(c, 4) for c in self.prompt[cut + 2:]]
            ids += [vocab.get(c, 4) for c in t]
            ids.append(self.tokenizer.sep_token_id)
            batch_ids.append(ids)
            mask_pos.append(mp)
        L = max(len(x) for x in batch_ids)
        pad = self.tokenizer.pad_token_id
        dev = next(self.model.parameters()).device
        input_ids = torch.tensor(
            [x + [pad] * (L - len(x)) for x in batch_ids], device=dev)
        out = self.model(input_ids,
                         mask_positions=torch.tensor(mask_pos, device=dev),
                         label_token_ids=self.label_token_ids.to(dev))
        preds = out.label_logits.argmax(-1)
        res = [{"label": int(p), "label_name": self.labels[int(p)]}
               for p in preds]
        return res[0] if single else res


Pipeline = TCBertPipeline
