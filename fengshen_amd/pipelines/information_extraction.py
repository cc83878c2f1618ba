"""Information extraction pipeline (UBERT wrapper) —
reference pipelines/information_extraction.py:19-60."""
from __future__ import annotations

import torch

from fengshen_amd.pipelines.base import BasePipeline


class InformationExtractionPipeline(BasePipeline):
    """UBERT span extraction (entity types as prompts)."""

    task_name = "information_extraction"

    def __init__(self, args=None, model=None, tokenizer=None, config=None):
        super().__init__(args, model, tokenizer)
        if self.model is None:
            from fengshen_amd.models.ubert.modeling_ubert import UbertModel
            if isinstance(model, str):
                self.model = UbertModel.from_pretrained(model)
            else:
                self.model = UbertModel(config)
        self.model.eval()

    def _encode(self, text: str, labels):
        rows = []
        for lab in labels:
            prompt = f"[CLS]{lab}[SEP]{text}[SEP]"
            ids = [self.tokenizer.get_vocab().get(c, 4) for c in prompt]
            rows.append(ids)
        L = max(len(r) for r in rows)
        pad = self.tokenizer.pad_token_id or 0
        return torch.tensor([r + [pad] * (L - len(r)) for r in rows],
                            dtype=torch.long)

    @torch.no_grad()
    def __call__(self, texts, entity_types=None):
        entity_types = entity_types or ["人名", "地名", "机构"]
        single = isinstance(texts, str)
        if single:
            texts = [texts]
        results = []
        for text in texts:
            ids = self._encode(text, entity_types).unsqueeze(0)
            dev = next(self.model.parameters()).device
            spans = self.model.extract(ids.to(dev))
            per_text = {}
            for li, lab in enumerate(entity_types):
                off = len(lab) + 10  # prompt chars before text
                per_text[lab] = [
                    {"span": (s, e), "score": sc}
                    for s, e, sc in spans[0][li]]
            results.append(per_text)
        return results[0] if single else results

    def predict(self, predict_data, batch_size: int = 8,
                max_length: int = 128, threshold: float = 0.5):
        """Reference UbertPipelines.predict (modeling_ubert.py:714-740):
        task-typed items ({task_type, subtask_type, text, choices}) are
        bucketed by choice count and decoded into entity structures."""
        from fengshen_amd.models.ubert.modeling_ubert import UbertExtractor
        extractor = UbertExtractor(self.model, self.tokenizer,
                                   max_length=max_length,
                                   threshold=threshold)
        result = []
        start = 0
        while start < len(predict_data):
            batch = predict_data[start:start + batch_size]
            start += batch_size
            buckets = {}
            for item in batch:
                buckets.setdefault(len(item["choices"]), []).append(item)
            for _n, items in buckets.items():
                result.extend(extractor.extract(items))
        return result


Pipeline = InformationExtractionPipeline
