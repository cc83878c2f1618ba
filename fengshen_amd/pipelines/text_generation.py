"""Text-generation pipeline (causal-LM serving and finetune).

The reference serves generation ad hoc (examples/mt5_summary/
fastapi_mt5_summary.py:28-40 wraps a model in a bare FastAPI app;
examples/wenzhong_qa scripts call model.generate directly).  Here
generation is a first-class pipeline so `fengshen-pipeline
text_generation` and the serving /generate endpoint work out of the
box.  On an MI355X with a bf16 fengshen LLaMA it routes through the
hipGraph GraphedDecoder (bf16 GEMV + fused decode-attention kernels,
6.1 ms/token for the 13B); for every other model/device it falls back
to HF generate() with the same sampling knobs.
"""
from __future__ import annotations

from typing import List, Optional, Union

import torch

from fengshen_amd.pipelines.base import BasePipeline


class _LMCollator:
    """Plain causal-LM collator: tokenize `text`, labels = input_ids
    (pad positions masked to -100)."""

    def __init__(self, tokenizer, max_length: int = 512,
                 text_key: str = "text"):
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.text_key = text_key

    def __call__(self, samples):
        texts = [s[self.text_key] for s in samples]
        batch = self.tokenizer(texts, padding=True, truncation=True,
                               max_length=self.max_length,
                               return_tensors="pt")
        labels = batch["input_ids"].clone()
        labels[batch["attention_mask"] == 0] = -100
        batch["labels"] = labels
        return dict(batch)


class Pipeline(BasePipeline):
    task_name = "text_generation"

    def __init__(self, args=None, model=None, tokenizer=None,
                 use_graph: bool = True, max_len: int = 1024,
                 max_new_tokens: int = 256, **kwargs):
        if isinstance(model, str):
            from transformers import AutoModelForCausalLM, AutoTokenizer
            if tokenizer is None:
                tokenizer = AutoTokenizer.from_pretrained(model)
            model = AutoModelForCausalLM.from_pretrained(model)
        super().__init__(args, model, tokenizer)
        if self.model is not None:
            self.model = self.model.to(self.device).eval()
        self.use_graph = use_graph
        self.max_len = max_len
        self.max_new_tokens = max_new_tokens
        self._decoder = None
        self._decoder_key = None

    def collator(self):
        return _LMCollator(
            self.tokenizer,
            max_length=getattr(self.args, "max_length", 512),
            text_key=getattr(self.args, "text_name", "text"))

    # ------------------------------------------------------------------
    def _graph_eligible(self) -> bool:
        if not (self.use_graph and torch.cuda.is_available()):
            return False
        try:
            from fengshen_amd.models.llama.modeling_llama import (
                LlamaForCausalLM)
        except Exception:
            return False
        return (isinstance(self.model, LlamaForCausalLM)
                and next(self.model.parameters()).dtype == torch.bfloat16)

    def _decoder_for(self, do_sample: bool, top_k: int, top_p: float,
                     temperature: float):
        from fengshen_amd.serving.graphed_decode import GraphedDecoder
        # top_k changes the captured graph shape; temperature/top_p are
        # device scalars and can be retuned on the live decoder.
        key = (do_sample, top_k)
        if self._decoder is None or self._decoder_key != key:
            self._decoder = GraphedDecoder(
                self.model, batch=1, max_len=self.max_len,
                max_new_tokens=self.max_new_tokens, do_sample=do_sample,
                top_k=top_k, top_p=top_p, temperature=temperature)
            self._decoder_key = key
        else:
            self._decoder.top_p.fill_(float(top_p))
            self._decoder.temperature.fill_(float(temperature))
        return self._decoder

    @torch.no_grad()
    def generate(self, text: str, max_new_tokens: Optional[int] = None,
                 do_sample: bool = False, top_k: int = 0,
                 top_p: float = 1.0, temperature: float = 1.0) -> str:
        n = min(max_new_tokens or self.max_new_tokens, self.max_new_tokens)
        ids = self.tokenizer(text, return_tensors="pt")["input_ids"]
        ids = ids.to(self.device)
        eos = getattr(self.tokenizer, "eos_token_id", None)
        if self._graph_eligible():
            dec = self._decoder_for(do_sample, top_k, top_p, temperature)
            out = dec.generate(ids, max_new_tokens=n, eos_token_id=eos)
        else:
            kw = {}
            if do_sample:
                kw = dict(do_sample=True, top_k=top_k, top_p=top_p,
                          temperature=temperature)
            else:
                kw = dict(do_sample=False)
            if eos is not None:
                kw["pad_token_id"] = eos
            out = self.model.generate(ids, max_new_tokens=n, **kw)
        return self.tokenizer.decode(out[0, ids.shape[1]:],
                                     skip_special_tokens=True)

    def __call__(self, texts: Union[str, List[str]], **kwargs):
        if isinstance(texts, str):
            return self.generate(texts, **kwargs)
        return [self.generate(t, **kwargs) for t in texts]
