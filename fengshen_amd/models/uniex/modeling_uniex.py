"""UniEX: unified information extraction via triaffine span-type scoring.

Behavioral parity with reference models/uniex/modeling_uniex.py:
- MLP start/end/cls projections + Triaffine scorer (:858-882): span logits
  [b, x, y, n_label] = einsum(start, W, end) x cls, where cls vectors are
  the hidden states at the in-sequence label-prompt token positions
  (label_token_idx);
- training path (:924-970): label 0 is the "index" head scored over the
  FULL sequence, labels 1.. are type heads scored over text tokens only
  (gathered by text_token_idx, span_gather :902-922); loss =
  1e5 * (BCE(index) + BCE(types)) with additive span_labels_mask;
- full extract (:972-984): sigmoid triaffine over every (start, end, type)
  for gathered text tokens;
- fast extract (:986-1025): stage 1 scores spans with the index head only,
  stage 2 types just the surviving spans.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
from transformers.utils import ModelOutput

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UniEXConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
    MegatronBertPreTrainedModel,
)


class MLPLayer(nn.Module):
    """Linear + GELU (ref :875-882)."""

    def __init__(self, input_size: int, output_size: int):
        super().__init__()
        self.mlp = nn.Sequential(nn.Linear(input_size, output_size),
                                 nn.GELU())

    def forward(self, hidden_state):
        return self.mlp(hidden_state)


class Triaffine(nn.Module):
    """span_logits[b,x,y,z] = (start_x W end_y) . cls_z (ref :858-872)."""

    def __init__(self, triaffine_hidden_size: int):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(
            triaffine_hidden_size, triaffine_hidden_size,
            triaffine_hidden_size))
        nn.init.normal_(self.weight, mean=0, std=0.1)

    def forward(self, start_logits, end_logits, cls_logits):
        span = torch.einsum("bxi,ioj,byj->bxyo",
                            start_logits.float(), self.weight.float(),
                            end_logits.float())
        return torch.einsum("bxyo,bzo->bxyz", span, cls_logits.float())


@dataclass
class UniEXOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    span_logits: Optional[torch.Tensor] = None
    span_labels: Optional[torch.Tensor] = None


def span_gather(span_labels: torch.Tensor,
                text_token_idx: torch.Tensor) -> torch.Tensor:
    """Subset a [b, s, s, L] span grid to the text tokens only ->
    [b, t, t, L] (ref :902-922)."""
    batch_size, seq_len, _, num_labels = span_labels.shape
    _, text_len = text_token_idx.shape
    e = torch.arange(seq_len, device=span_labels.device) * seq_len
    e = e.unsqueeze(0).unsqueeze(-1).repeat(batch_size, 1, text_len)
    e = e.gather(1, text_token_idx.unsqueeze(-1).repeat(1, 1, text_len))
    idx = text_token_idx.unsqueeze(1).repeat(1, text_len, 1) + e
    idx = idx.reshape(-1, text_len * text_len)
    flat = span_labels.reshape(-1, seq_len * seq_len, num_labels)
    out = flat.gather(1, idx.unsqueeze(-1).repeat(1, 1, num_labels))
    return out.reshape(-1, text_len, text_len, num_labels)


class UniEXModel(MegatronBertPreTrainedModel):
    """Ref UniEXBertModel (:885-1025) on the MI355X-native BERT stack."""

    config_class = UniEXConfig

    def __init__(self, config: UniEXConfig,
                 triaffine_hidden_size: int = 128):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        ths = getattr(config, "triaffine_hidden_size",
                      triaffine_hidden_size)
        self.mlp_start = MLPLayer(config.hidden_size, ths)
        self.mlp_end = MLPLayer(config.hidden_size, ths)
        self.mlp_cls = MLPLayer(config.hidden_size, ths)
        self.triaffine = Triaffine(ths)
        self.loss_sigmoid = nn.BCEWithLogitsLoss()
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                position_ids=None, span_labels=None, span_labels_mask=None,
                label_token_idx=None, text_token_idx=None,
                fast_ex_mode=False, threshold=0.5, **_kw):
        h = self.bert(input_ids, attention_mask, token_type_ids,
                      position_ids).last_hidden_state
        batch_size, seq_len, hidden_size = h.shape

        if span_labels_mask is not None:
            # ---- training (ref :944-969) --------------------------------
            start_logits = self.mlp_start(h)
            end_logits = self.mlp_end(h)
            cls_h = h.gather(
                1, label_token_idx.unsqueeze(-1).repeat(1, 1, hidden_size))
            cls_logits = self.mlp_cls(cls_h)

            # index head (label 0) over the full sequence
            idx_logits = self.triaffine(start_logits, end_logits,
                                        cls_logits[:, [0], :])
            idx_logits = idx_logits + span_labels_mask[:, :, :, [0]]
            index_loss = self.loss_sigmoid(idx_logits,
                                           span_labels[:, :, :, [0]].float())

            # type heads (labels 1..) over text tokens only
            ths = start_logits.shape[-1]
            st_t = start_logits.gather(
                1, text_token_idx.unsqueeze(-1).repeat(1, 1, ths))
            en_t = end_logits.gather(
                1, text_token_idx.unsqueeze(-1).repeat(1, 1, ths))
            span_logits = self.triaffine(st_t, en_t, cls_logits[:, 1:, :])
            span_labels_t = span_gather(span_labels[:, :, :, 1:].float(),
                                        text_token_idx)
            span_mask_t = span_gather(span_labels_mask[:, :, :, 1:],
                                      text_token_idx)
            span_logits = span_logits + span_mask_t
            span_loss = self.loss_sigmoid(span_logits, span_labels_t)
            all_loss = 100000 * span_loss + 100000 * index_loss
            return UniEXOutput(loss=all_loss, span_logits=span_logits,
                               span_labels=span_labels_t)

        if not fast_ex_mode:
            # ---- full extract (ref :972-984) ----------------------------
            text_h = h.gather(
                1, text_token_idx.unsqueeze(-1).repeat(1, 1, hidden_size))
            start_logits = self.mlp_start(text_h)
            end_logits = self.mlp_end(text_h)
            cls_h = h.gather(
                1, label_token_idx.unsqueeze(-1).repeat(1, 1, hidden_size))
            cls_logits = self.mlp_cls(cls_h)
            span_logits = torch.sigmoid(
                self.triaffine(start_logits, end_logits, cls_logits))
            return UniEXOutput(span_logits=span_logits)

        # ---- fast extract (ref :986-1025): index spans then type --------
        text_h = h.gather(
            1, text_token_idx.unsqueeze(-1).repeat(1, 1, hidden_size))
        start_logits = self.mlp_start(text_h)
        end_logits = self.mlp_end(text_h)
        cls_h = h.gather(
            1, label_token_idx.unsqueeze(-1).repeat(1, 1, hidden_size))
        cls_logits = self.mlp_cls(cls_h)
        index_logits = cls_logits[:, :1, :]
        type_logits = cls_logits[:, 1:, :]

        span_index = torch.sigmoid(
            self.triaffine(start_logits, end_logits,
                           index_logits)).squeeze(-1)  # [b, t, t]
        results = []
        for bi in range(batch_size):
            hits = (span_index[bi].triu() > threshold) \
                .nonzero(as_tuple=False)
            spans = []
            if hits.numel():
                st = start_logits[bi:bi + 1, hits[:, 0]]
                en = end_logits[bi:bi + 1, hits[:, 1]]
                # type each surviving span: score[z] for start/end pair
                tl = self.triaffine(st, en, type_logits[bi:bi + 1])
                # diagonal (start_i, end_i) pairs only
                n = hits.shape[0]
                diag = tl[0, torch.arange(n), torch.arange(n)]  # [n, T]
                probs = torch.sigmoid(diag)
                types = probs.argmax(-1)
                for j in range(n):
                    spans.append({
                        "span": (int(hits[j, 0]), int(hits[j, 1])),
                        "type": int(types[j]),
                        "score": float(span_index[bi, hits[j, 0],
                                                  hits[j, 1]]),
                        "type_score": float(probs[j, types[j]]),
                    })
            results.append(spans)
        return results


class UniEXExtractor:
    """Convenience extraction wrapper used by the IE pipeline: builds
    label-prompt + text sequences and decodes span/type structures."""

    def __init__(self, model: UniEXModel, tokenizer, max_length: int = 128):
        self.model = model
        self.tokenizer = tokenizer
        self.max_length = max_length

    @torch.no_grad()
    def extract(self, texts: List[str], entity_types: List[str],
                threshold: float = 0.5, fast: bool = True):
        tk = self.tokenizer
        dev = next(self.model.parameters()).device
        batch_ids, lab_idx, txt_idx = [], [], []
        for text in texts:
            ids = [tk.cls_token_id]
            li = [0]
            for et in entity_types:
                li.append(len(ids))
                ids += tk.encode(et, add_special_tokens=False)
                ids.append(tk.sep_token_id)
            ti = list(range(len(ids), len(ids)
                            + len(tk.encode(text,
                                            add_special_tokens=False))))
            ids += tk.encode(text, add_special_tokens=False)
            ids.append(tk.sep_token_id)
            batch_ids.append(ids[:self.max_length])
            lab_idx.append(li)
            txt_idx.append([i for i in ti if i < self.max_length])
        smax = max(len(x) for x in batch_ids)
        tmax = max(len(x) for x in txt_idx)
        ids_t = torch.zeros(len(texts), smax, dtype=torch.long)
        for i, x in enumerate(batch_ids):
            ids_t[i, :len(x)] = torch.tensor(x)
        lab_t = torch.tensor(lab_idx, dtype=torch.long)
        txt_t = torch.zeros(len(texts), tmax, dtype=torch.long)
        for i, x in enumerate(txt_idx):
            txt_t[i, :len(x)] = torch.tensor(x)
        out = self.model(ids_t.to(dev), label_token_idx=lab_t.to(dev),
                         text_token_idx=txt_t.to(dev),
                         fast_ex_mode=fast, threshold=threshold)
        if fast:
            return out
        # full mode: decode [b, t, t, 1+T] grid; label 0 is the index head
        results = []
        span_logits = out.span_logits
        for bi in range(len(texts)):
            hits = (span_logits[bi, :, :, 0].triu()
                    > threshold).nonzero(as_tuple=False)
            spans = []
            for st, en in hits.tolist():
                tscores = span_logits[bi, st, en, 1:]
                t = int(tscores.argmax())
                spans.append({"span": (st, en), "type": t,
                              "score": float(span_logits[bi, st, en, 0]),
                              "type_score": float(tscores[t])})
            results.append(spans)
        return results


# ---------------------------------------------------------------------------
# Task metrics (ref modeling_uniex.py:44-245) and data-side encoding
# ---------------------------------------------------------------------------
def _entity_key(e):
    return (e.get("entity_type"), tuple(map(tuple, e.get("entity_index",
                                                         []))) if
            isinstance(e.get("entity_index"), list)
            else e.get("entity_index"))


def get_entity_f1(test_data, pred_data):
    """(f1, recall, precision) over unique (type, index) entities
    (ref :44-100); falls back to spo subjects/objects when entity_list
    is empty."""
    corr = y_true = y_pred = 0
    for t, p in zip(test_data, pred_data):
        def collect(item):
            out = []
            for e in item.get("entity_list", []):
                k = _entity_key(e)
                if k not in out:
                    out.append(k)
            if not out:
                for spo in item.get("spo_list", []):
                    for side in ("subject", "object"):
                        k = _entity_key(spo[side])
                        if k not in out:
                            out.append(k)
            return out
        tl, pl = collect(t), collect(p)
        y_true += len(tl)
        y_pred += len(pl)
        corr += sum(1 for e in pl if e in tl)
    precise = corr / y_pred if y_pred > 0 else 0
    recall = corr / y_true if y_true > 0 else 0
    f1 = 2 * precise * recall / (precise + recall) \
        if precise + recall > 0 else 0
    return f1, recall, precise


def get_rel_f1(test_data, pred_data):
    """(f1, recall, precision) over (predicate, subject, object) triples
    (ref :157-162 wrapper over the strict entity matcher)."""
    corr = y_true = y_pred = 0
    for t, p in zip(test_data, pred_data):
        def collect(item):
            out = []
            for spo in item.get("spo_list", []):
                k = (spo.get("predicate"),
                     _entity_key(spo.get("subject", {})),
                     _entity_key(spo.get("object", {})))
                if k not in out:
                    out.append(k)
            return out
        tl, pl = collect(t), collect(p)
        y_true += len(tl)
        y_pred += len(pl)
        corr += sum(1 for e in pl if e in tl)
    precise = corr / y_pred if y_pred > 0 else 0
    recall = corr / y_true if y_true > 0 else 0
    f1 = 2 * precise * recall / (precise + recall) \
        if precise + recall > 0 else 0
    return f1, recall, precise


class UniEXDataEncoder:
    """Entity-task sample encoding (ref UniEXDataEncode :246-798, entity
    subset): sequence = [CLS][unused-index][type_1]..[type_T][SEP] text;
    label_token_idx covers the index token + type tokens; span labels
    [s, s, 1+T] carry an index-head hit and a type-head hit per entity
    span; span_labels_mask opens the text block for the index head and
    text×text for type heads."""

    def __init__(self, tokenizer, max_length: int = 128):
        self.tokenizer = tokenizer
        self.max_length = max_length

    def encode(self, item: dict, entity_type_list: List[str]) -> dict:
        tk = self.tokenizer
        T = len(entity_type_list)
        ids = [tk.cls_token_id]
        label_idx = [len(ids) - 1]  # the [CLS] doubles as the index token
        for et in entity_type_list:
            label_idx.append(len(ids))
            ids += tk.encode(et, add_special_tokens=False)
        ids.append(tk.sep_token_id)
        text_start = len(ids)
        text_ids = tk.encode(item["text"], add_special_tokens=False)
        ids += text_ids
        ids = ids[:self.max_length]
        text_token_idx = list(range(text_start,
                                    min(len(ids), self.max_length)))
        tlen = len(text_token_idx)

        s = len(ids)
        span_labels = torch.zeros(s, s, 1 + T)
        span_mask = torch.full((s, s, 1 + T), -10000.0)
        # index head: text block only
        span_mask[text_start:, text_start:, 0] = 0.0
        span_mask[text_start:, text_start:, 1:] = 0.0
        for e in item.get("entity_list", []):
            et = e["entity_type"]
            if et not in entity_type_list:
                continue
            ti = entity_type_list.index(et)
            for st_c, en_c in e.get("entity_index", []):
                st = text_start + len(tk.encode(item["text"][:st_c],
                                                add_special_tokens=False))
                en = text_start + len(tk.encode(item["text"][:en_c + 1],
                                                add_special_tokens=False)) - 1
                if st < s and en < s:
                    span_labels[st, en, 0] = 1        # index head
                    span_labels[st, en, 1 + ti] = 1   # type head
        return {
            "input_ids": torch.tensor(ids, dtype=torch.long),
            "label_token_idx": torch.tensor(label_idx, dtype=torch.long),
            "text_token_idx": torch.tensor(text_token_idx,
                                           dtype=torch.long),
            "span_labels": span_labels,
            "span_labels_mask": span_mask,
            "text_start": text_start,
            "tlen": tlen,
        }

    def collate(self, samples: List[dict]) -> dict:
        smax = max(x["input_ids"].shape[0] for x in samples)
        tmax = max(x["tlen"] for x in samples)
        nlab = samples[0]["span_labels"].shape[-1]
        b = len(samples)
        ids = torch.zeros(b, smax, dtype=torch.long)
        lab_idx = torch.stack([x["label_token_idx"] for x in samples])
        txt_idx = torch.zeros(b, tmax, dtype=torch.long)
        sl = torch.zeros(b, smax, smax, nlab)
        sm = torch.full((b, smax, smax, nlab), -10000.0)
        for i, x in enumerate(samples):
            n = x["input_ids"].shape[0]
            ids[i, :n] = x["input_ids"]
            txt_idx[i, :x["tlen"]] = x["text_token_idx"]
            sl[i, :n, :n] = x["span_labels"]
            sm[i, :n, :n] = x["span_labels_mask"]
        return {"input_ids": ids, "label_token_idx": lab_idx,
                "text_token_idx": txt_idx, "span_labels": sl,
                "span_labels_mask": sm}
