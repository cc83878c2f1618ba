"""UBERT: unified multi-task span extraction.

Behavioral parity with reference models/ubert/modeling_ubert.py:
- model (:256-310): [b, num_label, seq] stacked inputs, BERT encoder,
  query/key GELU projections, biaffine span scorer [b, num_label, s, s],
  additive span_labels_mask, loss = 10*(100*BCE + multilabel-CE over spans
  + multilabel-CE over labels);
- dataset encode (:56-190): per-choice prompt
  "task_type[SEP]subtask_type[SEP]entity_type" + text, span labels from
  char-level entity_idx, negative-label subsampling to num_labels rows;
- extraction decode (:436-740): OffsetMapping.rematch char<->token
  alignment, extract_index threshold scan, 抽取式阅读理解 top-k decode,
  分类任务 [0,0]-cell argmax.
"""
from __future__ import annotations

import unicodedata
from dataclasses import dataclass
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UbertConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
    MegatronBertPreTrainedModel,
)
from fengshen_amd.models.tagging_models.bert_for_tagging import Biaffine

TASK_CLS = "分类任务"  # classification: score lives in the [0,0] cell
SUBTASK_MRC = "抽取式阅读理解"  # extractive MRC: top-k single-span decode


@dataclass
class UbertOutput:
    loss: Optional[torch.Tensor] = None
    span_logits: Optional[torch.Tensor] = None


class MultilabelCrossEntropy(nn.Module):
    """Circle-loss-style multilabel CE (ref modeling_ubert.py:238-253)."""

    def forward(self, y_pred, y_true):
        y_true = y_true.float()
        y_pred = (1.0 - 2.0 * y_true) * y_pred
        y_pred_neg = y_pred - y_true * 1e12
        y_pred_pos = y_pred - (1.0 - y_true) * 1e12
        zeros = torch.zeros_like(y_pred[..., :1])
        neg = torch.logsumexp(torch.cat([y_pred_neg, zeros], dim=-1), dim=-1)
        pos = torch.logsumexp(torch.cat([y_pred_pos, zeros], dim=-1), dim=-1)
        return torch.mean(neg + pos)


class UbertModel(MegatronBertPreTrainedModel):
    config_class = UbertConfig

    def __init__(self, config: UbertConfig, biaffine_size: int = 128):
        super().__init__(config)
        self.bert = MegatronBertModel(config, add_pooling_layer=False)
        self.query_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.key_proj = nn.Sequential(
            nn.Linear(config.hidden_size, biaffine_size), nn.GELU())
        self.biaffine = Biaffine(biaffine_size, 1)
        self.loss_softmax = MultilabelCrossEntropy()
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                span_labels=None, span_labels_mask=None, span_mask=None,
                **_kw):
        """input_ids [b, num_label, s] (one row per candidate label prompt);
        span_labels [b, num_label, s, s]; span_labels_mask additive
        (0 valid / -10000 invalid, ref :298); span_mask multiplicative
        0/1 (back-compat simplified loss)."""
        b, nl, s = input_ids.shape
        flat = lambda t: t.reshape(b * nl, s) if t is not None else None
        h = self.bert(flat(input_ids), flat(attention_mask),
                      flat(token_type_ids)).last_hidden_state
        q = self.query_proj(h)
        k = self.key_proj(h)
        logits = self.biaffine(q, k).squeeze(-1).view(b, nl, s, s)
        if span_labels_mask is not None:
            logits = logits + span_labels_mask
        loss = None
        if span_labels is not None:
            if span_labels_mask is not None:
                # reference loss (modeling_ubert.py:301-309)
                soft1 = self.loss_softmax(
                    logits.reshape(-1, nl, s * s).float(),
                    span_labels.reshape(-1, nl, s * s))
                soft2 = self.loss_softmax(
                    logits.permute(0, 2, 3, 1).float(),
                    span_labels.permute(0, 2, 3, 1))
                sig = nn.functional.binary_cross_entropy_with_logits(
                    logits.float(), span_labels.float())
                loss = 10.0 * (100.0 * sig + soft1 + soft2)
            else:
                bce = nn.functional.binary_cross_entropy_with_logits(
                    logits.float(), span_labels.float(), reduction="none")
                if span_mask is not None:
                    bce = bce * span_mask.float()
                    loss = 10.0 * bce.sum() / \
                        span_mask.float().sum().clamp(min=1)
                else:
                    loss = 10.0 * bce.mean()
        return UbertOutput(loss=loss, span_logits=logits)

    @torch.no_grad()
    def extract(self, input_ids, attention_mask=None, token_type_ids=None,
                threshold: float = 0.5):
        """decode spans (start, end, prob) per (batch, label) above
        threshold (simple tensor-level decode; text-level decode lives in
        UbertExtractor)."""
        out = self.forward(input_ids, attention_mask, token_type_ids)
        probs = out.span_logits.sigmoid()
        b, nl, s, _ = probs.shape
        results = []
        for bi in range(b):
            per_label = []
            for li in range(nl):
                spans = (probs[bi, li] > threshold).nonzero(as_tuple=False)
                per_label.append([(int(st), int(en),
                                   float(probs[bi, li, st, en]))
                                  for st, en in spans if en >= st])
            results.append(per_label)
        return results


# ---------------------------------------------------------------------------
# data-side schema construction (ref UbertDataset.encode :56-190)
# ---------------------------------------------------------------------------
class UbertEncoder:
    """Builds [num_labels, s] prompt rows + [num_labels, s, s] span labels
    and masks from a task item:
    {"task_type": ..., "subtask_type": ..., "text": ...,
     "choices": [{"entity_type": ..., "label": 0/1,
                  "entity_list": [{"entity_idx": [[start, end], ...]}]}]}
    """

    def __init__(self, tokenizer, max_length: int = 128, num_labels: int = 8):
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.num_labels = num_labels

    def encode(self, item: dict, rng=None) -> dict:
        rng = rng or np.random
        tk = self.tokenizer
        ml = self.max_length
        pos, neg = [], []  # rows with/without positive spans
        for choice in item["choices"]:
            texta = (item["task_type"] + "[SEP]" + item["subtask_type"]
                     + "[SEP]" + choice["entity_type"])
            textb = item["text"]
            enc = _encode_pair(tk, texta, textb, ml)
            span_label = np.zeros((ml, ml))
            span_label_mask = np.zeros((ml, ml)) - 10000.0
            if item["task_type"] == TASK_CLS:
                span_label_mask[0, 0] = 0
                span_label[0, 0] = choice.get("label", 0)
            else:
                question_len = len(tk.encode(texta))
                span_label_mask[question_len:, question_len:] = 0.0
                for entity in choice.get("entity_list", []):
                    for eidx in entity.get("entity_idx", []):
                        if not eidx:
                            continue
                        start_idx = question_len + len(tk.encode(
                            item["text"][:eidx[0]],
                            add_special_tokens=False))
                        end_idx = question_len + len(tk.encode(
                            item["text"][:eidx[1] + 1],
                            add_special_tokens=False)) - 1
                        if start_idx < ml and end_idx < ml:
                            span_label[start_idx, end_idx] = 1
            row = (enc["input_ids"], enc["attention_mask"],
                   enc["token_type_ids"], span_label, span_label_mask)
            (pos if span_label.sum() >= 1 else neg).append(row)

        # negative subsampling to num_labels rows (ref :152-170)
        order = np.arange(len(neg))
        rng.shuffle(order)
        rows = list(pos)
        cur = 0
        while len(rows) < self.num_labels and cur < len(order):
            rows.append(neg[order[cur]])
            cur += 1
        while len(rows) < self.num_labels:
            rows.append(([0] * ml, [0] * ml, [0] * ml,
                         np.zeros((ml, ml)), np.zeros((ml, ml)) - 10000.0))
        rows = rows[:self.num_labels]

        span_labels = np.array([r[3] for r in rows])
        span_masks = np.array([r[4] for r in rows])
        if span_labels.sum() < 1:  # keep the loss well-defined (ref :181-183)
            span_labels[-1, -1, -1] = 1
            span_masks[-1, -1, -1] = 10000.0
        return {
            "input_ids": torch.tensor([r[0] for r in rows]).long(),
            "attention_mask": torch.tensor([r[1] for r in rows]).float(),
            "token_type_ids": torch.tensor([r[2] for r in rows]).long(),
            "span_labels": torch.tensor(span_labels).float(),
            "span_labels_mask": torch.tensor(span_masks).float(),
        }


def _encode_pair(tk, texta: str, textb: str, max_length: int):
    """tokenizer.encode_plus(texta, textb) with a fallback for minimal
    tokenizers that only take one text."""
    try:
        return tk.encode_plus(texta, textb, max_length=max_length,
                              padding="max_length",
                              truncation="longest_first")
    except TypeError:
        enc = tk.encode_plus(texta + "[SEP]" + textb, max_length=max_length,
                             padding="max_length",
                             truncation="longest_first")
        # token_type 1 over the textb region
        qlen = min(len(tk.encode(texta)), max_length)
        tt = list(enc["token_type_ids"])
        for i in range(qlen, len(tt)):
            if enc["attention_mask"][i]:
                tt[i] = 1
        enc["token_type_ids"] = tt
        return enc


def ubert_collate(batch: List[dict]) -> dict:
    return {k: torch.stack([b[k] for b in batch]) for k in batch[0]}


# ---------------------------------------------------------------------------
# extraction decode (ref OffsetMapping + extractModel :436-740)
# ---------------------------------------------------------------------------
class OffsetMapping:
    """char<->token alignment for decode (ref :436-484)."""

    def __init__(self):
        self._do_lower_case = True

    @staticmethod
    def stem(token):
        return token[2:] if token[:2] == "##" else token

    @staticmethod
    def _is_control(ch):
        return unicodedata.category(ch) in ("Cc", "Cf")

    @staticmethod
    def _is_special(ch):
        return bool(ch) and (ch[0] == "[") and (ch[-1] == "]")

    def rematch(self, text, tokens):
        if self._do_lower_case:
            text = text.lower()
        normalized_text, char_mapping = "", []
        for i, ch in enumerate(text):
            if self._do_lower_case:
                ch = unicodedata.normalize("NFD", ch)
                ch = "".join(c for c in ch
                             if unicodedata.category(c) != "Mn")
            ch = "".join(c for c in ch
                         if not (ord(c) == 0 or ord(c) == 0xfffd
                                 or self._is_control(c)))
            normalized_text += ch
            char_mapping.extend([i] * len(ch))
        text, token_mapping, offset = normalized_text, [], 0
        for token in tokens:
            if self._is_special(token):
                token_mapping.append([offset])
                offset += 1
            else:
                token = self.stem(token)
                try:
                    start = text[offset:].index(token) + offset
                except ValueError:
                    token_mapping.append([])
                    continue
                end = start + len(token)
                token_mapping.append(char_mapping[start:end])
                offset = end
        return token_mapping


class UbertExtractor:
    """Text-level decode of span logits into entity structures
    (ref extractModel :486-675)."""

    def __init__(self, model: UbertModel, tokenizer, max_length: int = 128,
                 threshold: float = 0.5):
        self.model = model
        self.tokenizer = tokenizer
        self.max_length = max_length
        self.threshold = threshold

    @staticmethod
    def extract_index(span_logits, sample_length, split_value=0.5):
        result = []
        n = min(sample_length, span_logits.shape[0])
        for i in range(n):
            for j in range(i, n):
                if span_logits[i, j] > split_value:
                    result.append((i, j, span_logits[i, j]))
        return result

    @staticmethod
    def extract_entity(text, entity_idx, text_start_id, text_mapping):
        i0, i1 = entity_idx[0] - text_start_id, entity_idx[1] - text_start_id
        start_split = text_mapping[i0] if 0 <= i0 < len(text_mapping) else []
        end_split = text_mapping[i1] if 0 <= i1 < len(text_mapping) else []
        if start_split and end_split:
            return text[start_split[0]:end_split[-1] + 1]
        return ""

    @torch.no_grad()
    def extract(self, batch_data: List[dict]) -> List[dict]:
        tk = self.tokenizer
        ml = self.max_length
        dev = next(self.model.parameters()).device
        ids_b, att_b, tt_b, mask_b = [], [], [], []
        for item in batch_data:
            ids0, att0, tt0, mask0 = [], [], [], []
            for choice in item["choices"]:
                texta = (item["task_type"] + "[SEP]" + item["subtask_type"]
                         + "[SEP]" + choice["entity_type"])
                enc = _encode_pair(tk, texta, item["text"], ml)
                slm = np.zeros((ml, ml)) - 10000.0
                if item["task_type"] == TASK_CLS:
                    slm[0, 0] = 0
                else:
                    qlen = len(tk.encode(texta))
                    slm[qlen:, qlen:] = 0.0
                ids0.append(enc["input_ids"])
                att0.append(enc["attention_mask"])
                tt0.append(enc["token_type_ids"])
                mask0.append(slm)
            ids_b.append(ids0)
            att_b.append(att0)
            tt_b.append(tt0)
            mask_b.append(mask0)

        out = self.model(
            input_ids=torch.tensor(ids_b).long().to(dev),
            attention_mask=torch.tensor(att_b).float().to(dev),
            token_type_ids=torch.tensor(tt_b).long().to(dev),
            span_labels_mask=torch.tensor(np.array(mask_b)).float().to(dev))
        span_logits = torch.sigmoid(out.span_logits).cpu().numpy()

        for i, item in enumerate(batch_data):
            if item["task_type"] == TASK_CLS:
                max_c = int(np.argmax(span_logits[i, :, 0, 0]))
                item["choices"][max_c]["label"] = 1
                item["choices"][max_c]["score"] = float(
                    span_logits[i, max_c, 0, 0])
                continue
            textb = item["text"]
            tokens = (tk.tokenize(textb) if hasattr(tk, "tokenize")
                      else list(textb))
            offset_mapping = OffsetMapping().rematch(textb, tokens)
            text_ids = tk.encode("[SEP]" + textb)[:ml]
            for c, choice in enumerate(item["choices"]):
                texta = (item["task_type"] + "[SEP]" + item["subtask_type"]
                         + "[SEP]" + choice["entity_type"])
                text_start_id = len(tk.encode(texta))
                logits = span_logits[i, c]
                entity_list, seen = [], []
                if item["subtask_type"] == SUBTASK_MRC:
                    top_k = max(int(choice.get("top_k", 1)), 1)
                    flat = torch.tensor(logits).flatten()
                    _, top_idx = torch.topk(flat, top_k)
                    for t in top_idx:
                        mi = np.unravel_index(int(t), logits.shape)
                        if logits[mi] > self.threshold:
                            name = self.extract_entity(
                                textb, (mi[0], mi[1]), text_start_id,
                                offset_mapping)
                            entity_list.append(
                                {"entity_name": name,
                                 "score": float(logits[mi])})
                else:
                    sample_length = text_start_id + len(text_ids)
                    for st, en, sc in self.extract_index(
                            logits, sample_length, self.threshold):
                        name = self.extract_entity(
                            textb, (st, en), text_start_id, offset_mapping)
                        if name and name not in seen:
                            seen.append(name)
                            entity_list.append(
                                {"entity_name": name, "score": float(sc)})
                choice["entity_list"] = entity_list
        return batch_data
