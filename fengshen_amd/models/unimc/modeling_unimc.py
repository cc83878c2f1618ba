"""UniMC: zero-shot label-as-option multiple choice.

Behavioral parity with reference models/unimc/modeling_unimc.py:
- option-isolation attention mask (ref :92-112 get_att_mask): every option
  span attends only to itself plus the shared question/text region, never to
  the other options;
- restarted position ids per option (ref :72-91 get_position_ids);
- classification logits read from the MLM head's yes-token column at every
  position, masked to the option anchor positions by an additive
  clslabels_mask of 0/-10000 (ref :297-332), CE target = anchor position of
  the correct option;
- MLM auxiliary loss via random masking of the text region (ref :114-137
  random_masking, rate sampled from {0,.1,.2,.3} with p=[.3,.3,.25,.15]).

The backbone is the MI355X-native MegatronBert MLM stack (3D per-sample
attention masks ride the fused scaled-masked-softmax HIP kernel's
[b,1,sq,sk] mask layout).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn

from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig as UniMCConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertForMaskedLM,
    MegatronBertPreTrainedModel,
)


@dataclass
class UniMCOutput:
    loss: Optional[torch.Tensor] = None
    mlm_logits: Optional[torch.Tensor] = None
    cls_logits: Optional[torch.Tensor] = None
    # back-compat alias used by earlier round-1 callers
    @property
    def option_logits(self):
        return self.cls_logits


class UniMCEncoder:
    """Data-side sample construction (ref UniMCDataset.encode, :139-236).

    Sequence layout: [MASK] opt_1 [MASK] opt_2 ... [SEP] question [SEP] text
    where each option is PRECEDED by its [MASK] anchor token (position
    label_idx[i]); targets put no_token at every anchor and yes_token at the
    correct option's anchor.
    """

    def __init__(self, tokenizer, yes_token: int, no_token: int,
                 max_length: int = 512, used_mask: bool = False):
        self.tokenizer = tokenizer
        self.yes_token = yes_token
        self.no_token = no_token
        self.max_length = max_length
        self.used_mask = used_mask

    # -- mask/position builders (pure, unit-testable) -------------------
    @staticmethod
    def get_att_mask(attention_mask: np.ndarray, label_idx: List[int],
                     question_len: int) -> np.ndarray:
        """Option isolation (ref :92-112): zero the option-region block,
        then turn each option's own diagonal block back on."""
        max_length = len(attention_mask)
        att = np.tile(np.asarray(attention_mask)[None, :], (max_length, 1))
        lo, hi = question_len, label_idx[-1]
        att[lo:hi, lo:hi] = 0
        for i in range(len(label_idx) - 1):
            a, b = label_idx[i], label_idx[i + 1]
            if b - a <= 0:
                continue
            att[a:b, a:b] = 1
        return att

    @staticmethod
    def get_position_ids(label_idx: List[int], max_length: int,
                         question_len: int) -> List[int]:
        """Per-option restarted positions (ref :72-91)."""
        question_position_ids = np.arange(question_len)
        label_position_ids = np.arange(question_len, label_idx[-1])
        for i in range(len(label_idx) - 1):
            label_position_ids[label_idx[i] - question_len:
                               label_idx[i + 1] - question_len] = np.arange(
                question_len, question_len + label_idx[i + 1] - label_idx[i])
        max_len_label = int(max(label_position_ids)) if len(
            label_position_ids) else question_len - 1
        text_position_ids = np.arange(
            max_len_label + 1, max_length + max_len_label + 1 - label_idx[-1])
        position_ids = (list(question_position_ids)
                        + list(label_position_ids) + list(text_position_ids))
        for i in range(512, max_length):
            if i < len(position_ids) and position_ids[i] > 511:
                position_ids[i] = 511
        return position_ids[:max_length]

    def random_masking(self, token_ids, mask_rate, mask_start_idx,
                       max_length, mask_id, rng=None):
        """BERT-style 80/10/10 masking of the text region (ref :114-137)."""
        rng = rng or np.random
        rands = rng.random(len(token_ids))
        source, target = [], []
        vocab_size = getattr(self.tokenizer, "vocab_size", 21128)
        for i, (r, t) in enumerate(zip(rands, token_ids)):
            if i < mask_start_idx:
                source.append(t)
                target.append(-100)
                continue
            if r < mask_rate * 0.8:
                source.append(mask_id)
                target.append(t)
            elif r < mask_rate * 0.9:
                source.append(t)
                target.append(t)
            elif r < mask_rate:
                source.append(int(rng.choice(vocab_size - 1)) + 1)
                target.append(t)
            else:
                source.append(t)
                target.append(-100)
        while len(source) < max_length:
            source.append(0)
            target.append(-100)
        return source[:max_length], target[:max_length]

    # -- full example encode --------------------------------------------
    def encode(self, item: dict) -> dict:
        tk = self.tokenizer
        mask_tok = tk.mask_token or "[MASK]"
        choice = list(item["choice"])
        while len(tk.encode(mask_tok.join(choice))) > self.max_length - 32:
            choice = [c[:max(len(c) // 2, 1)] for c in choice]

        parts = [mask_tok + mask_tok.join(choice)]
        if item.get("question"):
            parts.append(item["question"])
        parts.append(item["texta"])
        if item.get("textb"):
            parts.append(item["textb"])
        texta = "[SEP]".join(parts)

        enc = tk.encode_plus(texta, max_length=self.max_length,
                             padding="max_length", truncation="longest_first")
        encode_sent = enc["input_ids"]
        attention_mask = enc["attention_mask"]
        sample_max_length = int(sum(attention_mask))

        label = int(item.get("label", 0))
        question_len = 1  # [CLS]
        label_idx = [question_len]
        for c in choice:
            label_idx.append(label_idx[-1]
                             + len(tk.encode(c, add_special_tokens=False)) + 1)

        token_type_ids = ([0] * question_len
                          + [1] * (label_idx[-1] - label_idx[0] + 1)
                          + [0] * self.max_length)[:self.max_length]
        att2d = self.get_att_mask(attention_mask, label_idx, question_len)
        position_ids = self.get_position_ids(label_idx, self.max_length,
                                             question_len)

        clslabels_mask = np.zeros(len(encode_sent)) - 10000.0
        clslabels_mask[label_idx[:-1]] = 0.0
        mlmlabels_mask = np.zeros(len(encode_sent))
        mlmlabels_mask[label_idx[0]] = 1

        if self.used_mask:
            mask_rate = 0.1 * np.random.choice(4, p=[0.3, 0.3, 0.25, 0.15])
            source, target = self.random_masking(
                encode_sent, mask_rate, label_idx[-1], self.max_length,
                tk.mask_token_id)
        else:
            source, target = list(encode_sent), list(encode_sent)

        source = np.array(source)
        target = np.array(target)
        source[label_idx[:-1]] = tk.mask_token_id
        target[label_idx[:-1]] = self.no_token
        target[label_idx[label]] = self.yes_token

        n = sample_max_length
        return {
            "input_ids": torch.tensor(source[:n]).long(),
            "token_type_ids": torch.tensor(token_type_ids[:n]).long(),
            "attention_mask": torch.tensor(att2d[:n, :n]).float(),
            "position_ids": torch.tensor(position_ids[:n]).long(),
            "mlmlabels": torch.tensor(target[:n]).long(),
            "clslabels": torch.tensor(label_idx[label]).long(),
            "clslabels_mask": torch.tensor(clslabels_mask[:n]).float(),
            "mlmlabels_mask": torch.tensor(mlmlabels_mask[:n]).float(),
            "option_positions": torch.tensor(label_idx[:-1]).long(),
        }


def unimc_collate(batch: List[dict]) -> dict:
    """Pad a list of UniMCEncoder.encode outputs (ref collate_fn :258-294);
    2D attention masks pad into a [b, s, s] zero-filled block."""
    smax = max(b["input_ids"].shape[0] for b in batch)
    out = {}
    for k in batch[0]:
        if k == "clslabels":
            out[k] = torch.stack([b[k] for b in batch])
        elif k == "attention_mask":
            m = torch.zeros(len(batch), smax, smax)
            for i, b in enumerate(batch):
                n = b[k].shape[0]
                m[i, :n, :n] = b[k]
            out[k] = m
        elif k == "clslabels_mask":
            m = torch.full((len(batch), smax), -10000.0)
            for i, b in enumerate(batch):
                m[i, :b[k].shape[0]] = b[k]
            out[k] = m
        elif k == "option_positions":
            nopt = max(b[k].shape[0] for b in batch)
            m = torch.zeros(len(batch), nopt, dtype=torch.long)
            for i, b in enumerate(batch):
                m[i, :b[k].shape[0]] = b[k]
            out[k] = m
        else:
            out[k] = nn.utils.rnn.pad_sequence(
                [b[k] for b in batch], batch_first=True,
                padding_value=-100 if k == "mlmlabels" else 0)
    return out


class UniMCModel(MegatronBertPreTrainedModel):
    """Ref UniMCModel (:297-332): MLM backbone + yes-token option scoring."""

    config_class = UniMCConfig

    def __init__(self, config: UniMCConfig, yes_token_id: int = 1):
        super().__init__(config)
        self.mlm = MegatronBertForMaskedLM(config)
        self.yes_token_id = yes_token_id
        self.post_init()

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                position_ids=None, mlmlabels=None, clslabels=None,
                clslabels_mask=None, mlmlabels_mask=None,
                option_positions=None, labels=None, **_kw):
        """Two calling conventions:
        - reference-style: clslabels (anchor POSITION of the correct option)
          + clslabels_mask (additive 0/-10000 over positions) + mlmlabels;
        - simplified: option_positions [b, n_opt] + labels [b] (option index).
        """
        b, seq_len = input_ids.shape
        out = self.mlm(input_ids, attention_mask, token_type_ids,
                       labels=mlmlabels, position_ids=position_ids)
        mlm_logits = out.logits
        from fengshen_amd.parallel import groups
        from fengshen_amd.parallel.mappings import (
            gather_from_tensor_model_parallel_region)
        if (groups.get_tensor_model_parallel_world_size() > 1
                and mlmlabels is not None):
            mlm_logits = gather_from_tensor_model_parallel_region(mlm_logits)
        yes = mlm_logits[:, :, self.yes_token_id]  # [b, s]

        if clslabels_mask is not None:
            cls_logits = yes + clslabels_mask
            loss = None
            if clslabels is not None:
                cls_loss = nn.functional.cross_entropy(
                    cls_logits.float(), clslabels.view(-1))
                loss = cls_loss if out.loss is None else out.loss + cls_loss
            return UniMCOutput(loss=loss, mlm_logits=mlm_logits,
                               cls_logits=cls_logits)

        # simplified convention
        assert option_positions is not None
        option_logits = torch.gather(yes, 1, option_positions)
        loss = None
        if labels is not None:
            loss = nn.functional.cross_entropy(option_logits.float(), labels)
            if out.loss is not None:
                loss = loss + out.loss
        return UniMCOutput(loss=loss, mlm_logits=mlm_logits,
                           cls_logits=option_logits)

    @torch.no_grad()
    def predict(self, input_ids, attention_mask, token_type_ids,
                option_positions, position_ids=None):
        out = self.forward(input_ids, attention_mask, token_type_ids,
                           position_ids=position_ids,
                           option_positions=option_positions)
        return out.cls_logits.argmax(dim=-1)
