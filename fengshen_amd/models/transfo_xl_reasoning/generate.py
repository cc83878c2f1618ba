"""Causal-reasoning generation with fixed deduction/abduction prompts
(ref models/transfo_xl_reasoning/generate.py)."""
from __future__ import annotations

from typing import List, Union

import torch
from torch.nn.utils.rnn import pad_sequence

from fengshen_amd.utils.transfo_xl_utils import sample_sequence_batch


def en_to_zh(sentence: str) -> str:
    """Map ASCII punctuation to full-width Chinese (ref :12-18)."""
    en_pun = u",.!?[]()<>\"\"''"
    zh_pun = u"，。！？【】（）《》“”‘’"
    table = {ord(f): ord(t) for f, t in zip(en_pun, zh_pun)}
    return sentence.translate(table)


def _generate_with_prompt(model, tokenizer, prompts: List[str],
                          device, batch_size: int, temperature: float,
                          repetition_penalty: float, max_out_seq: int,
                          top_p: float, end_token_id: int) -> List[str]:
    dev = device if device is not None else next(model.parameters()).device
    model = model.eval().to(dev)
    enc = [tokenizer.encode(t) for t in prompts]
    input_ids = []
    for ids in enc:
        if ids and ids[-1] in (getattr(tokenizer, "sep_token_id", None),
                               getattr(tokenizer, "eos_token_id", None)):
            ids = ids[:-1]
        input_ids.append(torch.tensor(ids, dtype=torch.long))
    input_length = [len(ids) for ids in input_ids]

    output: List[str] = []
    for index in range(0, len(input_ids), batch_size):
        batch = pad_sequence(input_ids[index:index + batch_size],
                             batch_first=True,
                             padding_value=end_token_id).to(dev)
        lengths = torch.tensor(input_length[index:index + batch_size],
                               device=dev)
        res_ids, _probs = sample_sequence_batch(
            model, batch, lengths, end_token_id=end_token_id,
            top_k=0, top_p=top_p, max_out_seq=max_out_seq,
            repetition_penalty=repetition_penalty, temperature=temperature)
        output.extend(
            en_to_zh(tokenizer.decode(ids[length:])).replace(" ", "")
            for ids, length in zip(res_ids,
                                   input_length[index:index + batch_size]))
    return output


@torch.no_grad()
def deduction_generate(model, tokenizer,
                       input_text: Union[str, List[str]],
                       device=None, batch_size: int = 2,
                       temperature: float = 1.0,
                       repetition_penalty: float = 2.0,
                       max_out_seq: int = 512,
                       top_p: float = 0.6,
                       end_token_id: int = 50000) -> List[str]:
    """Cause -> effect with prompt "<bos>X，因而" (ref :21-70)."""
    if isinstance(input_text, str):
        input_text = [input_text]
    prompts = [f"<bos>{t}，因而" for t in input_text]
    return _generate_with_prompt(model, tokenizer, prompts, device,
                                 batch_size, temperature,
                                 repetition_penalty, max_out_seq, top_p,
                                 end_token_id)


@torch.no_grad()
def abduction_generate(model, tokenizer,
                       input_text: Union[str, List[str]],
                       device=None, batch_size: int = 2,
                       temperature: float = 1.0,
                       repetition_penalty: float = 2.0,
                       max_out_seq: int = 512,
                       top_p: float = 0.6,
                       end_token_id: int = 50000) -> List[str]:
    """Effect -> cause with prompt "<bos>之所以X，是因为" (ref :73-120)."""
    if isinstance(input_text, str):
        input_text = [input_text]
    prompts = [f"<bos>之所以{t}，是因为" for t in input_text]
    return _generate_with_prompt(model, tokenizer, prompts, device,
                                 batch_size, temperature,
                                 repetition_penalty, max_out_seq, top_p,
                                 end_token_id)
