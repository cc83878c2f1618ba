"""GPU end-to-end training smokes: tiny LLaMA + ZeRO on 1 MI355X, bf16,
with the HIP kernels on the hot path (extension required, no eager fallback)."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_llama_tiny_train_step_gpu():
    from fengshen_amd.ops import has_ext
    assert has_ext()
    from fengshen_amd.models.llama.configuration_llama import LlamaConfig
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    from fengshen_amd.parallel.zero import ZeroOptimizer

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=2048, hidden_size=512, num_hidden_layers=4,
                      num_attention_heads=8, intermediate_size=1408,
                      max_position_embeddings=512)
    m = LlamaForCausalLM(cfg).to(torch.bfloat16).to("cuda")
    m.gradient_checkpointing_enable()
    m.train()
    opt = ZeroOptimizer(m.parameters(), stage=2, lr=1e-3, weight_decay=0.01)
    ids = torch.randint(3, 2048, (4, 256), device="cuda")
    losses = []
    for _ in range(10):
        out = m(ids, labels=ids)
        opt.zero_grad()
        out.loss.backward()
        opt.step()
        losses.append(out.loss.item())
    assert all(torch.isfinite(torch.tensor(losses))), losses
    assert losses[-1] < losses[0] * 0.8, f"no learning: {losses}"


def test_bert_tiny_train_step_gpu():
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
        MegatronBertForPreTraining)
    from fengshen_amd.parallel.zero import ZeroOptimizer

    torch.manual_seed(0)
    cfg = bert_tiny_config(hidden_size=128, num_attention_heads=8,
                           intermediate_size=512)
    m = MegatronBertForPreTraining(cfg).to(torch.bfloat16).to("cuda")
    m.train()
    opt = ZeroOptimizer(m.parameters(), stage=1, lr=1e-3)
    ids = torch.randint(3, 256, (8, 64), device="cuda")
    labels = ids.clone()
    labels[:, ::2] = -100
    sop = torch.randint(0, 2, (8,), device="cuda")
    losses = []
    for _ in range(10):
        out = m(ids, attention_mask=torch.ones_like(ids), labels=labels,
                next_sentence_label=sop)
        opt.zero_grad()
        out.loss.backward()
        opt.step()
        losses.append(out.loss.item())
    assert losses[-1] < losses[0], losses


def test_generate_gpu():
    from fengshen_amd.models.llama.configuration_llama import LlamaConfig
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=2048, hidden_size=256, num_hidden_layers=2,
                      num_attention_heads=4, intermediate_size=704,
                      max_position_embeddings=256)
    m = LlamaForCausalLM(cfg).to(torch.bfloat16).to("cuda").eval()
    ids = torch.randint(3, 2048, (2, 8), device="cuda")
    gen = m.generate(ids, max_new_tokens=16, do_sample=False)
    assert gen.shape == (2, 24)
