"""GPU integration checks for round-1 additions: sparse attention,
HuBERT, selective activation checkpointing, ZeRO-offload."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_sparse_attention_gpu_matches_cpu():
    from fengshen_amd.ops.sparse_attention import (
        LocalSlidingWindowSparsityConfig,
        SparseSelfAttention,
    )
    torch.manual_seed(0)
    cfg = LocalSlidingWindowSparsityConfig(4, block=16,
                                           num_sliding_window_blocks=3)
    attn = SparseSelfAttention(cfg)
    q, k, v = [torch.randn(2, 4, 256, 64) for _ in range(3)]
    ref = attn(q, k, v)
    out32 = attn(q.cuda(), k.cuda(), v.cuda()).cpu()
    assert (out32 - ref).abs().max() < 1e-4
    out16 = attn(q.cuda().bfloat16(), k.cuda().bfloat16(),
                 v.cuda().bfloat16()).float().cpu()
    assert (out16 - ref).abs().max() < 0.15  # bf16 rounding only


def test_hubert_gpu_forward_backward():
    from fengshen_amd.models.hubert import (
        HubertForPreTraining,
        hubert_tiny_config,
    )
    torch.manual_seed(0)
    m = HubertForPreTraining(hubert_tiny_config()).cuda()
    src = torch.randn(2, 4000, device="cuda")
    T = m.hubert.frame_lengths(torch.tensor([4000]))[0].item()
    lab = torch.randint(0, 16, (2, T), device="cuda")
    out = m(src, labels=lab)
    assert out.loss.isfinite()
    out.loss.backward()
    assert m.label_embs.grad is not None


def test_selective_ckpt_gpu_grads_match():
    from fengshen_amd.models.llama.configuration_llama import LlamaConfig
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    cfg = LlamaConfig(vocab_size=512, hidden_size=512, num_hidden_layers=4,
                      num_attention_heads=4, intermediate_size=1024,
                      max_position_embeddings=256)
    torch.manual_seed(0)
    m1 = LlamaForCausalLM(cfg).bfloat16().cuda()
    m2 = LlamaForCausalLM(cfg).bfloat16().cuda()
    m2.load_state_dict(m1.state_dict())
    m1.gradient_checkpointing_enable()
    m2.gradient_checkpointing_enable(skip_interval=2)
    m1.train()
    m2.train()
    ids = torch.randint(0, 512, (2, 128), device="cuda")
    m1(ids, labels=ids).loss.backward()
    m2(ids, labels=ids).loss.backward()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        # atomic-accumulation ordering (norm wgrads) differs run to run,
        # so compare within bf16 noise rather than bitwise
        assert torch.allclose(p1.grad.float(), p2.grad.float(),
                              atol=1e-2, rtol=1e-2)


def test_zero_offload_gpu_step():
    from fengshen_amd.parallel.zero import ZeroOptimizer
    torch.manual_seed(0)
    net = torch.nn.Sequential(
        torch.nn.Linear(64, 64), torch.nn.Tanh(),
        torch.nn.Linear(64, 8)).bfloat16().cuda()
    ref = torch.nn.Sequential(
        torch.nn.Linear(64, 64), torch.nn.Tanh(),
        torch.nn.Linear(64, 8)).bfloat16().cuda()
    ref.load_state_dict(net.state_dict())
    opt = ZeroOptimizer(net.parameters(), stage=0, lr=1e-2,
                        weight_decay=0.0, cpu_offload=True)
    opt_ref = ZeroOptimizer(ref.parameters(), stage=0, lr=1e-2,
                            weight_decay=0.0)
    assert all(not b.master_shard.is_cuda for b in opt.buckets)
    x = torch.randn(16, 64, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        for o, m in [(opt, net), (opt_ref, ref)]:
            loss = m(x).float().pow(2).mean()
            o.zero_grad()
            loss.backward()
            o.step()
    for p1, p2 in zip(net.parameters(), ref.parameters()):
        # offload steps with the CPU eager AdamW, on-device with the HIP
        # kernel: same math, different rounding/fma order -> 1-ulp bf16 diffs
        assert torch.allclose(p1.float(), p2.float(), atol=1e-2, rtol=1e-2)


def test_graphed_decode_matches_eager_generate():
    """hipGraph-captured decode must produce the same greedy tokens as
    HF generate on the eager path (tiny llama)."""
    import torch
    from fengshen_amd.models.llama.configuration_llama import (
        llama_tiny_config)
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    from fengshen_amd.serving.graphed_decode import GraphedDecoder
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config()).to(
        torch.bfloat16).to("cuda").eval()
    ids = torch.randint(3, model.config.vocab_size, (2, 16), device="cuda")
    ref = model.generate(ids, max_new_tokens=12, do_sample=False)
    dec = GraphedDecoder(model, batch=2, max_len=64, max_new_tokens=12)
    out = dec.generate(ids, max_new_tokens=12)
    assert out.shape == ref.shape
    # bf16 decode paths can diverge after many steps; require a long
    # matching prefix (first 8 of 12 greedy tokens identical)
    assert torch.equal(out[:, 16:24], ref[:, 16:24]), (
        out[:, 16:].tolist(), ref[:, 16:].tolist())


def test_graphed_decode_sampling():
    """Sampled graphed decode: reproducible under a fixed torch seed
    (noise buffer is drawn with torch RNG outside the graph), valid ids,
    and diverse across different seeds."""
    import torch
    from fengshen_amd.models.llama.configuration_llama import (
        llama_tiny_config)
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    from fengshen_amd.serving.graphed_decode import GraphedDecoder
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config()).to(
        torch.bfloat16).to("cuda").eval()
    ids = torch.randint(3, model.config.vocab_size, (2, 16), device="cuda")
    dec = GraphedDecoder(model, batch=2, max_len=64, max_new_tokens=12,
                         do_sample=True, top_k=20, top_p=0.95,
                         temperature=0.8)
    torch.manual_seed(7)
    a = dec.generate(ids, max_new_tokens=12)
    torch.manual_seed(7)
    b = dec.generate(ids, max_new_tokens=12)
    assert torch.equal(a, b)
    torch.manual_seed(8)
    c = dec.generate(ids, max_new_tokens=12)
    assert not torch.equal(a, c)
    assert int(a.max()) < model.config.vocab_size and int(a.min()) >= 0


def test_text_generation_pipeline_graph_path():
    """text_generation pipeline routes a bf16 fengshen llama through the
    GraphedDecoder and produces the same text as the CPU/HF fallback."""
    import torch
    from fengshen_amd.models.llama.configuration_llama import (
        llama_tiny_config)
    from fengshen_amd.models.llama.modeling_llama import LlamaForCausalLM
    from fengshen_amd.pipelines import text_generation
    from fengshen_amd.tokenizer import SimpleCharTokenizer
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny_config()).to(torch.bfloat16).cuda()
    tok = SimpleCharTokenizer()
    pipe = text_generation.Pipeline(model=model, tokenizer=tok,
                                    max_len=64, max_new_tokens=8)
    assert pipe._graph_eligible()
    out = pipe.generate("你好世界", max_new_tokens=6)
    assert isinstance(out, str)
    assert pipe._decoder is not None  # graph path actually used
    # eager fallback (use_graph=False) should agree on greedy prefix
    pipe2 = text_generation.Pipeline(model=model, tokenizer=tok,
                                     use_graph=False, max_len=64,
                                     max_new_tokens=8)
    out2 = pipe2.generate("你好世界", max_new_tokens=6)
    assert out[:3] == out2[:3], (out, out2)
