"""HIP kernel numerics vs plain-PyTorch fp32 oracles (SURVEY.md §4 pattern).
All tests require an MI355X."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def _hip_ready():
    if not torch.cuda.is_available():
        return False
    from fengshen_amd.ops import has_ext
    return has_ext()


@pytest.fixture(autouse=True)
def _require_ext():
    assert _hip_ready(), "HIP extension missing on GPU box"


def _rand(*shape, dtype=torch.bfloat16, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    return torch.randn(*shape, generator=g, device="cuda", dtype=torch.float32) \
        .to(dtype)


def _close(a, b, tol=2e-2):
    """relative-to-range closeness (bf16 outputs vs fp32 oracle)."""
    a = a.float()
    b = b.float()
    scale = b.abs().max().clamp(min=1.0)
    err = (a - b).abs().max() / scale
    assert err.item() < tol, f"rel err {err.item():.4g} (scale {scale.item():.3g})"


def test_rms_norm_fwd_bwd():
    from fengshen_amd.ops import functional as F
    x = _rand(4, 33, 1024).requires_grad_(True)
    w = _rand(1024, seed=1).requires_grad_(True)
    y = F.rms_norm(x, w, 1e-6)
    ref = F.eager_rms_norm(x.detach().float(), w.detach().float(), 1e-6)
    _close(y, ref)

    gy = _rand(4, 33, 1024, seed=2)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    w2 = w.detach().float().requires_grad_(True)
    F.eager_rms_norm(x2, w2, 1e-6).backward(gy.float())
    _close(x.grad, x2.grad)
    rel = (w.grad.float() - w2.grad).abs().max() / w2.grad.abs().max()
    assert rel.item() < 2e-2


def test_layer_norm_fwd_bwd():
    from fengshen_amd.ops import functional as F
    x = _rand(6, 17, 768).requires_grad_(True)
    w = _rand(768, seed=1).requires_grad_(True)
    b = _rand(768, seed=2).requires_grad_(True)
    y = F.layer_norm(x, w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(
        x.detach().float(), (768,), w.detach().float(), b.detach().float(), 1e-5)
    _close(y, ref)

    gy = _rand(6, 17, 768, seed=3)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    w2 = w.detach().float().requires_grad_(True)
    b2 = b.detach().float().requires_grad_(True)
    torch.nn.functional.layer_norm(x2, (768,), w2, b2, 1e-5).backward(gy.float())
    _close(x.grad, x2.grad)
    assert ((w.grad.float() - w2.grad).abs().max()
            / w2.grad.abs().max()).item() < 2e-2
    assert ((b.grad.float() - b2.grad).abs().max()
            / b2.grad.abs().max()).item() < 2e-2


@pytest.mark.parametrize("sk", [128, 512, 2048, 4096, 100])
def test_scaled_masked_softmax(sk):
    from fengshen_amd.ops import functional as F
    b, np_, sq = 2, 4, 64
    x = _rand(b, np_, sq, sk).requires_grad_(True)
    mask = (torch.rand(b, 1, sq, sk, device="cuda") < 0.2)
    mask[..., 0] = False  # keep at least one position
    scale = 0.35
    y = F.scaled_masked_softmax(x, mask, scale)
    ref = F.eager_scaled_masked_softmax(x.detach().float(), mask, scale)
    _close(y, ref, 1e-2)

    gy = _rand(b, np_, sq, sk, seed=3)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    F.eager_scaled_masked_softmax(x2, mask, scale).backward(gy.float())
    _close(x.grad, x2.grad, 1e-2)


@pytest.mark.parametrize("s", [64, 1024, 2048])
def test_scaled_causal_softmax(s):
    from fengshen_amd.ops import functional as F
    ab = 8
    x = _rand(ab, s, s).requires_grad_(True)
    scale = 1.0 / math.sqrt(128)
    y = F.scaled_causal_softmax(x, scale)
    ref = F.eager_scaled_causal_softmax(x.detach().float(), scale)
    _close(y, ref, 1e-2)
    # strictly zero above diagonal
    assert y.float().triu(1).abs().max().item() == 0.0

    gy = _rand(ab, s, s, seed=3)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    F.eager_scaled_causal_softmax(x2, scale).backward(gy.float())
    _close(x.grad, x2.grad, 1e-2)


def test_rope_fwd_bwd():
    from fengshen_amd.ops import functional as F
    b, np_, s, hn = 2, 4, 128, 128
    q = _rand(b, np_, s, hn).requires_grad_(True)
    k = _rand(b, np_, s, hn, seed=1).requires_grad_(True)
    cos, sin = F.build_rope_cache(256, hn, device="cuda")
    qo, ko = F.apply_rotary(q, k, cos, sin, offset=7)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    rq, rk = F.eager_apply_rotary(q2, k2, cos, sin, offset=7)
    _close(qo, rq)
    _close(ko, rk)

    gq = _rand(b, np_, s, hn, seed=2)
    gk = _rand(b, np_, s, hn, seed=3)
    (qo.float() * gq.float()).sum().backward()
    (rq * gq.float()).sum().backward()
    _close(q.grad, q2.grad)


def test_swiglu_fwd_bwd():
    from fengshen_amd.ops import functional as F
    x = _rand(64, 2 * 1408).requires_grad_(True)
    y = F.swiglu(x)
    g, u = x.detach().float().chunk(2, -1)
    ref = torch.nn.functional.silu(g) * u
    _close(y, ref)

    gy = _rand(64, 1408, seed=5)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    g2, u2 = x2.chunk(2, -1)
    (torch.nn.functional.silu(g2) * u2).backward(gy.float())
    _close(x.grad, x2.grad)


def test_bias_gelu_fwd_bwd():
    from fengshen_amd.ops import functional as F
    x = _rand(128, 3072).requires_grad_(True)
    b = _rand(3072, seed=1).requires_grad_(True)
    y = F.bias_gelu(x, b)
    ref = F.eager_gelu(x.detach().float() + b.detach().float())
    _close(y, ref)

    gy = _rand(128, 3072, seed=2)
    y.backward(gy)
    x2 = x.detach().float().requires_grad_(True)
    b2 = b.detach().float().requires_grad_(True)
    F.eager_gelu(x2 + b2).backward(gy.float())
    _close(x.grad, x2.grad, 3e-2)
    assert ((b.grad.float() - b2.grad).abs().max()
            / b2.grad.abs().max()).item() < 3e-2


def test_fused_adamw_matches_torch():
    from fengshen_amd.ops.adamw import fused_adamw_flat_
    n = 1 << 20
    torch.manual_seed(0)
    master = torch.randn(n, device="cuda", dtype=torch.float32)
    ref_p = master.clone()
    grad = torch.randn(n, device="cuda", dtype=torch.bfloat16)
    m = torch.zeros_like(master)
    v = torch.zeros_like(master)
    out = torch.empty(n, device="cuda", dtype=torch.bfloat16)

    ref = torch.nn.Parameter(ref_p.clone())
    opt = torch.optim.AdamW([ref], lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=0.01)
    for step in range(1, 4):
        fused_adamw_flat_(master, grad, m, v, out, lr=1e-3, beta1=0.9,
                          beta2=0.999, eps=1e-8, weight_decay=0.01, step=step)
        ref.grad = grad.float()
        opt.step()
    assert (master - ref.detach()).abs().max().item() < 1e-5
    rel = (out.float() - master).abs() / master.abs().clamp(min=1.0)
    assert rel.max().item() < 1e-2  # bf16 roundoff


def test_attention_vs_sdpa():
    """composite attention path (bmm + fused softmax) vs torch eager oracle."""
    from fengshen_amd.ops import functional as F
    b, np_, s, hn = 2, 8, 512, 128
    q = _rand(b, np_, s, hn)
    k = _rand(b, np_, s, hn, seed=1)
    v = _rand(b, np_, s, hn, seed=2)
    out = F.attention(q, k, v, causal=True, scale=1.0 / math.sqrt(hn))
    # fp32 oracle
    qf, kf, vf = q.float(), k.float(), v.float()
    scores = qf @ kf.transpose(-1, -2) / math.sqrt(hn)
    causal = torch.ones(s, s, device="cuda", dtype=torch.bool).triu(1)
    scores = scores.masked_fill(causal, -1e9)
    ref = torch.softmax(scores, -1) @ vf
    _close(out, ref)


@pytest.mark.parametrize("s", [64, 512, 2048])
def test_flash_attention_fwd(s):
    from fengshen_amd.ops.flash import flash_attention
    from fengshen_amd.ops import functional as F
    b, h, d = 2, 4, 128
    q = _rand(b, h, s, d)
    k = _rand(b, h, s, d, seed=1)
    v = _rand(b, h, s, d, seed=2)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale)
    qf, kf, vf = q.float(), k.float(), v.float()
    scores = qf @ kf.transpose(-1, -2) * scale
    causal = torch.ones(s, s, device="cuda", dtype=torch.bool).triu(1)
    scores = scores.masked_fill(causal, float("-inf"))
    ref = torch.softmax(scores, -1) @ vf
    _close(out, ref)


def test_flash_attention_bwd():
    from fengshen_amd.ops.flash import flash_attention
    b, h, s, d = 2, 4, 256, 128
    q = _rand(b, h, s, d).requires_grad_(True)
    k = _rand(b, h, s, d, seed=1).requires_grad_(True)
    v = _rand(b, h, s, d, seed=2).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale)
    gy = _rand(b, h, s, d, seed=3)
    out.backward(gy)

    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    scores = q2 @ k2.transpose(-1, -2) * scale
    causal = torch.ones(s, s, device="cuda", dtype=torch.bool).triu(1)
    scores = scores.masked_fill(causal, float("-inf"))
    ref = torch.softmax(scores, -1) @ v2
    ref.backward(gy.float())
    _close(q.grad, q2.grad, 3e-2)
    _close(k.grad, k2.grad, 3e-2)
    _close(v.grad, v2.grad, 3e-2)


def test_w8_gemv_matches_dequant():
    from fengshen_amd.ops import get_ext
    torch.manual_seed(0)
    out_f, in_f = 1024, 2048
    w = torch.randn(out_f, in_f, device="cuda")
    scale = w.abs().amax(1, keepdim=True) / 127.0
    q8 = torch.round(w / scale).clamp(-127, 127).to(torch.int8)
    x = _rand(3, in_f)
    y = get_ext().w8_gemv(q8, scale.view(-1).float(), x)
    ref = x.float() @ (q8.float() * scale).t()
    _close(y, ref)


# ---------------------------------------------------------------------------
# generalized flash: head dims 40-160, non-causal, klens, cross-attention
# ---------------------------------------------------------------------------
def _flash_oracle(q, k, v, scale, causal=False, klens=None):
    qf, kf, vf = q.float(), k.float(), v.float()
    sq, sk = q.shape[-2], k.shape[-2]
    scores = qf @ kf.transpose(-1, -2) * scale
    if causal:
        cm = torch.ones(sq, sk, device=q.device, dtype=torch.bool).triu(1)
        scores = scores.masked_fill(cm, float("-inf"))
    if klens is not None:
        ar = torch.arange(sk, device=q.device)
        pm = ar.unsqueeze(0) >= klens.long().unsqueeze(1)  # [b, sk]
        scores = scores.masked_fill(pm[:, None, None, :], float("-inf"))
    return torch.softmax(scores, -1) @ vf


@pytest.mark.parametrize("d", [40, 64, 80, 96, 128, 160])
def test_flash_headdims_causal_or_bidir(d):
    from fengshen_amd.ops.flash import flash_attention
    b, h, s = 2, 3, 256
    q = _rand(b, h, s, d)
    k = _rand(b, h, s, d, seed=1)
    v = _rand(b, h, s, d, seed=2)
    scale = 1.0 / math.sqrt(d)
    out_c = flash_attention(q, k, v, scale, causal=True)
    _close(out_c, _flash_oracle(q, k, v, scale, causal=True))
    out_b = flash_attention(q, k, v, scale, causal=False)
    _close(out_b, _flash_oracle(q, k, v, scale))


@pytest.mark.parametrize("d", [64, 96, 160])
def test_flash_bwd_headdims(d):
    from fengshen_amd.ops.flash import flash_attention
    b, h, s = 2, 2, 128
    q = _rand(b, h, s, d).requires_grad_(True)
    k = _rand(b, h, s, d, seed=1).requires_grad_(True)
    v = _rand(b, h, s, d, seed=2).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale, causal=(d != 64))
    gy = _rand(b, h, s, d, seed=3)
    out.backward(gy)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ref = _flash_oracle(q2, k2, v2, scale, causal=(d != 64))
    ref.backward(gy.float())
    _close(q.grad, q2.grad, 3e-2)
    _close(k.grad, k2.grad, 3e-2)
    _close(v.grad, v2.grad, 3e-2)


def test_flash_klens_padding_fwd_bwd():
    """BERT-style suffix padding via klens: fwd matches masked oracle,
    dK/dV rows at pad keys are zero."""
    from fengshen_amd.ops.flash import flash_attention
    b, h, s, d = 3, 2, 128, 64
    klens = torch.tensor([128, 70, 33], device="cuda", dtype=torch.int32)
    q = _rand(b, h, s, d).requires_grad_(True)
    k = _rand(b, h, s, d, seed=1).requires_grad_(True)
    v = _rand(b, h, s, d, seed=2).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale, causal=False, klens=klens)
    gy = _rand(b, h, s, d, seed=3)
    out.backward(gy)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ref = _flash_oracle(q2, k2, v2, scale, klens=klens)
    ref.backward(gy.float())
    _close(out, ref)
    _close(q.grad, q2.grad, 3e-2)
    _close(k.grad, k2.grad, 3e-2)
    _close(v.grad, v2.grad, 3e-2)
    assert k.grad[1, :, 70:, :].abs().max() == 0
    assert v.grad[2, :, 33:, :].abs().max() == 0


@pytest.mark.parametrize("d,sk", [(40, 77), (80, 77), (160, 77), (64, 100)])
def test_flash_cross_attention(d, sk):
    """SD UNet cross-attention: sq != sk, ragged sk (77 text tokens)."""
    from fengshen_amd.ops.flash import flash_attention
    b, h, sq = 2, 4, 256
    q = _rand(b, h, sq, d).requires_grad_(True)
    k = _rand(b, h, sk, d, seed=1).requires_grad_(True)
    v = _rand(b, h, sk, d, seed=2).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale, causal=False)
    gy = _rand(b, h, sq, d, seed=3)
    out.backward(gy)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ref = _flash_oracle(q2, k2, v2, scale)
    ref.backward(gy.float())
    _close(out, ref)
    _close(q.grad, q2.grad, 3e-2)
    _close(k.grad, k2.grad, 3e-2)
    _close(v.grad, v2.grad, 3e-2)


def test_attention_routes_bert_mask_to_flash():
    """functional.attention with a [b,1,1,s] suffix pad mask must hit the
    flash path (mask_to_klens) and match the eager oracle."""
    from fengshen_amd.ops import functional as F
    b, h, s, d = 2, 4, 128, 64
    q = _rand(b, h, s, d)
    k = _rand(b, h, s, d, seed=1)
    v = _rand(b, h, s, d, seed=2)
    klens = torch.tensor([128, 50], device="cuda")
    ar = torch.arange(s, device="cuda")
    mask = (ar.unsqueeze(0) >= klens.unsqueeze(1))[:, None, None, :]
    out = F.attention(q, k, v, causal=False, mask=mask,
                      scale=1.0 / math.sqrt(d))
    ref = _flash_oracle(q, k, v, 1.0 / math.sqrt(d),
                        klens=klens.to(torch.int32))
    # pad-query rows: flash normalizes over the klen window too, so only
    # compare real content
    _close(out, ref)


def test_flash_dropout_determinism_and_rate():
    """Same seed -> identical output; drop rate ~ p; p=0 path unchanged."""
    from fengshen_amd.ops.flash import _FlashAttention
    b, h, s, d = 2, 2, 128, 64
    q = _rand(b, h, s, d)
    k = _rand(b, h, s, d, seed=1)
    v = torch.ones(b, h, s, d, device="cuda", dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(d)
    o1 = _FlashAttention.apply(q, k, v, scale, False, None, 0.5, 1234)
    o2 = _FlashAttention.apply(q, k, v, scale, False, None, 0.5, 1234)
    assert torch.equal(o1, o2)
    o3 = _FlashAttention.apply(q, k, v, scale, False, None, 0.5, 999)
    assert not torch.equal(o1, o3)
    # with V = ones, output rows = sum(dropped P)/l: mean ~ 1 (keep_scale
    # compensates), but high variance per row; check global mean
    assert abs(o1.float().mean().item() - 1.0) < 0.05


def test_flash_dropout_bwd_matches_mask_extracted_oracle():
    """fwd and bwd must regenerate the SAME dropout mask.  Extract the
    effective dropped-normalized attention matrix M by running the fixed-
    seed forward with V = I, derive the keep mask D = (M > 0), and compare
    flash backward against an eager graph using exactly that D."""
    from fengshen_amd.ops.flash import _FlashAttention
    b, h, s, d = 1, 2, 64, 64
    scale = 1.0 / math.sqrt(d)
    seed, p = 77, 0.3
    q = _rand(b, h, s, d).requires_grad_(True)
    k = _rand(b, h, s, d, seed=1).requires_grad_(True)
    v = _rand(b, h, s, d, seed=2).requires_grad_(True)
    eye = torch.eye(s, device="cuda", dtype=torch.bfloat16) \
        .expand(b, h, s, s).contiguous()
    with torch.no_grad():
        m_mat = _FlashAttention.apply(q, k, eye, scale, True, None, p,
                                      seed).float()  # = D*A row-normalized
    keep = (m_mat > 0)
    # eager with the extracted mask
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    scores = q2 @ k2.transpose(-1, -2) * scale
    cm = torch.ones(s, s, device="cuda", dtype=torch.bool).triu(1)
    scores = scores.masked_fill(cm, float("-inf"))
    a = torch.softmax(scores, -1)
    ref = (a * keep.float() / (1 - p)) @ v2
    out = _FlashAttention.apply(q, k, v, scale, True, None, p, seed)
    _close(out, ref)
    gy = _rand(b, h, s, d, seed=3)
    out.backward(gy)
    ref.backward(gy.float())
    _close(q.grad, q2.grad, 4e-2)
    _close(k.grad, k2.grad, 4e-2)
    _close(v.grad, v2.grad, 4e-2)


@pytest.mark.parametrize("s", [64, 512, 2048])
def test_flash_fwd_v3_matches_oracle(s):
    """v3 swapped-QK^T 32x32 schedule vs fp32 oracle (causal d=128)."""
    from fengshen_amd.ops import get_ext
    b, h, d = 2, 3, 128
    q = _rand(b, h, s, d)
    k = _rand(b, h, s, d, seed=1)
    v = _rand(b, h, s, d, seed=2)
    scale = 1.0 / math.sqrt(d)
    o, lse = get_ext().flash_attn_fwd_v3(q, k, v, scale, True, None,
                                         0.0, 0)
    _close(o, _flash_oracle(q, k, v, scale, causal=True))
    # LSE must match the general kernel's (used by the shared backward)
    o2, lse2 = get_ext().flash_attn_fwd(q, k, v, scale, True, None, 0.0, 0)
    assert (lse - lse2).abs().max().item() < 1e-3


@pytest.mark.parametrize("s", [64, 512, 2048])
def test_flash_bwd_v3_matches_oracle(s):
    """v3 backward (dq/dkv 32x32 swapped schedule) vs fp32 autograd."""
    from fengshen_amd.ops.flash import flash_attention
    b, h, d = 2, 3, 128
    q = _rand(b, h, s, d).requires_grad_(True)
    k = _rand(b, h, s, d, seed=1).requires_grad_(True)
    v = _rand(b, h, s, d, seed=2).requires_grad_(True)
    scale = 1.0 / math.sqrt(d)
    out = flash_attention(q, k, v, scale, causal=True)  # routes to v3
    gy = _rand(b, h, s, d, seed=3)
    out.backward(gy)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ref = _flash_oracle(q2, k2, v2, scale, causal=True)
    ref.backward(gy.float())
    _close(out, ref)
    _close(q.grad, q2.grad, 3e-2)
    _close(k.grad, k2.grad, 3e-2)
    _close(v.grad, v2.grad, 3e-2)


def test_fused_vocab_ce_matches_composite():
    """Fused CE (per-row stats, bwd recompute) vs the composite
    vocab-parallel CE and torch.nn CE, incl. ignored (-100) rows."""
    from fengshen_amd.parallel.cross_entropy import (
        _FusedVocabParallelCrossEntropy, _VocabParallelCrossEntropy,
        _hip_ext)
    torch.manual_seed(0)
    b, s, v = 2, 64, 1024
    logits = _rand(b, s, v, seed=4).requires_grad_(True)
    target = torch.randint(0, v, (b, s), device="cuda")
    target[0, :5] = -100
    ext = _hip_ext(logits)
    assert ext is not None
    loss_f = _FusedVocabParallelCrossEntropy.apply(logits, target, ext)
    valid = (target != -100)
    total_f = (loss_f * valid).sum() / valid.sum()
    total_f.backward()
    g_fused = logits.grad.clone()

    logits2 = logits.detach().clone().requires_grad_(True)
    loss_c = _VocabParallelCrossEntropy.apply(logits2,
                                              target.clamp(min=0))
    total_c = (loss_c * valid).sum() / valid.sum()
    total_c.backward()
    # clamp(min=0) gives ignored rows a fake target: compare valid only
    assert torch.allclose((loss_f * valid), (loss_c * valid), atol=2e-2,
                          rtol=1e-2)
    _close(g_fused[valid], logits2.grad[valid], 3e-2)

    # absolute check vs torch CE
    ref = torch.nn.functional.cross_entropy(
        logits.detach().float().reshape(-1, v), target.reshape(-1),
        ignore_index=-100, reduction="none").view(b, s)
    assert torch.allclose(loss_f * valid, ref * valid, atol=2e-2, rtol=1e-2)


def test_bf16_gemv_matches_eager():
    """Decode GEMV vs fp32 eager for all batch sizes 1..8 and the 13B
    shapes (qkv 15360x5120, down 5120x13824, lm_head-ish 4096x5120)."""
    import torch
    from fengshen_amd.ops import get_ext
    ext = get_ext()
    assert ext is not None
    torch.manual_seed(0)
    for (out, inn) in [(15360, 5120), (5120, 13824), (4096, 5120), (64, 128)]:
        w = torch.randn(out, inn, device="cuda", dtype=torch.bfloat16) * 0.02
        for b in (1, 3, 8):
            x = torch.randn(b, inn, device="cuda", dtype=torch.bfloat16)
            y = ext.bf16_gemv(w, x)
            ref = (x.float() @ w.float().t())
            err = (y.float() - ref).abs().max() / ref.abs().max()
            assert err < 2e-2, (out, inn, b, float(err))


def test_fast_linear_routes_gemv():
    """F_ops.linear uses the GEMV kernel for decode shapes and matches
    F.linear output within bf16 tolerance."""
    import torch
    from fengshen_amd.ops import functional as F_ops
    torch.manual_seed(1)
    w = torch.randn(256, 512, device="cuda", dtype=torch.bfloat16) * 0.05
    x = torch.randn(2, 1, 512, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        y = F_ops.linear(x, w)
    ref = torch.nn.functional.linear(x.float(), w.float())
    assert y.shape == (2, 1, 256)
    assert (y.float() - ref).abs().max() < 0.5
    # grad-enabled path must stay on F.linear (autograd works)
    xg = x.clone().requires_grad_(True)
    y2 = F_ops.linear(xg, w)
    y2.sum().backward()
    assert xg.grad is not None


def test_decode_attn_matches_eager():
    """Fused rope+cache+attention decode kernel vs fp32 eager reference
    (cache write AND context output), d in {64,128}, rope on/off,
    pos at chunk boundaries."""
    import math
    import torch
    from fengshen_amd.ops import get_ext
    from fengshen_amd.ops import functional as F_ops
    ext = get_ext()
    assert ext is not None
    torch.manual_seed(0)
    for (nh, d, L, pos_i, b, rope) in [(4, 128, 256, 130, 2, True),
                                       (4, 64, 128, 0, 1, True),
                                       (2, 128, 192, 127, 1, True),
                                       (2, 128, 64, 63, 1, False)]:
        H = nh * d
        qkv = (torch.randn(b, 3 * H, device="cuda") * 0.5).bfloat16()
        kc = (torch.randn(b, nh, L, d, device="cuda") * 0.5).bfloat16()
        vc = (torch.randn(b, nh, L, d, device="cuda") * 0.5).bfloat16()
        cos, sin = F_ops.build_rope_cache(L, d)
        cos, sin = cos.cuda(), sin.cuda()
        pos = torch.tensor([pos_i], device="cuda")
        scale = 1 / math.sqrt(d)
        kc2, vc2 = kc.clone(), vc.clone()
        ctx = ext.decode_attn(qkv, kc2, vc2, cos, sin, pos, scale, rope)
        q, k, v = qkv.float().view(b, 3, nh, d).unbind(1)
        if rope:
            c, s = cos[pos_i], sin[pos_i]

            def rot(x):
                x1, x2 = x.chunk(2, -1)
                return torch.cat((-x2, x1), -1)
            q = q * c + rot(q) * s
            k = k * c + rot(k) * s
        assert (kc2[:, :, pos_i].float() - k).abs().max() < 0.02
        assert (vc2[:, :, pos_i].float() - v).abs().max() < 0.02
        kr = kc2[:, :, :pos_i + 1].float()
        vr = vc2[:, :, :pos_i + 1].float()
        att = torch.einsum("bhd,bhld->bhl", q, kr) * scale
        ref = torch.einsum("bhl,bhld->bhd", att.softmax(-1), vr)
        err = (ctx.view(b, nh, d).float() - ref).abs().max()
        assert err < 0.03, (nh, d, L, pos_i, float(err))


def test_add_rms_norm_matches_eager():
    """Fused residual-add + RMSNorm (decode) vs fp32 eager."""
    import torch
    from fengshen_amd.ops import get_ext
    ext = get_ext()
    assert ext is not None
    torch.manual_seed(0)
    for (rows, H) in [(1, 5120), (8, 5120), (3, 256)]:
        a = (torch.randn(rows, H, device="cuda") * 0.7).bfloat16()
        b = (torch.randn(rows, H, device="cuda") * 0.7).bfloat16()
        w = (torch.randn(H, device="cuda") * 0.1 + 1).bfloat16()
        s, y = ext.add_rms_norm(a, b, w, 1e-6)
        sf = a.float() + b.float()
        ref = sf * torch.rsqrt(sf.pow(2).mean(-1, keepdim=True) + 1e-6) \
            * w.float()
        assert (s.float() - sf).abs().max() < 0.02
        assert (y.float() - ref).abs().max() < 0.05, (rows, H)
