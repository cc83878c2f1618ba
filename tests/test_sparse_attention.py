"""Block-sparse attention: layouts + numerics vs dense reference."""
import math
import types

import pytest
import torch

from fengshen_amd.ops.sparse_attention import (
    BigBirdSparsityConfig,
    BSLongformerSparsityConfig,
    FixedSparsityConfig,
    LocalSlidingWindowSparsityConfig,
    SparseSelfAttention,
    VariableSparsityConfig,
    configure_sparse_attention,
)


def dense_ref(q, k, v, block_mask, block, causal=True):
    """fp32 dense attention under the expanded block layout."""
    b, np_, s, hn = q.shape
    scores = (q.float() @ k.float().transpose(-1, -2)) / math.sqrt(hn)
    mask = block_mask.repeat_interleave(block, 1).repeat_interleave(block, 2)
    scores = scores.masked_fill(~mask[None], torch.finfo(torch.float32).min)
    if causal:
        cm = torch.triu(torch.ones(s, s, dtype=torch.bool), 1)
        scores = scores.masked_fill(cm[None, None], torch.finfo(torch.float32).min)
    return (torch.softmax(scores, -1) @ v.float()).to(v.dtype)


@pytest.mark.parametrize("cfg_cls,kw", [
    (LocalSlidingWindowSparsityConfig, dict(num_sliding_window_blocks=2)),
    (FixedSparsityConfig, dict(num_local_blocks=2, num_global_blocks=1)),
    (BigBirdSparsityConfig, dict(num_random_blocks=1,
                                 num_sliding_window_blocks=2)),
    (BSLongformerSparsityConfig, dict(num_sliding_window_blocks=2,
                                      global_block_indices=[0])),
    (VariableSparsityConfig, dict(local_window_blocks=[2],
                                  global_block_indices=[0])),
])
def test_sparse_matches_dense_under_same_layout(cfg_cls, kw):
    torch.manual_seed(0)
    cfg = cfg_cls(num_heads=2, block=8, **kw)
    attn = SparseSelfAttention(cfg)
    b, np_, s, hn = 2, 2, 64, 16
    q, k, v = [torch.randn(b, np_, s, hn) for _ in range(3)]
    out = attn(q, k, v)
    layout = cfg.make_layout(s)
    # dense reference per head (layouts can differ per head)
    ref = dense_ref(q, k, v, layout, cfg.block)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_layout_causal_and_diag():
    for cfg in [LocalSlidingWindowSparsityConfig(4, block=8,
                                                 num_sliding_window_blocks=3),
                BigBirdSparsityConfig(4, block=8),
                FixedSparsityConfig(4, block=8, num_local_blocks=4)]:
        lay = cfg.make_layout(64)
        assert lay.shape == (4, 8, 8)
        # unidirectional: no block above the diagonal
        assert not lay.triu(1).any()
        # every query block can at least see its own diagonal block
        assert bool(lay.diagonal(dim1=-2, dim2=-1).all())


def test_sparse_is_actually_sparse_and_full_matches_dense():
    cfg = LocalSlidingWindowSparsityConfig(2, block=8,
                                           num_sliding_window_blocks=2)
    assert SparseSelfAttention(cfg).density(128) < 0.25
    # a window covering everything == plain causal attention
    full = LocalSlidingWindowSparsityConfig(2, block=8,
                                            num_sliding_window_blocks=64)
    attn = SparseSelfAttention(full)
    q, k, v = [torch.randn(1, 2, 64, 16) for _ in range(3)]
    out = attn(q, k, v)
    ones = torch.ones(2, 8, 8, dtype=torch.bool).tril()
    ref = dense_ref(q, k, v, ones, 8)
    assert torch.allclose(out, ref, atol=1e-5)


def test_padding_mask():
    cfg = LocalSlidingWindowSparsityConfig(2, block=8,
                                           num_sliding_window_blocks=8)
    attn = SparseSelfAttention(cfg)
    b, s = 2, 32
    q, k, v = [torch.randn(b, 2, s, 16) for _ in range(3)]
    am = torch.ones(b, s, dtype=torch.long)
    am[:, -8:] = 0
    out = attn(q, k, v, attention_mask=am)
    # masked-out key positions must not influence earlier queries:
    v2 = v.clone()
    v2[:, :, -8:] = 99.0
    out2 = attn(q, k, v2, attention_mask=am)
    assert torch.allclose(out[:, :, :-8], out2[:, :, :-8], atol=1e-5)


def test_configure_sparse_attention_parity_surface():
    config = types.SimpleNamespace(
        sparsity_config={"block": 8, "num_local_blocks": 2},
        max_position_embeddings=128)
    for t in ["sparse_fixed", "sparse_variable", "local", "bigbird",
              "bslongformer"]:
        m = configure_sparse_attention(config, t, num_attention_heads=4)
        assert isinstance(m, SparseSelfAttention)
        lay = m.layout(64, "cpu")
        assert lay.shape[0] == 4
    with pytest.raises(ValueError):
        configure_sparse_attention(config, "nope", 4)


def test_sparse_attention_backward():
    """blockwise path is autograd-clean and matches dense-masked grads."""
    torch.manual_seed(0)
    cfg = LocalSlidingWindowSparsityConfig(2, block=8,
                                           num_sliding_window_blocks=2)
    attn = SparseSelfAttention(cfg)
    q1, k1, v1 = [torch.randn(1, 2, 32, 8, requires_grad=True)
                  for _ in range(3)]
    attn(q1, k1, v1).sum().backward()
    q2, k2, v2 = [t.detach().clone().requires_grad_(True)
                  for t in (q1, k1, v1)]
    dense_ref(q2, k2, v2, cfg.make_layout(32), cfg.block).sum().backward()
    for a, b in [(q1, q2), (k1, k2), (v1, v2)]:
        assert torch.allclose(a.grad, b.grad, atol=1e-5)
