"""hipGraph-captured greedy decode for LLaMA serving.

Eager decode of a 13B model launches ~400 kernels per token (40 layers x
~10 ops); at ~8 us a launch that is several ms of pure launch overhead on
a path that is otherwise weight-bandwidth-bound.  This engine captures ONE
self-advancing decode step into a hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) and replays it per token:

- static KV cache [b, heads, max_len, hd] per layer, in-place index_copy_
  at a device position tensor;
- rope cos/sin gathered by the position tensor (no host sync);
- attention over the full static window with an additive (pos-driven) mask;
- greedy argmax feeds the token buffer consumed by the NEXT replay, so a
  whole max_new_tokens decode is just N graph replays with zero host
  round-trips.

Ref context: examples/ziya_inference (the reference's only throughput
table) and PERF_ROADMAP item 5 (hipGraph capture of the decode loop).
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch

from fengshen_amd.ops import functional as F_ops


class GraphedDecoder:
    def __init__(self, model, batch: int = 1, max_len: int = 512,
                 max_new_tokens: int = 256):
        self.model = model.eval()
        cfg = model.config
        dev = next(model.parameters()).device
        dt = next(model.parameters()).dtype
        self.dev, self.dt = dev, dt
        self.batch = batch
        self.max_len = max_len
        self.max_new = max_new_tokens
        L = cfg.num_hidden_layers
        nh = cfg.num_attention_heads
        hd = cfg.hidden_size // nh
        self.nh, self.hd = nh, hd
        self.scale = 1.0 / math.sqrt(hd)

        # static state
        self.k_cache = [torch.zeros(batch, nh, max_len, hd, device=dev,
                                    dtype=dt) for _ in range(L)]
        self.v_cache = [torch.zeros(batch, nh, max_len, hd, device=dev,
                                    dtype=dt) for _ in range(L)]
        self.tok = torch.zeros(batch, 1, device=dev, dtype=torch.long)
        self.pos = torch.zeros(1, device=dev, dtype=torch.long)
        self.step_i = torch.zeros(1, device=dev, dtype=torch.long)
        self.out_tokens = torch.zeros(batch, max_new_tokens, device=dev,
                                      dtype=torch.long)
        self.ar = torch.arange(max_len, device=dev)
        cos, sin = F_ops.build_rope_cache(
            max_len, hd, base=getattr(cfg, "rope_base", 10000.0))
        self.cos = cos.to(dev)
        self.sin = sin.to(dev)
        self._graph: Optional[torch.cuda.CUDAGraph] = None

    # ------------------------------------------------------------------
    @staticmethod
    def _rot_half(x):
        x1, x2 = x.chunk(2, dim=-1)
        return torch.cat((-x2, x1), dim=-1)

    def _step(self):
        """Graph-safe single-token decode; advances tok/pos/step_i."""
        m = self.model.model
        b, nh, hd = self.batch, self.nh, self.hd
        pos = self.pos
        h = m.embed_tokens(self.tok)  # [b, 1, H]
        c = self.cos.index_select(0, pos).to(self.dt)  # [1, hd]
        s = self.sin.index_select(0, pos).to(self.dt)
        # additive mask over the static window: visible iff idx <= pos
        amask = torch.where(self.ar <= pos,
                            torch.zeros((), device=self.dev),
                            torch.full((), float("-inf"), device=self.dev))
        for i, layer in enumerate(m.layers):
            res = h
            x = layer.input_norm(h)
            qkv = layer.attention.qkv_proj(x)
            if isinstance(qkv, tuple):
                qkv = qkv[0]
            q, k, v = qkv.chunk(3, dim=-1)
            q = q.view(b, 1, nh, hd).transpose(1, 2)  # [b, nh, 1, hd]
            k = k.view(b, 1, nh, hd).transpose(1, 2)
            v = v.view(b, 1, nh, hd).transpose(1, 2)
            if layer.attention.rotary:
                q = q * c + self._rot_half(q) * s
                k = k * c + self._rot_half(k) * s
            self.k_cache[i].index_copy_(2, pos, k)
            self.v_cache[i].index_copy_(2, pos, v)
            scores = (q.float() @
                      self.k_cache[i].float().transpose(-1, -2)) * self.scale
            probs = torch.softmax(scores + amask, dim=-1)
            ctx = (probs.to(self.dt) @ self.v_cache[i])  # [b, nh, 1, hd]
            ctx = ctx.transpose(1, 2).reshape(b, 1, nh * hd)
            out = layer.attention.out_proj(ctx)
            if isinstance(out, tuple):
                out = out[0]
            h = res + out
            res = h
            x = layer.post_attention_norm(h)
            mlp = layer.mlp(x)
            if isinstance(mlp, tuple):
                mlp = mlp[0]
            h = res + mlp
        h = m.norm(h)
        logits = self.model.lm_head(h)
        if isinstance(logits, tuple):
            logits = logits[0]
        nxt = logits[:, -1, :].argmax(dim=-1, keepdim=True)  # [b, 1]
        self.out_tokens.index_copy_(1, self.step_i, nxt)
        self.tok.copy_(nxt)
        self.pos.add_(1)
        self.step_i.add_(1)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _prefill(self, prompt_ids: torch.Tensor):
        """Eager prefill through the model's normal cache path, then copy
        the per-layer K/V into the static buffers."""
        from transformers.cache_utils import DynamicCache
        assert prompt_ids.shape[0] == self.batch, (
            f"decoder captured for batch {self.batch}, "
            f"got {prompt_ids.shape[0]}")
        plen = prompt_ids.shape[1]
        assert plen + self.max_new <= self.max_len, "window too small"
        cache = DynamicCache()
        out = self.model(input_ids=prompt_ids.to(self.dev),
                         past_key_values=cache, use_cache=True)
        for i in range(len(self.k_cache)):
            layer_cache = cache.layers[i]  # transformers >= 5 DynamicCache
            k, v = layer_cache.keys, layer_cache.values
            self.k_cache[i][:, :, :plen].copy_(k)
            self.v_cache[i][:, :, :plen].copy_(v)
            self.k_cache[i][:, :, plen:].zero_()
            self.v_cache[i][:, :, plen:].zero_()
        self.pos.fill_(plen)
        self.step_i.zero_()
        self.tok.copy_(out.logits[:, -1, :].argmax(-1, keepdim=True))

    def _capture(self):
        # warm up twice on a side stream (allocator + kernels), then capture
        st = torch.cuda.Stream()
        st.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(st):
            for _ in range(2):
                self._step()
        torch.cuda.current_stream().wait_stream(st)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._step()
        self._graph = g

    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate(self, prompt_ids: torch.Tensor,
                 max_new_tokens: Optional[int] = None,
                 eos_token_id: Optional[int] = None) -> torch.Tensor:
        """Greedy decode via graph replays.  Returns [b, plen + n]."""
        n = min(max_new_tokens or self.max_new, self.max_new)
        if self._graph is None:
            # capture against scratch state, then restore via real prefill
            self.pos.fill_(prompt_ids.shape[1])
            self._capture()
        self._prefill(prompt_ids)
        # first generated token came from prefill logits
        self.out_tokens.index_copy_(
            1, self.step_i, self.tok)
        self.step_i.add_(1)
        for _ in range(n - 1):
            self._graph.replay()
        toks = self.out_tokens[:, :n].clone()
        if eos_token_id is not None:
            # trim after the first EOS per row (host-side, post hoc)
            done = (toks == eos_token_id).cumsum(dim=1) > 0
            toks = toks.masked_fill(done, eos_token_id)
        return torch.cat([prompt_ids.to(self.dev), toks], dim=1)
