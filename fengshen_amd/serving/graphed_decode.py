"""hipGraph-captured decode for LLaMA serving (greedy or sampled).

Eager decode of a 13B model launches ~400 kernels per token (40 layers x
~10 ops); at ~8 us a launch that is several ms of pure launch overhead on
a path that is otherwise weight-bandwidth-bound.  This engine captures ONE
self-advancing decode step into a hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) and replays it per token:

- static KV cache [b, heads, max_len, hd] per layer, position driven by a
  device tensor (no host sync anywhere in the loop);
- inside the graph the hot path is hand-written HIP (kernels.hip):
  bf16_gemv (weights streamed at 5.7-6.8 TB/s vs hipBLASLt's ~2.5-3 at
  M=1), decode_attn (rope + cache write + online-softmax attention in one
  launch per layer) and add_rms_norm (residual add fused into the norm);
  eager torch fallback when the extension/dtype/head-dim doesn't fit;
- greedy argmax — or gumbel-max top-k/top-p sampling over a pre-generated
  seeded noise buffer indexed by the step tensor — feeds the token buffer
  consumed by the NEXT replay, so a whole max_new_tokens decode is just N
  graph replays with zero host round-trips.

Measured (13B, 1x MI355X): 5.92 ms/token bf16, 5.06 int8, 0.99
ms/seq/token at batch 8 (profiles/decode_kernels_r2.md).

Ref context: examples/ziya_inference (the reference's only throughput
table) and PERF_ROADMAP item 5 (hipGraph capture of the decode loop).
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from fengshen_amd.ops import functional as F_ops


def filter_logits(logits: torch.Tensor, temperature: torch.Tensor,
                  top_k: int, top_p: torch.Tensor) -> torch.Tensor:
    """HF-convention sampling filter, graph-safe (static shapes only).

    temperature / top_p are device scalars so they can be retuned between
    generates without recapturing; top_k is shape-affecting and fixed at
    capture.  Order matches the reference's sample_sequence helpers
    (fengshen/utils/transfo_xl_utils.py top_k_logits): temperature, then
    top-k, then nucleus.
    """
    # NOTE: no torch.tensor(..., device=...) here — an H2D copy is
    # hipErrorStreamCaptureUnsupported inside graph capture; masked_fill
    # with a python scalar lowers to a capture-safe fill kernel.
    lg = logits.float() / temperature
    if top_k > 0:
        kth = lg.topk(top_k, dim=-1).values[..., -1:]
        lg = lg.masked_fill(lg < kth, float("-inf"))
    # nucleus: drop tokens once the cumulative prob BEFORE them >= top_p
    # (the top-1 token always survives).  top_p >= 1 keeps everything.
    srt, idx = lg.sort(dim=-1, descending=True)
    probs = torch.softmax(srt, dim=-1)
    shifted_cum = probs.cumsum(dim=-1) - probs
    srt = srt.masked_fill(shifted_cum >= top_p, float("-inf"))
    return torch.full_like(lg, float("-inf")).scatter(-1, idx, srt)


class GraphedDecoder:
    def __init__(self, model, batch: int = 1, max_len: int = 512,
                 max_new_tokens: int = 256, do_sample: bool = False,
                 top_k: int = 0, top_p: float = 1.0,
                 temperature: float = 1.0):
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
        # sampling state: gumbel-max with a pre-generated noise buffer
        # indexed by step_i — no RNG state inside the graph, and a fixed
        # torch seed gives reproducible samples.
        self.do_sample = do_sample
        self.top_k = top_k
        self.top_p = torch.tensor(float(top_p), device=dev)
        self.temperature = torch.tensor(float(temperature), device=dev)
        if do_sample:
            self.gumbel = torch.zeros(max_new_tokens, batch,
                                      cfg.vocab_size, device=dev)
        # fused rope+cache+attention kernel (one launch per layer instead
        # of ~10 eager ops); requires bf16 and head dim 64/128
        from fengshen_amd.ops import get_ext
        import os
        self._ext = get_ext() if dev.type == "cuda" else None
        self._fused_attn = (
            self._ext is not None and hd in (64, 128)
            and dt == torch.bfloat16
            and os.environ.get("FENGSHEN_AMD_FORCE_EAGER") != "1"
            and os.environ.get("FENGSHEN_DECODE_FUSED", "1") == "1")
        self._graph: Optional[torch.cuda.CUDAGraph] = None

    # ------------------------------------------------------------------
    @staticmethod
    def _rot_half(x):
        x1, x2 = x.chunk(2, dim=-1)
        return torch.cat((-x2, x1), dim=-1)

    def _add_norm(self, a, b_, norm):
        """(a + b_, norm(a + b_)) — fused into one kernel when norm is an
        RMSNorm on the HIP path (the 13B decode graph has 81 such pairs
        per token)."""
        if self._fused_attn and type(norm).__name__ == "RMSNorm":
            s, y = self._ext.add_rms_norm(a.contiguous(), b_.contiguous(),
                                          norm.weight, norm.eps)
            return s.view_as(a), y.view_as(a)
        s = a + b_
        return s, norm(s)

    def _step(self):
        """Graph-safe single-token decode; advances tok/pos/step_i."""
        m = self.model.model
        b, nh, hd = self.batch, self.nh, self.hd
        pos = self.pos
        h = m.embed_tokens(self.tok)  # [b, 1, H]
        if not self._fused_attn:
            c = self.cos.index_select(0, pos).to(self.dt)  # [1, hd]
            s = self.sin.index_select(0, pos).to(self.dt)
            # additive mask over the static window: visible iff idx <= pos
            amask = torch.where(self.ar <= pos,
                                torch.zeros((), device=self.dev),
                                torch.full((), float("-inf"),
                                           device=self.dev))
        if self._fused_attn:
            # residual adds ride the next norm's kernel (add_rms_norm),
            # including across layer boundaries and into the final norm
            pend = None  # (residual, branch) awaiting add+norm
            for i, layer in enumerate(m.layers):
                if pend is None:
                    x = layer.input_norm(h)
                else:
                    h, x = self._add_norm(pend[0], pend[1],
                                          layer.input_norm)
                qkv = layer.attention.qkv_proj(x)
                if isinstance(qkv, tuple):
                    qkv = qkv[0]
                ctx = self._ext.decode_attn(
                    qkv.view(b, 3 * nh * hd), self.k_cache[i],
                    self.v_cache[i], self.cos, self.sin, pos,
                    self.scale, bool(layer.attention.rotary))
                out = layer.attention.out_proj(ctx.view(b, 1, nh * hd))
                if isinstance(out, tuple):
                    out = out[0]
                h, x = self._add_norm(h, out, layer.post_attention_norm)
                mlp = layer.mlp(x)
                if isinstance(mlp, tuple):
                    mlp = mlp[0]
                pend = (h, mlp)
            _, h = self._add_norm(pend[0], pend[1], m.norm)
            self._emit(h)
            return
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
        self._emit(h)

    def _emit(self, h):
        """lm_head -> pick -> record token and advance pos/step_i."""
        logits = self.model.lm_head(h)
        if isinstance(logits, tuple):
            logits = logits[0]
        nxt = self._pick(logits[:, -1, :], self.step_i)  # [b, 1]
        self.out_tokens.index_copy_(1, self.step_i, nxt)
        self.tok.copy_(nxt)
        self.pos.add_(1)
        self.step_i.add_(1)

    def _pick(self, last_logits: torch.Tensor,
              step: torch.Tensor) -> torch.Tensor:
        """Greedy argmax, or gumbel-max sample over the filtered logits.
        `step` indexes the pre-generated noise buffer; graph-safe."""
        if not self.do_sample:
            return last_logits.argmax(dim=-1, keepdim=True)
        lg = filter_logits(last_logits, self.temperature,
                           self.top_k, self.top_p)
        # clamp: warmup/capture advance step_i past the real range
        g = self.gumbel.index_select(
            0, step.clamp(max=self.max_new - 1)).squeeze(0)  # [b, vocab]
        return (lg + g).argmax(dim=-1, keepdim=True)

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
        self.tok.copy_(self._pick(out.logits[:, -1, :], self.step_i))

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
        if self.do_sample:
            # fresh Gumbel(0,1) noise per call: -log(Exp(1)); seeded via
            # torch.manual_seed for reproducible sampling.
            self.gumbel.exponential_().log_().neg_()
        if self._graph is None:
            # capture against scratch state, then restore via real prefill.
            # The 2 warmup steps + capture step advance pos by 3 total, so
            # start low enough that every scratch write stays inside the
            # static window even for tiny max_new_tokens.
            self.pos.fill_(min(prompt_ids.shape[1], self.max_len - 3))
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
