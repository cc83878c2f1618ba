"""Randeng-Della deepVAE: recursive layer-wise latent variables.

Behavioral parity with reference models/deepVAE/deep_vae.py:
- latent_layer recursion (:44-53): z_{<l} evolves by
  tanh(W_hh z_{<l-1} + W_ih z_{l-1});
- AverageSelfAttention pooling of each ENCODER LAYER's hidden states
  (:56-75) — not the CLS token;
- learned priors: posterior q(z_l | pooled_l, z_{<l}) vs prior
  p(z_l | z_{<l}), per-layer Gaussian-vs-Gaussian KL summed (:101-155);
- CVAE mode (:119-127): condition representation pooled separately and
  concatenated into both nets; decoder loss masks the condition prefix;
- inference (:188-222): latents from priors (CVAE) or posteriors, then
  top-k/top-p sampling with repetition penalty.
Decoder injection: each z_l is projected and added to decoder layer l's
input (Della layer-wise latent fusion).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
from transformers import PretrainedConfig, PreTrainedModel
from transformers.utils import ModelOutput

from fengshen_amd.models.gpt2.configuration_gpt2 import GPT2Config
from fengshen_amd.models.gpt2.modeling_gpt2 import GPT2Model
from fengshen_amd.models.layers import parallel_lm_logits
from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
    MegatronBertConfig,
)
from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
    MegatronBertModel,
)
from fengshen_amd.parallel.cross_entropy import vocab_parallel_cross_entropy


class DeepVAEConfig(PretrainedConfig):
    model_type = "fengshen_deep_vae"

    def __init__(self, latent_dim: int = 32, beta_kl: float = 1.0,
                 cvae: bool = False,
                 bos_token_id: int = 1, eos_token_id: int = 2,
                 pad_token_id: int = 0,
                 encoder_config: Optional[dict] = None,
                 decoder_config: Optional[dict] = None, **kw):
        self.latent_dim = latent_dim
        self.beta_kl = beta_kl
        self.cvae = cvae
        self.encoder_config = encoder_config or {}
        self.decoder_config = decoder_config or {}
        super().__init__(bos_token_id=bos_token_id,
                         eos_token_id=eos_token_id,
                         pad_token_id=pad_token_id, **kw)


def deep_vae_tiny_config(**over):
    enc = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, intermediate_size=128,
               max_position_embeddings=64)
    dec = dict(vocab_size=256, hidden_size=64, num_hidden_layers=2,
               num_attention_heads=4, max_position_embeddings=64)
    cfg = dict(latent_dim=16, encoder_config=enc, decoder_config=dec)
    cfg.update(over)
    return DeepVAEConfig(**cfg)


@dataclass
class DeepVAEOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    rec_loss: Optional[torch.Tensor] = None
    kl_loss: Optional[torch.Tensor] = None
    layer_kl: Optional[List[torch.Tensor]] = None


class LatentLayer(nn.Module):
    """z_{<l} = tanh(W_hh z_{<l-1} + W_ih z_{l-1}) (ref :44-53)."""

    def __init__(self, dim: int):
        super().__init__()
        self.W_hh = nn.Linear(dim, dim, bias=False)
        self.W_ih = nn.Linear(dim, dim, bias=False)

    def forward(self, z_lt, z_prev):
        return torch.tanh(self.W_hh(z_lt) + self.W_ih(z_prev))


class AverageSelfAttention(nn.Module):
    """Learned softmax pooling over the sequence (ref :56-75)."""

    def __init__(self, hidden_dim: int):
        super().__init__()
        self.attention_weights = nn.Parameter(torch.empty(hidden_dim))
        nn.init.normal_(self.attention_weights, std=0.02)

    def forward(self, x, mask=None):
        scores = torch.tanh(x.float()) @ self.attention_weights.float()
        if mask is not None:
            scores = scores.masked_fill(~mask, float("-inf"))
        w = torch.softmax(scores, dim=-1)
        pooled = (x.float() * w.unsqueeze(-1)).sum(dim=1).to(x.dtype)
        return pooled, w


def _connect(mean, logvar, sample: bool, beta_logvar: float = 1.0):
    if sample:
        return mean + torch.exp(0.5 * logvar * beta_logvar) \
            * torch.randn_like(mean)
    return mean


def _gaussian_kl(mu_q, logvar_q, mu_p, logvar_p):
    """KL(q || p) for diagonal Gaussians, summed over latent dims."""
    return 0.5 * (
        logvar_p - logvar_q
        + (logvar_q.exp() + (mu_q - mu_p).pow(2)) / logvar_p.exp()
        - 1.0).sum(dim=-1)


class DeepVAEModel(PreTrainedModel):
    config_class = DeepVAEConfig

    def _init_weights(self, module):
        pass

    def __init__(self, config: DeepVAEConfig):
        super().__init__(config)
        enc_cfg = MegatronBertConfig(**config.encoder_config)
        dec_cfg = GPT2Config(**config.decoder_config)
        self.encoder = MegatronBertModel(enc_cfg, add_pooling_layer=False)
        self.decoder = GPT2Model(dec_cfg)
        L = dec_cfg.num_hidden_layers
        d = config.latent_dim
        hdim = enc_cfg.hidden_size
        self.layer_num = L
        self.latent_dim = d
        self.cvae = config.cvae
        # first latent recursion depends on the zero vector: L-1 nets
        self.latent_nets = nn.ModuleList(
            [LatentLayer(d) for _ in range(L - 1)])
        post_in = (hdim + d) if not config.cvae else (2 * hdim + d)
        prior_in = d if not config.cvae else (hdim + d)
        self.posterior_nets = nn.ModuleList(
            [nn.Linear(post_in, 2 * d, bias=False) for _ in range(L)])
        self.prior_nets = nn.ModuleList(
            [nn.Linear(prior_in, 2 * d, bias=False) for _ in range(L)])
        self.pooling = nn.ModuleList(
            [AverageSelfAttention(hdim) for _ in range(L)])
        self.injections = nn.ModuleList(
            [nn.Linear(d, dec_cfg.hidden_size) for _ in range(L)])
        self.post_init()

    # ------------------------------------------------------------------
    def get_latent_vecs(self, layer_hidden_states, sample=True,
                        beta_logvar=1.0, cond_inputs=None):
        """Recursive per-layer posterior/prior latents (ref :112-142)."""
        b = layer_hidden_states[0].shape[0]
        dev = layer_hidden_states[0].device
        z = torch.zeros(b, self.latent_dim, device=dev,
                        dtype=torch.float32)
        prior_zs, post_zs, prior_outs, post_outs = [], [], [], []
        for li in range(self.layer_num):
            hs = layer_hidden_states[li]
            if self.cvae:
                clen = cond_inputs.shape[-1]
                cond_repr, _ = self.pooling[li](hs[:, :clen])
                sent_repr, _ = self.pooling[li](hs[:, clen:])
                prior_in = torch.cat([cond_repr.float(), z], dim=1)
                post_in = torch.cat(
                    [cond_repr.float(), sent_repr.float(), z], dim=1)
            else:
                sent_repr, _ = self.pooling[li](hs)
                prior_in = z
                post_in = torch.cat([sent_repr.float(), z], dim=1)
            prior_out = self.prior_nets[li](prior_in.to(
                self.prior_nets[li].weight.dtype)).float()
            post_out = self.posterior_nets[li](post_in.to(
                self.posterior_nets[li].weight.dtype)).float()
            d = self.latent_dim
            prior_z = _connect(prior_out[:, :d], prior_out[:, d:], sample)
            post_z = _connect(post_out[:, :d], post_out[:, d:], sample,
                              beta_logvar)
            if li != self.layer_num - 1:
                z = self.latent_nets[li](z, post_z)
            prior_zs.append(prior_z)
            post_zs.append(post_z)
            prior_outs.append(prior_out)
            post_outs.append(post_out)
        return prior_zs, post_zs, prior_outs, post_outs

    def get_cond_prior_vecs(self, layer_hidden_states, cond_inputs,
                            sample=True, beta_logvar=1.0):
        """CVAE inference latents from priors only (ref :168-186)."""
        b = layer_hidden_states[0].shape[0]
        dev = layer_hidden_states[0].device
        z = torch.zeros(b, self.latent_dim, device=dev,
                        dtype=torch.float32)
        prior_zs = []
        for li in range(self.layer_num):
            clen = cond_inputs.shape[-1]
            cond_repr, _ = self.pooling[li](
                layer_hidden_states[li][:, :clen])
            prior_out = self.prior_nets[li](
                torch.cat([cond_repr.float(), z], dim=1).to(
                    self.prior_nets[li].weight.dtype)).float()
            d = self.latent_dim
            prior_z = _connect(prior_out[:, :d], prior_out[:, d:], sample,
                               beta_logvar)
            if li != self.layer_num - 1:
                z = self.latent_nets[li](z, prior_z)
            prior_zs.append(prior_z)
        return prior_zs

    def _decode_logits(self, input_ids, latent_vecs):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        h = self.decoder.wte(input_ids) + self.decoder.wpe(pos)
        h = self.decoder.drop(h)
        for layer, z, inj in zip(self.decoder.h, latent_vecs,
                                 self.injections):
            h = h + inj(z.to(inj.weight.dtype)).unsqueeze(1).to(h.dtype)
            h = layer(h)
        h = self.decoder.ln_f(h)
        return parallel_lm_logits(h, self.decoder.wte.weight,
                                  parallel_output=True)

    # ------------------------------------------------------------------
    def forward(self, input_ids, attention_mask=None, labels=None,
                cond_inputs=None, beta_kl: Optional[float] = None, **_kw):
        enc_inputs = (torch.cat([cond_inputs, input_ids], dim=1)
                      if self.cvae else input_ids)
        enc = self.encoder(enc_inputs, output_hidden_states=True)
        # per-layer states, embedding output excluded (ref :159-161)
        layer_states = enc.hidden_states[1:]
        prior_zs, post_zs, prior_outs, post_outs = self.get_latent_vecs(
            layer_states, sample=self.training, cond_inputs=cond_inputs)

        # per-layer Gaussian KL against the LEARNED prior (ref :144-155)
        layer_kl = []
        kl = None
        d = self.latent_dim
        for po, qo in zip(prior_outs, post_outs):
            k = _gaussian_kl(qo[:, :d], qo[:, d:], po[:, :d], po[:, d:])
            layer_kl.append(k.mean())
            kl = k if kl is None else kl + k
        kl = kl.mean()

        dec_inputs = (torch.cat([cond_inputs, input_ids], dim=1)
                      if self.cvae else input_ids)
        logits = self._decode_logits(dec_inputs, post_zs)
        loss = rec = None
        if labels is not None:
            tgt = (torch.cat([torch.full_like(cond_inputs, -100), labels],
                             dim=1) if self.cvae else labels)
            shift_logits = logits[:, :-1].contiguous()
            shift_labels = tgt[:, 1:].contiguous()
            per_token = vocab_parallel_cross_entropy(
                shift_logits, shift_labels.clamp(min=0))
            valid = (shift_labels != -100)
            rec = (per_token * valid).sum() / valid.sum().clamp(min=1)
            beta = self.config.beta_kl if beta_kl is None else beta_kl
            loss = rec + beta * kl
        return DeepVAEOutput(loss=loss, rec_loss=rec, kl_loss=kl,
                             layer_kl=layer_kl)

    # ------------------------------------------------------------------
    @torch.no_grad()
    def inference(self, inputs, top_p=0.9, max_length=32, top_k=0,
                  temperature=1.0, repetition_penalty=1.0, sample=False,
                  beta_logvar=1.0):
        """Ref :188-222: latents from priors (CVAE) / posteriors, then
        sampled autoregressive decode."""
        from fengshen_amd.utils.transfo_xl_utils import (
            enforce_repetition_penalty, top_k_logits)
        enc = self.encoder(inputs, output_hidden_states=True)
        layer_states = enc.hidden_states[1:]
        if self.cvae:
            latents = self.get_cond_prior_vecs(
                layer_states, inputs, sample=sample,
                beta_logvar=beta_logvar)
            generated = inputs
        else:
            _pz, post_zs, _po, _qo = self.get_latent_vecs(
                layer_states, sample=sample, beta_logvar=beta_logvar)
            latents = post_zs
            generated = torch.full((inputs.shape[0], 1),
                                   self.config.bos_token_id,
                                   dtype=torch.long, device=inputs.device)
        for _ in range(max_length):
            logits = self._decode_logits(generated, latents)
            nxt = logits[:, -1, :].float() / temperature
            nxt = top_k_logits(nxt, top_k=top_k, top_p=top_p)
            probs = torch.softmax(nxt, dim=-1)
            if repetition_penalty != 1.0:
                for bi in range(probs.shape[0]):
                    enforce_repetition_penalty(
                        probs[bi], generated[bi].tolist(),
                        repetition_penalty)
            tok = torch.multinomial(probs.clamp(min=0), 1)
            generated = torch.cat([generated, tok], dim=1)
            if (tok.squeeze(1) == self.config.eos_token_id).all():
                break
        return generated
