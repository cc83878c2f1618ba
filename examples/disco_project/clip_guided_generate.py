"""CLIP-guided diffusion generation (disco_project equivalent).

Behavioral parity: reference examples/disco_project vendors OpenAI
guided-diffusion for CLIP-guided art generation.  Same mechanism here on
our own stack: at every reverse-DDPM step, the predicted clean image x0
is scored by Taiyi-CLIP against the text prompt and the sample is nudged
along the similarity gradient (classifier-guidance style), using
models/taiyi_sd's UNet+DDPMScheduler and models/clip's dual tower.

Run:  python clip_guided_generate.py [--steps 20 --guidance_scale 50]
Tiny random-weight models (smoke mode; real checkpoints load via
from_pretrained).
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(
    os.path.dirname(__file__), "..", "..")))


import argparse

import torch
import torch.nn.functional as F

from fengshen_amd.models.clip.modeling_taiyi_clip import (
    TaiyiCLIPModel,
    taiyi_clip_tiny_config,
)
from fengshen_amd.models.taiyi_sd.scheduler import DDPMScheduler
from fengshen_amd.models.taiyi_sd.unet import UNet2DConditionModel, unet_tiny_config
from fengshen_amd.tokenizer import SimpleCharTokenizer


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--prompt", default="山水画")
    parser.add_argument("--steps", default=20, type=int)
    parser.add_argument("--guidance_scale", default=50.0, type=float)
    parser.add_argument("--size", default=32, type=int)
    args = parser.parse_args()

    torch.manual_seed(0)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    unet = UNet2DConditionModel(unet_tiny_config()).float().to(device).eval()
    clip = TaiyiCLIPModel(taiyi_clip_tiny_config()).float().to(device).eval()
    tokenizer = SimpleCharTokenizer()
    sched = DDPMScheduler(num_train_timesteps=1000)

    ids = torch.tensor([tokenizer.encode(args.prompt)], device=device)
    with torch.no_grad():
        text_feat = F.normalize(clip.get_text_features(ids), dim=-1)
        # conditioning for the UNet's cross attention
        cond = torch.randn(1, 8, unet.config.cross_attention_dim,
                           device=device)

    x = torch.randn(1, unet.config.in_channels, args.size, args.size,
                    device=device)
    timesteps = torch.linspace(999, 0, args.steps).long()
    for t in timesteps:
        with torch.no_grad():
            eps = unet(x, t[None].to(device), cond)
        # CLIP guidance: grade the predicted x0, push x toward similarity
        ac = sched.alphas_cumprod[t]
        with torch.enable_grad():
            xg = x.detach().requires_grad_(True)
            x0 = (xg - (1 - ac).sqrt() * eps) / ac.sqrt()
            img = F.interpolate(x0[:, :3].clamp(-1, 1),
                                size=clip.config.image_size)
            img_feat = F.normalize(clip.get_image_features(img), dim=-1)
            sim = (img_feat * text_feat).sum()
            (grad,) = torch.autograd.grad(sim, xg)
            sim = sim.detach()
        x = x + args.guidance_scale * (1 - ac) * grad
        x = sched.step(eps, int(t), x)
        if int(t) % 200 == 0:
            print(f"t={int(t):4d}  clip_sim={float(sim):+.4f}  "
                  f"x std={float(x.std()):.3f}")
    print("final sample stats: mean %.4f std %.4f" %
          (float(x.mean()), float(x.std())))


if __name__ == "__main__":
    main()
