"""Taiyi-SD path: UNet fwd/bwd shapes, scheduler roundtrip, VAE, full
training step (text encoder -> add_noise -> UNet -> mse)."""
import torch


def test_unet_shapes_and_grad():
    from fengshen_amd.models.taiyi_sd.unet import (
        UNet2DConditionModel, unet_tiny_config)
    torch.manual_seed(0)
    m = UNet2DConditionModel(unet_tiny_config())
    x = torch.randn(2, 4, 16, 16)
    t = torch.randint(0, 1000, (2,))
    ctx = torch.randn(2, 7, 64)
    out = m(x, t, ctx)
    assert out.shape == x.shape
    out.square().mean().backward()
    assert m.conv_in.weight.grad is not None


def test_ddpm_scheduler():
    from fengshen_amd.models.taiyi_sd.scheduler import DDPMScheduler
    s = DDPMScheduler(num_train_timesteps=100)
    x0 = torch.randn(2, 4, 8, 8)
    noise = torch.randn_like(x0)
    t = torch.tensor([0, 99])
    xt = s.add_noise(x0, noise, t)
    # t=0: nearly clean; t=99: mostly noise
    assert (xt[0] - x0[0]).abs().mean() < (xt[1] - x0[1]).abs().mean()
    # one reverse step runs
    out = s.step(noise[0:1], 50, xt[1:2])
    assert out.shape == x0[0:1].shape


def test_vae_roundtrip_shapes():
    from fengshen_amd.models.taiyi_sd.vae import AutoencoderKL
    torch.manual_seed(0)
    vae = AutoencoderKL()
    px = torch.randn(2, 3, 32, 32)
    z = vae.encode(px)
    assert z.shape == (2, 4, 4, 4)  # 8x downsample
    rec = vae.decode(z)
    assert rec.shape == px.shape
    _, loss = vae(px)
    assert loss.isfinite()
    loss.backward()


def test_sd_training_step_end_to_end():
    """the finetune.py:112-152 hot loop: text enc -> noise -> UNet -> mse."""
    from fengshen_amd.models.taiyi_sd import (
        DDPMScheduler, UNet2DConditionModel)
    from fengshen_amd.models.taiyi_sd.unet import unet_tiny_config
    from fengshen_amd.models.megatron_bert.configuration_megatron_bert import (
        bert_tiny_config)
    from fengshen_amd.models.megatron_bert.modeling_megatron_bert import (
        MegatronBertModel)
    torch.manual_seed(0)
    text_encoder = MegatronBertModel(bert_tiny_config(),
                                     add_pooling_layer=False)
    unet = UNet2DConditionModel(unet_tiny_config())
    sched = DDPMScheduler()
    ids = torch.randint(3, 256, (2, 12))
    latents = torch.randn(2, 4, 16, 16)
    noise = torch.randn_like(latents)
    t = torch.randint(0, 1000, (2,))
    noisy = sched.add_noise(latents, noise, t)
    ctx = text_encoder(ids).last_hidden_state
    pred = unet(noisy, t, ctx)
    loss = torch.nn.functional.mse_loss(pred.float(), noise.float())
    assert loss.isfinite()
    loss.backward()
    assert text_encoder.embeddings.word_embeddings.weight.grad is not None
