from fengshen_amd.models.taiyi_sd.unet import UNet2DConditionModel, UNetConfig  # noqa: F401
from fengshen_amd.models.taiyi_sd.scheduler import DDPMScheduler  # noqa: F401
from fengshen_amd.models.taiyi_sd.vae import AutoencoderKL  # noqa: F401
