from fengshen_amd.models.clip.modeling_taiyi_clip import (  # noqa: F401
    TaiyiCLIPConfig,
    TaiyiCLIPModel,
    clip_contrastive_loss,
)
