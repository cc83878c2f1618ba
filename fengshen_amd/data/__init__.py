from fengshen_amd.data.universal_datamodule import UniversalDataModule  # noqa: F401
from fengshen_amd.data.universal_sampler import (  # noqa: F401
    PretrainingSampler,
    PretrainingRandomSampler,
)
