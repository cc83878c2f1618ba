from fengshen_amd.trainer.module import FengshenModule  # noqa: F401
from fengshen_amd.trainer.trainer import Trainer  # noqa: F401
from fengshen_amd.trainer.callbacks import (  # noqa: F401
    Callback,
    LearningRateMonitor,
    ThroughputMonitor,
)
