"""fengshen_amd — MI355X-native Chinese foundation-model training framework.

A ground-up rebuild of the capabilities of IDEA-CCNL/Fengshenbang-LM for AMD
Instinct MI355X (gfx950, CDNA4): PyTorch-ROCm + hand-written HIP kernels for
the hot ops + RCCL over xGMI for all collectives.  Unlike the reference
(which composes PyTorch-Lightning + DeepSpeed + HF), this framework owns its
training loop, its ZeRO-1/2/3 implementation, its tensor-parallel layer
library and its fused kernels.

Layer map (mirrors reference SURVEY.md §1):
  ops/       — HIP/CDNA4 fused kernels + eager fallbacks (ref: fused_kernels/)
  parallel/  — process groups, TP layers, native ZeRO    (ref: mpu/ + DeepSpeed)
  trainer/   — module API, fit loop, callbacks           (ref: PL Trainer)
  data/      — UniversalDataModule, samplers, collators  (ref: fengshen/data)
  models/    — model zoo                                  (ref: fengshen/models)
  pipelines/ — task pipelines                             (ref: fengshen/pipelines)
  cli/       — fengshen-pipeline CLI                      (ref: fengshen/cli)
  serving/   — FastAPI serving                            (ref: fengshen/API)
"""

__version__ = "0.1.0"

from fengshen_amd.trainer.module import FengshenModule  # noqa: F401
from fengshen_amd.trainer.trainer import Trainer  # noqa: F401
from fengshen_amd.data.universal_datamodule import UniversalDataModule  # noqa: F401
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint  # noqa: F401
