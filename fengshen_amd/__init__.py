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

# Reference-style top-level model exports (the reference README documents
# `from fengshen import LongformerModel` etc.) — lazy so importing
# fengshen_amd stays light.
_LAZY_EXPORTS = {
    "LongformerModel": "fengshen_amd.models.longformer.modeling_longformer",
    "LongformerConfig": "fengshen_amd.models.longformer.modeling_longformer",
    "LongformerForMaskedLM":
        "fengshen_amd.models.longformer.modeling_longformer",
    "RoFormerModel": "fengshen_amd.models.roformer.modeling_roformer",
    "RoFormerConfig": "fengshen_amd.models.roformer.modeling_roformer",
    "LlamaForCausalLM": "fengshen_amd.models.llama.modeling_llama",
    "LlamaConfig": "fengshen_amd.models.llama.configuration_llama",
    "MegatronBertModel":
        "fengshen_amd.models.megatron_bert.modeling_megatron_bert",
    "MegatronBertForPreTraining":
        "fengshen_amd.models.megatron_bert.modeling_megatron_bert",
    "GPT2LMHeadModel": "fengshen_amd.models.gpt2.modeling_gpt2",
    "T5ForConditionalGeneration": "fengshen_amd.models.t5.modeling_t5",
    "UbertModel": "fengshen_amd.models.ubert.modeling_ubert",
    "UniMCModel": "fengshen_amd.models.unimc.modeling_unimc",
}


def __getattr__(name):
    mod = _LAZY_EXPORTS.get(name)
    if mod is not None:
        import importlib
        return getattr(importlib.import_module(mod), name)
    raise AttributeError(f"module 'fengshen_amd' has no attribute {name!r}")


def __dir__():
    return sorted(list(globals()) + list(_LAZY_EXPORTS))
