from fengshen_amd.metric.metric import (  # noqa: F401
    metrics_mlm_acc,
    EntityScore,
    SeqEntityScore,
)
from fengshen_amd.metric.utils_ner import get_entities  # noqa: F401
