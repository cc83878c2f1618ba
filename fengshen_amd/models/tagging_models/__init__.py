from fengshen_amd.models.tagging_models.crf import CRF  # noqa: F401
from fengshen_amd.models.tagging_models.bert_for_tagging import (  # noqa: F401
    BertLinear,
    BertCrf,
    BertSpan,
    BertBiaffine,
)
