from fengshen_amd.utils.utils import report_memory, chinese_char_tokenize  # noqa: F401
from fengshen_amd.utils.universal_checkpoint import UniversalCheckpoint  # noqa: F401
