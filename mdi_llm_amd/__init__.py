"""mdi_llm_amd — an MI355X-native model-distributed inference framework.

A from-scratch rebuild of the capabilities of MDI-LLM (davmacario/MDI-LLM,
surveyed in /root/repo/SURVEY.md): recurrent pipeline-parallel LLM inference
across the GPUs of one 8×MI355X node, with hand-written CDNA4 HIP kernels
for the decode path, RCCL point-to-point activation passing over xGMI, and
litGPT-compatible checkpoints/chunks, plus single-device generation, chat,
and DDP training.
"""

__version__ = "0.1.0"

from .config import ModelConfig, name_to_config  # noqa: F401
from .models.model import GPT, KVCachePool  # noqa: F401
from .models.stages import StarterStage, SecondaryStage, build_stage  # noqa: F401
from .models.sampling import sample  # noqa: F401
