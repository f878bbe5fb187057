from .model import (  # noqa: F401
    GPT,
    Block,
    CausalSelfAttention,
    KVCachePool,
    RMSNorm,
    apply_rope,
    build_rope_cache,
)
from .sampling import sample, sample_top_p  # noqa: F401
from .stages import SecondaryStage, StageBase, StarterStage, build_stage  # noqa: F401
