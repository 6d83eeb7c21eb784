from torchdistx_amd.models.configs import (  # noqa: F401
    LLAMA3_405B,
    CONFIGS,
    GPT2_XL,
    LLAMA3_8B,
    LLAMA3_70B,
    MIXTRAL_8X22B,
    TINY,
    TINY_GPT2,
    TINY_MOE,
    TransformerConfig,
)
from torchdistx_amd.models.transformer import (  # noqa: F401
    TransformerLM,
    build_model,
)
