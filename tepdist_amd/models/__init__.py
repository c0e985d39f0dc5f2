from tepdist_amd.models.configs import (  # noqa: F401
    GPT2_CONFIGS,
    MOE_CONFIGS,
    WIDE_RESNET_CONFIGS,
    GPT2Config,
    MoEConfig,
    WideResNetConfig,
)
from tepdist_amd.models.gpt2 import GPT2, GPT2Block  # noqa: F401
from tepdist_amd.models.llama import LLAMA_CONFIGS, Llama, LlamaConfig  # noqa: F401
