from .llama import LlamaConfig, LlamaForCausalLM, LlamaModel, get_config, CONFIGS
from .mixtral import (MixtralConfig, MixtralForCausalLM, MixtralModel,
                      get_moe_config, MOE_CONFIGS)
from .gpt_neox import (GPTNeoXConfig, GPTNeoXForCausalLM, get_neox_config,
                       NEOX_CONFIGS)
