from .llama import LlamaConfig, LlamaForCausalLM, LlamaModel, get_config, CONFIGS
