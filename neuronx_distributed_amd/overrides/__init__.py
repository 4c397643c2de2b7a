from .transformer_overrides import (ATTN_IMPL_NAME,
                                    convert_hf_llama_state_dict,
                                    convert_to_hf_llama_state_dict,
                                    nxda_flash_attention,
                                    register_flash_attention)

__all__ = ["ATTN_IMPL_NAME", "register_flash_attention",
           "nxda_flash_attention", "convert_hf_llama_state_dict",
           "convert_to_hf_llama_state_dict"]
