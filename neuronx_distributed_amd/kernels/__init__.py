from .flash_attn import flash_attn_func, nki_flash_attn_func  # noqa: F401
