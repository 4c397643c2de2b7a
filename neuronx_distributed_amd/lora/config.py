"""LoRA configuration (reference modules/lora/config.py)."""

from dataclasses import dataclass
from typing import List, Optional


@dataclass
class LoraConfig:
    lora_rank: int = 16
    lora_alpha: float = 32.0
    lora_dropout: float = 0.0
    bias: str = "none"
    lora_verbose: bool = False
    target_modules: Optional[List[str]] = None
    save_lora_base: bool = False
    merge_lora: bool = False
    load_lora_from_ckpt: bool = False
    save_lora_config_adapter: bool = True
