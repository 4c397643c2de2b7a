"""LoraModel wrapper (reference modules/lora/model.py:74): swaps target
modules for LoRA-adapted versions, freezes the base, exposes adapter
save/load/merge."""

from typing import Dict

import torch
import torch.nn as nn

from ..parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..parallel.qkv_linear import GQAQKVColumnParallelLinear
from ..utils.logger import get_logger
from .config import LoraConfig
from .layer import LoraEmbedding, LoraLinear
from .tp_layer import LoraGQAQKVParallelLinear, LoraParallelLinear

logger = get_logger(__name__)

# reference model.py:35-68 target-module mapping table
DEFAULT_TARGET_MODULES = ["q_proj", "k_proj", "v_proj", "o_proj", "qkv_proj",
                          "gate_up_proj", "down_proj", "c_attn", "c_proj"]


class LoraModel(nn.Module):
    def __init__(self, module: nn.Module, config: LoraConfig):
        super().__init__()
        self.module = module
        self.lora_config = config
        # freeze the whole base model; only adapters (and optionally biases)
        # train (reference model.py:74 mark_only_lora_as_trainable)
        for p in module.parameters():
            p.requires_grad = False
        targets = config.target_modules or DEFAULT_TARGET_MODULES
        self._swap(module, targets)
        if config.bias == "all":
            for n, p in module.named_parameters():
                if n.endswith("bias"):
                    p.requires_grad = True

    def _make(self, child):
        cfg = self.lora_config
        kw = dict(rank=cfg.lora_rank, alpha=cfg.lora_alpha,
                  dropout=cfg.lora_dropout)
        if isinstance(child, GQAQKVColumnParallelLinear):
            return LoraGQAQKVParallelLinear(child, **kw)
        if isinstance(child, (ColumnParallelLinear, RowParallelLinear)):
            return LoraParallelLinear(child, **kw)
        if isinstance(child, nn.Linear):
            return LoraLinear(child, **kw)
        if isinstance(child, nn.Embedding):
            return LoraEmbedding(child, **kw)
        return None

    def _swap(self, root, targets):
        n = 0
        for name, parent in root.named_modules():
            for cname, child in list(parent.named_children()):
                if cname in targets:
                    new = self._make(child)
                    if new is not None:
                        setattr(parent, cname, new)
                        n += 1
        logger.info("LoRA: adapted %d modules (rank=%d)", n,
                    self.lora_config.lora_rank)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def run_train(self, *args, **kwargs):
        return self.module.run_train(*args, **kwargs)

    # -- adapter state -----------------------------------------------------
    def get_adapter_state_dict(self) -> Dict[str, torch.Tensor]:
        return {k: v for k, v in self.state_dict().items() if "lora_" in k}

    def load_adapter_state_dict(self, sd):
        missing, unexpected = self.load_state_dict(sd, strict=False)
        assert not unexpected, unexpected

    def merge_lora(self):
        """Fold adapters into the base weights (reference adapter merge)."""
        for name, parent in self.module.named_modules():
            for cname, child in list(parent.named_children()):
                if hasattr(child, "merge") and hasattr(child, "base_layer"):
                    setattr(parent, cname, child.merge())
        return self.module
