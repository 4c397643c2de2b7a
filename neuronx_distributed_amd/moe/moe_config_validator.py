"""MoE configuration validation (reference
modules/moe/moe_config_validator.py:13-120): dropless/capacity/activation
consistency rules, for both HF-config-backed and explicit model configs."""

import json
import logging

logger = logging.getLogger(__name__)


class MoeConfigValidator:
    """Validates a training config's MoE block.

    ``cfg`` duck-types the reference's shape: ``cfg.model_source`` in
    {"hf", "megatron"}, ``cfg.model.moe`` with ``dropless`` /
    ``capacity_factor`` / ``glu_mlp``, and for HF sources
    ``cfg.model.model_config`` = path of the HF config.json."""

    def __init__(self, cfg):
        self.cfg = cfg
        self.hf_model_config = {}

    def _load_hf_config(self):
        with open(self.cfg.model.model_config) as f:
            return json.load(f)

    def _validate_hf_activation(self, dropless: bool) -> None:
        if not dropless:
            return
        if self.hf_model_config.get("model_type") == "dbrx":
            act = self.hf_model_config.get("ffn_config", {}) \
                .get("ffn_act_fn", {}).get("name")
            if act != "silu":
                raise ValueError(
                    "For DBRX models, dropless mode requires SiLU; got "
                    f"{act}")
        elif self.hf_model_config.get("hidden_act") != "silu":
            raise ValueError(
                "Dropless mode is only supported with SiLU activation; got "
                f"{self.hf_model_config.get('hidden_act')}")

    def _validate_megatron_activation(self, dropless: bool) -> None:
        if not dropless:
            return
        act = getattr(self.cfg.model, "activation", None)
        if act not in ("silu", "swiglu"):
            raise ValueError(
                "For Megatron models, dropless mode requires SiLU/SwiGLU; "
                f"got {act}")

    def validate_moe_config(self) -> None:
        if not hasattr(self.cfg.model, "moe"):
            raise AttributeError(
                "MoE configuration missing: 'moe' attribute not present in "
                "the model configuration")
        moe = self.cfg.model.moe
        dropless = getattr(moe, "dropless", False)
        capacity_factor = moe.capacity_factor
        glu_mlp = getattr(moe, "glu_mlp", True)

        if self.cfg.model_source == "hf":
            self.hf_model_config = self._load_hf_config()
            self._validate_hf_activation(dropless)
        elif self.cfg.model_source == "megatron":
            self._validate_megatron_activation(dropless)

        if dropless:
            if not glu_mlp:
                raise ValueError("Dropless mode requires GLU_MLP to be True.")
            if capacity_factor is None or capacity_factor > 0.0:
                logger.warning(
                    "Dropless mode expects capacity_factor 0.0 (got %s); "
                    "setting it to 0.0", capacity_factor)
                self.cfg.model.moe.capacity_factor = 0.0
        else:
            if capacity_factor is not None and capacity_factor <= 0.0:
                raise ValueError(
                    "Dropping requires a capacity factor greater than 0.0; "
                    "adjust your configuration.")
