"""Trainer facade: config factory + model/optimizer initialization.

Parity with reference ``trainer/trainer.py``: ``neuronx_distributed_config``
(:32-144 nested-dict config with defaulting), ``initialize_parallel_model``
(:147-234, 6 phases), ``initialize_parallel_optimizer`` (:237-315 with ZeRO-1
plumbing).
"""

from typing import Any, Callable, Dict, Optional

import torch

from ..parallel import parallel_state as ps
from ..parallel.pad import pad_model
from ..optimizer import NeuronZero1Optimizer
from ..utils.logger import get_logger
from ..utils.activation_checkpoint import apply_activation_checkpointing
from ..utils.model_utils import init_on_device, get_model_sequential
from .model import NxDModel
from .optimizer import NxDOptimizer

logger = get_logger(__name__)


def neuronx_distributed_config(
    tensor_parallel_size: int = 1,
    pipeline_parallel_size: int = 1,
    expert_parallel_size: int = 1,
    context_parallel_size: int = 1,
    pipeline_config: Optional[Dict] = None,
    optimizer_config: Optional[Dict] = None,
    activation_checkpoint_config=None,
    pad_model_flag: bool = False,
    sequence_parallel: bool = False,
    model_init_config: Optional[Dict] = None,
    lora_config=None,
    mixed_precision_config: Optional[Dict] = None,
    lnc_size: int = 1,
) -> Dict[str, Any]:
    """Build + validate the config dict (reference trainer.py:32-144) and
    initialize model parallelism if not already done.

    ``lnc_size`` is accepted for API compatibility only: logical-neuron-
    core grouping is trn2 hardware topology with no MI355X analogue (one
    process drives one GPU)."""
    optimizer_config = dict(optimizer_config or {})
    optimizer_config.setdefault("zero_one_enabled", True)
    optimizer_config.setdefault("grad_clipping", True)
    optimizer_config.setdefault("max_grad_norm", 1.0)

    model_init_config = dict(model_init_config or {})
    model_init_config.setdefault("meta_device_init", False)
    model_init_config.setdefault("param_init_fn", None)
    model_init_config.setdefault("sequential_move_factor", 11)

    mixed_precision_config = dict(mixed_precision_config or {})
    mixed_precision_config.setdefault("use_master_weights", True)
    mixed_precision_config.setdefault("use_fp32_grad_acc", False)
    mixed_precision_config.setdefault("use_master_weights_in_ckpt", False)

    config = {
        "tensor_parallel_size": tensor_parallel_size,
        "pipeline_parallel_size": pipeline_parallel_size,
        "expert_parallel_size": expert_parallel_size,
        "context_parallel_size": context_parallel_size,
        "pipeline_config": pipeline_config,
        "optimizer_config": optimizer_config,
        "activation_checkpoint_config": activation_checkpoint_config,
        "pad_model": pad_model_flag,
        "sequence_parallel": sequence_parallel,
        "model_init_config": model_init_config,
        "lora_config": lora_config,
        "mixed_precision_config": mixed_precision_config,
        "lnc_size": lnc_size,
    }

    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(
            tensor_model_parallel_size=tensor_parallel_size,
            pipeline_model_parallel_size=pipeline_parallel_size,
            expert_model_parallel_size=expert_parallel_size,
            context_parallel_size=context_parallel_size,
        )
    return config


nxd_config = neuronx_distributed_config


def initialize_parallel_model(nxd_config: Dict, model_fn: Callable, *model_args,
                              **model_kwargs) -> "NxDModel":
    """6 phases (reference trainer.py:147-234): build (meta/cpu) -> PP wrap
    -> materialize/move -> LoRA -> pad -> NxDModel + activation ckpt."""
    init_cfg = nxd_config["model_init_config"]
    pp_size = nxd_config["pipeline_parallel_size"]
    device = torch.device("cuda", torch.cuda.current_device()) \
        if torch.cuda.is_available() else torch.device("cpu")

    # phase 1: build
    if init_cfg.get("meta_device_init"):
        with init_on_device(torch.device("meta")):
            model = model_fn(*model_args, **model_kwargs)
    else:
        model = model_fn(*model_args, **model_kwargs)

    # phase 2: pipeline wrap
    if pp_size > 1:
        from ..pipeline import NxDPPModel

        pipeline_cfg = dict(nxd_config.get("pipeline_config") or {})
        model = NxDPPModel(model, **pipeline_cfg)

    # phase 3: materialize + move — host-RAM-bounded: ranks move in
    # groups of sequential_move_factor with a rendezvous between groups
    # (reference model_utils.py:335-358)
    if init_cfg.get("meta_device_init"):
        from ..utils.model_utils import reinit_model

        model = reinit_model(model, device, init_cfg.get("param_init_fn"))
    elif pp_size == 1:
        model = get_model_sequential(
            lambda: model, device,
            sequential_move_factor=init_cfg.get("sequential_move_factor",
                                                11))

    # phase 4: LoRA
    if nxd_config.get("lora_config") is not None:
        from ..lora import LoraModel

        model = LoraModel(model, nxd_config["lora_config"])

    # phase 5: pad (noop for natively-built models)
    # phase 6: wrap + activation checkpointing
    ac = nxd_config.get("activation_checkpoint_config")
    target = model.local_module() if pp_size > 1 and hasattr(model, "local_module") else model
    if ac == "full":
        from ..models.llama import LlamaDecoderLayer

        apply_activation_checkpointing(
            target, activation_checkpoint_classes=[LlamaDecoderLayer])
    elif ac is not None:
        classes = ac if isinstance(ac, (list, tuple)) else [ac]
        apply_activation_checkpointing(target,
                                       activation_checkpoint_classes=classes)

    from .post_partition_hooks import run_post_partition_hooks

    run_post_partition_hooks(model, nxd_config)
    return NxDModel(model, nxd_config)


def initialize_parallel_optimizer(nxd_config: Dict, optimizer_class,
                                  parameters, **defaults) -> "NxDOptimizer":
    """reference trainer.py:237-315."""
    opt_cfg = nxd_config["optimizer_config"]
    mp_cfg = nxd_config.get("mixed_precision_config") or {}
    if opt_cfg.get("zero_one_enabled", True):
        optimizer = NeuronZero1Optimizer(
            parameters, optimizer_class,
            grad_clipping=opt_cfg.get("grad_clipping", True),
            max_norm=opt_cfg.get("max_grad_norm", 1.0),
            use_fp32_grad_acc=mp_cfg.get("use_fp32_grad_acc", False),
            **defaults)
    else:
        optimizer = optimizer_class(parameters, **defaults)
    return NxDOptimizer(optimizer, nxd_config)
