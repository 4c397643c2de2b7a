"""Attention-head padding to TP divisibility (reference parallel_layers/pad.py:
``pad_model`` :32-111).  Pads Q heads (and the model's head bookkeeping) up to
the next multiple of tp; the padded heads are zero-initialized and masked out
of attention outputs by construction (their output projection rows are zero).
"""

import torch

from . import parallel_state as ps
from ..utils.logger import get_logger

logger = get_logger(__name__)


def get_number_of_extra_heads(num_heads: int, tp_degree: int) -> int:
    if num_heads % tp_degree == 0:
        return 0
    return tp_degree - (num_heads % tp_degree)


def pad_model(model: torch.nn.Module, tp_degree: int, n_heads: int,
              wrapped_classes=(), pad_hook_fn=None):
    """Reference-compatible entry point; models built from this package's
    parallel layers compute padded sizes at construction time, so this is
    a validation + hook pass."""
    extra = get_number_of_extra_heads(n_heads, tp_degree)
    if extra == 0:
        return model
    if pad_hook_fn is not None:
        pad_hook_fn(model, tp_degree)
    logger.warning(
        "pad_model: %d heads padded by %d to divide tp=%d; ensure the model "
        "was constructed with the padded head count", n_heads, extra, tp_degree)
    return model


def pad_llama_config(config, tp_degree: int):
    """Return a COPY of a llama-style config with ``num_attention_heads``
    (and KV heads when needed) padded up to divisibility by ``tp_degree``
    (reference pad.py:32-111 applied at config level — this package's
    parallel layers shard at construction, so padding happens before the
    model is built)."""
    import dataclasses

    extra = get_number_of_extra_heads(config.num_attention_heads, tp_degree)
    kv = config.num_key_value_heads
    # KV heads must divide tp or tp divide them (replication); pad them to
    # the same granularity when neither holds
    if kv % tp_degree != 0 and tp_degree % kv != 0:
        kv = kv + get_number_of_extra_heads(kv, tp_degree)
    if extra == 0 and kv == config.num_key_value_heads:
        return config
    return dataclasses.replace(
        config,
        num_attention_heads=config.num_attention_heads + extra,
        num_key_value_heads=kv,
        head_dim_override=config.head_dim)


def pad_attention_state_dict(state_dict, config, padded_config):
    """Pad an UNPADDED checkpoint's attention weights to a padded model's
    shapes: new Q (and KV) head rows are ZERO in wq/wk/wv and the matching
    o_proj COLUMNS are zero, so padded heads contribute nothing — the
    padded model computes exactly the unpadded model's function.

    Keys handled (llama naming): ``*.qkv_proj.weight_q/k/v`` (rows =
    heads*head_dim) and ``*.o_proj.weight`` (columns = heads*head_dim).
    Returns a new dict; non-attention keys pass through."""
    D = config.head_dim
    hq_old = config.num_attention_heads
    hq_new = padded_config.num_attention_heads
    kv_old = config.num_key_value_heads
    kv_new = padded_config.num_key_value_heads
    out = {}
    for key, t in state_dict.items():
        if key.endswith("qkv_proj.weight_q") and t.shape[0] == hq_old * D:
            pad = t.new_zeros((hq_new - hq_old) * D, t.shape[1])
            out[key] = torch.cat([t, pad], dim=0)
        elif (key.endswith("qkv_proj.weight_k")
              or key.endswith("qkv_proj.weight_v")) and \
                t.shape[0] == kv_old * D:
            pad = t.new_zeros((kv_new - kv_old) * D, t.shape[1])
            out[key] = torch.cat([t, pad], dim=0)
        elif key.endswith("o_proj.weight") and t.shape[1] == hq_old * D:
            pad = t.new_zeros(t.shape[0], (hq_new - hq_old) * D)
            out[key] = torch.cat([t, pad], dim=1)
        else:
            out[key] = t
    return out
