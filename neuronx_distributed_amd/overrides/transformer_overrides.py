"""HuggingFace `transformers` integration (reference
overrides/transformer_overrides.py:1-60 — there, monkey-patched
rotate_half / apply_rotary_pos_emb glue so HF llama runs on the NKI
flash kernel).  MI355X-first equivalent: instead of monkey-patching
module internals, the HIP flash kernel is registered as a first-class
HF *attention implementation* via `AttentionInterface`, and HF <->
native state-dict converters map checkpoints in both directions.

Usage::

    from neuronx_distributed_amd.overrides import register_flash_attention
    impl = register_flash_attention()          # -> "nxda_flash"
    model = AutoModelForCausalLM.from_config(cfg, attn_implementation=impl)

Padding/arbitrary additive masks are NOT supported by the fused kernel —
the implementation asserts the causal (mask-free) decode/training case
and falls back to the composed fp32 reference off-GPU, so CPU runs of HF
models validate numerics against sdpa.
"""

from typing import Dict

import torch

ATTN_IMPL_NAME = "nxda_fused"


def nxda_flash_attention(module, query, key, value, attention_mask=None,
                         dropout: float = 0.0, scaling=None, is_causal=None,
                         **kwargs):
    """HF AttentionInterface entry: query (B,Hq,S,D), key/value (B,Hkv,S,D)
    NOT yet GQA-repeated — the kernel handles grouping natively.  Returns
    (attn_out (B,S,Hq,D), None)."""
    from ..kernels.flash_attn import flash_attn_func

    if dropout:
        raise NotImplementedError("nxda_fused: attention dropout "
                                  "unsupported (train with dropout 0)")
    causal = True if is_causal is None else bool(is_causal)
    window = getattr(module, "sliding_window", None)
    out = flash_attn_func(query, key, value, causal=causal,
                          softmax_scale=scaling, window=window)
    return out.transpose(1, 2).contiguous(), None


def register_flash_attention() -> str:
    """Register the HIP flash kernel as the HF attention implementation
    ``"nxda_fused"``; returns the name to pass as ``attn_implementation``."""
    from transformers.modeling_utils import ALL_ATTENTION_FUNCTIONS

    if ATTN_IMPL_NAME not in ALL_ATTENTION_FUNCTIONS:
        ALL_ATTENTION_FUNCTIONS.register(ATTN_IMPL_NAME,
                                         nxda_flash_attention)
    return ATTN_IMPL_NAME


# ---------------------------------------------------------------------------
# HF llama-family checkpoint <-> native state-dict converters
# ---------------------------------------------------------------------------

def convert_hf_llama_state_dict(hf_sd: Dict[str, torch.Tensor]
                                ) -> Dict[str, torch.Tensor]:
    """HF `LlamaForCausalLM` names -> this package's names: q/k/v_proj
    become the GQA qkv_proj weight_q/k/v entries; gate_proj+up_proj fuse
    into the single [gate; up] gate_up_proj weight.  Unsharded (tp=1)
    tensors — shard with scripts/checkpoint_converter.py afterwards."""
    out: Dict[str, torch.Tensor] = {}
    gates: Dict[str, torch.Tensor] = {}
    ups: Dict[str, torch.Tensor] = {}
    for k, v in hf_sd.items():
        if k.endswith("self_attn.q_proj.weight"):
            out[k.replace("q_proj.weight", "qkv_proj.weight_q")] = v
        elif k.endswith("self_attn.k_proj.weight"):
            out[k.replace("k_proj.weight", "qkv_proj.weight_k")] = v
        elif k.endswith("self_attn.v_proj.weight"):
            out[k.replace("v_proj.weight", "qkv_proj.weight_v")] = v
        elif k.endswith("self_attn.q_proj.bias"):
            out[k.replace("q_proj.bias", "qkv_proj.bias_q")] = v
        elif k.endswith("self_attn.k_proj.bias"):
            out[k.replace("k_proj.bias", "qkv_proj.bias_k")] = v
        elif k.endswith("self_attn.v_proj.bias"):
            out[k.replace("v_proj.bias", "qkv_proj.bias_v")] = v
        elif k.endswith("mlp.gate_proj.weight"):
            gates[k.rsplit("gate_proj.weight", 1)[0]] = v
        elif k.endswith("mlp.up_proj.weight"):
            ups[k.rsplit("up_proj.weight", 1)[0]] = v
        elif k.endswith("rotary_emb.inv_freq"):
            continue  # recomputed natively (precompute_rope_freqs)
        else:
            out[k] = v
    for prefix, g in gates.items():
        if prefix not in ups:
            raise KeyError(f"gate_proj without up_proj under {prefix!r}")
        out[prefix + "gate_up_proj.weight"] = torch.cat([g, ups.pop(prefix)],
                                                        dim=0)
    if ups:
        raise KeyError(f"up_proj without gate_proj: {sorted(ups)[:3]}")
    return out


def convert_to_hf_llama_state_dict(sd: Dict[str, torch.Tensor]
                                   ) -> Dict[str, torch.Tensor]:
    """Inverse of :func:`convert_hf_llama_state_dict` (full tensors)."""
    out: Dict[str, torch.Tensor] = {}
    for k, v in sd.items():
        if k.endswith("qkv_proj.weight_q"):
            out[k.replace("qkv_proj.weight_q", "q_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_k"):
            out[k.replace("qkv_proj.weight_k", "k_proj.weight")] = v
        elif k.endswith("qkv_proj.weight_v"):
            out[k.replace("qkv_proj.weight_v", "v_proj.weight")] = v
        elif k.endswith("qkv_proj.bias_q"):
            out[k.replace("qkv_proj.bias_q", "q_proj.bias")] = v
        elif k.endswith("qkv_proj.bias_k"):
            out[k.replace("qkv_proj.bias_k", "k_proj.bias")] = v
        elif k.endswith("qkv_proj.bias_v"):
            out[k.replace("qkv_proj.bias_v", "v_proj.bias")] = v
        elif k.endswith("mlp.gate_up_proj.weight"):
            half = v.shape[0] // 2
            base = k.rsplit("gate_up_proj.weight", 1)[0]
            out[base + "gate_proj.weight"] = v[:half]
            out[base + "up_proj.weight"] = v[half:]
        else:
            out[k] = v
    return out
