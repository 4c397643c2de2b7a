"""Mixtral-style MoE causal LM on the MI355X parallel layers.

Model-zoo parity with the reference's MoE helpers
(``experimental/nxd_model_zoo/mixtral_model.py``, ``llama4_moe.py``; see
SURVEY.md §2.4): GQA attention + top-k routed expert MLPs built from this
repo's MoE stack (RouterTopK, ExpertMLPs with all-experts /
capacity-factor / blockwise dispatch, optional SharedExperts, EP/TP
sharding).  ``moe_frequency`` interleaves dense and MoE layers
(llama4-style: every Nth layer is MoE); ``num_shared_experts`` adds an
always-on shared expert branch (llama4/deepseek style).

Aux (load-balancing) loss: router logits from every MoE layer are
collected by the model forward and added to the CE loss scaled by
``router_aux_loss_coef`` (reference loss_function.py parity).  For
pipeline partitioning run with ``router_aux_loss_coef = 0`` so layers
return plain tensors.
"""

from dataclasses import dataclass
from typing import Optional

import os

import torch
import torch.nn as nn

from ..moe.expert_mlps import ExpertMLPs
from ..moe.loss_function import load_balancing_loss_func
from ..moe.model import MoE
from ..moe.routing import (GroupLimitedRouter, RouterSinkhorn, RouterTopK)
from ..moe.shared_experts import SharedExperts
from ..parallel.layers import ColumnParallelLinear, ParallelEmbedding
from ..parallel.loss_functions import parallel_cross_entropy
from ..parallel.mappings import (
    gather_from_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)
from .. import ops
from .llama import (LlamaAttention, LlamaMLP, RMSNorm, _init_method,
                    fused_norm_train_ok)

torch.fx.wrap("parallel_cross_entropy")


@dataclass
class MixtralConfig:
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    vocab_size: int = 32000
    max_position_embeddings: int = 4096
    rope_theta: float = 1e6
    rms_norm_eps: float = 1e-5
    initializer_range: float = 0.02
    sequence_parallel_enabled: bool = False
    tie_word_embeddings: bool = False
    # MoE
    num_local_experts: int = 8
    num_experts_per_tok: int = 2
    router_aux_loss_coef: float = 0.02
    capacity_factor: Optional[float] = None  # None = all-experts dispatch
    moe_frequency: int = 1       # every Nth layer is MoE (1 = all layers)
    num_shared_experts: int = 0  # >0 adds a SharedExperts branch
    glu_mlp: bool = True
    # "topk" (Mixtral) | "group_limited" (DeepSeek-V3 style) | "sinkhorn"
    router_type: str = "topk"
    n_groups: int = 4
    topk_group: int = 2

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


MOE_CONFIGS = {
    "mixtral-8x7b": MixtralConfig(),
    "tiny-moe": MixtralConfig(
        hidden_size=64, intermediate_size=128, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, vocab_size=256,
        max_position_embeddings=128, num_local_experts=4,
        num_experts_per_tok=2),
}


def get_moe_config(name: str, **overrides) -> MixtralConfig:
    import dataclasses

    return dataclasses.replace(MOE_CONFIGS[name], **overrides)


def build_moe_layer(config: MixtralConfig) -> MoE:
    """Assemble router + expert MLPs (+ shared experts) — the reference's
    ``initialize_moe_module`` equivalent (mixtral_model.py)."""
    sp = config.sequence_parallel_enabled
    if config.router_type == "group_limited":
        router = GroupLimitedRouter(
            config.num_local_experts, config.num_experts_per_tok,
            config.hidden_size, n_groups=config.n_groups,
            topk_group=config.topk_group, sequence_parallel_enabled=sp)
    elif config.router_type == "sinkhorn":
        router = RouterSinkhorn(config.num_local_experts, 1,
                                config.hidden_size,
                                sequence_parallel_enabled=sp)
    else:
        router = RouterTopK(config.num_local_experts,
                            config.num_experts_per_tok, config.hidden_size,
                            sequence_parallel_enabled=sp)
    expert_mlps = ExpertMLPs(
        config.num_local_experts, config.hidden_size,
        config.intermediate_size, config.num_experts_per_tok,
        capacity_factor=config.capacity_factor, glu_mlp=config.glu_mlp,
        init_method=_init_method(config.initializer_range))
    shared = None
    if config.num_shared_experts > 0:
        shared = SharedExperts(config.hidden_size, config.intermediate_size,
                               config.num_shared_experts,
                               init_method=_init_method(
                                   config.initializer_range))
    return MoE(router, expert_mlps, shared_experts=shared,
               return_router_logits=config.router_aux_loss_coef > 0,
               sequence_parallel_enabled=sp)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, config: MixtralConfig, layer_idx: int = 0):
        super().__init__()
        sp = config.sequence_parallel_enabled
        self.self_attn = LlamaAttention(config)
        self.is_moe = (layer_idx + 1) % config.moe_frequency == 0
        if self.is_moe:
            self.block_sparse_moe = build_moe_layer(config)
        else:  # llama4-style interleaving: dense layers in between
            self.mlp = LlamaMLP(config)
        self.input_layernorm = RMSNorm(config.hidden_size,
                                       config.rms_norm_eps,
                                       sequence_parallel_enabled=sp)
        self.post_attention_layernorm = RMSNorm(
            config.hidden_size, config.rms_norm_eps,
            sequence_parallel_enabled=sp)

    def forward(self, hidden, cos, sin, pos_offset=0, kv_cache=None):
        h = hidden + self.self_attn(self.input_layernorm(hidden), cos, sin,
                                    pos_offset, kv_cache)
        normed = self.post_attention_layernorm(h)
        if not self.is_moe:
            return h + self.mlp(normed), None
        out = self.block_sparse_moe(normed)
        if isinstance(out, tuple):
            moe_out, router_logits = out
        else:
            moe_out, router_logits = out, None
        return h + moe_out, router_logits

    def forward_fused_train(self, residual, delta, cos, sin, pos_offset=0):
        """Training path carrying (residual, delta): every residual add
        fuses into the next RMSNorm via ops.add_rmsnorm_train (same
        machinery as LlamaDecoderLayer.forward_fused_train)."""
        if delta is None:
            normed = self.input_layernorm(residual)
        else:
            residual, normed = ops.add_rmsnorm_train(
                delta, residual, self.input_layernorm.weight,
                self.input_layernorm.variance_epsilon)
        attn_out = self.self_attn(normed, cos, sin, pos_offset, None)
        residual, normed2 = ops.add_rmsnorm_train(
            attn_out, residual, self.post_attention_layernorm.weight,
            self.post_attention_layernorm.variance_epsilon)
        if not self.is_moe:
            return residual, self.mlp(normed2), None
        out = self.block_sparse_moe(normed2)
        if isinstance(out, tuple):
            moe_out, router_logits = out
        else:
            moe_out, router_logits = out, None
        return residual, moe_out, router_logits


class MixtralModel(nn.Module):
    def __init__(self, config: MixtralConfig):
        super().__init__()
        self.config = config
        self.embed_tokens = ParallelEmbedding(
            config.vocab_size, config.hidden_size,
            init_method=_init_method(config.initializer_range),
            sequence_parallel_enabled=False)
        self.layers = nn.ModuleList(
            MixtralDecoderLayer(config, i)
            for i in range(config.num_hidden_layers))
        self.norm = RMSNorm(
            config.hidden_size, config.rms_norm_eps,
            sequence_parallel_enabled=config.sequence_parallel_enabled)
        cos, sin = ops.precompute_rope_freqs(config.max_position_embeddings,
                                             config.head_dim,
                                             config.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def reset_parameters(self):
        # meta materialization clobbers non-persistent buffers: recompute
        cos, sin = ops.precompute_rope_freqs(
            self.config.max_position_embeddings, self.config.head_dim,
            self.config.rope_theta, device=self.rope_cos.device)
        with torch.no_grad():
            self.rope_cos.copy_(cos)
            self.rope_sin.copy_(sin)

    def forward(self, input_ids, pos_offset=0, kv_caches=None):
        hidden = self.embed_tokens(input_ids)  # (B,S,H)
        if self.config.sequence_parallel_enabled:
            hidden = hidden.transpose(0, 1).contiguous()
            hidden = scatter_to_sequence_parallel_region(hidden, seq_dim=0)
        all_router_logits = []
        if fused_norm_train_ok(self.config, hidden, self.training) \
                and kv_caches is None \
                and all(hasattr(l, "forward_fused_train")
                        and not l._forward_hooks
                        and not l._forward_pre_hooks
                        for l in self.layers):
            # fused residual+norm training path (see models/llama.py;
            # SP-safe — per-row ops on the local seq shard, norm-weight
            # SP grad marking is a param attribute)
            residual, delta = hidden, None
            for layer in self.layers:
                residual, delta, router_logits = layer.forward_fused_train(
                    residual, delta, self.rope_cos, self.rope_sin,
                    pos_offset)
                if router_logits is not None:
                    all_router_logits.append(router_logits)
            _, hidden = ops.add_rmsnorm_train(delta, residual,
                                              self.norm.weight,
                                              self.norm.variance_epsilon)
        else:
            for i, layer in enumerate(self.layers):
                kc = kv_caches[i] if kv_caches is not None else None
                hidden, router_logits = layer(hidden, self.rope_cos,
                                              self.rope_sin, pos_offset, kc)
                if router_logits is not None:
                    all_router_logits.append(router_logits)
            hidden = self.norm(hidden)
        if self.config.sequence_parallel_enabled:
            hidden = gather_from_sequence_parallel_region(
                hidden, seq_dim=0, to_model_parallel=True)
            hidden = hidden.transpose(0, 1)
        return hidden, all_router_logits


class MixtralForCausalLM(nn.Module):
    # attention is LlamaAttention -> tensor-position decode works
    supports_tensor_position = True

    def __init__(self, config: MixtralConfig):
        super().__init__()
        self.config = config
        self.model = MixtralModel(config)
        self.lm_head = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, bias=False,
            gather_output=False,
            init_method=_init_method(config.initializer_range))
        if config.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, labels=None, pos_offset=0, kv_caches=None):
        hidden, router_logits = self.model(input_ids, pos_offset, kv_caches)
        logits = self.lm_head(hidden)  # (B,S,V/tp)
        if labels is None:
            return logits
        shift = labels[:, 1:].contiguous()
        loss = parallel_cross_entropy(logits[:, :-1, :].contiguous(),
                                      shift).sum() / \
            (shift != -100).sum().clamp(min=1)
        if router_logits and self.config.router_aux_loss_coef > 0:
            aux = load_balancing_loss_func(
                torch.cat(router_logits, dim=0),
                self.config.num_local_experts,
                self.config.num_experts_per_tok)
            loss = loss + self.config.router_aux_loss_coef * aux
        return loss
