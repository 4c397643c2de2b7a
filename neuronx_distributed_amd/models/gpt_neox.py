"""GPT-NeoX family on the parallel layers (the reference's
``gpt_neox_20B`` integration model; see SURVEY.md §4 integration list).

NeoX specifics vs GPT-2/llama: PARALLEL residual
(x + attn(ln1(x)) + mlp(ln2(x))), LayerNorm with bias, fused QKV with
bias, partial rotary embeddings (``rotary_pct`` of head_dim), no GQA."""

from dataclasses import dataclass

import torch
import torch.nn as nn

from .. import ops
from ..kernels.flash_attn import flash_attn_func
from ..parallel import parallel_state as ps
from ..parallel.layer_norm import LayerNorm
from ..parallel.layers import (
    ColumnParallelLinear,
    ParallelEmbedding,
    RowParallelLinear,
)
from ..parallel.loss_functions import parallel_cross_entropy

torch.fx.wrap("parallel_cross_entropy")


@dataclass
class GPTNeoXConfig:
    hidden_size: int = 6144
    intermediate_size: int = 24576
    num_hidden_layers: int = 44
    num_attention_heads: int = 64
    vocab_size: int = 50432
    max_position_embeddings: int = 2048
    rotary_pct: float = 0.25
    rope_theta: float = 10000.0
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02
    use_parallel_residual: bool = True

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @property
    def num_key_value_heads(self):
        return self.num_attention_heads  # no GQA in NeoX


NEOX_CONFIGS = {
    "gpt-neox-20b": GPTNeoXConfig(),
    "gpt-neox-tiny": GPTNeoXConfig(hidden_size=64, intermediate_size=256,
                                   num_hidden_layers=2,
                                   num_attention_heads=4, vocab_size=256,
                                   max_position_embeddings=128),
}


def get_neox_config(name: str, **overrides) -> GPTNeoXConfig:
    import dataclasses

    return dataclasses.replace(NEOX_CONFIGS[name], **overrides)


def _init(std):
    return lambda t: nn.init.normal_(t, mean=0.0, std=std)


def _rotate_half(x):
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat((-x2, x1), dim=-1)


def _partial_rope(q, k, cos, sin, rot_dim, pos_offset=0):
    """Apply rotary embedding to the first ``rot_dim`` dims of the head,
    pass the rest through (NeoX rotary_pct).  cos/sin are the (S, rot/2)
    half tables from ops.precompute_rope_freqs; rows taken at
    [pos_offset, pos_offset+S) for KV-cache decode."""
    q_rot, q_pass = q[..., :rot_dim], q[..., rot_dim:]
    k_rot, k_pass = k[..., :rot_dim], k[..., rot_dim:]
    S = q.shape[2]
    half = rot_dim // 2
    cs = cos[pos_offset:pos_offset + S, :half]
    sn = sin[pos_offset:pos_offset + S, :half]
    c = torch.cat([cs, cs], -1).to(q.dtype)
    s = torch.cat([sn, sn], -1).to(q.dtype)
    q_rot = q_rot * c + _rotate_half(q_rot) * s
    k_rot = k_rot * c + _rotate_half(k_rot) * s
    return (torch.cat([q_rot, q_pass], dim=-1),
            torch.cat([k_rot, k_pass], dim=-1))


class GPTNeoXAttention(nn.Module):
    def __init__(self, cfg: GPTNeoXConfig):
        super().__init__()
        tp = ps.get_tensor_model_parallel_size()
        self.n_local = cfg.num_attention_heads // tp
        self.head_dim = cfg.head_dim
        self.rot_dim = int(cfg.head_dim * cfg.rotary_pct)
        self.query_key_value = ColumnParallelLinear(
            cfg.hidden_size, 3 * cfg.hidden_size, bias=True,
            gather_output=False, stride=3,
            init_method=_init(cfg.initializer_range))
        self.dense = RowParallelLinear(
            cfg.hidden_size, cfg.hidden_size, bias=True,
            input_is_parallel=True, init_method=_init(cfg.initializer_range))

    def forward(self, x, cos, sin, pos_offset=0, kv_cache=None):
        B, S, _ = x.shape
        qkv = self.query_key_value(x)
        q, k, v = qkv.chunk(3, dim=-1)
        q = q.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        k = k.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        v = v.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        q, k = _partial_rope(q, k, cos, sin, self.rot_dim, pos_offset)
        if kv_cache is not None:
            k, v = kv_cache.update(k.contiguous(), v.contiguous(), pos_offset)
        # flash dispatcher routes D != 128 / rectangular shapes through the
        # composed batched-GEMM reference path
        out = flash_attn_func(q.contiguous(), k.contiguous(), v.contiguous(),
                              causal=True)
        out = out.transpose(1, 2).reshape(B, S, -1)
        return self.dense(out)


class GPTNeoXMLP(nn.Module):
    def __init__(self, cfg: GPTNeoXConfig):
        super().__init__()
        self.dense_h_to_4h = ColumnParallelLinear(
            cfg.hidden_size, cfg.intermediate_size, bias=True,
            gather_output=False, init_method=_init(cfg.initializer_range))
        self.dense_4h_to_h = RowParallelLinear(
            cfg.intermediate_size, cfg.hidden_size, bias=True,
            input_is_parallel=True, init_method=_init(cfg.initializer_range))

    def forward(self, x):
        return self.dense_4h_to_h(
            torch.nn.functional.gelu(self.dense_h_to_4h(x)))


class GPTNeoXLayer(nn.Module):
    def __init__(self, cfg: GPTNeoXConfig):
        super().__init__()
        self.use_parallel_residual = cfg.use_parallel_residual
        self.input_layernorm = LayerNorm(cfg.hidden_size,
                                         eps=cfg.layer_norm_epsilon)
        self.post_attention_layernorm = LayerNorm(cfg.hidden_size,
                                                  eps=cfg.layer_norm_epsilon)
        self.attention = GPTNeoXAttention(cfg)
        self.mlp = GPTNeoXMLP(cfg)

    def forward(self, x, cos, sin, pos_offset=0, kv_cache=None):
        attn_out = self.attention(self.input_layernorm(x), cos, sin,
                                  pos_offset, kv_cache)
        if self.use_parallel_residual:
            # x + attn(ln1(x)) + mlp(ln2(x))  — NeoX parallel residual
            return x + attn_out + self.mlp(self.post_attention_layernorm(x))
        h = x + attn_out
        return h + self.mlp(self.post_attention_layernorm(h))


class GPTNeoXForCausalLM(nn.Module):
    def __init__(self, cfg: GPTNeoXConfig):
        super().__init__()
        self.config = cfg
        self.embed_in = ParallelEmbedding(
            cfg.vocab_size, cfg.hidden_size,
            init_method=_init(cfg.initializer_range))
        self.layers = nn.ModuleList(GPTNeoXLayer(cfg)
                                    for _ in range(cfg.num_hidden_layers))
        self.final_layer_norm = LayerNorm(cfg.hidden_size,
                                          eps=cfg.layer_norm_epsilon)
        self.embed_out = ColumnParallelLinear(
            cfg.hidden_size, cfg.vocab_size, bias=False, gather_output=False,
            init_method=_init(cfg.initializer_range))
        cos, sin = ops.precompute_rope_freqs(
            cfg.max_position_embeddings,
            max(2, 2 * (int(cfg.head_dim * cfg.rotary_pct) // 2)),
            cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def reset_parameters(self):
        cfg = self.config
        cos, sin = ops.precompute_rope_freqs(
            cfg.max_position_embeddings,
            max(2, 2 * (int(cfg.head_dim * cfg.rotary_pct) // 2)),
            cfg.rope_theta, device=self.rope_cos.device)
        with torch.no_grad():
            self.rope_cos.copy_(cos)
            self.rope_sin.copy_(sin)

    def forward(self, input_ids, labels=None, pos_offset=0, kv_caches=None):
        x = self.embed_in(input_ids)
        for i, layer in enumerate(self.layers):
            kc = kv_caches[i] if kv_caches is not None else None
            x = layer(x, self.rope_cos, self.rope_sin, pos_offset, kc)
        x = self.final_layer_norm(x)
        logits = self.embed_out(x)
        if labels is None:
            return logits
        shift = labels[:, 1:].contiguous()
        loss = parallel_cross_entropy(logits[:, :-1, :].contiguous(), shift)
        return loss.sum() / (shift != -100).sum().clamp(min=1)
