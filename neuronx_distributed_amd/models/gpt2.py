"""GPT-2 family on the parallel layers — the reference's second model
family (test vehicle for PP partitioning, test/unit_test/pipeline/
test_base.py, and BASELINE.json config #1: 2-layer GPT-2-small TP=2
CPU/gloo plumbing check)."""

from dataclasses import dataclass

import torch
import torch.nn as nn

from ..parallel.layers import (
    ColumnParallelLinear,
    RowParallelLinear,
    ParallelEmbedding,
)
from ..parallel.layer_norm import LayerNorm
from ..parallel.loss_functions import parallel_cross_entropy

# fx.wrap patches name lookups in THIS module's globals (the defining
# module's wrap does not cover imported references) so the pipeline tracer
# treats the collective-bearing loss as a leaf call
torch.fx.wrap("parallel_cross_entropy")
from ..parallel import parallel_state as ps
from ..kernels.flash_attn import flash_attn_func


@dataclass
class GPT2Config:
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    vocab_size: int = 50304  # padded to 64 multiple (50257 original)
    max_position_embeddings: int = 1024
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @property
    def num_key_value_heads(self):
        return self.num_attention_heads  # MHA


CONFIGS = {
    "gpt2-small": GPT2Config(),
    "gpt2-small-2l": GPT2Config(num_hidden_layers=2),
    "gpt2-tiny": GPT2Config(hidden_size=64, num_hidden_layers=2,
                            num_attention_heads=4, vocab_size=256,
                            max_position_embeddings=128),
}


def get_gpt2_config(name: str, **overrides) -> GPT2Config:
    import dataclasses

    return dataclasses.replace(CONFIGS[name], **overrides)


def _init(std):
    return lambda t: nn.init.normal_(t, mean=0.0, std=std)


class GPT2Attention(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        tp = ps.get_tensor_model_parallel_size()
        self.n_local = cfg.num_attention_heads // tp
        self.head_dim = cfg.hidden_size // cfg.num_attention_heads
        self.c_attn = ColumnParallelLinear(
            cfg.hidden_size, 3 * cfg.hidden_size, bias=True,
            gather_output=False, stride=3,
            init_method=_init(cfg.initializer_range))
        self.c_proj = RowParallelLinear(
            cfg.hidden_size, cfg.hidden_size, bias=True,
            input_is_parallel=True, init_method=_init(cfg.initializer_range))

    def forward(self, x, pos_offset=0, kv_cache=None):
        B, S, _ = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.chunk(3, dim=-1)
        q = q.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        k = k.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        v = v.reshape(B, S, self.n_local, self.head_dim).transpose(1, 2)
        if kv_cache is not None:
            k, v = kv_cache.update(k.contiguous(), v.contiguous(), pos_offset)
        # dispatcher: HIP kernel for D=128 square shapes, composed
        # batched-GEMM reference otherwise (odd head dims, decode)
        out = flash_attn_func(q.contiguous(), k.contiguous(), v.contiguous(),
                              causal=True)
        out = out.transpose(1, 2).reshape(B, S, -1)
        return self.c_proj(out)


class GPT2MLP(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.c_fc = ColumnParallelLinear(
            cfg.hidden_size, 4 * cfg.hidden_size, bias=True,
            gather_output=False, init_method=_init(cfg.initializer_range))
        self.c_proj = RowParallelLinear(
            4 * cfg.hidden_size, cfg.hidden_size, bias=True,
            input_is_parallel=True, init_method=_init(cfg.initializer_range))

    def forward(self, x):
        return self.c_proj(torch.nn.functional.gelu(self.c_fc(x)))


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.ln_1 = LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.attn = GPT2Attention(cfg)
        self.ln_2 = LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.mlp = GPT2MLP(cfg)

    def forward(self, x, pos_offset=0, kv_cache=None):
        x = x + self.attn(self.ln_1(x), pos_offset, kv_cache)
        return x + self.mlp(self.ln_2(x))


class GPT2LMHeadModel(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.config = cfg
        self.wte = ParallelEmbedding(cfg.vocab_size, cfg.hidden_size,
                                     init_method=_init(cfg.initializer_range))
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        nn.init.normal_(self.wpe.weight, std=cfg.initializer_range)
        self.h = nn.ModuleList(GPT2Block(cfg)
                               for _ in range(cfg.num_hidden_layers))
        self.ln_f = LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.lm_head = ColumnParallelLinear(
            cfg.hidden_size, cfg.vocab_size, bias=False, gather_output=False,
            init_method=_init(cfg.initializer_range))

    def forward(self, input_ids, labels=None, pos_offset=0, kv_caches=None):
        B, S = input_ids.shape
        pos = torch.arange(pos_offset, pos_offset + S,
                           device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)
        for i, block in enumerate(self.h):
            kc = kv_caches[i] if kv_caches is not None else None
            x = block(x, pos_offset, kc)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        shift = labels[:, 1:].contiguous()
        loss = parallel_cross_entropy(logits[:, :-1, :].contiguous(), shift)
        return loss.sum() / (shift != -100).sum().clamp(min=1)
