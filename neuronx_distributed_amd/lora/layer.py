"""Dense LoRA adapters (reference modules/lora/layer.py)."""

import math

import torch
import torch.nn as nn


class LoraLayerBase(nn.Module):
    def __init__(self, base: nn.Module, rank: int, alpha: float,
                 dropout: float):
        super().__init__()
        self.base_layer = base
        self.lora_rank = rank
        self.scaling = alpha / rank
        self.lora_dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
        for p in self.base_layer.parameters():
            p.requires_grad = False

    def merge(self):
        raise NotImplementedError


class LoraLinear(LoraLayerBase):
    def __init__(self, base: nn.Linear, rank=16, alpha=32.0, dropout=0.0):
        super().__init__(base, rank, alpha, dropout)
        dtype = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(torch.zeros(rank, base.in_features,
                                               dtype=dtype, device=dev))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, rank,
                                               dtype=dtype, device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))

    def forward(self, x):
        out = self.base_layer(x)
        lora = self.lora_dropout(x) @ self.lora_A.t() @ self.lora_B.t()
        return out + lora * self.scaling

    @torch.no_grad()
    def merge(self):
        self.base_layer.weight += (self.lora_B @ self.lora_A) * self.scaling
        return self.base_layer


class LoraEmbedding(LoraLayerBase):
    def __init__(self, base: nn.Embedding, rank=16, alpha=32.0, dropout=0.0):
        super().__init__(base, rank, alpha, dropout)
        dtype = base.weight.dtype
        dev = base.weight.device
        self.lora_A = nn.Parameter(torch.zeros(rank, base.num_embeddings,
                                               dtype=dtype, device=dev))
        self.lora_B = nn.Parameter(torch.zeros(base.embedding_dim, rank,
                                               dtype=dtype, device=dev))
        nn.init.normal_(self.lora_A)

    def forward(self, x):
        out = self.base_layer(x)
        onehot_emb = torch.nn.functional.embedding(x, self.lora_A.t())
        return out + (onehot_emb @ self.lora_B.t()) * self.scaling

    @torch.no_grad()
    def merge(self):
        self.base_layer.weight += (self.lora_A.t() @ self.lora_B.t()) * self.scaling
        return self.base_layer
