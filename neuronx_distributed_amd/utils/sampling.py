"""On-device sampling (reference utils/sampling.py:6 ``Sampler``): greedy /
top-k / top-p over vocab-parallel logits via the distributed operators."""

import torch

from ..operators import topk as dist_topk, argmax as dist_argmax


class Sampler:
    def __init__(self, do_sample: bool = False, top_k: int = 50,
                 top_p: float = 1.0, temperature: float = 1.0):
        self.do_sample = do_sample
        self.top_k = top_k
        self.top_p = top_p
        self.temperature = temperature

    def __call__(self, vocab_parallel_logits: torch.Tensor) -> torch.Tensor:
        """logits (B, V/tp) -> token ids (B,)."""
        if not self.do_sample:
            return dist_argmax(vocab_parallel_logits, dim=-1, gather_dim=-1)
        logits = vocab_parallel_logits / max(self.temperature, 1e-5)
        vals, idx = dist_topk(logits, self.top_k, dim=-1, gather_dim=-1)
        probs = torch.softmax(vals.float(), dim=-1)
        if self.top_p < 1.0:
            sorted_probs, order = probs.sort(dim=-1, descending=True)
            cum = sorted_probs.cumsum(-1)
            mask = cum - sorted_probs > self.top_p
            sorted_probs = sorted_probs.masked_fill(mask, 0.0)
            sorted_probs = sorted_probs / sorted_probs.sum(-1, keepdim=True)
            pick = torch.multinomial(sorted_probs, 1)
            pick = order.gather(-1, pick)
        else:
            pick = torch.multinomial(probs, 1)
        return idx.gather(-1, pick).squeeze(-1)
