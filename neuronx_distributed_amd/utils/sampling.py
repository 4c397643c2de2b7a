"""On-device sampling (reference utils/sampling.py:6 ``Sampler``): greedy /
top-k / top-p over vocab-parallel logits via the distributed operators."""

import torch

from ..operators import topk as dist_topk, argmax as dist_argmax


class Sampler:
    """``on_device_multinomial`` replaces torch.multinomial with the
    CDF-subtract-count formulation (reference sampling.py:28-77): a FIXED
    op sequence with no data-dependent control flow, so a sampling step is
    hipGraph-capturable (the reference needed it because multinomial's
    validation breaks tracing; here it keeps the decode graph whole).
    ``return_topk_indices`` returns the top-k index tensor instead of one
    sample (Medusa tree verification, reference :76)."""

    def __init__(self, do_sample: bool = False, top_k: int = 50,
                 top_p: float = 1.0, temperature: float = 1.0,
                 on_device_multinomial: bool = False,
                 return_topk_indices: bool = False):
        self.do_sample = do_sample
        self.top_k = top_k
        self.top_p = top_p
        self.temperature = temperature
        self.on_device_multinomial = on_device_multinomial
        self.return_topk_indices = return_topk_indices

    @staticmethod
    def _multinomial_cdf(probs: torch.Tensor) -> torch.Tensor:
        """Sample one index per row via inverse-CDF: count how many
        cumulative probabilities fall below a uniform draw."""
        cdf = probs.cumsum(dim=-1)
        u = torch.rand(probs.shape[0], 1, device=probs.device)
        return (cdf < u).sum(dim=-1, keepdim=True).clamp(
            max=probs.shape[-1] - 1)

    def __call__(self, vocab_parallel_logits: torch.Tensor) -> torch.Tensor:
        """logits (B, V/tp) -> token ids (B,) (or (B, top_k) indices in
        Medusa mode)."""
        if not self.do_sample and not self.return_topk_indices:
            return dist_argmax(vocab_parallel_logits, dim=-1, gather_dim=-1)
        logits = vocab_parallel_logits / max(self.temperature, 1e-5)
        vals, idx = dist_topk(logits, self.top_k, dim=-1, gather_dim=-1)
        if self.return_topk_indices:
            return idx
        probs = torch.softmax(vals.float(), dim=-1)
        if self.top_p < 1.0:
            sorted_probs, order = probs.sort(dim=-1, descending=True)
            cum = sorted_probs.cumsum(-1)
            mask = cum - sorted_probs > self.top_p
            sorted_probs = sorted_probs.masked_fill(mask, 0.0)
            sorted_probs = sorted_probs / sorted_probs.sum(-1, keepdim=True)
            pick = self._multinomial_cdf(sorted_probs) \
                if self.on_device_multinomial \
                else torch.multinomial(sorted_probs, 1)
            pick = order.gather(-1, pick)
        else:
            pick = self._multinomial_cdf(probs) \
                if self.on_device_multinomial \
                else torch.multinomial(probs, 1)
        return idx.gather(-1, pick).squeeze(-1)
