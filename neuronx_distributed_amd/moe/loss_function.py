"""MoE auxiliary load-balancing loss (reference modules/moe/loss_function.py:5):
loss = E * sum_e( fraction_of_tokens_e * mean_router_prob_e )."""

import torch


def load_balancing_loss_func(router_logits, num_experts: int, top_k: int):
    """router_logits: (T,E) or tuple/list of per-layer (T,E)."""
    if isinstance(router_logits, (tuple, list)):
        router_logits = torch.cat([r.reshape(-1, num_experts)
                                   for r in router_logits], dim=0)
    probs = torch.softmax(router_logits.float(), dim=-1)
    _, idx = torch.topk(probs, top_k, dim=-1)
    onehot = torch.nn.functional.one_hot(idx, num_experts).float().sum(1)
    fraction = onehot.mean(0) / top_k
    prob_mean = probs.mean(0)
    return num_experts * torch.sum(fraction * prob_mean)
