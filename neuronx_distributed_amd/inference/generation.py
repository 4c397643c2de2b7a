"""Autoregressive generation loop (reference examples/inference runner +
utils/sampling): prefill through the flash kernel, decode through the
KV-cache path, sampling over vocab-parallel logits."""

from typing import Optional

import torch

from ..utils.sampling import Sampler
from .kv_cache import build_kv_caches


@torch.no_grad()
def generate(model, input_ids: torch.Tensor, max_new_tokens: int = 32,
             sampler: Optional[Sampler] = None, eos_token_id: int = -1,
             use_cuda_graph: bool = True):
    """model: LlamaForCausalLM-compatible (vocab-parallel logits out).
    input_ids (B, S) on the model's device; returns (B, S+new).

    On GPU the per-token decode step is captured in a hipGraph and
    replayed (launch-bound otherwise); pass ``use_cuda_graph=False`` for
    the eager loop."""
    from ..parallel import parallel_state as ps
    from .decode_graph import GraphDecoder

    sampler = sampler or Sampler(do_sample=False)
    cfg = model.config
    tp = ps.get_tensor_model_parallel_size()
    B, S = input_ids.shape
    kv_mult = max(1, tp // cfg.num_key_value_heads)
    n_kv_local = cfg.num_key_value_heads * kv_mult // tp
    model_dtype = next(model.parameters()).dtype
    caches = build_kv_caches(cfg.num_hidden_layers, B, n_kv_local,
                             S + max_new_tokens, cfg.head_dim,
                             dtype=model_dtype, device=input_ids.device,
                             window=getattr(cfg, "sliding_window", None))

    # prefill (eager, flash kernel)
    logits = model(input_ids, kv_caches=caches, pos_offset=0)
    next_tok = sampler(logits[:, -1, :])
    out = [input_ids, next_tok.unsqueeze(1)]

    if input_ids.is_cuda and getattr(model, "supports_tensor_position",
                                     False):
        dec = GraphDecoder(model, caches, start_pos=S, batch=B,
                           device=input_ids.device)
        if use_cuda_graph:
            dec.capture()
        for _ in range(max_new_tokens - 1):
            logits = dec.step(next_tok)
            next_tok = sampler(logits[:, -1, :])
            out.append(next_tok.unsqueeze(1))
            if eos_token_id >= 0 and bool((next_tok == eos_token_id).all()):
                break
        return torch.cat(out, dim=1)

    # eager decode: tensor positions when supported — the tensor-pos
    # decode step masks by cache position_index(), which RollingKVCache
    # (sliding-window models) requires; int positions assume ordered
    # cache rows
    tensor_pos = getattr(model, "supports_tensor_position", False)
    pos = S
    for _ in range(max_new_tokens - 1):
        step_in = next_tok.unsqueeze(1)
        po = torch.tensor([pos], device=input_ids.device) if tensor_pos \
            else pos
        logits = model(step_in, kv_caches=caches, pos_offset=po)
        next_tok = sampler(logits[:, -1, :])
        out.append(next_tok.unsqueeze(1))
        pos += 1
        if eos_token_id >= 0 and bool((next_tok == eos_token_id).all()):
            break
    return torch.cat(out, dim=1)
