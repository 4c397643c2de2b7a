"""Speculative decoding (draft-and-verify; reference: spec-draft groups in
parallel_state.py:1533 + the NxDI speculation flow).

Greedy acceptance: the draft proposes ``spec_len`` tokens autoregressively;
the target scores the whole chunk in ONE forward and the longest prefix
agreeing with the target's own greedy choices is accepted, plus one bonus
token from the target distribution.  With greedy sampling the output is
IDENTICAL to target-only decoding — only the number of target forwards
changes (that invariant is what the test asserts).

KV-cache bookkeeping: both caches may hold STALE rows beyond the accepted
position after a rejection; every attention path here masks by position
and rewrites rows in place, so stale rows are simply overwritten."""


import torch

from .kv_cache import build_kv_caches


def _greedy(logits: torch.Tensor) -> torch.Tensor:
    return logits.argmax(dim=-1)


@torch.no_grad()
def speculative_generate(target, draft, input_ids: torch.Tensor,
                         max_new_tokens: int = 32, spec_len: int = 4):
    """Greedy speculative generation.  target/draft:
    LlamaForCausalLM-compatible; input_ids (B, S).  Returns
    (tokens (B, S+new), acceptance_rate)."""
    B, S = input_ids.shape

    def make_caches(model):
        cfg = model.config
        from ..parallel import parallel_state as ps

        tp = ps.get_tensor_model_parallel_size()
        kv_mult = max(1, tp // cfg.num_key_value_heads)
        n_kv_local = cfg.num_key_value_heads * kv_mult // tp
        return build_kv_caches(cfg.num_hidden_layers, B, n_kv_local,
                               S + max_new_tokens + spec_len + 1,
                               cfg.head_dim, device=input_ids.device)

    t_caches = make_caches(target)
    d_caches = make_caches(draft)

    # prefill both; the draft's prefill logits are unused
    t_logits = target(input_ids, kv_caches=t_caches, pos_offset=0)
    if S > 1:
        draft(input_ids[:, :-1], kv_caches=d_caches, pos_offset=0)

    tok = _greedy(t_logits[:, -1, :])  # (B,)
    out = [input_ids, tok.unsqueeze(1)]
    # target cache holds rows [0, S); draft cache rows [0, S-1); the last
    # prompt token goes through the draft with its first proposal step
    pos = S
    d_prev = input_ids[:, -1]
    produced = 1
    n_rounds = 0
    n_accepted = 0

    while produced < max_new_tokens:
        k = min(spec_len, max_new_tokens - produced)
        # ---- draft proposes k tokens ---------------------------------
        proposals = []
        cur = tok
        dl = draft(torch.stack([d_prev, cur], 1), kv_caches=d_caches,
                   pos_offset=pos - 1)
        proposals.append(_greedy(dl[:, -1, :]))
        for i in range(1, k):
            dl = draft(proposals[-1].unsqueeze(1), kv_caches=d_caches,
                       pos_offset=pos + i)
            proposals.append(_greedy(dl[:, -1, :]))
        dmat = torch.stack(proposals, dim=1)  # (B, k)

        # ---- target verifies the chunk in one forward ----------------
        chunk = torch.cat([tok.unsqueeze(1), dmat], dim=1)  # (B, k+1)
        tl = target(chunk, kv_caches=t_caches, pos_offset=pos)
        tmat = _greedy(tl)  # (B, k+1): target choice AFTER each prefix

        # longest agreeing prefix (whole batch must agree to advance a
        # slot — per-sequence acceptance would need ragged caches)
        agree = (dmat == tmat[:, :-1]).all(dim=0)  # (k,)
        j = int(agree.cumprod(dim=0).sum().item())

        accepted = [dmat[:, i] for i in range(j)]
        bonus = tmat[:, j]
        for a in accepted:
            out.append(a.unsqueeze(1))
        out.append(bonus.unsqueeze(1))
        produced += j + 1
        n_rounds += 1
        n_accepted += j
        d_prev = dmat[:, j - 1] if j > 0 else tok
        tok = bonus
        pos += j + 1

    tokens = torch.cat(out, dim=1)[:, : S + max_new_tokens]
    rate = n_accepted / max(1, n_rounds * spec_len)
    return tokens, rate


@torch.no_grad()
def medusa_generate(target, medusa_heads, input_ids: torch.Tensor,
                    max_new_tokens: int = 32):
    """Medusa-style speculation: ``medusa_heads`` (iterable of
    utils.medusa_utils.MedusaHead, head i predicting the token at offset
    i+2) propose a chain from the LAST HIDDEN STATE in one shot — no
    autoregressive draft model — and the target verifies the chunk exactly
    like :func:`speculative_generate`.  Greedy output is identical to
    plain generation."""
    from ..utils.sampling import Sampler

    sampler = Sampler(do_sample=False)
    heads = list(medusa_heads)
    k = len(heads)
    B, S = input_ids.shape

    from ..parallel import parallel_state as ps

    cfg = target.config
    tp = ps.get_tensor_model_parallel_size()
    kv_mult = max(1, tp // cfg.num_key_value_heads)
    n_kv_local = cfg.num_key_value_heads * kv_mult // tp
    caches = build_kv_caches(cfg.num_hidden_layers, B, n_kv_local,
                             S + max_new_tokens + k + 1, cfg.head_dim,
                             device=input_ids.device)

    hidden = target.model(input_ids, pos_offset=0, kv_caches=caches)
    tok = sampler(target.lm_head(hidden)[:, -1, :])
    h_last = hidden[:, -1, :]
    out = [input_ids, tok.unsqueeze(1)]
    pos = S
    produced = 1
    n_rounds = 0
    n_accepted = 0

    while produced < max_new_tokens:
        kk = min(k, max_new_tokens - produced)
        props = [sampler(heads[i](h_last)) for i in range(kk)]
        dmat = torch.stack(props, dim=1)  # (B, kk)
        chunk = torch.cat([tok.unsqueeze(1), dmat], dim=1)
        hidden = target.model(chunk, pos_offset=pos, kv_caches=caches)
        tl = target.lm_head(hidden)
        tmat = torch.stack([sampler(tl[:, i, :])
                            for i in range(kk + 1)], dim=1)
        agree = (dmat == tmat[:, :-1]).all(dim=0)
        j = int(agree.cumprod(dim=0).sum().item())
        for i in range(j):
            out.append(dmat[:, i].unsqueeze(1))
        out.append(tmat[:, j].unsqueeze(1))
        h_last = hidden[:, j, :]
        tok = tmat[:, j]
        produced += j + 1
        pos += j + 1
        n_rounds += 1
        n_accepted += j

    tokens = torch.cat(out, dim=1)[:, : S + max_new_tokens]
    rate = n_accepted / max(1, n_rounds * k)
    return tokens, rate
