"""Flash attention wrapper (reference kernels/flash_attn.py parity).

``flash_attn_func(q, k, v, causal=True)`` with q (B,Hq,S,D), k/v (B,Hkv,S,D)
bf16 — GQA supported (Hq a multiple of Hkv).  On GPU this dispatches to the
hand-written CDNA4 MFMA kernel (ops.flash_attn); on CPU it runs the plain
fp32 torch reference the kernel is tested against.

Set NXDA_ALLOW_TORCH_FALLBACK=1 to permit the composed-torch path on GPU
(bring-up only — the HIP kernel is the production path and GPU calls fail
loudly without it otherwise).
"""

import math
import os

import torch


def _torch_reference(q, k, v, causal=True, scale=None, window=None):
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    if rep > 1:
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    scale = scale or 1.0 / math.sqrt(D)
    scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    Sk = k.shape[2]
    if causal:
        mask = torch.ones(S, Sk, dtype=torch.bool, device=q.device).tril(
            diagonal=Sk - S)
        if window is not None:
            # sliding window: row i (global Sk-S+i) attends the last
            # ``window`` positions only
            mask &= torch.ones(S, Sk, dtype=torch.bool,
                               device=q.device).triu(
                                   diagonal=Sk - S - window + 1)
        scores = scores.masked_fill(~mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    return torch.matmul(probs, v.float()).to(q.dtype)


def _banded_window_attn(q, k, v, scale, window, block=512):
    """Causal sliding-window attention in O(S*(window+block)) memory:
    q-row blocks attend only the kv band [q0 - window + 1, q1); each
    block is gradient-checkpointed so training memory stays per-block
    (the full composed path materializes S x S scores and OOMs at
    production shapes).  Numerically identical to _torch_reference with
    the same window."""
    import torch.utils.checkpoint as ckpt

    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    scale = scale or 1.0 / math.sqrt(D)

    def run_block(qb, kb, vb, q0, k0, q1):
        # qb (B,Hq,bq,D); kb/vb (B,Hq,bk,D) already GQA-expanded
        scores = torch.matmul(qb.float(), kb.float().transpose(-1, -2))
        scores = scores * scale
        qi = torch.arange(q0, q1, device=qb.device).unsqueeze(1)
        kj = torch.arange(k0, k0 + kb.shape[2], device=qb.device)
        keep = (kj <= qi) & (kj > qi - window)
        scores = scores.masked_fill(~keep, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        return torch.matmul(probs, vb.float()).to(qb.dtype)

    if rep > 1:
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    outs = []
    for q0 in range(0, S, block):
        q1 = min(q0 + block, S)
        k0 = max(0, q0 - window + 1)
        qb = q[:, :, q0:q1]
        kb = k[:, :, k0:q1]
        vb = v[:, :, k0:q1]
        if torch.is_grad_enabled() and (qb.requires_grad
                                        or kb.requires_grad):
            o = ckpt.checkpoint(run_block, qb, kb, vb, q0, k0, q1,
                                use_reentrant=False)
        else:
            o = run_block(qb, kb, vb, q0, k0, q1)
        outs.append(o)
    return torch.cat(outs, dim=2)


def flash_attn_func(q, k, v, causal=True, softmax_scale=None, window=None):
    if window is not None and causal and window >= k.shape[2]:
        # a window covering the whole KV length masks nothing beyond the
        # causal triangle (row i attends [i-window+1, i]) — run the plain
        # causal MFMA kernel, fwd AND bwd.  Makes Mistral-style models
        # train on the fused path whenever seq_len <= sliding_window
        # (the composed fallback materializes S x S scores and OOMs at
        # production batch sizes).
        window = None
    if q.is_cuda:
        from .. import ops

        if (window is not None and causal and not torch.is_grad_enabled()
                and q.shape[2] == k.shape[2] and q.shape[3] == 128
                and q.dtype == torch.bfloat16
                and ops.flash_attn_window_available()):
            # sliding-window MFMA kernel, no-grad fast path (training
            # windows route through ops.flash_attn(window=) below)
            return ops.flash_attn_windowed(q, k, v, int(window),
                                           softmax_scale)
        if (window is not None and causal and q.shape[2] == k.shape[2]
                and q.shape[3] == 128 and q.dtype == torch.bfloat16
                and ops.flash_attn_available()):
            # window < S TRAINING on the MFMA kernels: fwd tile-skip +
            # band masks and the windowed dkdv/dq backward
            return ops.flash_attn(q, k, v, causal=True,
                                  softmax_scale=softmax_scale,
                                  window=int(window))
        if window is not None and causal and q.shape[2] == k.shape[2] \
                and q.shape[2] > 1024:
            # window < S training/prefill at real seq lengths without the
            # kernel: banded blocks in O(S*window) memory (the full
            # composed path materializes S x S scores and OOMs)
            return _banded_window_attn(q, k, v, softmax_scale, int(window))
        if q.shape[2] != k.shape[2] or q.shape[3] != 128 or \
                window is not None or q.dtype != torch.bfloat16:
            # rectangular attention (KV-cache decode), head_dim != 128,
            # sliding-window masking, or non-bf16 dtype (fp32 golden runs):
            # composed path — the HIP MFMA kernel is bf16 D=128
            # full-causal (the production training shapes)
            return _torch_reference(q, k, v, causal, softmax_scale, window)
        if hasattr(ops, "flash_attn") and ops.flash_attn_available():
            return ops.flash_attn(q, k, v, causal=causal,
                                  softmax_scale=softmax_scale)
        if os.environ.get("NXDA_ALLOW_TORCH_FALLBACK", "0") == "1":
            return _torch_reference(q, k, v, causal, softmax_scale)
        raise RuntimeError(
            "HIP flash-attention kernel unavailable on GPU; build "
            "neuronx_distributed_amd.ops (or set NXDA_ALLOW_TORCH_FALLBACK=1 "
            "for bring-up)")
    return _torch_reference(q, k, v, causal, softmax_scale, window)


# reference-compatible name (kernels/flash_attn.py:162)
nki_flash_attn_func = flash_attn_func
