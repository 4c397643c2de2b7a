"""Ring attention for context parallelism (reference K3,
kernels/ring_attention_kernel.py:13-115 — NKI kernel with in-kernel ring).

MI355X-native design: the flash kernel stays a per-block
primitive; the ring is HOST-DRIVEN — K/V blocks rotate around the CP group
over RCCL P2P (xGMI), double-buffered so the next block's transfer overlaps
the current block's flash kernel.  Causal load balance: rank r skips blocks
b > r entirely (upper triangle), so with contiguous S/cp slices rank r does
r+1 of cp block-pairs (the reference's zigzag pairing is a further
balance refinement; contiguous slices match batch_utils slicing,
reference utils/batch_utils.py:19).

Forward merges per-block (O, lse) online; backward re-rotates K/V and
rides the (dK, dV) accumulator around the same ring so each block's grads
arrive home after cp steps — no all-reduce, no extra memory.
"""

import math

import torch
import torch.distributed as dist

from ..parallel import parallel_state as ps


def _band_mask(Sq, Sk, diag_off, window, device):
    """Boolean keep-mask for q row ql vs kv row kl with GLOBAL offset
    q_global - k_global = diag_off + ql - kl: causal (>= 0 guaranteed by
    the caller for off-diagonal blocks) + sliding window (< window)."""
    ql = torch.arange(Sq, device=device).unsqueeze(1)
    kl = torch.arange(Sk, device=device)
    d = diag_off + ql - kl
    keep = d >= 0
    if window is not None:
        keep &= d < window
    return keep


def _block_fwd_masked(q, k, v, scale, diag_off, window):
    """Composed (O, lse) for a PARTIALLY in-window block pair (the band
    crosses the block boundary, which the square kernel cannot express);
    runs on CPU or GPU tensors."""
    rep = q.shape[1] // k.shape[1]
    kk = k.repeat_interleave(rep, 1).float() if rep > 1 else k.float()
    vv = v.repeat_interleave(rep, 1).float() if rep > 1 else v.float()
    scores = q.float() @ kk.transpose(-1, -2) * scale
    keep = _band_mask(q.shape[2], k.shape[2], diag_off, window, q.device)
    scores = scores.masked_fill(~keep, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)  # -inf for all-masked rows
    # rows with NO in-window kv in THIS block contribute nothing: zero
    # them (softmax of all -inf is NaN) — _merge weights them out via
    # the -inf lse
    probs = torch.nan_to_num(torch.softmax(scores, -1), nan=0.0)
    out = (probs @ vv).to(q.dtype)
    return out, lse


def _block_bwd_masked(q, k, v, out, dout, lse, scale, diag_off, window):
    rep = q.shape[1] // k.shape[1]
    Hkv = k.shape[1]
    kk = k.repeat_interleave(rep, 1).float() if rep > 1 else k.float()
    vv = v.repeat_interleave(rep, 1).float() if rep > 1 else v.float()
    qf, of, dof = q.float(), out.float(), dout.float()
    scores = qf @ kk.transpose(-1, -2) * scale
    keep = _band_mask(q.shape[2], k.shape[2], diag_off, window, q.device)
    scores = scores.masked_fill(~keep, float("-inf"))
    P = torch.exp(scores - lse.unsqueeze(-1))
    delta = (dof * of).sum(-1, keepdim=True)
    dP = dof @ vv.transpose(-1, -2)
    dS = P * (dP - delta)
    dq = (dS @ kk) * scale
    dk_full = (dS.transpose(-1, -2) @ qf) * scale
    dv_full = P.transpose(-1, -2) @ dof
    if rep > 1:
        B = q.shape[0]
        dk_full = dk_full.view(B, Hkv, rep, *dk_full.shape[2:]).sum(2)
        dv_full = dv_full.view(B, Hkv, rep, *dv_full.shape[2:]).sum(2)
    return dq.to(q.dtype), dk_full.to(q.dtype), dv_full.to(q.dtype)


def _block_fwd(q, k, v, causal, scale, window=None):
    """(O normalized, lse natural-log) for one q-block x kv-block pair.
    ``window`` here is the SQUARE same-offset case (diagonal block)."""
    if q.is_cuda and window is not None:
        from .. import ops

        if causal and ops.flash_attn_window_available():
            import ctypes
            from ..ops import _ptr, _require_lib, _stream

            lib = _require_lib()
            B, Hq, S, D = q.shape
            out = torch.empty_like(q)
            lse = torch.empty(B, Hq, S, dtype=torch.float32,
                              device=q.device)
            lib.flash_attn_fwd_window(
                _ptr(q.contiguous()), _ptr(k.contiguous()),
                _ptr(v.contiguous()), _ptr(out), _ptr(lse),
                ctypes.c_int(B), ctypes.c_int(Hq),
                ctypes.c_int(k.shape[1]), ctypes.c_int(S),
                ctypes.c_float(scale), ctypes.c_int(int(window)),
                _stream())
            return out, lse
        return _block_fwd_masked(q, k, v, scale, 0, window)
    if window is not None and not q.is_cuda:
        return _block_fwd_masked(q, k, v, scale, 0, window)
    if q.is_cuda:
        from .. import ops
        import ctypes
        from ..ops import _ptr, _require_lib, _stream

        lib = _require_lib()
        B, Hq, S, D = q.shape
        Hkv = k.shape[1]
        out = torch.empty_like(q)
        lse = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        lib.flash_attn_fwd(_ptr(q.contiguous()), _ptr(k.contiguous()),
                           _ptr(v.contiguous()), _ptr(out), _ptr(lse),
                           ctypes.c_int(B), ctypes.c_int(Hq),
                           ctypes.c_int(Hkv), ctypes.c_int(S),
                           ctypes.c_float(scale),
                           ctypes.c_int(1 if causal else 0), _stream())
        return out, lse
    B, Hq, S, D = q.shape
    rep = Hq // k.shape[1]
    kk = k.repeat_interleave(rep, 1).float() if rep > 1 else k.float()
    vv = v.repeat_interleave(rep, 1).float() if rep > 1 else v.float()
    scores = q.float() @ kk.transpose(-1, -2) * scale
    if causal:
        mask = torch.ones(S, k.shape[2], dtype=torch.bool,
                          device=q.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)
    out = (torch.softmax(scores, -1) @ vv).to(q.dtype)
    return out, lse


def _block_bwd(q, k, v, out, dout, lse, causal, scale, window=None):
    """dq, dk, dv for one block pair given the GLOBAL lse (and delta from
    the final out/dout) — the FA2 decomposition the HIP bwd kernels use.
    ``window`` is the square same-offset (diagonal) case."""
    if window is not None and not q.is_cuda:
        return _block_bwd_masked(q, k, v, out, dout, lse, scale, 0, window)
    if q.is_cuda:
        import ctypes
        from ..ops import _fa_strides, _ptr, _require_lib, _stream

        lib = _require_lib()
        B, Hq, S, D = q.shape
        Hkv = k.shape[1]
        delta = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        dq = torch.empty_like(q)
        dk_pq = torch.empty(B, Hq, S, D, dtype=q.dtype, device=q.device)
        dv_pq = torch.empty(B, Hq, S, D, dtype=q.dtype, device=q.device)
        qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
        oc, doc = out.contiguous(), dout.contiguous()
        lib.flash_attn_bwd_strided(
            _ptr(qc), _ptr(kc), _ptr(vc), _ptr(oc), _ptr(doc),
            _ptr(lse.contiguous()), _ptr(delta), _ptr(dq), _ptr(dk_pq),
            _ptr(dv_pq), ctypes.c_int(B), ctypes.c_int(Hq),
            ctypes.c_int(Hkv), ctypes.c_int(S), ctypes.c_float(scale),
            ctypes.c_int(1 if causal else 0),
            ctypes.c_int(int(window) if window else 0),
            _fa_strides(qc, kc, vc, oc, doc, dq, dk_pq, dv_pq), _stream())
        rep = Hq // Hkv
        if rep > 1:
            dk = dk_pq.view(B, Hkv, rep, S, D).float().sum(2).to(q.dtype)
            dv = dv_pq.view(B, Hkv, rep, S, D).float().sum(2).to(q.dtype)
        else:
            dk, dv = dk_pq, dv_pq
        return dq, dk, dv
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    kk = k.repeat_interleave(rep, 1).float() if rep > 1 else k.float()
    vv = v.repeat_interleave(rep, 1).float() if rep > 1 else v.float()
    qf, of, dof = q.float(), out.float(), dout.float()
    scores = qf @ kk.transpose(-1, -2) * scale
    if causal:
        mask = torch.ones(S, k.shape[2], dtype=torch.bool,
                          device=q.device).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
    P = torch.exp(scores - lse.unsqueeze(-1))
    delta = (dof * of).sum(-1, keepdim=True)
    dP = dof @ vv.transpose(-1, -2)
    dS = P * (dP - delta)
    dq = (dS @ kk) * scale
    dk_full = (dS.transpose(-1, -2) @ qf) * scale
    dv_full = P.transpose(-1, -2) @ dof
    if rep > 1:
        dk_full = dk_full.view(B, Hkv, rep, *dk_full.shape[2:]).sum(2)
        dv_full = dv_full.view(B, Hkv, rep, *dv_full.shape[2:]).sum(2)
    return dq.to(q.dtype), dk_full.to(q.dtype), dv_full.to(q.dtype)


def _merge(o1, l1, o2, l2):
    """Merge two normalized partial attentions with their lses."""
    l_new = torch.logaddexp(l1, l2)
    w1 = torch.exp(l1 - l_new).unsqueeze(-1)
    w2 = torch.exp(l2 - l_new).unsqueeze(-1)
    return (o1.float() * w1 + o2.float() * w2).to(o1.dtype), l_new


def _ring_neighbors():
    cp = ps.get_group_info("cp")
    ranks = cp.ranks_of(dist.get_rank())
    i = ranks.index(dist.get_rank())
    nxt = ranks[(i + 1) % len(ranks)]
    prv = ranks[(i - 1) % len(ranks)]
    return i, len(ranks), nxt, prv, cp


def _rot(t_pair, nxt, prv):
    """Send current pair to next rank, receive previous rank's (blocking,
    posted receive-first to stay deadlock-free on a ring)."""
    outs = [torch.empty_like(x) for x in t_pair]
    reqs = []
    for o in outs:
        reqs.append(dist.irecv(o, prv))
    sends = [dist.isend(x.contiguous(), nxt) for x in t_pair]
    for r in reqs + sends:
        r.wait()
    return outs


def _block_kind(i, blk, C, window, causal):
    """Classify a (rank i, kv block blk) pair under a sliding window.
    Returns one of: "skip", "diag" (square same-offset; window may
    apply), "full" (entirely in-window off-diagonal), "partial"
    (band crosses the block edge -> composed masked math).
    Offsets: q_global - k_global = D + ql - kl with D = (i - blk) * C."""
    if causal and blk > i:
        return "skip"
    if window is None:
        return "diag" if blk == i else "full"
    D = (i - blk) * C
    if blk == i:
        return "diag"
    if D - (C - 1) >= window:
        return "skip"          # whole block older than the window
    if D + (C - 1) < window:
        return "full"          # whole block inside the window
    return "partial"


class _RingAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, window):
        i, cp, nxt, prv, _ = _ring_neighbors()
        C = q.shape[2]
        o_acc, l_acc = None, None
        cur_k, cur_v = k, v
        for step in range(cp):
            blk = (i - step) % cp  # owner of the current kv block
            nxt_kv = None
            if step + 1 < cp:
                nxt_kv = _rot((cur_k, cur_v), nxt, prv)
            kind = _block_kind(i, blk, C, window, causal)
            if kind != "skip":
                if kind == "partial":
                    o, l = _block_fwd_masked(q, cur_k, cur_v, scale,
                                             (i - blk) * C, window)
                elif kind == "diag" and window is not None and window < C:
                    o, l = _block_fwd(q, cur_k, cur_v, True, scale,
                                      window=int(window))
                else:
                    o, l = _block_fwd(q, cur_k, cur_v,
                                      causal and blk == i, scale)
                if o_acc is None:
                    o_acc, l_acc = o, l
                else:
                    o_acc, l_acc = _merge(o_acc, l_acc, o, l)
            if nxt_kv is not None:
                cur_k, cur_v = nxt_kv
        ctx.save_for_backward(q, k, v, o_acc, l_acc)
        ctx.causal = causal
        ctx.scale = scale
        ctx.window = window
        return o_acc

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        causal, scale = ctx.causal, ctx.scale
        window = getattr(ctx, "window", None)
        i, cp, nxt, prv, _ = _ring_neighbors()
        C = q.shape[2]
        dq_acc = torch.zeros_like(q, dtype=torch.float32)
        # rotate (k, v, dk_acc, dv_acc) together: after cp steps each
        # block's grad accumulator is home
        cur_k, cur_v = k, v
        dk_acc = torch.zeros_like(k, dtype=torch.float32)
        dv_acc = torch.zeros_like(v, dtype=torch.float32)
        for step in range(cp):
            blk = (i - step) % cp
            kind = _block_kind(i, blk, C, window, causal)
            if kind != "skip":
                if kind == "partial":
                    dq_b, dk_b, dv_b = _block_bwd_masked(
                        q, cur_k, cur_v, out, dout, lse, scale,
                        (i - blk) * C, window)
                elif kind == "diag" and window is not None and window < C:
                    dq_b, dk_b, dv_b = _block_bwd(q, cur_k, cur_v, out,
                                                  dout, lse, True, scale,
                                                  window=int(window))
                else:
                    dq_b, dk_b, dv_b = _block_bwd(q, cur_k, cur_v, out,
                                                  dout, lse,
                                                  causal and blk == i,
                                                  scale)
                dq_acc += dq_b.float()
                dk_acc += dk_b.float()
                dv_acc += dv_b.float()
            if step + 1 < cp:
                cur_k, cur_v, dk_acc, dv_acc = _rot(
                    (cur_k, cur_v, dk_acc, dv_acc), nxt, prv)
        # after cp-1 rotations the accumulators hold grads for block
        # (i - (cp-1)) % cp = i+1; one more rotation brings them home
        if cp > 1:
            cur_k, cur_v, dk_acc, dv_acc = _rot(
                (cur_k, cur_v, dk_acc, dv_acc), nxt, prv)
        return (dq_acc.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                None, None, None)


def ring_attn_func(q, k, v, causal=True, softmax_scale=None, window=None):
    """q,k,v (B,H,S/cp,D) — this rank's contiguous sequence slice.
    ``window``: Mistral-style sliding window over GLOBAL positions —
    out-of-window kv blocks are skipped entirely (ring steps still run
    for the rotation, but no compute)."""
    scale = softmax_scale or 1.0 / math.sqrt(q.shape[-1])
    if window is not None and causal and window >= q.shape[2] * max(
            ps.get_context_model_parallel_size(), 1):
        window = None  # covers the whole (global) sequence
    if ps.get_context_model_parallel_size() == 1:
        from .flash_attn import flash_attn_func

        return flash_attn_func(q, k, v, causal=causal, softmax_scale=scale,
                               window=window)
    # the K/V ring (irecv) and the block kernels need contiguous tensors
    return _RingAttnFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                             causal, scale, window)


# reference-compatible name (kernels/ring_attention_kernel.py)
nki_ring_attn_func = ring_attn_func
