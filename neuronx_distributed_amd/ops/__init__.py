"""Hand-written CDNA4 HIP kernels — python wrappers + autograd.

The native library (``libnxd_ops.so``, built in-tree by ``ops/build.py``)
replaces the reference's NKI kernels: RMSNorm, RoPE,
SwiGLU, flash attention, fused AdamW.  On a GPU box the HIP path is
MANDATORY — if the library is missing, GPU calls raise instead of silently
falling back to eager torch.  CPU tensors use plain-torch reference
implementations (the numerics baseline the GPU kernels are tested against).
"""

import ctypes
import math
import os
from typing import Optional

import torch

_LIB = None
_LIB_ERR = None


def _load():
    global _LIB, _LIB_ERR
    if _LIB is not None or _LIB_ERR is not None:
        return _LIB
    from .build import LIB

    if not os.path.exists(LIB):
        _LIB_ERR = f"{LIB} not built — run neuronx_distributed_amd.ops.build"
        return None
    try:
        lib = ctypes.CDLL(LIB)
    except OSError as e:  # pragma: no cover
        _LIB_ERR = str(e)
        return None
    _LIB = lib
    return _LIB


def is_available() -> bool:
    return _load() is not None


def _require_lib():
    lib = _load()
    if lib is None:
        raise RuntimeError(
            f"nxd_ops HIP library unavailable on a GPU tensor: {_LIB_ERR}. "
            "The MI355X-native kernels are mandatory on GPU — build with "
            "python -m neuronx_distributed_amd.ops.build")
    return lib


def _stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _ptr(t: torch.Tensor) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr())


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------

class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ctx.eps = eps
        if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0:
            lib = _require_lib()
            x2 = x.contiguous()
            rows = x2.numel() // x2.shape[-1]
            out = torch.empty_like(x2)
            rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
            lib.rmsnorm_fwd(_ptr(x2), _ptr(weight.contiguous()), _ptr(out),
                            _ptr(rstd), ctypes.c_int(rows),
                            ctypes.c_int(x2.shape[-1]), ctypes.c_float(eps),
                            _stream())
            ctx.save_for_backward(x2, weight, rstd)
            return out
        # CPU reference (fp32 math like the kernel)
        xf = x.float()
        rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        ctx.save_for_backward(x, weight, rstd.squeeze(-1).reshape(-1))
        return (xf * rstd * weight.float()).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        H = x.shape[-1]
        if x.is_cuda and x.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _require_lib()
            dy2 = dy.contiguous()
            rows = x.numel() // H
            dx = torch.empty_like(x)
            dw32 = torch.zeros(H, dtype=torch.float32, device=x.device)
            lib.rmsnorm_bwd(_ptr(x), _ptr(weight.contiguous()), _ptr(dy2),
                            _ptr(rstd), _ptr(dx), _ptr(dw32),
                            ctypes.c_int(rows), ctypes.c_int(H), _stream())
            return dx, dw32.to(weight.dtype), None
        xf = x.float()
        dyf = dy.float()
        wf = weight.float()
        r = rstd.reshape(x.shape[:-1]).unsqueeze(-1).to(torch.float32)
        dot = (dyf * wf * xf).sum(-1, keepdim=True)
        dx = r * wf * dyf - r.pow(3) / H * xf * dot
        dw = (dyf * xf * r).reshape(-1, H).sum(0)
        return dx.to(x.dtype), dw.to(weight.dtype), None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6):
    return _RMSNormFn.apply(x, weight, eps)


def add_rmsnorm_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "add_rmsnorm_fwd")


def add_rmsnorm(residual: torch.Tensor, delta: torch.Tensor,
                weight: torch.Tensor, eps: float = 1e-6):
    """Fused ``res_out = residual + delta; normed = rmsnorm(res_out)*w``
    (inference only, no autograd).  bf16 GPU rows; H % 8 == 0."""
    lib = _require_lib()
    H = residual.shape[-1]
    rows = residual.numel() // H
    r = residual.contiguous()
    d = delta.contiguous()
    res_out = torch.empty_like(r)
    normed = torch.empty_like(r)
    lib.add_rmsnorm_fwd(_ptr(r), _ptr(d), _ptr(weight.contiguous()),
                        _ptr(res_out), _ptr(normed), rows, H,
                        ctypes.c_float(eps), _stream())
    return res_out, normed


class _AddRMSNormFn(torch.autograd.Function):
    """Training fused residual-add + RMSNorm: (delta, res) -> (h, normed)
    with h = bf16(delta + res), normed = rmsnorm(h) * w.  The backward
    folds the residual fork's pass-through gradient dh into the RMSNorm
    backward pass (``rmsnorm_bwd_add``), so the fused site costs one
    kernel in each direction instead of norm + a 3-pass eager add
    (CUDAFunctor_add was ~2%% of the training step)."""

    @staticmethod
    def forward(ctx, delta, res, weight, eps):
        ctx.eps = eps
        if delta.is_cuda and delta.dtype == torch.bfloat16 \
                and delta.shape[-1] % 8 == 0:
            lib = _require_lib()
            H = delta.shape[-1]
            rows = delta.numel() // H
            dc, rc = delta.contiguous(), res.contiguous()
            h = torch.empty_like(dc)
            normed = torch.empty_like(dc)
            rstd = torch.empty(rows, dtype=torch.float32,
                               device=delta.device)
            lib.add_rmsnorm_fwd_train(_ptr(rc), _ptr(dc),
                                      _ptr(weight.contiguous()), _ptr(h),
                                      _ptr(normed), _ptr(rstd), rows, H,
                                      ctypes.c_float(eps), _stream())
            ctx.save_for_backward(h, weight, rstd)
            return h, normed
        # composed reference (fp32 math, norm stats on the rounded sum —
        # matches the kernel and an unfused bf16 add -> rmsnorm chain)
        h = (delta.float() + res.float()).to(delta.dtype)
        hf = h.float()
        rstd = torch.rsqrt(hf.pow(2).mean(-1, keepdim=True) + eps)
        ctx.save_for_backward(h, weight, rstd.squeeze(-1).reshape(-1))
        return h, (hf * rstd * weight.float()).to(delta.dtype)

    @staticmethod
    def backward(ctx, dh, dnormed):
        h, weight, rstd = ctx.saved_tensors
        H = h.shape[-1]
        if dnormed is None:
            # normed output unused: the op reduces to the residual add —
            # grads pass straight through, no weight grad
            return dh, dh, None, None
        if h.is_cuda and h.dtype == torch.bfloat16 and H % 8 == 0:
            lib = _require_lib()
            rows = h.numel() // H
            dy = dnormed.contiguous()
            dx = torch.empty_like(h)
            dw32 = torch.zeros(H, dtype=torch.float32, device=h.device)
            if dh is None:
                lib.rmsnorm_bwd(_ptr(h), _ptr(weight.contiguous()),
                                _ptr(dy), _ptr(rstd), _ptr(dx), _ptr(dw32),
                                rows, H, _stream())
            else:
                lib.rmsnorm_bwd_add(_ptr(h), _ptr(weight.contiguous()),
                                    _ptr(dy), _ptr(rstd),
                                    _ptr(dh.contiguous()), _ptr(dx),
                                    _ptr(dw32), rows, H, _stream())
            return dx, dx, dw32.to(weight.dtype), None
        hf = h.float()
        dyf = dnormed.float()
        wf = weight.float()
        r = rstd.reshape(h.shape[:-1]).unsqueeze(-1).to(torch.float32)
        dot = (dyf * wf * hf).sum(-1, keepdim=True)
        dx = r * wf * dyf - r.pow(3) / H * hf * dot
        if dh is not None:
            dx = dx + dh.float()
        dw = (dyf * hf * r).reshape(-1, H).sum(0)
        dxc = dx.to(h.dtype)
        return dxc, dxc, dw.to(weight.dtype), None


def add_rmsnorm_train(delta: torch.Tensor, res: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-6):
    """Autograd fused ``h = delta + res; normed = rmsnorm(h) * w`` —
    returns (h, normed); d delta == d res == rmsnorm_bwd + dh (fused)."""
    return _AddRMSNormFn.apply(delta, res, weight, eps)


def add_rmsnorm_train_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "add_rmsnorm_fwd_train")


# ---------------------------------------------------------------------------
# RoPE (neox rotate-half), in-place on clones
# ---------------------------------------------------------------------------

def _rope_torch(x, cos, sin, sign=1.0):
    # x (B,S,h,D); cos/sin (S, D/2) f32
    half = x.shape[-1] // 2
    x0 = x[..., :half].float()
    x1 = x[..., half:].float()
    c = cos.view(1, cos.shape[0], 1, half)
    s = sin.view(1, sin.shape[0], 1, half) * sign
    o0 = x0 * c - x1 * s
    o1 = x1 * c + x0 * s
    return torch.cat([o0, o1], dim=-1).to(x.dtype)


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin, pos_offset):
        # bounds-check HERE: an out-of-range position would be an
        # out-of-bounds global read inside the HIP kernel (GPU fault)
        S = q.shape[1]
        off = int(pos_offset) if not isinstance(pos_offset, torch.Tensor) \
            else 0  # device-tensor positions are decode-path (S == 1)
        if off + S > cos.shape[0]:
            raise ValueError(
                f"RoPE table too short: seq {off}+{S} > table "
                f"{cos.shape[0]} (raise max_position_embeddings)")
        ctx.pos_offset = pos_offset
        ctx.save_for_backward(cos, sin)
        if q.is_cuda and q.dtype == torch.bfloat16:
            lib = _require_lib()
            B, S, Hq, D = q.shape
            Hk = k.shape[2]
            qc, kc = q.contiguous(), k.contiguous()
            qo = torch.empty_like(qc)
            ko = torch.empty_like(kc)
            lib.rope_fwd(_ptr(qc), _ptr(kc), _ptr(qo), _ptr(ko), _ptr(cos),
                         _ptr(sin),
                         ctypes.c_int(B), ctypes.c_int(S), ctypes.c_int(Hq),
                         ctypes.c_int(Hk), ctypes.c_int(D),
                         ctypes.c_int(pos_offset), ctypes.c_int(0), _stream())
            return qo, ko
        S = q.shape[1]
        c = cos[pos_offset:pos_offset + S]
        s = sin[pos_offset:pos_offset + S]
        return _rope_torch(q, c, s), _rope_torch(k, c, s)

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        off = ctx.pos_offset
        if dq.is_cuda and dq.dtype == torch.bfloat16:
            lib = _require_lib()
            B, S, Hq, D = dq.shape
            Hk = dk.shape[2]
            dqc, dkc = dq.contiguous(), dk.contiguous()
            dqo = torch.empty_like(dqc)
            dko = torch.empty_like(dkc)
            lib.rope_fwd(_ptr(dqc), _ptr(dkc), _ptr(dqo), _ptr(dko),
                         _ptr(cos), _ptr(sin),
                         ctypes.c_int(B), ctypes.c_int(S), ctypes.c_int(Hq),
                         ctypes.c_int(Hk), ctypes.c_int(D), ctypes.c_int(off),
                         ctypes.c_int(1), _stream())
            return dqo, dko, None, None, None
        S = dq.shape[1]
        c = cos[off:off + S]
        s = sin[off:off + S]
        return (_rope_torch(dq, c, s, -1.0), _rope_torch(dk, c, s, -1.0),
                None, None, None)


def apply_rotary_pos_emb(q, k, cos, sin, pos_offset: int = 0):
    """q (B,S,Hq,D), k (B,S,Hk,D); cos/sin (S_max, D/2) fp32 tables."""
    return _RoPEFn.apply(q, k, cos, sin, pos_offset)


def _llama3_scale_freqs(inv_freqs: torch.Tensor, scaling: dict):
    """Llama-3.1 long-context frequency scaling (reference
    attention/utils.py apply_scaling): high-frequency components (short
    wavelength vs the original context) pass through, low-frequency ones
    divide by ``factor``, the band between interpolates smoothly."""
    import math as _math

    factor = float(scaling["factor"])
    lo = float(scaling.get("low_freq_factor", 1.0))
    hi = float(scaling.get("high_freq_factor", 4.0))
    orig = float(scaling.get("original_max_position_embeddings", 8192))
    wavelen = 2 * _math.pi / inv_freqs
    smooth = ((orig / wavelen - lo) / (hi - lo)).clamp(0.0, 1.0)
    blended = (1.0 - smooth) * inv_freqs / factor + smooth * inv_freqs
    return torch.where(wavelen < orig / hi, inv_freqs,
                       torch.where(wavelen > orig / lo, inv_freqs / factor,
                                   blended))


def precompute_rope_freqs(seq_len: int, dim: int, theta: float = 10000.0,
                          device=None, rope_scaling: dict = None):
    """cos/sin tables (S, D/2) fp32 (reference attention/utils.py
    precompute_freqs_cis; ``rope_scaling`` with rope_type \"llama3\"
    applies the 3.1 long-context wavelength interpolation)."""
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device,
                                        dtype=torch.float32) / dim))
    if rope_scaling:
        rt = rope_scaling.get("rope_type", rope_scaling.get("type",
                                                            "llama3"))
        if rt == "llama3":
            inv = _llama3_scale_freqs(inv, rope_scaling)
        elif rt == "linear":
            inv = inv / float(rope_scaling["factor"])
        else:
            raise NotImplementedError(f"rope_scaling type {rt!r}")
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return freqs.cos(), freqs.sin()


def apply_rotary_polar_compatible(query, key, freqs):
    """Meta-llama INTERLEAVED rotary (reference attention/utils.py:50
    apply_rotary_polar_compatible): dims pair as (2i, 2i+1) complex
    components — used for Meta-format checkpoints, whereas the model
    stack and HIP kernel use the HF neox rotate-half pairing.
    query/key (B, S, H, D); freqs (S, D/2) fp32 angles."""
    if freqs.dtype != torch.float32:
        raise ValueError("freqs must be fp32 for accuracy")
    phase = torch.polar(torch.ones_like(freqs), freqs)  # e^{i*theta}
    phase = phase.view(1, freqs.shape[0], 1, freqs.shape[1])

    def rot(x):
        xc = torch.view_as_complex(
            x.float().reshape(*x.shape[:-1], x.shape[-1] // 2, 2))
        return torch.view_as_real(xc * phase).reshape(x.shape).to(x.dtype)

    return rot(query), rot(key)


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------

class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        I = x.shape[-1] // 2
        if x.is_cuda and x.dtype == torch.bfloat16 and I % 8 == 0:
            lib = _require_lib()
            x2 = x.contiguous()
            N = x2.numel() // x2.shape[-1]
            out = torch.empty(x2.shape[:-1] + (I,), dtype=x2.dtype,
                              device=x2.device)
            lib.swiglu_fwd(_ptr(x2), _ptr(out), ctypes.c_long(N),
                           ctypes.c_int(I), _stream())
            return out
        g, u = x[..., :I].float(), x[..., I:].float()
        return (torch.nn.functional.silu(g) * u).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        I = x.shape[-1] // 2
        if x.is_cuda and x.dtype == torch.bfloat16 and I % 8 == 0:
            lib = _require_lib()
            dy2 = dy.contiguous()
            N = x.numel() // x.shape[-1]
            dx = torch.empty_like(x)
            lib.swiglu_bwd(_ptr(x), _ptr(dy2), _ptr(dx), ctypes.c_long(N),
                           ctypes.c_int(I), _stream())
            return dx
        g, u = x[..., :I].float(), x[..., I:].float()
        dyf = dy.float()
        sig = torch.sigmoid(g)
        s = g * sig
        dg = dyf * u * (sig + s * (1 - sig))
        du = dyf * s
        return torch.cat([dg, du], dim=-1).to(x.dtype)


def swiglu(x: torch.Tensor) -> torch.Tensor:
    """x = [gate; up] on the last dim -> silu(gate)*up."""
    return _SwiGLUFn.apply(x)


# ---------------------------------------------------------------------------
# Flash attention (MFMA kernel in csrc/flash_attn.hip)
# ---------------------------------------------------------------------------

def flash_attn_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "flash_attn_fwd")


def _fa_ok_layout(t) -> bool:
    """The flash kernels take any (batch, head, seq) strides with a DENSE
    last dim — transpose views of BSHD activations go in with no copy."""
    return t.stride(3) == 1


def _fa_strides(*tensors):
    vals = []
    for t in tensors:
        vals += [t.stride(0), t.stride(1), t.stride(2)]
    arr = (ctypes.c_long * len(vals))(*vals)
    return arr


def _fa_alloc_like(t):
    """Allocate an empty tensor with t's LAYOUT (so a BSHD-view input
    produces a BSHD-view output and the downstream reshape stays a view)."""
    B, H, S, D = t.shape
    if t.stride(1) == D and t.stride(2) == H * D:  # BSHD transpose view
        return torch.empty(B, S, H, D, dtype=t.dtype,
                           device=t.device).permute(0, 2, 1, 3)
    return torch.empty_like(t, memory_format=torch.contiguous_format)


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, window=0):
        lib = _require_lib()
        B, Hq, S, D = q.shape
        Hkv = k.shape[1]
        assert D == 128, "flash kernel supports D=128"
        assert q.dtype == torch.bfloat16
        if not _fa_ok_layout(q):
            q = q.contiguous()
        if not _fa_ok_layout(k):
            k = k.contiguous()
        if not _fa_ok_layout(v):
            v = v.contiguous()
        out = _fa_alloc_like(q)
        lse = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        lib.flash_attn_fwd_strided(
            _ptr(q), _ptr(k), _ptr(v), _ptr(out), _ptr(lse),
            ctypes.c_int(B), ctypes.c_int(Hq), ctypes.c_int(Hkv),
            ctypes.c_int(S), ctypes.c_float(scale),
            ctypes.c_int(1 if causal else 0), ctypes.c_int(window or 0),
            _fa_strides(q, k, v, out), _stream())
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.causal = causal
        ctx.scale = scale
        ctx.window = window or 0
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        lib = _require_lib()
        if not hasattr(lib, "flash_attn_bwd"):
            raise RuntimeError("flash_attn_bwd kernel not built")
        B, Hq, S, D = q.shape
        Hkv = k.shape[1]
        if not _fa_ok_layout(dout):
            dout = dout.contiguous()
        delta = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        # grads are allocated in their producer's LAYOUT so autograd's
        # accumulation into BSHD views stays copy-free
        dq = _fa_alloc_like(q)
        # dk/dv are written bf16 PER Q-HEAD (B,Hq,S,D); GQA replicas are
        # reduced here (each replica covers distinct Q heads -> SUM)
        dk_pq = torch.empty(B, Hq, S, D, dtype=q.dtype, device=q.device)
        dv_pq = torch.empty(B, Hq, S, D, dtype=q.dtype, device=q.device)
        lib.flash_attn_bwd_strided(
            _ptr(q), _ptr(k), _ptr(v), _ptr(out), _ptr(dout),
            _ptr(lse), _ptr(delta), _ptr(dq), _ptr(dk_pq), _ptr(dv_pq),
            ctypes.c_int(B), ctypes.c_int(Hq), ctypes.c_int(Hkv),
            ctypes.c_int(S), ctypes.c_float(ctx.scale),
            ctypes.c_int(1 if ctx.causal else 0),
            ctypes.c_int(getattr(ctx, "window", 0)),
            _fa_strides(q, k, v, out, dout, dq, dk_pq, dv_pq), _stream())
        rep = Hq // Hkv
        if rep > 1:
            dk = dk_pq.view(B, Hkv, rep, S, D).float().sum(2).to(q.dtype)
            dv = dv_pq.view(B, Hkv, rep, S, D).float().sum(2).to(q.dtype)
        else:
            dk = dk_pq
            dv = dv_pq
        return dq, dk, dv, None, None, None


def flash_attn(q, k, v, causal=True, softmax_scale=None, window=0):
    """q (B,Hq,S,D=128), k/v (B,Hkv,S,D) bf16 on GPU.  window > 0:
    sliding-window causal (Mistral), fwd AND bwd on the MFMA kernels."""
    scale = softmax_scale or 1.0 / math.sqrt(q.shape[-1])
    return _FlashAttnFn.apply(q, k, v, causal, scale, window)


def flash_attn_window_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "flash_attn_fwd_window")


def flash_attn_windowed(q, k, v, window: int, softmax_scale=None):
    """Sliding-window causal flash attention (inference/no-grad): q row i
    attends kv rows [i-window+1, i].  q (B,Hq,S,128), k/v (B,Hkv,S,128)
    bf16; out-of-band tiles are skipped entirely (Mistral-style)."""
    lib = _require_lib()
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    assert D == 128 and q.dtype == torch.bfloat16
    scale = softmax_scale or 1.0 / math.sqrt(D)
    q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
    out = torch.empty_like(q)
    lse = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
    lib.flash_attn_fwd_window(_ptr(q), _ptr(k), _ptr(v), _ptr(out),
                              _ptr(lse), ctypes.c_int(B), ctypes.c_int(Hq),
                              ctypes.c_int(Hkv), ctypes.c_int(S),
                              ctypes.c_float(scale), ctypes.c_int(window),
                              _stream())
    return out


# ---------------------------------------------------------------------------
# Fused AdamW (ZeRO-1 master-shard update; csrc/adamw.hip)
# ---------------------------------------------------------------------------

def adamw_step(master: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               grad_bf16: torch.Tensor, param_bf16: torch.Tensor,
               clip: Optional[torch.Tensor], lr: float, beta1: float,
               beta2: float, eps: float, weight_decay: float, step: int):
    """One fused pass: bf16 grad -> fp32 m/v/master update -> bf16 param."""
    lib = _require_lib()
    n = master.numel()
    assert m.numel() == n and v.numel() == n and grad_bf16.numel() == n
    assert grad_bf16.dtype == torch.bfloat16 and param_bf16.dtype == torch.bfloat16
    inv_bc1 = 1.0 / (1.0 - beta1 ** step)
    inv_bc2 = 1.0 / (1.0 - beta2 ** step)
    lib.adamw_step(ctypes.c_long(n), _ptr(master), _ptr(m), _ptr(v),
                   _ptr(grad_bf16), _ptr(param_bf16),
                   _ptr(clip) if clip is not None else None,
                   ctypes.c_float(lr), ctypes.c_float(beta1),
                   ctypes.c_float(beta2), ctypes.c_float(eps),
                   ctypes.c_float(weight_decay), ctypes.c_float(inv_bc1),
                   ctypes.c_float(inv_bc2), _stream())


# ---------------------------------------------------------------------------
# Fused MoE decode (K9 parity; csrc/moe_decode.hip)
# ---------------------------------------------------------------------------

_MOE_MB = 16


def moe_decode_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "moe_decode_glu")


def moe_decode_glu(hidden: torch.Tensor, gate_up_w: torch.Tensor,
                   down_w: torch.Tensor, expert_affinities: torch.Tensor,
                   expert_index: torch.Tensor) -> torch.Tensor:
    """Fused decode MoE (inference): hidden (T,H) bf16, gate_up_w
    (E,H,2I) fused [gate|up], down_w (E,I,H), affinities (T,E) float,
    expert_index (T,k) long -> (T,H) bf16 (TP-partial, like ExpertMLPs).

    Slots (token, expert hit) are grouped per expert into blocks of 16 so
    each block streams its expert's weights once; gather, SwiGLU, affinity
    scaling and the scatter-add all happen inside the two kernels."""
    lib = _require_lib()
    T, H = hidden.shape
    E, _, twoI = gate_up_w.shape
    I = twoI // 2
    assert H % 64 == 0 and I % 64 == 0, (H, I)
    k = expert_index.shape[1]
    dev = hidden.device

    flat_e = expert_index.reshape(-1)
    order = torch.argsort(flat_e, stable=True)
    sorted_e = flat_e[order].to(torch.long)
    sorted_tok = (order // k).to(torch.int32)

    counts = torch.bincount(flat_e, minlength=E)
    blocks_per_e = (counts + _MOE_MB - 1) // _MOE_MB
    nb = int(blocks_per_e.sum().item())
    if nb == 0:
        return hidden.new_zeros(T, H)
    blockbase = torch.cumsum(torch.nn.functional.pad(blocks_per_e, (1, 0)),
                             0)  # (E+1,)
    countbase = torch.cumsum(torch.nn.functional.pad(counts, (1, 0)), 0)

    within = torch.arange(T * k, device=dev) - countbase[sorted_e]
    padded_idx = (blockbase[sorted_e] + within // _MOE_MB) * _MOE_MB + \
        within % _MOE_MB

    slot_token = torch.zeros(nb * _MOE_MB, dtype=torch.int32, device=dev)
    slot_token[padded_idx] = sorted_tok
    aff = torch.zeros(nb * _MOE_MB, dtype=torch.float32, device=dev)
    aff[padded_idx] = expert_affinities.float()[
        sorted_tok.long(), sorted_e]

    block_expert = torch.repeat_interleave(
        torch.arange(E, device=dev), blocks_per_e).to(torch.int32)
    bi = torch.arange(nb, device=dev)
    block_ord = bi - blockbase[block_expert.long()]
    block_len = torch.clamp(counts[block_expert.long()] -
                            _MOE_MB * block_ord, 0, _MOE_MB).to(torch.int32)

    act = torch.empty(nb * _MOE_MB, I, dtype=torch.bfloat16, device=dev)
    out = torch.zeros(T, H, dtype=torch.float32, device=dev)
    hidden = hidden.contiguous()
    gate_up_w = gate_up_w.contiguous()
    down_w = down_w.contiguous()
    lib.moe_decode_glu(_ptr(hidden), _ptr(gate_up_w), _ptr(down_w),
                       _ptr(slot_token), _ptr(block_expert), _ptr(block_len),
                       _ptr(aff), _ptr(act), _ptr(out), nb, H, I, _stream())
    return out.to(hidden.dtype)


# ---------------------------------------------------------------------------
# Skinny-M decode GEMM (csrc/skinny_gemm.hip)
# ---------------------------------------------------------------------------

def skinny_gemm_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "skinny_gemm")


def skinny_linear(x2: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """x2 (M<=32, K) bf16 @ w (N, K)^T -> (M, N) bf16 via the split-K
    weight-streaming kernel (decode shapes run ~10x hipBLASLt's skinny
    kernels; see profiles/README.md)."""
    lib = _require_lib()
    M, K = x2.shape
    N = w.shape[0]
    out = torch.zeros(M, N, dtype=torch.float32, device=x2.device)
    lib.skinny_gemm(_ptr(x2), _ptr(w), _ptr(out), M, N, K, _stream())
    return out.to(torch.bfloat16)


def use_skinny_linear(inp: torch.Tensor, weight: torch.Tensor,
                      sequence_parallel_enabled: bool = False) -> bool:
    """Gate for the decode GEMM fast path (opt-in via NXDA_SKINNY_GEMM=1):
    measured on MI355X, hipBLASLt's skinny kernels already run the decode
    shapes at 1.8-6.6 TB/s — this kernel is kept for shapes/layouts where
    the library falls over, not as the default."""
    if os.environ.get("NXDA_SKINNY_GEMM", "0") != "1":
        return False
    return (not torch.is_grad_enabled() and not sequence_parallel_enabled
            and inp.is_cuda and inp.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and inp.numel() // inp.shape[-1] <= 32
            and weight.shape[-1] % 8 == 0 and inp.shape[-1] == weight.shape[-1]
            and skinny_gemm_available())


# ---------------------------------------------------------------------------
# Fused decode attention (csrc/decode_attn.hip)
# ---------------------------------------------------------------------------

def decode_attn_available() -> bool:
    lib = _load()
    return lib is not None and hasattr(lib, "decode_attn")


def decode_attn_step(q2: torch.Tensor, k2: torch.Tensor, v2: torch.Tensor,
                     kcache: torch.Tensor, vcache: torch.Tensor,
                     cos: torch.Tensor, sin: torch.Tensor,
                     pos_t: torch.Tensor, Hq: int, Hkv: int,
                     scale: float) -> torch.Tensor:
    """One fused decode-attention step: RoPE(q,k) + cache append at the
    device position ``pos_t`` + flash-decode over the cache + GQA.
    q2 (B, Hq*128), k2/v2 (B, Hkv*128) bf16 pre-rope; kcache/vcache
    (B, Hkv, Smax, 128); cos/sin (Smax, 64) fp32 -> out (B, Hq*128).

    q2/k2/v2 may be row-strided views (e.g. slices of one fused-QKV GEMM
    output): only the last dim must be dense (stride 1)."""
    lib = _require_lib()
    B = q2.shape[0]
    Smax = kcache.shape[2]
    assert Hq // Hkv in (1, 2, 4, 8), "GQA rep must be 1/2/4/8"
    assert cos.dtype == torch.float32 and pos_t.dtype == torch.int64
    assert cos.shape[0] >= Smax, (
        f"RoPE table ({cos.shape[0]}) shorter than KV cache ({Smax})")
    assert q2.stride(1) == 1 and k2.stride(1) == 1 and v2.stride(1) == 1
    assert k2.stride(0) == v2.stride(0)
    rep = Hq // Hkv
    # split-KV workspace (4 splits; see decode_attn.hip) — cached-alloc
    # tensors with static shapes, so the pair of launches is
    # hipGraph-capturable
    part_o = torch.empty(B * Hkv * 4, rep, 128, dtype=torch.float32,
                         device=q2.device)
    part_ml = torch.empty(B * Hkv * 4, rep, 2, dtype=torch.float32,
                          device=q2.device)
    out = torch.empty(B, Hq * 128, dtype=torch.bfloat16, device=q2.device)
    lib.decode_attn(_ptr(q2), _ptr(k2), _ptr(v2), _ptr(kcache), _ptr(vcache),
                    _ptr(cos), _ptr(sin), _ptr(pos_t), _ptr(part_o),
                    _ptr(part_ml), _ptr(out), B, Hq, Hkv,
                    Smax, ctypes.c_float(scale), int(q2.stride(0)),
                    int(k2.stride(0)), _stream())
    return out
