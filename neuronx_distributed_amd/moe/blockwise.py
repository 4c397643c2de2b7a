"""Blockwise MoE matmul (reference modules/moe/blockwise.py K4/K5 +
expert_mlps_v2.py:691 blockwise strategy).

Tokens are routed into fixed-size blocks (DEFAULT_BLOCK_SIZE=512, reference
model_utils.py:94), each block owned by ONE expert; ``token_position_to_id``
(-1 = padding) and ``block_to_expert`` drive a gather -> per-block expert
MLP -> scatter-accumulate with affinity scaling.  No token dropping.

This module provides the routing-index computation (replacing NKI
K6 find_nonzero_indices/indexed_flatten with torch ops) and a
grouped-GEMM execution that batches the blocks of each expert into one
hipBLASLt GEMM (equivalent math to the reference's blockwise NKI kernel;
a persistent-CTA HIP grouped GEMM is the planned follow-up and slots in
behind the same interface)."""

import math

import torch

DEFAULT_BLOCK_SIZE = 512


def compute_block_indices(expert_index: torch.Tensor, num_experts: int,
                          block_size: int = DEFAULT_BLOCK_SIZE,
                          num_blocks: int = None,
                          max_blocks_per_expert: int = None):
    """expert_index (T, k) -> (token_position_to_id (N*B,), block_to_expert
    (N,), n_used_blocks).  Assignment order is arrival order per expert
    (stable sort), padding slots are -1.

    ``max_blocks_per_expert`` enables the DROPPING variant (reference K5
    blockwise_mm_shard_intermediate_dropping + SkipMode): each expert gets
    a static block budget; assignments beyond ``budget * block_size`` are
    dropped (their slots never enter the scatter, so they contribute zero
    like the capacity-factor strategy's over-capacity tokens)."""
    T, k = expert_index.shape
    flat_expert = expert_index.reshape(-1)
    token_of = (torch.arange(T * k, device=expert_index.device) // k)
    order = torch.argsort(flat_expert, stable=True)
    sorted_expert = flat_expert[order]
    sorted_token = token_of[order]

    counts = torch.bincount(flat_expert, minlength=num_experts)
    blocks_per_e = torch.ceil(counts.float() / block_size).long()
    if max_blocks_per_expert is not None:
        blocks_per_e = blocks_per_e.clamp(max=max_blocks_per_expert)
    n_blocks = int(blocks_per_e.sum().item())
    if num_blocks is not None:
        assert n_blocks <= num_blocks, "static block budget exceeded"
    else:
        num_blocks = n_blocks

    block_to_expert = torch.repeat_interleave(
        torch.arange(num_experts, device=expert_index.device), blocks_per_e)
    if block_to_expert.numel() < num_blocks:
        pad = torch.zeros(num_blocks - block_to_expert.numel(),
                          dtype=torch.long, device=expert_index.device)
        block_to_expert = torch.cat([block_to_expert, pad])

    # slot of each sorted assignment: expert-local position + padded offset
    within = torch.arange(T * k, device=expert_index.device) - \
        torch.cumsum(torch.nn.functional.pad(counts, (1, 0)), 0)[sorted_expert]
    block_start = torch.cumsum(
        torch.nn.functional.pad(blocks_per_e, (1, 0)), 0)[sorted_expert]
    slot = block_start * block_size + within

    keep = within < blocks_per_e[sorted_expert] * block_size
    token_position_to_id = torch.full((num_blocks * block_size,), -1,
                                      dtype=torch.long,
                                      device=expert_index.device)
    token_position_to_id[slot[keep]] = sorted_token[keep]
    return token_position_to_id, block_to_expert, n_blocks


def blockwise_mm_mx(hidden: torch.Tensor, expert_affinities: torch.Tensor,
                    gate_up_q: torch.Tensor, gate_up_scales: torch.Tensor,
                    down_q: torch.Tensor, down_scales: torch.Tensor,
                    token_position_to_id: torch.Tensor,
                    block_to_expert: torch.Tensor,
                    expert_index: torch.Tensor,
                    block_size: int = DEFAULT_BLOCK_SIZE,
                    glu: bool = True, fmt: str = "fp4_e2m1") -> torch.Tensor:
    """Blockwise MoE with OCP-MX-quantized expert weights (reference K5
    ``bwmm_shard_on_block_mx``, blockwise.py:50-61): per-expert weights
    stored as MX fp4/fp8 blocks + shared scales, dequantized on the fly
    into the grouped-GEMM path.  CDNA4 has native MX MFMA; this emulation
    establishes the format/accuracy contract (quantization/microscaling)."""
    from ..quantization.microscaling import dequantize_mx

    gu_w = dequantize_mx(gate_up_q, gate_up_scales, axis=1,
                         dtype=hidden.dtype)
    d_w = dequantize_mx(down_q, down_scales, axis=1, dtype=hidden.dtype)
    return blockwise_mm(hidden, expert_affinities, gu_w, d_w,
                        token_position_to_id, block_to_expert, expert_index,
                        block_size, glu=glu)


def blockwise_mm(hidden: torch.Tensor, expert_affinities: torch.Tensor,
                 gate_up_w: torch.Tensor, down_w: torch.Tensor,
                 token_position_to_id: torch.Tensor,
                 block_to_expert: torch.Tensor, expert_index: torch.Tensor,
                 block_size: int = DEFAULT_BLOCK_SIZE,
                 glu: bool = True) -> torch.Tensor:
    """The K4 math: for each block b -> gather its tokens, run expert
    block_to_expert[b]'s MLP, scatter-accumulate affinity-scaled output.
    Blocks of one expert are batched into a single GEMM."""
    T, H = hidden.shape
    E = gate_up_w.shape[0]
    device = hidden.device

    safe_ids = token_position_to_id.clamp(min=0)
    x = hidden[safe_ids]                       # (N*B, H), pads garbage
    valid = (token_position_to_id >= 0).unsqueeze(-1)
    x = x * valid.to(x.dtype)

    nb = block_to_expert.numel()
    xb = x.reshape(nb, block_size, H)
    from .. import ops as _ops

    def _act_fn(gu):
        if glu:
            if gu.is_cuda and gu.dtype == torch.bfloat16 and \
                    _ops.is_available():
                return _ops.swiglu(gu)  # fused HIP silu(gate)*up
            I = gu.shape[-1] // 2
            return torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]
        return torch.nn.functional.gelu(gu)

    if (x.is_cuda and x.dtype == torch.bfloat16
            and hasattr(torch, "_grouped_mm")
            and gate_up_w.dtype == torch.bfloat16
            # grouped-GEMM needs expert-contiguous rows; a static
            # num_blocks budget appends padding blocks of expert 0 at the
            # END, breaking monotonicity -> use the per-expert loop there
            and bool((block_to_expert.diff() >= 0).all())):
        # ONE grouped GEMM per projection over all experts (hipBLASLt
        # grouped kernels; the blockwise layout is already expert-sorted
        # and block-padded, so group r = rows [offs[r-1], offs[r]) )
        E_w = gate_up_w.shape[0]
        counts = torch.bincount(block_to_expert, minlength=E_w)
        offs = torch.cumsum(counts * block_size, 0).to(torch.int32)
        gu = torch._grouped_mm(x, gate_up_w, offs=offs)
        out_flat2 = torch._grouped_mm(_act_fn(gu), down_w, offs=offs)
        out_b = out_flat2.reshape(nb, block_size, H)
    else:
        out_b = torch.zeros_like(xb)
        # SkipMode: blocks with no valid token (padding-only) never touch
        # the GEMMs (reference K5 SkipMode DMA/compute skip)
        nonempty = (token_position_to_id.reshape(nb, block_size) >= 0).any(1)
        for e in torch.unique(block_to_expert[nonempty]).tolist():
            sel = ((block_to_expert == e) & nonempty).nonzero().reshape(-1)
            xe = xb[sel].reshape(-1, H)
            act = _act_fn(xe @ gate_up_w[e])
            out_b[sel] = (act @ down_w[e]).reshape(-1, block_size, H)

    # affinity of (token, owning expert of its slot)
    aff = expert_affinities[safe_ids.clamp(max=T - 1),
                            torch.repeat_interleave(block_to_expert,
                                                    block_size)]
    out_flat = out_b.reshape(-1, H) * (aff.unsqueeze(-1) *
                                       valid.to(out_b.dtype))
    result = hidden.new_zeros(T, H).float()
    result.index_add_(0, safe_ids * valid.reshape(-1).long(),
                      out_flat.float() * valid.to(torch.float32))
    return result.to(hidden.dtype)
