"""Round-2 MoE depth: selective loading, expert_indices sub-selection,
blockwise dropping + SkipMode, whole-layer fused TKG facade, SP-replicated
shared experts (reference expert_mlps_v2.py:595-689,1407-1499,
moe_parallel_layers.py:263-276, moe_fused_tkg.py:24-250,
shared_experts.py:73-207)."""

import pytest
import torch
import torch.nn.functional as F

from dist_utils import run_distributed


def _mk_mlps(E=8, H=16, I=32, k=2):
    from neuronx_distributed_amd.moe import ExpertMLPs

    torch.manual_seed(0)
    mlps = ExpertMLPs(E, H, I, k, capacity_factor=None, dtype=torch.float32)
    return mlps


def _router_masked_aff(T, E, k, seed=3):
    torch.manual_seed(seed)
    probs = torch.softmax(torch.randn(T, E), -1)
    vals, idx = torch.topk(probs, k, dim=-1)
    aff = torch.zeros(T, E).scatter(-1, idx, vals)
    return aff, idx


def _selective_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, k = 8, 16, 32, 2
    mlps = _mk_mlps(E, H, I, k)
    mlps.eval()
    T = 3  # T*k/E = 0.75 < SELECTIVE_LOADING_THRESHOLD -> selective
    torch.manual_seed(1)
    x = torch.randn(T, H)
    aff, idx = _router_masked_aff(T, E, k)

    sel = mlps.forward_selective(x, aff, idx)
    ref = mlps.forward_all_experts(x, aff)
    assert torch.allclose(sel, ref, atol=1e-5), (sel - ref).abs().max()

    # the dispatch itself must route small decode batches to selective
    called = {}
    orig = mlps.forward_selective

    def spy(*a, **kw):
        called["yes"] = True
        return orig(*a, **kw)

    mlps.forward_selective = spy
    out = mlps(x, aff, idx)
    assert called.get("yes"), "dispatch did not choose selective loading"
    assert torch.allclose(out, ref, atol=1e-5)
    return 0.0


def test_selective_loading_matches_all_experts():
    run_distributed(_selective_worker, world_size=1)


def _expert_indices_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    mlps = _mk_mlps()
    idx = torch.tensor([5, 2, 2, 7])
    x = torch.randn(4, 3, 16)
    out = mlps.gate_up_proj(x, expert_indices=idx)
    ref = torch.bmm(x, mlps.gate_up_proj.weight.index_select(0, idx))
    assert torch.equal(out, ref)
    return 0.0


def test_expert_indices_subselection():
    run_distributed(_expert_indices_worker, world_size=1)


def _blockwise_drop_worker(rank, world):
    from neuronx_distributed_amd.moe.blockwise import (blockwise_mm,
                                                       compute_block_indices)
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, k, T, B = 4, 16, 32, 1, 10, 4
    torch.manual_seed(0)
    hidden = torch.randn(T, H)
    gu_w = torch.randn(E, H, 2 * I) * 0.1
    d_w = torch.randn(E, I, H) * 0.1
    # 7 tokens on expert 1 (needs 2 blocks; budget 1 -> 3 dropped), rest on 3
    eidx = torch.tensor([1, 1, 1, 1, 3, 1, 1, 1, 3, 3]).unsqueeze(1)
    aff = torch.zeros(T, E)
    aff[torch.arange(T), eidx.squeeze(1)] = 1.0

    tpi, b2e, _ = compute_block_indices(eidx, E, B, max_blocks_per_expert=1)
    out = blockwise_mm(hidden, aff, gu_w, d_w, tpi, b2e, eidx, B)

    # golden: first B tokens per expert IN ARRIVAL ORDER survive
    kept = {}
    ref = torch.zeros(T, H)
    for t in range(T):
        e = int(eidx[t])
        kept.setdefault(e, 0)
        if kept[e] >= B:
            continue  # dropped
        kept[e] += 1
        gu = hidden[t] @ gu_w[e]
        act = F.silu(gu[:I]) * gu[I:]
        ref[t] = act @ d_w[e]
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    return 0.0


def test_blockwise_dropping_budget():
    run_distributed(_blockwise_drop_worker, world_size=1)


def _skipmode_worker(rank, world):
    """Padding-only blocks never hit the GEMMs but the result is identical
    to the math without skipping."""
    from neuronx_distributed_amd.moe.blockwise import (blockwise_mm,
                                                       compute_block_indices)
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, B = 4, 16, 32, 8
    torch.manual_seed(0)
    T = 6
    hidden = torch.randn(T, H)
    gu_w = torch.randn(E, H, 2 * I) * 0.1
    d_w = torch.randn(E, I, H) * 0.1
    eidx = torch.full((T, 1), 2)  # all on one expert; experts 0/1/3 empty
    aff = torch.zeros(T, E)
    aff[:, 2] = 1.0
    # force a static num_blocks with padding blocks mapped to expert 0
    tpi, b2e, used = compute_block_indices(eidx, E, B, num_blocks=4)
    out = blockwise_mm(hidden, aff, gu_w, d_w, tpi, b2e, eidx, B)
    gu = hidden @ gu_w[2]
    ref = F.silu(gu[:, :I]) * gu[:, I:] @ d_w[2]
    assert torch.allclose(out, ref, atol=1e-5)
    return 0.0


def test_blockwise_skipmode_empty_blocks():
    run_distributed(_skipmode_worker, world_size=1)


def _tkg_worker(rank, world):
    from neuronx_distributed_amd.models.llama import RMSNorm
    from neuronx_distributed_amd.moe import ExpertMLPs, RouterTopK
    from neuronx_distributed_amd.moe.moe_fused_tkg import MoEFusedTKG
    from neuronx_distributed_amd.moe.shared_experts import SharedExperts
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, k = 4, 16, 32, 2
    torch.manual_seed(0)
    router = RouterTopK(E, k, H, dtype=torch.float32)
    mlps = ExpertMLPs(E, H, I, k, capacity_factor=None, dtype=torch.float32)
    norm = RMSNorm(H)
    shared = SharedExperts(H, I, dtype=torch.float32)
    tkg = MoEFusedTKG(router, mlps, norm=norm, shared_experts=shared).eval()

    torch.manual_seed(2)
    x = torch.randn(2, 1, H)
    out = tkg(x)

    # golden: manual compose
    h = x.reshape(-1, H)
    n = norm(h)
    _, aff, idx = router(n)
    ref = mlps.forward_all_experts(n, aff) + shared(n) + h
    assert torch.allclose(out.reshape(-1, H), ref, atol=1e-5), \
        (out.reshape(-1, H) - ref).abs().max()
    return 0.0


def test_moe_fused_tkg_facade_cpu():
    run_distributed(_tkg_worker, world_size=1)


def _shared_replicated_worker(rank, world):
    """replicate_for_sp: identical weights on every rank, output equals
    the plain dense MLP, and is rank-local (no collective involved)."""
    from neuronx_distributed_amd.moe.shared_experts import SharedExperts
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    H, I = 16, 32
    torch.manual_seed(0)
    se = SharedExperts(H, I, replicate_for_sp=True, dtype=torch.float32)
    torch.manual_seed(100 + rank)  # per-rank (SP-slice) tokens
    x = torch.randn(5, H)
    out = se(x)
    gu = F.linear(x, se.gate_up_proj)
    ref = F.linear(F.silu(gu[:, :I]) * gu[:, I:], se.down_proj)
    assert torch.allclose(out, ref, atol=1e-6)
    # weights replicated -> same checksum on every rank
    wsum = se.gate_up_proj.sum()
    mx = wsum.clone()
    torch.distributed.all_reduce(mx, op=torch.distributed.ReduceOp.MAX)
    assert torch.allclose(wsum, mx)
    return float(out.sum())


def test_shared_experts_replicated_tp2():
    run_distributed(_shared_replicated_worker, world_size=2)


@pytest.mark.gpu
def test_blockwise_grouped_mm_matches_loop(monkeypatch):
    """The single grouped-GEMM blockwise path (torch._grouped_mm over
    hipBLASLt grouped kernels) must match the per-expert-loop path in
    forward AND gradients."""
    from neuronx_distributed_amd.moe import blockwise as bw
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import os
    import torch.distributed as dist

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29764")
        dist.init_process_group("nccl", rank=0, world_size=1)
    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)

    T, H, I, E, k, B = 512, 256, 192, 8, 2, 64
    dev = "cuda"
    torch.manual_seed(0)
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    gu_w = (torch.randn(E, H, 2 * I, device=dev, dtype=torch.bfloat16)
            * 0.05).requires_grad_(True)
    d_w = (torch.randn(E, I, H, device=dev, dtype=torch.bfloat16)
           * 0.05).requires_grad_(True)
    vals, idx = torch.topk(torch.softmax(torch.randn(T, E, device=dev), -1),
                           k, -1)
    aff = torch.zeros(T, E, device=dev).scatter(-1, idx, vals)
    tpi, b2e, _ = bw.compute_block_indices(idx, E, B)

    xg = x.clone().requires_grad_(True)
    out_g = bw.blockwise_mm(xg, aff, gu_w, d_w, tpi, b2e, idx, B)
    out_g.float().pow(2).sum().backward()
    grads_g = (xg.grad.clone(), gu_w.grad.clone(), d_w.grad.clone())

    gu_w.grad = None
    d_w.grad = None
    monkeypatch.delattr(torch, "_grouped_mm")  # force the loop path
    xl = x.clone().requires_grad_(True)
    out_l = bw.blockwise_mm(xl, aff, gu_w, d_w, tpi, b2e, idx, B)
    out_l.float().pow(2).sum().backward()

    assert (out_g.float() - out_l.float()).abs().max() < 2e-2
    for a, b in zip(grads_g, (xl.grad, gu_w.grad, d_w.grad)):
        rel = (a.float() - b.float()).norm() / (b.float().norm() + 1e-9)
        assert rel < 2e-2, rel


@pytest.mark.gpu
def test_moe_fused_tkg_gpu_fused_matches_unfused():
    from neuronx_distributed_amd.models.llama import RMSNorm
    from neuronx_distributed_amd.moe import ExpertMLPs, RouterTopK
    from neuronx_distributed_amd.moe.moe_fused_tkg import MoEFusedTKG
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import os
    import torch.distributed as dist

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29763")
        dist.init_process_group("nccl", rank=0, world_size=1)
    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, k = 8, 256, 192, 2
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    torch.manual_seed(0)
    with torch.device("cuda"):
        router = RouterTopK(E, k, H)
        mlps = ExpertMLPs(E, H, I, k, capacity_factor=None)
        norm = RMSNorm(H)
    torch.set_default_dtype(prev)
    tkg = MoEFusedTKG(router, mlps, norm=norm).eval()
    torch.manual_seed(2)
    x = torch.randn(32, 1, H, device="cuda", dtype=torch.bfloat16)
    assert tkg.can_fuse(x)
    out = tkg._fused(x)
    ref = tkg._unfused(x)
    err = (out.float() - ref.float()).abs().max()
    assert err < 3e-2, err


def _shuffle_grad_worker(rank, world):
    """Token shuffling must be autograd-aware: gradients ride the inverse
    all-to-all back, so shuffle -> f -> unshuffle has exact identity-like
    gradients (reference _AllToAllForTokenShuffle)."""
    from neuronx_distributed_amd.moe.token_shuffling import (token_shuffle,
                                                             token_unshuffle)
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1,
                                 token_shuffle_group_size=world)
    torch.manual_seed(10 + rank)
    x = torch.randn(8, 4, requires_grad=True)
    h, perm = token_shuffle(x, seed=42)
    out = token_unshuffle(h * 2.0, perm)
    torch.manual_seed(20 + rank)
    g = torch.randn_like(out)
    out.backward(g)
    assert torch.allclose(x.grad, 2.0 * g, atol=1e-6), \
        (x.grad - 2.0 * g).abs().max()
    return 0.0


def test_token_shuffle_gradients_world2():
    run_distributed(_shuffle_grad_worker, world_size=2)


def test_sampler_cdf_multinomial_and_medusa():
    from neuronx_distributed_amd.utils.sampling import Sampler

    torch.manual_seed(0)
    # a concentrated distribution: CDF sampling must pick inside top mass
    probs = torch.tensor([[0.9, 0.05, 0.03, 0.02]])
    picks = [int(Sampler._multinomial_cdf(probs)) for _ in range(200)]
    assert picks.count(0) > 140  # ~90%
    assert max(picks) <= 3

    from neuronx_distributed_amd.parallel import parallel_state as ps
    if not ps.model_parallel_is_initialized():
        import torch.distributed as dist
        import os
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29767")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        if not dist.is_initialized():
            dist.init_process_group("gloo", rank=0, world_size=1)
        ps.initialize_model_parallel(tensor_model_parallel_size=1)
    logits = torch.randn(3, 64)
    s = Sampler(do_sample=True, top_k=8, on_device_multinomial=True)
    ids = s(logits)
    assert ids.shape == (3,)
    m = Sampler(do_sample=True, top_k=5, return_topk_indices=True)
    topk_idx = m(logits)
    assert topk_idx.shape == (3, 5)
    ref = logits.topk(5, dim=-1).indices
    assert torch.equal(topk_idx, ref)


def _mx_blockwise_worker(rank, world):
    """K5 MX variant: blockwise_mm_mx with fp4-e2m1 MX weights tracks the
    full-precision blockwise output within fp4 quantization error."""
    from neuronx_distributed_amd.moe.blockwise import (blockwise_mm,
                                                       blockwise_mm_mx,
                                                       compute_block_indices)
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.quantization.microscaling import quantize_mx

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    E, H, I, k, T, B = 4, 64, 32, 2, 24, 8
    torch.manual_seed(0)
    hidden = torch.randn(T, H)
    gu_w = torch.randn(E, H, 2 * I) * 0.1
    d_w = torch.randn(E, I, H) * 0.1
    vals, idx = torch.topk(torch.softmax(torch.randn(T, E), -1), k, -1)
    aff = torch.zeros(T, E).scatter(-1, idx, vals)
    tpi, b2e, _ = compute_block_indices(idx, E, B)

    ref = blockwise_mm(hidden, aff, gu_w, d_w, tpi, b2e, idx, B)

    gq, gs = quantize_mx(gu_w, fmt="fp4_e2m1", axis=1)
    dq, ds = quantize_mx(d_w, fmt="fp4_e2m1", axis=1)
    out = blockwise_mm_mx(hidden, aff, gq, gs, dq, ds, tpi, b2e, idx, B)
    rel = (out - ref).norm() / (ref.norm() + 1e-9)
    assert rel < 0.35, rel  # fp4 resolution
    # fp8 variant is much tighter
    gq8, gs8 = quantize_mx(gu_w, fmt="fp8_e4m3", axis=1)
    dq8, ds8 = quantize_mx(d_w, fmt="fp8_e4m3", axis=1)
    out8 = blockwise_mm_mx(hidden, aff, gq8, gs8, dq8, ds8, tpi, b2e, idx, B)
    rel8 = (out8 - ref).norm() / (ref.norm() + 1e-9)
    assert rel8 < 0.08, rel8
    return 0.0


def test_blockwise_mx_variants():
    run_distributed(_mx_blockwise_worker, world_size=1)


def test_moe_configs_and_validator(tmp_path):
    """moe_configs dataclasses + MoeConfigValidator rules (reference
    moe_configs.py / moe_config_validator.py)."""
    import json
    import types

    import pytest as _pytest

    from neuronx_distributed_amd.moe import (BlockwiseMatmulConfig,
                                             GLUType, MoeConfigValidator,
                                             RoutedExpertsMLPOpsConfig)

    c = RoutedExpertsMLPOpsConfig(num_experts=8, top_k=2, hidden_size=64,
                                  intermediate_size=128)
    assert c.glu_type == GLUType.SWIGLU
    with _pytest.raises(ValueError):
        RoutedExpertsMLPOpsConfig(num_experts=2, top_k=4, hidden_size=8,
                                  intermediate_size=8)
    b = BlockwiseMatmulConfig.from_kwargs(block_size=128, bogus_key=1)
    assert b.block_size == 128

    hf = tmp_path / "config.json"
    hf.write_text(json.dumps({"hidden_act": "silu"}))
    moe = types.SimpleNamespace(dropless=True, capacity_factor=1.5,
                                glu_mlp=True)
    cfg = types.SimpleNamespace(
        model_source="hf",
        model=types.SimpleNamespace(moe=moe, model_config=str(hf)))
    MoeConfigValidator(cfg).validate_moe_config()
    assert cfg.model.moe.capacity_factor == 0.0  # dropless forces 0

    hf.write_text(json.dumps({"hidden_act": "gelu"}))
    with _pytest.raises(ValueError):
        MoeConfigValidator(cfg).validate_moe_config()

    moe2 = types.SimpleNamespace(dropless=False, capacity_factor=-1.0,
                                 glu_mlp=True)
    cfg2 = types.SimpleNamespace(model_source="megatron",
                                 model=types.SimpleNamespace(moe=moe2))
    with _pytest.raises(ValueError):
        MoeConfigValidator(cfg2).validate_moe_config()


def _hybrid_group_worker(rank, world):
    """Hybrid CTE/TKG sharding groups: prefill pair (tp=world) vs decode
    pair (ep=world) expose different meshes by phase."""
    from neuronx_distributed_amd.moe import (
        destroy_moe_model_parallel, get_moe_ep_group, get_moe_tp_ep_group,
        init_tensor_expert_parallel_moe_process_groups)
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    init_tensor_expert_parallel_moe_process_groups(
        tkg_tp_degree=1, tkg_ep_degree=world,
        cte_tp_degree=world, cte_ep_degree=1)
    cte_tp = get_moe_tp_ep_group(prefill=True)
    tkg_tp = get_moe_tp_ep_group(prefill=False)
    tkg_ep = get_moe_ep_group(prefill=False)
    assert cte_tp.size == world
    assert tkg_tp.size == 1
    assert tkg_ep.size == world
    # the groups are usable communicators
    import torch.distributed as dist
    t = torch.ones(1)
    dist.all_reduce(t, group=cte_tp.group)
    assert float(t) == world
    destroy_moe_model_parallel()
    return 0.0


def test_moe_hybrid_process_groups():
    run_distributed(_hybrid_group_worker, world_size=2)
