"""Sequence-parallel comm/GEMM overlap machinery (NXDA_SP_OVERLAP=1):
ring-pipelined all-gather producer in ColumnParallel and chunked
reduce-scatter consumer in RowParallel must be numerically identical to
the unfused paths, forward AND backward (SURVEY §7 hard-parts; reference
layers_utils.py:91-103)."""

import os

import torch

from dist_utils import run_distributed


def _overlap_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.layers import (
        ColumnParallelLinear, RowParallelLinear)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    S, B, H, I = 8, 2, 16, 32  # S is the per-rank (sharded) seq len
    torch.manual_seed(0)
    cpl = ColumnParallelLinear(H, I, bias=True, gather_output=False,
                               sequence_parallel_enabled=True,
                               dtype=torch.float32)
    rpl = RowParallelLinear(I, H, bias=True, input_is_parallel=True,
                            sequence_parallel_enabled=True,
                            dtype=torch.float32)

    torch.manual_seed(100 + rank)
    x = torch.randn(S, B, H)

    results = {}
    for mode in ("0", "1"):
        os.environ["NXDA_SP_OVERLAP"] = mode
        xg = x.clone().requires_grad_(True)
        mid = cpl(xg)               # (world*S, B, I/tp)
        out = rpl(mid)              # (S, B, H) back in SP form
        loss = out.float().pow(2).sum()
        loss.backward()
        results[mode] = (out.detach().clone(), xg.grad.clone(),
                         cpl.weight.grad.clone(), rpl.weight.grad.clone(),
                         float(loss))
        cpl.weight.grad = None
        rpl.weight.grad = None
        cpl.bias.grad = None
        rpl.bias.grad = None

    o0, g0, cw0, rw0, l0 = results["0"]
    o1, g1, cw1, rw1, l1 = results["1"]
    assert torch.allclose(o0, o1, atol=1e-5), (o0 - o1).abs().max()
    assert torch.allclose(g0, g1, atol=1e-5), (g0 - g1).abs().max()
    assert torch.allclose(cw0, cw1, atol=1e-4), (cw0 - cw1).abs().max()
    assert torch.allclose(rw0, rw1, atol=1e-4), (rw0 - rw1).abs().max()
    assert abs(l0 - l1) < 1e-3 * (1 + abs(l0))
    return l0


def test_sp_overlap_matches_unfused_tp2():
    run_distributed(_overlap_worker, world_size=2)


def test_sp_overlap_matches_unfused_tp4():
    run_distributed(_overlap_worker, world_size=4)


def _qkv_overlap_worker(rank, world):
    """Ring-overlapped SP QKV == plain all-gather + 3 GEMMs, fwd + bwd."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.qkv_linear import (
        GQAQKVColumnParallelLinear)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    S, B, H = 8, 2, 16
    torch.manual_seed(0)
    qkv = GQAQKVColumnParallelLinear(
        H, [32, 16], head_dim=4, num_attention_heads=8,
        num_key_value_heads=4, sequence_parallel_enabled=True,
        dtype=torch.float32)
    qkv.train()
    torch.manual_seed(100 + rank)
    x = torch.randn(S, B, H)

    results = {}
    for mode in ("0", "1"):
        os.environ["NXDA_SP_OVERLAP"] = mode
        xg = x.clone().requires_grad_(True)
        with torch.enable_grad():
            q, k, v = qkv(xg)
            loss = (q.float().pow(2).sum() + k.float().pow(2).sum()
                    + v.float().pow(2).sum())
        loss.backward()
        results[mode] = (q.detach().clone(), k.detach().clone(),
                         v.detach().clone(), xg.grad.clone(),
                         qkv.weight_q.grad.clone(),
                         qkv.weight_k.grad.clone())
        for w in (qkv.weight_q, qkv.weight_k, qkv.weight_v):
            w.grad = None
    for a, b in zip(results["0"], results["1"]):
        assert torch.allclose(a, b, atol=1e-4), (a - b).abs().max()
    return 0.0


def test_sp_overlap_qkv_tp4():
    run_distributed(_qkv_overlap_worker, world_size=4)
