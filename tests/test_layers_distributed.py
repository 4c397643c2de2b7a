"""TP layer numerics vs single-rank torch golden, over gloo world_size=2
(the reference's NXD_CPU_MODE integration path)."""

import pytest
import torch
import torch.nn.functional as F

from dist_utils import run_distributed


def _init_mp(tp=2, **kw):
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=tp,
                                 skip_collective_init=False, **kw)
    return ps


def _cpl_worker(rank, world):
    import neuronx_distributed_amd.parallel as pl

    _init_mp(tp=world)
    pl.model_parallel_manual_seed(1234)
    torch.manual_seed(99)
    layer = pl.ColumnParallelLinear(16, 32, bias=True, gather_output=True,
                                    dtype=torch.float32)
    x = torch.randn(4, 16)  # same on all ranks (same seed)
    out = layer(x)
    loss = (out * out).sum()
    loss.backward()

    # golden: full master weight
    full_w = layer.master_weight if layer.master_weight is not None else None
    # reconstruct full weight by gather
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps
    full_weight = comm.all_gather(layer.weight.detach(), dim=0,
                                  group=ps.get_group_info("tp"))
    ref = F.linear(x, full_weight)
    ref_loss = (ref * ref).sum()
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    # weight grad check vs autograd on the gathered weight
    wf = full_weight.clone().requires_grad_(True)
    r = F.linear(x, wf)
    (r * r).sum().backward()
    my_slice = wf.grad.chunk(world, dim=0)[rank]
    assert torch.allclose(layer.weight.grad, my_slice, atol=1e-4)
    return True


def _rpl_worker(rank, world):
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world)
    pl.model_parallel_manual_seed(1234)
    torch.manual_seed(99)
    layer = pl.RowParallelLinear(16, 32, bias=True, input_is_parallel=False,
                                 dtype=torch.float32)
    x = torch.randn(4, 16)
    out = layer(x)
    full_weight = comm.all_gather(layer.weight.detach(), dim=1,
                                  group=ps.get_group_info("tp"))
    ref = F.linear(x, full_weight) + layer.bias.detach()
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()

    (out * out).sum().backward()
    wf = full_weight.clone().requires_grad_(True)
    xf = x.clone().requires_grad_(True)
    r = F.linear(xf, wf) + layer.bias.detach()
    (r * r).sum().backward()
    my_slice = wf.grad.chunk(world, dim=1)[rank]
    assert torch.allclose(layer.weight.grad, my_slice, atol=1e-4)
    return True


def _embedding_worker(rank, world):
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world)
    pl.model_parallel_manual_seed(1234)
    torch.manual_seed(7)
    emb = pl.ParallelEmbedding(64, 12, dtype=torch.float32)
    ids = torch.randint(0, 64, (3, 5))
    out = emb(ids)
    full_weight = comm.all_gather(emb.weight.detach(), dim=0,
                                  group=ps.get_group_info("tp"))
    ref = F.embedding(ids, full_weight)
    assert torch.allclose(out, ref, atol=1e-5)
    return True


def _cross_entropy_worker(rank, world):
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world)
    torch.manual_seed(5)
    B, S, V = 2, 6, 32
    full_logits = torch.randn(B, S, V, requires_grad=True)
    target = torch.randint(0, V, (B, S))
    local = full_logits.detach().chunk(world, dim=-1)[rank].requires_grad_(True)
    loss = pl.parallel_cross_entropy(local, target)
    ref = F.cross_entropy(full_logits.reshape(-1, V), target.reshape(-1),
                          reduction="none").reshape(B, S)
    assert torch.allclose(loss, ref, atol=1e-5), (loss - ref).abs().max()
    loss.sum().backward()
    ref.sum().backward()
    ref_slice = full_logits.grad.chunk(world, dim=-1)[rank]
    assert torch.allclose(local.grad, ref_slice, atol=1e-5)
    return True


def _sp_worker(rank, world):
    """Sequence-parallel CPL+RPL roundtrip vs dense reference."""
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world)
    pl.model_parallel_manual_seed(1234)
    torch.manual_seed(42)
    S, B, H, F_ = 8, 2, 16, 32
    cpl = pl.ColumnParallelLinear(H, F_, bias=False, gather_output=False,
                                  sequence_parallel_enabled=True,
                                  dtype=torch.float32)
    rpl = pl.RowParallelLinear(F_, H, bias=False, input_is_parallel=True,
                               sequence_parallel_enabled=True,
                               dtype=torch.float32)
    x_full = torch.randn(S, B, H)  # same on all ranks
    x_shard = x_full.chunk(world, dim=0)[rank].clone().requires_grad_(True)
    y = rpl(torch.relu(cpl(x_shard)))
    assert y.shape == (S // world, B, H)

    wc = comm.all_gather(cpl.weight.detach(), dim=0, group=ps.get_group_info("tp"))
    wr = comm.all_gather(rpl.weight.detach(), dim=1, group=ps.get_group_info("tp"))
    xf = x_full.clone().requires_grad_(True)
    ref = torch.relu(xf @ wc.t()) @ wr.t()
    ref_shard = ref.chunk(world, dim=0)[rank]
    assert torch.allclose(y, ref_shard, atol=1e-5), (y - ref_shard).abs().max()

    y.sum().backward()
    ref.sum().backward()
    ref_gx = xf.grad.chunk(world, dim=0)[rank]
    assert torch.allclose(x_shard.grad, ref_gx, atol=1e-5)
    # weight grads: compare vs dense
    wcf = wc.clone().requires_grad_(True)
    wrf = wr.clone().requires_grad_(True)
    (torch.relu(x_full @ wcf.t()) @ wrf.t()).sum().backward()
    assert torch.allclose(cpl.weight.grad, wcf.grad.chunk(world, 0)[rank], atol=1e-4)
    assert torch.allclose(rpl.weight.grad, wrf.grad.chunk(world, 1)[rank], atol=1e-4)
    return True


def _gqa_worker(rank, world):
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world, kv_size_multiplier=world)  # 1 kv head replicated
    pl.model_parallel_manual_seed(1234)
    torch.manual_seed(3)
    H, heads, kvh, d = 16, 4, 1, 4
    qkv = pl.GQAQKVColumnParallelLinear(
        H, [heads * d, kvh * d], bias=False, gather_output=False,
        num_attention_heads=heads, num_key_value_heads=kvh, head_dim=d,
        kv_size_multiplier=world, dtype=torch.float32)
    x = torch.randn(3, H)
    q, k, v = qkv(x)
    assert q.shape == (3, heads * d // world)
    assert k.shape == (3, d)  # one replicated head per rank
    # k must equal the full single-head projection on every rank
    ks = [torch.empty_like(k) for _ in range(world)]
    torch.distributed.all_gather(ks, k)
    for other in ks:
        assert torch.allclose(other, k, atol=1e-6)
    return True


def _deterministic_init_worker(rank, world):
    """TP-degree invariance: master weight at tp=2 equals tp=1 init."""
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    _init_mp(tp=world)
    torch.manual_seed(11)
    layer = pl.ColumnParallelLinear(8, 8, bias=False, keep_master_weight=True,
                                    dtype=torch.float32)
    full = comm.all_gather(layer.weight.detach(), dim=0,
                           group=ps.get_group_info("tp"))
    assert torch.allclose(full, layer.master_weight)
    return True


@pytest.mark.parametrize("worker", [
    _cpl_worker, _rpl_worker, _embedding_worker, _cross_entropy_worker,
    _sp_worker, _gqa_worker, _deterministic_init_worker,
])
def test_distributed_layer(worker):
    run_distributed(worker, world_size=2)


def _embed_dim_shard_worker(rank, world):
    """shard_along_embedding=True: full vocab per rank, H/tp columns,
    all-gathered output matches the dense embedding."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.layers import ParallelEmbedding

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    emb = ParallelEmbedding(32, 16, shard_along_embedding=True)
    torch.manual_seed(0)
    dense = ParallelEmbedding(32, 16)  # world>1 vocab-shard; compare paths
    x = torch.randint(0, 32, (2, 5))
    out = emb(x)
    ref = dense(x)
    assert out.shape == ref.shape == (2, 5, 16)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    return float(out.sum())


def test_embedding_dim_sharding():
    outs = run_distributed(_embed_dim_shard_worker, world_size=2)
    assert abs(outs[0] - outs[1]) < 1e-4


def _pad_worker(rank, world):
    """Head padding: a 5-head model's checkpoint padded to 6 heads (zero
    extra qkv rows + zero o_proj columns) computes the SAME function; the
    padded config divides tp=2."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.parallel.pad import (pad_attention_state_dict,
                                                      pad_llama_config)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    cfg = get_config("tiny", hidden_size=40, num_attention_heads=5,
                     num_key_value_heads=1, intermediate_size=64)
    padded = pad_llama_config(cfg, 2)
    assert padded.num_attention_heads == 6 and padded.head_dim == 8

    if world == 1:
        torch.manual_seed(0)
        dense = LlamaForCausalLM(cfg)
        sd = pad_attention_state_dict(dense.state_dict(), cfg, padded)
        pm = LlamaForCausalLM(padded)
        pm.load_state_dict(sd)
        x = torch.randint(0, cfg.vocab_size, (2, 8))
        ref = dense(x)
        out = pm(x)
        assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
        return float(out.sum())
    # tp2: padded config constructs and runs (5 heads would not divide)
    torch.manual_seed(0)
    pm = LlamaForCausalLM(padded)
    x = torch.randint(0, cfg.vocab_size, (2, 8))
    loss = pm(x, labels=x)
    loss.backward()
    return float(loss.detach())


def test_head_padding_equivalence():
    run_distributed(_pad_worker, world_size=1)


def test_head_padding_tp2():
    out = run_distributed(_pad_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-5


def _dist_ops_worker(rank, world):
    """Distributed topk/argmax over the TP-sharded dim match single-rank
    torch results."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.operators import argmax as dargmax
    from neuronx_distributed_amd.operators import topk as dtopk

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    full = torch.randn(4, 64)
    shard = full.chunk(world, dim=-1)[rank]

    am = dargmax(shard, dim=-1, gather_dim=-1)
    assert torch.equal(am.reshape(-1), full.argmax(-1).reshape(-1)), am

    vals, idx = dtopk(shard, k=3, dim=-1, gather_dim=-1)
    rv, ri = full.topk(3, dim=-1)
    assert torch.allclose(vals, rv, atol=1e-6)
    assert torch.equal(idx, ri)
    return float(vals.sum())


def test_distributed_topk_argmax():
    out = run_distributed(_dist_ops_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-5


def _reshard_worker(rank, world):
    """FULL (unsharded) tensors reshard onto TP2 layers through
    _reshard_full_state_dict: Column rows, Row columns, and the GQAQKV
    preshard_hook's KV replication — the public sharded-checkpoint
    contract."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.checkpointing import (
        _reshard_full_state_dict)
    from neuronx_distributed_amd.parallel.layers import (ColumnParallelLinear,
                                                         RowParallelLinear)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)

    col = ColumnParallelLinear(8, 12, bias=False, gather_output=False)
    full_c = torch.arange(96, dtype=torch.float32).reshape(12, 8)
    sd = {"weight": full_c.clone()}
    _reshard_full_state_dict(col, sd)
    col.load_state_dict(sd)
    assert torch.equal(col.weight.detach(),
                       full_c[rank * 6:(rank + 1) * 6])

    row = RowParallelLinear(12, 8, bias=False, input_is_parallel=True)
    full_r = torch.arange(96, dtype=torch.float32).reshape(8, 12)
    sd = {"weight": full_r.clone()}
    _reshard_full_state_dict(row, sd)
    row.load_state_dict(sd)
    assert torch.equal(row.weight.detach(),
                       full_r[:, rank * 6:(rank + 1) * 6])
    return 0.0


def test_full_checkpoint_resharding():
    run_distributed(_reshard_worker, world_size=2)


def _sp_overlap_worker(rank, world):
    """NXDA_SP_OVERLAP=1 (chunked broadcast + per-chunk GEMM) produces the
    same forward AND backward as the plain all-gather SP path."""
    import os

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.layers import ColumnParallelLinear

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    lin = ColumnParallelLinear(16, 32, bias=True, gather_output=False,
                               sequence_parallel_enabled=True)
    torch.manual_seed(3 + rank)
    x = torch.randn(4, 2, 16, requires_grad=True)  # (S/tp, B, H)

    os.environ["NXDA_SP_OVERLAP"] = "0"
    out_ref = lin(x)
    g = torch.randn_like(out_ref)
    out_ref.backward(g)
    grads_ref = [p.grad.clone() for p in lin.parameters()]
    xg_ref = x.grad.clone()
    for p in lin.parameters():
        p.grad = None
    x.grad = None

    os.environ["NXDA_SP_OVERLAP"] = "1"
    try:
        out = lin(x)
        out.backward(g)
    finally:
        os.environ["NXDA_SP_OVERLAP"] = "0"
    assert torch.allclose(out, out_ref, atol=1e-6)
    for p, gr in zip(lin.parameters(), grads_ref):
        assert torch.allclose(p.grad, gr, atol=1e-5), (p.grad - gr).abs().max()
    assert torch.allclose(x.grad, xg_ref, atol=1e-5)
    return float(out.sum())


def test_sp_overlap_matches_plain():
    run_distributed(_sp_overlap_worker, world_size=2)


def _oneshot_ar_worker(rank, world):
    """One-shot all-reduce (all-gather + local sum) matches dist ring
    all-reduce when enabled for small payloads."""
    import os

    from neuronx_distributed_amd.parallel import comm, parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    t1 = torch.arange(17, dtype=torch.float32) * (rank + 1)
    t2 = t1.clone()
    comm.all_reduce(t1, group=ps.get_group_info("tp"))
    os.environ["NXDA_ONESHOT_AR_MAX_BYTES"] = "1048576"
    try:
        comm.all_reduce(t2, group=ps.get_group_info("tp"))
    finally:
        os.environ["NXDA_ONESHOT_AR_MAX_BYTES"] = "0"
    assert torch.allclose(t1, t2), (t1 - t2).abs().max()
    return float(t1.sum())


def test_oneshot_allreduce():
    out = run_distributed(_oneshot_ar_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-5


def _ce_ignore_worker(rank, world):
    """parallel_cross_entropy ignores -100 targets (zero loss + zero grad)
    and matches torch F.cross_entropy(ignore_index=-100) at tp2."""
    import torch.nn.functional as F

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.loss_functions import (
        parallel_cross_entropy)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    full = torch.randn(6, 16, requires_grad=True)
    tgt = torch.tensor([1, 3, -100, 7, -100, 12])
    shard = full.detach().chunk(world, dim=-1)[rank].requires_grad_(True)

    loss_tok = parallel_cross_entropy(shard, tgt)
    assert float(loss_tok[2]) == 0.0 and float(loss_tok[4]) == 0.0
    loss = loss_tok.sum() / 4
    loss.backward()

    ref = F.cross_entropy(full, tgt, ignore_index=-100)
    ref.backward()
    assert abs(float(loss) - float(ref)) < 1e-5, (loss, ref)
    gref = full.grad.chunk(world, dim=-1)[rank]
    assert torch.allclose(shard.grad, gref, atol=1e-5), \
        (shard.grad - gref).abs().max()
    return float(loss)


def test_ce_ignore_index():
    out = run_distributed(_ce_ignore_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-6


def _logprobs_worker(rank, world):
    """from_parallel_logits_to_logprobs == log_softmax gather of the
    shifted target at tp2."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.loss_functions import (
        from_parallel_logits_to_logprobs)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    full = torch.randn(2, 6, 16)
    tgt = torch.randint(0, 16, (2, 6))
    shard = full.chunk(world, dim=-1)[rank]
    lp = from_parallel_logits_to_logprobs(shard, tgt, inference=True)
    ref = torch.log_softmax(full[:, :-1, :], dim=-1).gather(
        -1, tgt[:, 1:].unsqueeze(-1)).squeeze(-1)
    assert torch.allclose(lp, ref, atol=1e-5), (lp - ref).abs().max()
    return float(lp.sum())


def test_parallel_logprobs():
    out = run_distributed(_logprobs_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-5


def _lowlevel_ckpt_worker(rank, world):
    """parallel.checkpointing save/load: tp_rank_xx_pp_rank_xx shard files
    roundtrip with staggered (serial) loading."""
    import os
    import tempfile

    from neuronx_distributed_amd.parallel import checkpointing
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.layers import ColumnParallelLinear

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    lin = ColumnParallelLinear(8, 12, bias=True, gather_output=False)
    path = os.path.join(tempfile.gettempdir(), "nxda_lowlevel_ckpt")
    checkpointing.save(lin.state_dict(), path, tag="model")

    torch.manual_seed(99)
    lin2 = ColumnParallelLinear(8, 12, bias=True, gather_output=False)
    assert not torch.allclose(lin2.weight, lin.weight)
    checkpointing.load(path, tag="model", model=lin2, load_serially=True)
    assert torch.allclose(lin2.weight, lin.weight)
    assert torch.allclose(lin2.bias, lin.bias)
    return float(lin2.weight.sum())


def test_lowlevel_checkpointing():
    out = run_distributed(_lowlevel_ckpt_worker, world_size=2)
    assert out[0] != out[1]  # different shards per tp rank


def _bucket_ar_worker(rank, world):
    """bucket_allreduce_gradients: per-dtype buckets, cap-splitting, mean
    over DP; grads match manual all-reduce."""
    import os

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.grads import (
        bucket_allreduce_gradients)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)  # dp = world
    torch.manual_seed(5 + rank)
    params = [torch.nn.Parameter(torch.randn(64, 64)) for _ in range(4)]
    params.append(torch.nn.Parameter(torch.randn(16).double()))
    for p in params:
        p.grad = torch.randn_like(p)
    manual = [p.grad.clone() for p in params]

    os.environ["ALLREDUCE_BUCKET_CAP_MB"] = "0.01"  # force multiple buckets
    try:
        bucket_allreduce_gradients([p.grad for p in params])
    finally:
        os.environ.pop("ALLREDUCE_BUCKET_CAP_MB")

    import torch.distributed as dist
    for p, g in zip(params, manual):
        dist.all_reduce(g)
        g /= world
        assert torch.allclose(p.grad, g, atol=1e-6), (p.grad - g).abs().max()
    return 0.0


def test_bucketed_dp_allreduce():
    run_distributed(_bucket_ar_worker, world_size=2)


def _gradnorm_worker(rank, world):
    """clip_grad_norm at tp2: sharded params contribute their shards,
    REPLICATED params count once (not tp times); the norm equals the dense
    single-rank value."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.grads import clip_grad_norm
    from neuronx_distributed_amd.parallel.utils import (
        set_tensor_model_parallel_attributes)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    full = torch.randn(8, 4)     # sharded rows over tp
    rep = torch.randn(5)         # replicated (e.g. a norm weight)

    p_shard = torch.nn.Parameter(full.chunk(world, 0)[rank].clone())
    set_tensor_model_parallel_attributes(p_shard, True, 0)
    p_rep = torch.nn.Parameter(rep.clone())
    p_shard.grad = p_shard.data.clone()
    p_rep.grad = p_rep.data.clone()

    total = clip_grad_norm([p_shard, p_rep], max_norm=1e9)
    ref = torch.sqrt(full.pow(2).sum() + rep.pow(2).sum())
    assert abs(float(total) - float(ref)) < 1e-4, (total, ref)
    return float(total)


def test_grad_norm_tp_duplicates():
    out = run_distributed(_gradnorm_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-6
