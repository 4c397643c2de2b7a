"""ZeRO-1 numerics: sharded AdamW over 2 gloo ranks must match plain AdamW
on the same (DP-averaged) gradients."""

import torch
import torch.nn as nn

from dist_utils import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(16, 31), nn.Tanh(), nn.Linear(31, 5))


def _zero1_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer

    ps.initialize_model_parallel(tensor_model_parallel_size=1)

    model = _make_model()
    ref_model = _make_model()
    for p, q in zip(model.parameters(), ref_model.parameters()):
        assert torch.equal(p, q)

    opt = NeuronZero1Optimizer(model.parameters(), torch.optim.AdamW, lr=1e-2,
                               weight_decay=0.01, grad_clipping=True,
                               max_norm=1.0)
    ref_opt = torch.optim.AdamW(ref_model.parameters(), lr=1e-2,
                                weight_decay=0.01)

    for step in range(5):
        torch.manual_seed(100 + step * world + rank)
        x = torch.randn(8, 16)
        y = model(x).pow(2).mean()
        opt.zero_grad()
        y.backward()

        # reference: average grads over DP by hand, clip, step
        xs = []
        for r in range(world):
            torch.manual_seed(100 + step * world + r)
            xs.append(torch.randn(8, 16))
        ref_opt.zero_grad()
        ref_loss = sum(ref_model(xr).pow(2).mean() for xr in xs) / world
        ref_loss.backward()
        torch.nn.utils.clip_grad_norm_(ref_model.parameters(), 1.0)
        ref_opt.step()

        opt.step()

        for p, q in zip(model.parameters(), ref_model.parameters()):
            assert torch.allclose(p, q, atol=1e-5), \
                f"step {step}: {(p - q).abs().max()}"
    return True


def test_zero1_matches_adamw():
    run_distributed(_zero1_worker, world_size=2)


def test_adamw_fp32_optim_params():
    from neuronx_distributed_amd.optimizer import AdamW_FP32OptimParams

    torch.manual_seed(0)
    m1 = nn.Linear(8, 8)
    m2 = nn.Linear(8, 8)
    m2.load_state_dict(m1.state_dict())
    o1 = AdamW_FP32OptimParams(m1.parameters(), lr=1e-2)
    o2 = torch.optim.AdamW(m2.parameters(), lr=1e-2)
    for i in range(5):
        x = torch.randn(4, 8)
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            m(x).pow(2).sum().backward()
            o.step()
    for p, q in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p, q, atol=1e-5)


def _dcp_worker(rank, world):
    """DCP (torch.distributed.checkpoint) zero1 save/load roundtrip over
    DTensor shards."""
    import os
    import tempfile

    import torch
    import torch.distributed as dist
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer
    from neuronx_distributed_amd.optimizer.zero_dcp_utils import (
        load_zero1_optimizer_dcp, save_zero1_optimizer_dcp)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(64, 64),
                                torch.nn.Linear(64, 32))
    opt = NeuronZero1Optimizer(model.parameters(), torch.optim.AdamW,
                               lr=1e-2, grad_clipping=False)
    for _ in range(2):
        opt.zero_grad()
        loss = model(torch.randn(8, 64)).pow(2).mean()
        loss.backward()
        opt.step()

    path = os.path.join(tempfile.gettempdir(), "nxda_dcp_test")
    if rank == 0 and os.path.exists(path):
        import shutil
        shutil.rmtree(path)
    dist.barrier()
    save_zero1_optimizer_dcp(opt, path)

    before = [b.master.clone() for b in opt.buckets]
    for b in opt.buckets:
        b.master.data.zero_()
    load_zero1_optimizer_dcp(opt, path)
    for b, ref in zip(opt.buckets, before):
        assert torch.equal(b.master, ref)
    dist.barrier()
    if rank == 0:
        import shutil
        shutil.rmtree(path, ignore_errors=True)
    return 0.0


def test_zero1_dcp_roundtrip():
    run_distributed(_dcp_worker, world_size=2)


def _multi_group_worker(rank, world):
    """Two param groups (different lr / weight_decay) through zero1 match
    plain AdamW with the same groups."""
    import torch
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(32, 32),
                             torch.nn.LayerNorm(32))
    torch.manual_seed(0)
    m2 = torch.nn.Sequential(torch.nn.Linear(32, 32),
                             torch.nn.LayerNorm(32))

    def groups(m):
        decay = [p for n, p in m.named_parameters() if "0." in n]
        no_decay = [p for n, p in m.named_parameters() if "1." in n]
        return [{"params": decay, "weight_decay": 0.1, "lr": 1e-2},
                {"params": no_decay, "weight_decay": 0.0, "lr": 5e-3}]

    o1 = NeuronZero1Optimizer(groups(m1), torch.optim.AdamW,
                              grad_clipping=False)
    o2 = torch.optim.AdamW(groups(m2))
    for _ in range(3):
        x = torch.randn(8, 32)
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            m(x).pow(2).mean().backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()
    return 0.0


def test_zero1_param_groups():
    run_distributed(_multi_group_worker, world_size=2)


def _fp32_acc_worker(rank, world):
    """use_fp32_grad_acc: grads accumulate in fp32 over microbatches and
    the result matches plain AdamW with fp32-accumulated grads (tolerance
    far tighter than bf16 accumulation allows)."""
    import torch
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m1 = torch.nn.Linear(64, 64, dtype=torch.bfloat16)
    torch.manual_seed(0)
    m2 = torch.nn.Linear(64, 64, dtype=torch.bfloat16)

    o1 = NeuronZero1Optimizer(m1.parameters(), torch.optim.AdamW, lr=1e-2,
                              grad_clipping=False, use_fp32_grad_acc=True)
    o2 = torch.optim.AdamW([{"params": list(m2.parameters())}], lr=1e-2)

    torch.manual_seed(5)
    xs = [torch.randn(8, 64, dtype=torch.bfloat16) for _ in range(6)]
    o1.zero_grad()
    fp32_grads = [torch.zeros(p.shape) for p in m2.parameters()]
    for x in xs:
        m1(x).float().pow(2).mean().backward()
        m2(x).float().pow(2).mean().backward()
        for g, p in zip(fp32_grads, m2.parameters()):
            g.add_(p.grad.float())
            p.grad = None
    # zero1 side should now hold the fp32 sum in its flat buffer
    flat = o1.buckets[0].flat_grad
    assert flat.dtype == torch.float32
    ref_flat = torch.cat([g.reshape(-1) for g in fp32_grads])
    assert torch.allclose(flat[:ref_flat.numel()], ref_flat, atol=1e-5), \
        (flat[:ref_flat.numel()] - ref_flat).abs().max()
    return 0.0


def test_zero1_fp32_grad_acc():
    run_distributed(_fp32_acc_worker, world_size=1)


def _lr_propagation_worker(rank, world):
    """Scheduler-updated lr reaches BOTH the fused path (reads
    param_groups directly) and the base optimizer (mirrored groups):
    stepping with lr=0 must leave weights unchanged."""
    import torch
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m = torch.nn.Linear(16, 16)
    opt = NeuronZero1Optimizer(m.parameters(), torch.optim.AdamW, lr=1e-2,
                               grad_clipping=False)
    opt.param_groups[0]["lr"] = 0.0  # what a scheduler would do
    before = [p.detach().clone() for p in m.parameters()]
    opt.zero_grad()
    m(torch.randn(4, 16)).pow(2).mean().backward()
    opt.step()
    for p, b in zip(m.parameters(), before):
        assert torch.equal(p.detach(), b), (p - b).abs().max()
    return 0.0


def test_zero1_lr_propagation():
    run_distributed(_lr_propagation_worker, world_size=1)
