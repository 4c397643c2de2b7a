"""Round-2 pipeline machinery: cross-stage pass-through wires (skip
connections), non-tensor stage IO, output deallocation + custom_backward
(reference pipeline/model.py:1163-1215, partition.py:132-223,
scheduler.py:281-293)."""

import torch
import torch.nn as nn

from dist_utils import run_distributed


class _SkipBlock(nn.Module):
    def __init__(self, h):
        super().__init__()
        self.lin = nn.Linear(h, h)

    def forward(self, x):
        return torch.relu(self.lin(x))


class SkipNet(nn.Module):
    """b1's INPUT skips over b2 into the final sum — when cut at b2 and b3
    the value produced at stage 0 is consumed at stage 2, exercising the
    pass-through wire (forwarded through stage 1's P2P messages, with the
    pass-through gradient route on the way back)."""

    def __init__(self, h=16):
        super().__init__()
        self.b1 = _SkipBlock(h)
        self.b2 = _SkipBlock(h)
        self.b3 = _SkipBlock(h)

    def forward(self, x):
        h1 = self.b1(x)
        h2 = self.b2(h1)
        h3 = self.b3(h2 + 0.5 * h1)  # h1 skips stage 1 -> consumed stage 2
        return h3.pow(2).mean()


def _skip_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.pipeline import NxDPPModel

    ps.initialize_model_parallel(tensor_model_parallel_size=1,
                                 pipeline_model_parallel_size=world)
    torch.manual_seed(0)
    model = SkipNet()
    torch.manual_seed(0)
    golden = SkipNet()

    pp = NxDPPModel(model, transformer_layer_cls=_SkipBlock,
                    num_microbatches=2, input_names=["x"])
    torch.manual_seed(5)
    x = torch.randn(4, 16)
    loss = pp.run_train(x=x)

    gl = (golden(x[:2]) + golden(x[2:])) / 2
    gl.backward()
    assert abs(float(loss) - float(gl)) < 1e-6, (float(loss), float(gl))

    g_named = dict(golden.named_parameters())
    n_checked = 0
    for n, p in pp.local_named_parameters():
        # split_module flattens module paths with "_": b1.lin.weight ->
        # b1_lin.weight; normalize both to dots for matching
        norm = n.replace("_", ".")
        matches = [gp for gn, gp in g_named.items()
                   if norm.endswith(gn.replace("_", "."))]
        assert len(matches) == 1, (n, norm)
        gp = matches[0]
        assert p.grad is not None, n
        assert torch.allclose(p.grad, gp.grad, atol=1e-6), \
            (n, (p.grad - gp.grad).abs().max())
        n_checked += 1
    assert n_checked == 2  # weight + bias of this rank's stage
    return float(loss)


def test_pp3_skip_connection_pass_through():
    res = run_distributed(_skip_worker, world_size=3)
    assert abs(res[0] - res[1]) < 1e-6  # loss broadcast to all ranks


class _ObjStage0(nn.Module):
    def __init__(self):
        super().__init__()
        self.lin = nn.Linear(8, 8)

    def forward(self, x):
        # returns a tensor AND python metadata crossing the boundary
        return self.lin(x), {"scale": 2.0, "tag": "stage0"}


class _ObjStage1(nn.Module):
    def forward(self, h, meta):
        assert meta["tag"] == "stage0"
        return (h * meta["scale"]).pow(2).mean()


def _obj_io_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.pipeline import NxDPPModel
    from neuronx_distributed_amd.pipeline.manual_pipe_stage import \
        PipelineStageModule

    ps.initialize_model_parallel(tensor_model_parallel_size=1,
                                 pipeline_model_parallel_size=world)
    torch.manual_seed(0)
    stages = [_ObjStage0(), _ObjStage1()]
    stage = PipelineStageModule(stages, world, ps.get_pipeline_model_parallel_rank())
    pp = NxDPPModel(stage, num_microbatches=2, input_names=["x"])
    torch.manual_seed(3)
    x = torch.randn(4, 8)
    loss = pp.run_train(x=x)

    torch.manual_seed(0)
    g0, g1 = _ObjStage0(), _ObjStage1()
    gl = (g1(*g0(x[:2])) + g1(*g0(x[2:]))) / 2
    gl.backward()
    assert abs(float(loss) - float(gl)) < 1e-6
    if rank == 0:
        mine = dict(pp.local_named_parameters())
        for n, gp in g0.named_parameters():
            ours = next(v for k, v in mine.items() if k.endswith(n))
            assert torch.allclose(ours.grad, gp.grad, atol=1e-6), n
    return float(loss)


def test_manual_pp_non_tensor_stage_io():
    run_distributed(_obj_io_worker, world_size=2)


def test_custom_backward_with_deallocated_output():
    """custom_backward must run the graph even after output.data was
    replaced by a 1-element stub (torch.autograd.backward would reject the
    shape mismatch)."""
    from neuronx_distributed_amd.pipeline.model import NxDPPModel

    x = torch.randn(4, 8, requires_grad=True)
    w = torch.randn(8, 8, requires_grad=True)
    out = (x @ w).relu()
    g = torch.randn_like(out)

    ref_x = x.detach().clone().requires_grad_(True)
    ref_w = w.detach().clone().requires_grad_(True)
    ref = (ref_x @ ref_w).relu()
    ref.backward(g)

    out.data = torch.empty(1, dtype=out.dtype)  # deallocate
    NxDPPModel._custom_backward([out], [g])
    assert torch.allclose(x.grad, ref_x.grad, atol=1e-6)
    assert torch.allclose(w.grad, ref_w.grad, atol=1e-6)


def _dealloc_worker(rank, world):
    """With deallocation ON (default), sent stage outputs shrink to stub
    size after their send drains; loss/grads still match the flag-off
    run (engine uses custom_backward)."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.pipeline import NxDPPModel

    ps.initialize_model_parallel(tensor_model_parallel_size=1,
                                 pipeline_model_parallel_size=world)
    results = []
    for dealloc in (True, False):
        torch.manual_seed(0)
        model = SkipNet()
        pp = NxDPPModel(model, transformer_layer_cls=_SkipBlock,
                        num_microbatches=2, input_names=["x"],
                        deallocate_pipeline_outputs=dealloc)
        torch.manual_seed(5)
        x = torch.randn(4, 16)
        loss = pp.run_train(x=x)
        grads = sorted((n, p.grad.clone())
                       for n, p in pp.local_named_parameters())
        results.append((float(loss), grads))
    (l1, g1), (l2, g2) = results
    assert abs(l1 - l2) < 1e-7
    for (n1, a), (n2, b) in zip(g1, g2):
        assert n1 == n2 and torch.allclose(a, b, atol=1e-7), n1
    return l1


def test_deallocate_outputs_matches_undeallocated():
    run_distributed(_dealloc_worker, world_size=2)
