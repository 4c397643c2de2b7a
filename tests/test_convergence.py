"""Short-convergence tests on NON-repeated data (reference
test/integration/llama2_7B convergence-run approach): the full NxD stack's
loss curve must track a plain fp32 torch golden step for step — catches
optimizer/kernel/grad-sync drift that per-op numerics tests cannot see."""

import pytest
import torch

from dist_utils import run_distributed

STEPS = 12
LR = 1e-3


def _make_batches(vocab, batch, seqlen, steps):
    """Fresh (non-repeated) batches with LEARNABLE structure: arithmetic
    token sequences x[t+1] = (x[t] + stride) % vocab with random starts —
    pure-random tokens would leave nothing to converge on."""
    torch.manual_seed(1234)
    out = []
    for _ in range(steps):
        start = torch.randint(0, vocab, (batch, 1))
        stride = torch.randint(1, 5, (batch, 1))
        pos = torch.arange(seqlen).unsqueeze(0)
        out.append((start + stride * pos) % vocab)
    return out


def _conv_dp2_worker(rank, world):
    """dp=2 + ZeRO-1 over fresh data each step == single-process fp32 AdamW
    on the combined batch: losses must match to float tolerance."""
    import copy

    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    cfg = get_config("tiny")
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg)
    golden = copy.deepcopy(model)
    opt = NeuronZero1Optimizer(model.parameters(), torch.optim.AdamW,
                               grad_clipping=True, max_norm=1.0, lr=LR,
                               use_fused_kernel=False)
    gopt = torch.optim.AdamW(golden.parameters(), lr=LR)

    batches = _make_batches(cfg.vocab_size, 4, 16, STEPS)
    my_losses, golden_losses = [], []
    for x in batches:
        mine = x.chunk(world)[rank]
        loss = model(mine, labels=mine)
        loss.backward()
        opt.step()
        opt.zero_grad()

        halves = [golden(h, labels=h) for h in x.chunk(world)]
        gl = torch.stack(halves).mean()
        gl.backward()
        torch.nn.utils.clip_grad_norm_(golden.parameters(), 1.0)
        gopt.step()
        gopt.zero_grad()
        # dp-mean of the local losses for comparison
        lm = loss.detach().clone()
        torch.distributed.all_reduce(lm)
        my_losses.append(float(lm) / world)
        golden_losses.append(float(gl))

    for i, (a, b) in enumerate(zip(my_losses, golden_losses)):
        assert abs(a - b) < 5e-3 + 0.002 * abs(b), (i, a, b)
    assert my_losses[-1] < my_losses[0] - 0.1, my_losses  # actually learned
    return my_losses[-1]


def test_convergence_dp2_zero1_matches_fp32_golden():
    run_distributed(_conv_dp2_worker, world_size=2)


@pytest.mark.gpu
def test_convergence_gpu_bf16_tracks_fp32():
    """1-GPU bf16 training with EVERY HIP kernel in the hot path (flash,
    RMSNorm, RoPE, SwiGLU, fused CE, fused AdamW) vs the same architecture
    in fp32 composed torch ops: the loss curves must track within bf16
    noise on fresh data each step (no memorization)."""
    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer
    from neuronx_distributed_amd.parallel import parallel_state as ps
    import torch.distributed as dist
    import os

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29761")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("nccl", rank=0, world_size=1)
    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)

    cfg = get_config("tiny", num_hidden_layers=2, hidden_size=256,
                     intermediate_size=512, num_attention_heads=2,
                     num_key_value_heads=2, head_dim_override=128,
                     max_position_embeddings=512)
    steps = 25

    def build(dtype):
        prev = torch.get_default_dtype()
        torch.set_default_dtype(dtype)
        torch.manual_seed(0)
        with torch.device("cuda"):
            m = LlamaForCausalLM(cfg)
        torch.set_default_dtype(prev)
        return m

    bf = build(torch.bfloat16)
    fp = build(torch.float32)
    # identical init across dtypes
    with torch.no_grad():
        for p, q in zip(bf.parameters(), fp.parameters()):
            q.data.copy_(p.data.float())

    opt_bf = NeuronZero1Optimizer(bf.parameters(), torch.optim.AdamW,
                                  grad_clipping=True, max_norm=1.0, lr=LR,
                                  use_fused_kernel=True)
    opt_fp = torch.optim.AdamW(fp.parameters(), lr=LR)

    batches = _make_batches(cfg.vocab_size, 8, 256, steps)
    curve_bf, curve_fp = [], []
    for x in batches:
        x = x.cuda()
        lb = bf(x, labels=x)
        lb.backward()
        opt_bf.step()
        opt_bf.zero_grad()
        lf = fp(x, labels=x)
        lf.backward()
        torch.nn.utils.clip_grad_norm_(fp.parameters(), 1.0)
        opt_fp.step()
        opt_fp.zero_grad()
        curve_bf.append(float(lb))
        curve_fp.append(float(lf))

    # both curves must descend, and track each other within bf16 noise
    assert curve_fp[-1] < curve_fp[0] - 0.3, curve_fp
    assert curve_bf[-1] < curve_bf[0] - 0.3, curve_bf
    for i, (a, b) in enumerate(zip(curve_bf, curve_fp)):
        assert abs(a - b) < 0.05 + 0.05 * abs(b), (i, a, b, curve_bf,
                                                   curve_fp)


@pytest.mark.gpu
def test_convergence_gpu_fp8_forward_tracks_bf16():
    """NXDA_FP8_LINEAR=1 (selective fp8 forward GEMMs on the wide
    gate_up-class shapes) must track the bf16 loss curve."""
    import os

    import torch.distributed as dist

    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer
    from neuronx_distributed_amd.parallel import parallel_state as ps

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29762")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        dist.init_process_group("nccl", rank=0, world_size=1)
    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)

    cfg = get_config("tiny", num_hidden_layers=2, hidden_size=256,
                     intermediate_size=512, num_attention_heads=2,
                     num_key_value_heads=2, head_dim_override=128,
                     max_position_embeddings=512)
    steps = 15

    def run(fp8: bool):
        os.environ["NXDA_FP8_LINEAR"] = "1" if fp8 else "0"
        prev = torch.get_default_dtype()
        torch.set_default_dtype(torch.bfloat16)
        torch.manual_seed(0)
        with torch.device("cuda"):
            m = LlamaForCausalLM(cfg)
        torch.set_default_dtype(prev)
        opt = NeuronZero1Optimizer(m.parameters(), torch.optim.AdamW,
                                   grad_clipping=True, max_norm=1.0, lr=LR)
        curve = []
        for x in _make_batches(cfg.vocab_size, 8, 256, steps):
            x = x.cuda()
            loss = m(x, labels=x)
            loss.backward()
            opt.step()
            opt.zero_grad()
            curve.append(float(loss))
        return curve

    try:
        c8 = run(True)
        cb = run(False)
    finally:
        os.environ["NXDA_FP8_LINEAR"] = "0"
    # MEASURED trade-off (this test documents it): on this tiny 256-hidden
    # model the fp8 forward costs real learning speed (tail loss ~20-25%
    # above bf16 at step 15) — quantization error is proportionally huge
    # at small width.  At 7B the bench loss matches bf16 to 3 decimals
    # (profiles/README.md fp8 notes).  The guard here catches CATASTROPHIC
    # breakage: NaNs, no-learning, or runaway divergence.
    assert all(map(lambda v: v == v, c8)), c8  # no NaNs
    assert cb[-1] < cb[0] - 1.0, cb
    assert c8[-1] < c8[0] - 1.0, c8  # fp8 still learns strongly
    tail8 = sum(c8[-3:]) / 3
    tailb = sum(cb[-3:]) / 3
    assert tail8 < tailb * 1.5, (tail8, tailb)
