"""End-to-end tiny-Llama tests on CPU: TP=2 must reproduce TP=1 numerics
(same deterministic init), SP on/off must agree, and a training step must
run under ZeRO-1."""

import os

import pytest
import torch

from dist_utils import run_distributed


def _build_model(sp=False, seed=1234):
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    pl.model_parallel_manual_seed(seed)
    torch.manual_seed(seed)
    cfg = get_config("tiny", sequence_parallel_enabled=sp)
    return LlamaForCausalLM(cfg)


def _tp1_reference_loss(seed=1234):
    """Single-process tp=1 golden loss, run in-process via AOT-free path."""
    import neuronx_distributed_amd.parallel as pl
    from neuronx_distributed_amd.parallel import parallel_state as ps

    model = _build_model()
    torch.manual_seed(7)
    x = torch.randint(0, 256, (2, 16))
    loss = model(x, labels=x)
    return loss.item()


def _tp2_worker(rank, world, sp):
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    model = _build_model(sp=sp)
    torch.manual_seed(7)
    x = torch.randint(0, 256, (2, 16))
    loss = model(x, labels=x)
    loss.backward()
    return loss.item()


def _tp1_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    model = _build_model()
    torch.manual_seed(7)
    x = torch.randint(0, 256, (2, 16))
    return model(x, labels=x).item()


def test_tp2_matches_tp1():
    ref = run_distributed(_tp1_worker, world_size=1)[0]
    out = run_distributed(_tp2_worker, world_size=2, args=(False,))
    assert abs(out[0] - ref) < 2e-2, (out, ref)
    assert abs(out[0] - out[1]) < 1e-6


def test_sp_matches_dense():
    dense = run_distributed(_tp2_worker, world_size=2, args=(False,))
    sp = run_distributed(_tp2_worker, world_size=2, args=(True,))
    assert abs(dense[0] - sp[0]) < 2e-2, (dense, sp)


def _train_step_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.optimizer import NeuronZero1Optimizer

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    model = _build_model()
    opt = NeuronZero1Optimizer(model.parameters(), torch.optim.AdamW, lr=1e-3)
    losses = []
    for step in range(3):
        torch.manual_seed(step)
        x = torch.randint(0, 256, (2, 16))
        opt.zero_grad()
        loss = model(x, labels=x)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] + 1.0  # training proceeds, finite
    for l in losses:
        assert l == l  # no NaN
    return losses


def test_train_step_zero1():
    out = run_distributed(_train_step_worker, world_size=2)
    assert out[0] == pytest.approx(out[1], abs=1e-5)


def _sliding_window_worker(rank, world):
    """Sliding-window attention (mistral-style): the window mask matches a
    brute-force reference, and windowed KV-cache generation matches full
    re-forward greedy decoding."""
    import math as _math

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.kernels.flash_attn import _torch_reference
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.inference.generation import generate

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    B, H, S, D, W = 1, 2, 16, 8, 5
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    out = _torch_reference(q, k, v, causal=True, window=W)
    # brute force
    scale = 1.0 / _math.sqrt(D)
    sc = (q.float() @ k.float().transpose(-1, -2)) * scale
    for i in range(S):
        for j in range(S):
            if j > i or j <= i - W:
                sc[:, :, i, j] = float("-inf")
    ref = (torch.softmax(sc, -1) @ v.float()).to(q.dtype)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()

    cfg = get_config("tiny", sliding_window=6)
    torch.manual_seed(0)
    m = LlamaForCausalLM(cfg).eval()
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 10))
    gen = generate(m, x, max_new_tokens=5)
    seq = x
    for _ in range(5):
        logits = m(seq)
        seq = torch.cat([seq, logits[:, -1, :].argmax(-1, keepdim=True)], 1)
    assert torch.equal(gen, seq), (gen, seq)
    return 0.0


def test_sliding_window():
    run_distributed(_sliding_window_worker, world_size=1)


def _qwen_worker(rank, world):
    """Qwen2-style attention bias: trains at tp2, TP parity with tp1."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    m = LlamaForCausalLM(get_config("tiny", attention_bias=True))
    assert m.model.layers[0].self_attn.qkv_proj.bias_q is not None
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 16))
    loss = m(x, labels=x)
    loss.backward()
    assert m.model.layers[0].self_attn.qkv_proj.bias_q.grad is not None
    return float(loss.detach())


def test_qwen_attention_bias():
    tp1 = run_distributed(_qwen_worker, world_size=1)[0]
    tp2 = run_distributed(_qwen_worker, world_size=2)
    assert abs(tp2[0] - tp2[1]) < 1e-5
    assert abs(tp1 - tp2[0]) < 5e-3, (tp1, tp2)


def _chunked_loss_worker(rank, world):
    """chunked_lm_loss == plain lm_head+CE loss AND produces identical
    grads (recompute correctness) at tp2."""
    import copy

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.utils.chunked_loss import chunked_lm_loss

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    m = LlamaForCausalLM(get_config("tiny"))
    ref = copy.deepcopy(m)
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 17))
    labels = x.clone()
    labels[0, 5:8] = -100

    hidden = m.model(x)
    loss = chunked_lm_loss(hidden, m.lm_head, labels, num_chunks=3)
    loss.backward()

    rl = ref(x, labels=labels)
    rl.backward()
    assert abs(float(loss) - float(rl)) < 1e-5, (loss, rl)
    for p1, p2 in zip(m.parameters(), ref.parameters()):
        if p2.grad is None:
            continue
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5), \
            (p1.grad - p2.grad).abs().max()
    return float(loss.detach())


def test_chunked_lm_loss():
    out = run_distributed(_chunked_loss_worker, world_size=2)
    assert abs(out[0] - out[1]) < 1e-6


def _loss_chunks_cfg_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    m1 = LlamaForCausalLM(get_config("tiny"))
    torch.manual_seed(0)
    m2 = LlamaForCausalLM(get_config("tiny", loss_chunks=4))
    torch.manual_seed(1)
    x = torch.randint(0, 256, (2, 16))
    assert abs(float(m1(x, labels=x)) - float(m2(x, labels=x))) < 1e-5
    return 0.0


def test_loss_chunks_config():
    run_distributed(_loss_chunks_cfg_worker, world_size=1)


def _dp2_worker(rank, world):
    """Pure DP (tp=1, world 2): bucketed grad averaging over different
    per-rank batches equals the single-process run on the combined batch
    (loss-mean scaling matched)."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.grads import (
        bucket_allreduce_gradients)
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM

    ps.initialize_model_parallel(tensor_model_parallel_size=1)  # dp=2
    torch.manual_seed(0)
    m = LlamaForCausalLM(get_config("tiny"))
    torch.manual_seed(0)
    golden = LlamaForCausalLM(get_config("tiny"))

    torch.manual_seed(42)
    xfull = torch.randint(0, 256, (4, 16))
    x = xfull.chunk(world, 0)[rank]
    loss = m(x, labels=x)
    loss.backward()
    bucket_allreduce_gradients([p.grad for p in m.parameters()
                                if p.grad is not None])

    gl = golden(xfull, labels=xfull)
    gl.backward()
    for p, g in zip(m.parameters(), golden.parameters()):
        if g.grad is None:
            continue
        assert torch.allclose(p.grad, g.grad, atol=2e-4), \
            (p.grad - g.grad).abs().max()
    return float(loss.detach())


def test_dp2_grad_parity():
    run_distributed(_dp2_worker, world_size=2)


def _fused_norm_sp_worker(rank, world):
    """tp2 + SP: forced fused residual+RMSNorm training path (composed
    CPU fallback via NXDA_FUSED_NORM_FORCE=1) vs the plain path — loss
    and norm-weight/embedding grads must match, and the norm weights
    must keep their SP grad marking."""
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(7)
    x = torch.randint(0, 256, (2, 16))

    results = {}
    for force in ("0", "1"):
        os.environ["NXDA_FUSED_NORM_FORCE"] = force
        try:
            model = _build_model(sp=True)
            model.train()
            loss = model(x, labels=x)
            loss.backward()
            lyr = model.model.layers[0]
            assert lyr.input_layernorm.weight.sequence_parallel_enabled
            results[force] = (
                float(loss),
                lyr.input_layernorm.weight.grad.clone(),
                lyr.post_attention_layernorm.weight.grad.clone(),
                model.model.norm.weight.grad.clone(),
                model.model.embed_tokens.weight.grad.clone(),
            )
        finally:
            os.environ["NXDA_FUSED_NORM_FORCE"] = "0"
    l0, g0a, g0b, g0n, g0e = results["0"]
    l1, g1a, g1b, g1n, g1e = results["1"]
    assert abs(l0 - l1) < 1e-5 * (1 + abs(l0)), (l0, l1)
    assert torch.allclose(g0a, g1a, atol=1e-5), (g0a - g1a).abs().max()
    assert torch.allclose(g0b, g1b, atol=1e-5), (g0b - g1b).abs().max()
    assert torch.allclose(g0n, g1n, atol=1e-5), (g0n - g1n).abs().max()
    assert torch.allclose(g0e, g1e, atol=1e-4), (g0e - g1e).abs().max()
    return l1


def test_fused_norm_train_sp_matches_plain_tp2():
    run_distributed(_fused_norm_sp_worker, world_size=2)


def _fused_norm_ckpt_worker(rank, world):
    """Activation checkpointing wraps the decoder layers in
    CheckpointWrapper (no forward_fused_train): the fused-norm gate must
    fall back to the plain path instead of crashing, and training must
    still work."""
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.utils.activation_checkpoint import (
        apply_activation_checkpointing)

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    os.environ["NXDA_FUSED_NORM_FORCE"] = "1"
    try:
        model = _build_model()
        from neuronx_distributed_amd.models.llama import LlamaDecoderLayer

        apply_activation_checkpointing(
            model, check_fn=lambda m: isinstance(m, LlamaDecoderLayer))
        model.train()
        torch.manual_seed(7)
        x = torch.randint(0, 256, (2, 16))
        loss = model(x, labels=x)
        loss.backward()
        assert torch.isfinite(loss)
        assert model.model.layers[0].module.input_layernorm.weight.grad \
            is not None
    finally:
        os.environ["NXDA_FUSED_NORM_FORCE"] = "0"
    return float(loss)


def test_fused_norm_gate_respects_activation_checkpoint():
    run_distributed(_fused_norm_ckpt_worker, world_size=1)


def _fused_norm_hooks_worker(rank, world):
    """Module-level forward hooks on decoder layers (tensor capture
    style) must keep firing: the fused path bypasses __call__, so the
    gate falls back to the plain path when hooks are registered."""
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    os.environ["NXDA_FUSED_NORM_FORCE"] = "1"
    try:
        model = _build_model()
        fired = []
        for lyr in model.model.layers:
            lyr.register_forward_hook(
                lambda m, i, o: fired.append(type(m).__name__))
        model.train()
        torch.manual_seed(7)
        x = torch.randint(0, 256, (2, 16))
        loss = model(x, labels=x)
        loss.backward()
        assert len(fired) == len(model.model.layers), fired
        assert torch.isfinite(loss)
    finally:
        os.environ["NXDA_FUSED_NORM_FORCE"] = "0"
    return float(loss)


def test_fused_norm_gate_respects_layer_hooks():
    run_distributed(_fused_norm_hooks_worker, world_size=1)
