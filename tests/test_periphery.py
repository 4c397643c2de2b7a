"""Conv2d parallel layers, medusa buffers, safetensors dedup, MX quant."""

import torch

from dist_utils import run_distributed


def _conv_worker(rank, world):
    import torch.nn.functional as F
    from neuronx_distributed_amd.parallel import (
        OutputChannelParallelConv2d, InputChannelParallelConv2d,
        parallel_state as ps, comm)

    ps.initialize_model_parallel(tensor_model_parallel_size=world)
    torch.manual_seed(0)
    oc = OutputChannelParallelConv2d(4, 8, 3, padding=1, bias=False,
                                     gather_output=True, dtype=torch.float32)
    torch.manual_seed(1)
    x = torch.randn(2, 4, 8, 8)
    out = oc(x)
    wfull = comm.all_gather(oc.weight.detach(), dim=0,
                            group=ps.get_group_info("tp"))
    ref = F.conv2d(x, wfull, padding=1)
    assert torch.allclose(out, ref, atol=1e-5)

    torch.manual_seed(0)
    ic = InputChannelParallelConv2d(4, 8, 3, padding=1, bias=False,
                                    input_is_parallel=False,
                                    dtype=torch.float32)
    out2 = ic(x)
    wfull2 = comm.all_gather(ic.weight.detach(), dim=1,
                             group=ps.get_group_info("tp"))
    ref2 = F.conv2d(x, wfull2, padding=1)
    assert torch.allclose(out2, ref2, atol=1e-5)
    return True


def test_parallel_conv2d_tp2():
    run_distributed(_conv_worker, world_size=2)


def test_medusa_buffers():
    from neuronx_distributed_amd.utils.medusa_utils import (
        generate_medusa_buffers)

    buf = generate_medusa_buffers([[0], [1], [0, 0], [0, 1], [1, 0]])
    n = 6
    assert buf["medusa_attn_mask"].shape == (n, n)
    # every node attends to root and itself
    assert buf["medusa_attn_mask"][:, 0].all()
    assert buf["medusa_attn_mask"].diagonal().all()
    # node [0,0] attends to [0]
    assert buf["medusa_attn_mask"][3, 1]
    assert not buf["medusa_attn_mask"][3, 2]
    assert buf["medusa_position_ids"].tolist() == [0, 1, 1, 2, 2, 2]
    # leaves: [0,0],[0,1],[1,0] -> 3 retrieve paths
    assert buf["retrieve_indices"].shape[0] == 3


def test_safetensors_dedup(tmp_path):
    from neuronx_distributed_amd.utils.safetensors_utils import (
        save_safetensors, load_safetensors)

    w = torch.randn(4, 4)
    sd = {"a.weight": w, "tied.weight": w, "b": torch.ones(2)}
    p = str(tmp_path / "x.safetensors")
    save_safetensors(sd, p)
    out = load_safetensors(p)
    assert torch.equal(out["a.weight"], w)
    assert torch.equal(out["tied.weight"], w)
    assert set(out) == set(sd)


def test_mx_quantization():
    from neuronx_distributed_amd.quantization.microscaling import (
        quantize_mx, dequantize_mx)

    torch.manual_seed(0)
    w = torch.randn(64, 128)
    # e4m3: 3 mantissa bits -> ~6% per-element; e2m1 grid -> ~17%
    for fmt, tol in (("fp8_e4m3", 0.08), ("fp4_e2m1", 0.25)):
        q, s = quantize_mx(w, fmt)
        wd = dequantize_mx(q, s, dtype=torch.float32)
        rel = (wd - w).abs().max() / w.abs().max()
        assert rel < tol, (fmt, rel.item())


def test_quantization_observer():
    import torch.nn as nn
    from neuronx_distributed_amd.quantization.observer import (
        attach_observers, collect_activation_scales)

    m = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 4))
    obs = attach_observers(m, nn.Linear)
    assert len(obs) == 2
    x = torch.randn(32, 8) * 3
    m(x)
    scales = collect_activation_scales(obs, qmax=448.0)
    assert set(scales) == set(obs)
    assert abs(scales["0"].item() - x.abs().max().item() / 448.0) < 1e-5


def test_tensor_replacement():
    import torch.nn as nn
    from neuronx_distributed_amd.utils.tensor_replacement import TensorReplacer

    m = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 4))
    x = torch.randn(2, 4)
    base = m(x)
    fixed = torch.ones(2, 4)
    with TensorReplacer(m).replace("0", fixed):
        out = m(x)
    assert torch.allclose(out, m[1](fixed))
    # scaling callable + restoration after exit
    with TensorReplacer(m).replace("0", lambda o: o * 0):
        out0 = m(x)
    assert torch.allclose(out0, m[1](torch.zeros(2, 4)))
    assert torch.allclose(m(x), base)


def test_pad_batch_to_multiple():
    from neuronx_distributed_amd.utils.batch_utils import (
        pad_batch_to_multiple)

    b = {"input_ids": torch.ones(2, 10, dtype=torch.long),
         "labels": torch.ones(2, 10, dtype=torch.long),
         "meta": 3}
    out, pad = pad_batch_to_multiple(b, 8)
    assert pad == 6
    assert out["input_ids"].shape == (2, 16)
    assert (out["input_ids"][:, 10:] == 0).all()
    assert (out["labels"][:, 10:] == -100).all()
    assert out["meta"] == 3
    out2, pad2 = pad_batch_to_multiple(b, 5)
    assert pad2 == 0 and out2["input_ids"].shape == (2, 10)


def test_memory_stats_cpu():
    from neuronx_distributed_amd.utils.memory import (log_memory_stats,
                                                      memory_stats)

    s = memory_stats()
    assert set(s) >= {"allocated_gib", "reserved_gib", "peak_allocated_gib"}
    log_memory_stats("test")


def test_yaml_converter(tmp_path):
    """reference scripts/yaml_converter.py parity: YAML training config ->
    converter JSON (geometry keys + MoE expert count)."""
    import json

    from neuronx_distributed_amd.scripts.yaml_converter import \
        convert_yaml_to_json

    y = tmp_path / "train.yaml"
    y.write_text(
        "model:\n  num_layers: 12\n  num_attention_heads: 16\n"
        "  hidden_size: 1024\n  num_kv_heads: 4\n"
        "  moe:\n    num_experts: 8\n")
    out = convert_yaml_to_json(str(y), str(tmp_path / "cfg.json"))
    cfg = json.load(open(out))
    assert cfg == {"num_hidden_layers": 12, "num_attention_heads": 16,
                   "hidden_size": 1024, "num_key_value_heads": 4,
                   "num_local_experts": 8}


def test_flash_stride_layout_helpers():
    """CPU coverage of the strided-flash layout plumbing: transpose views
    are accepted without copies, outputs are allocated in the producer's
    layout, and the ctypes stride array matches tensor.stride()."""
    import torch

    from neuronx_distributed_amd.ops import (_fa_alloc_like, _fa_ok_layout,
                                             _fa_strides)

    B, H, S, D = 2, 4, 16, 128
    bhsd = torch.zeros(B, H, S, D)
    bshd_view = torch.zeros(B, S, H, D).permute(0, 2, 1, 3)
    assert _fa_ok_layout(bhsd) and _fa_ok_layout(bshd_view)
    assert not _fa_ok_layout(torch.zeros(B, H, S, 2 * D)[..., ::2])

    # layout preservation: BSHD view in -> BSHD storage out (the model's
    # downstream .transpose(1,2).reshape stays a view)
    o1 = _fa_alloc_like(bshd_view)
    assert o1.shape == (B, H, S, D)
    assert o1.transpose(1, 2).is_contiguous()
    o2 = _fa_alloc_like(bhsd)
    assert o2.is_contiguous()

    arr = _fa_strides(bshd_view, bhsd)
    assert list(arr) == list(bshd_view.stride()[:3]) + \
        list(bhsd.stride()[:3])


def test_add_rmsnorm_train_cpu_fallback():
    """Composed CPU path of the fused training add+RMSNorm autograd op
    (the GPU kernel path is covered by test_ops_gpu)."""
    import torch

    from neuronx_distributed_amd import ops

    torch.manual_seed(3)
    d = torch.randn(8, 64, requires_grad=True)
    r = torch.randn(8, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    h, n = ops.add_rmsnorm_train(d, r, w, 1e-5)
    (n.pow(2).sum() + 0.5 * h.sum()).backward()

    d2 = d.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    h2 = d2 + r2
    n2 = h2 * torch.rsqrt(h2.pow(2).mean(-1, keepdim=True) + 1e-5) * w2
    (n2.pow(2).sum() + 0.5 * h2.sum()).backward()
    assert torch.allclose(h, h2, atol=1e-6)
    assert torch.allclose(n, n2, atol=1e-5)
    assert torch.allclose(d.grad, d2.grad, atol=1e-4)
    assert torch.allclose(r.grad, r2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-3)


def test_parallel_layer_norm_sp_tagging():
    """parallel/layer_norm.py: numerics equal nn.LayerNorm and the
    weight/bias carry the sequence_parallel_enabled attribute grads.py's
    SP all-reduce looks for (reference parallel_layers/layer_norm.py)."""
    import torch

    from neuronx_distributed_amd.parallel.layer_norm import LayerNorm

    ln = LayerNorm(32, sequence_parallel_enabled=True)
    assert ln.weight.sequence_parallel_enabled
    assert ln.bias.sequence_parallel_enabled
    ref = torch.nn.LayerNorm(32)
    with torch.no_grad():
        ref.weight.copy_(ln.weight)
        ref.bias.copy_(ln.bias)
    x = torch.randn(4, 6, 32)
    assert torch.allclose(ln(x), ref(x), atol=1e-6)

    ln2 = LayerNorm(16)
    assert not ln2.weight.sequence_parallel_enabled


def test_tensor_utils_cumsum():
    """utils/tensor_utils.py cumsum: fp64-accumulated dim-0 scan with the
    reference's signature (tril_size accepted, 2-D/dim-0 contract)."""
    import pytest
    import torch

    from neuronx_distributed_amd.utils.tensor_utils import cumsum

    x = torch.randn(64, 8, dtype=torch.float32)
    out = cumsum(x, tril_size=4)
    ref = torch.cumsum(x.double(), dim=0).float()
    assert torch.allclose(out, ref, atol=1e-6)
    assert out.dtype == x.dtype
    with pytest.raises(ValueError):
        cumsum(torch.randn(3))
    with pytest.raises(NotImplementedError):
        cumsum(x, dim=1)


def test_llama3_rope_scaling_matches_transformers():
    """ops.precompute_rope_freqs(rope_scaling=...) vs the HF transformers
    llama3 rope init — an INDEPENDENT implementation of the same
    published scheme (reference attention/utils.py apply_scaling)."""
    import pytest
    import torch

    transformers = pytest.importorskip("transformers")
    from transformers import LlamaConfig
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

    from neuronx_distributed_amd import ops

    scaling = {"rope_type": "llama3", "factor": 8.0,
               "low_freq_factor": 1.0, "high_freq_factor": 4.0,
               "original_max_position_embeddings": 64}
    dim, theta, S = 128, 500000.0, 256
    cfg = LlamaConfig(hidden_size=dim, num_attention_heads=1,
                      rope_theta=theta, max_position_embeddings=S,
                      rope_scaling=dict(scaling))
    hf_inv, att = ROPE_INIT_FUNCTIONS["llama3"](cfg, device="cpu")
    t = torch.arange(S, dtype=torch.float32)
    hf_freqs = torch.outer(t, hf_inv.float() * att)

    cos, sin = ops.precompute_rope_freqs(S, dim, theta,
                                         rope_scaling=scaling)
    assert torch.allclose(cos, hf_freqs.cos(), atol=1e-5)
    assert torch.allclose(sin, hf_freqs.sin(), atol=1e-5)

    # linear scaling + unknown type
    c2, _ = ops.precompute_rope_freqs(
        16, 8, rope_scaling={"rope_type": "linear", "factor": 2.0})
    c_ref, _ = ops.precompute_rope_freqs(16, 8)
    assert torch.allclose(c2[2], c_ref[1], atol=1e-6)  # half-speed angles
    with pytest.raises(NotImplementedError):
        ops.precompute_rope_freqs(8, 8, rope_scaling={"rope_type": "yarn"})


def test_llama31_config_builds():
    from neuronx_distributed_amd.models import get_config

    cfg = get_config("llama3.1-8b")
    assert cfg.rope_scaling["factor"] == 8.0
    assert cfg.max_position_embeddings == 131072


def test_apply_rotary_polar_compatible():
    """Interleaved (Meta 'polar') rotary vs an independent construction:
    de-interleave -> neox rotate-half (_rope_torch) -> re-interleave."""
    import torch

    from neuronx_distributed_amd.ops import (_rope_torch,
                                             apply_rotary_polar_compatible,
                                             precompute_rope_freqs)

    torch.manual_seed(5)
    B, S, H, D = 2, 16, 3, 32
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    cos, sin = precompute_rope_freqs(S, D)
    freqs = torch.atan2(sin, cos)  # recover angles

    qo, ko = apply_rotary_polar_compatible(q, k, freqs)

    # independent golden: permute even dims first, neox-rotate, invert
    perm = torch.cat([torch.arange(0, D, 2), torch.arange(1, D, 2)])
    inv_perm = torch.argsort(perm)
    q_ref = _rope_torch(q[..., perm], cos, sin)[..., inv_perm]
    k_ref = _rope_torch(k[..., perm], cos, sin)[..., inv_perm]
    assert torch.allclose(qo, q_ref, atol=1e-5), (qo - q_ref).abs().max()
    assert torch.allclose(ko, k_ref, atol=1e-5)

    import pytest
    with pytest.raises(ValueError):
        apply_rotary_polar_compatible(q, k, freqs.double())


def test_flash_window_covering_seq_is_causal():
    """window >= kv len masks nothing beyond causal: flash_attn_func
    must short-circuit to the plain causal path (CPU reference here;
    keeps Mistral seq<=window training off the O(S^2) composed path)."""
    import torch

    from neuronx_distributed_amd.kernels.flash_attn import flash_attn_func

    torch.manual_seed(9)
    q = torch.randn(1, 2, 32, 128)
    k = torch.randn(1, 2, 32, 128)
    v = torch.randn(1, 2, 32, 128)
    w_eq = flash_attn_func(q, k, v, causal=True, window=32)
    w_big = flash_attn_func(q, k, v, causal=True, window=1000)
    plain = flash_attn_func(q, k, v, causal=True)
    assert torch.allclose(w_eq, plain, atol=1e-6)
    assert torch.allclose(w_big, plain, atol=1e-6)
    # a genuinely smaller window must differ
    w_small = flash_attn_func(q, k, v, causal=True, window=4)
    assert not torch.allclose(w_small, plain, atol=1e-3)


def test_banded_window_attention_matches_reference():
    """O(S*window) banded sliding-window attention vs the full composed
    reference: identical forward and gradients (per-block checkpointing
    path included)."""
    import torch

    from neuronx_distributed_amd.kernels.flash_attn import (
        _banded_window_attn, _torch_reference)

    torch.manual_seed(11)
    B, Hq, Hkv, S, D, W = 1, 4, 2, 192, 32, 64
    q = torch.randn(B, Hq, S, D, requires_grad=True)
    k = torch.randn(B, Hkv, S, D, requires_grad=True)
    v = torch.randn(B, Hkv, S, D, requires_grad=True)

    out = _banded_window_attn(q, k, v, None, W, block=48)
    dy = torch.randn_like(out)
    out.backward(dy)
    g = (q.grad.clone(), k.grad.clone(), v.grad.clone())
    for t in (q, k, v):
        t.grad = None

    ref = _torch_reference(q, k, v, causal=True, window=W)
    ref.backward(dy)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    for a, b in zip(g, (q.grad, k.grad, v.grad)):
        assert torch.allclose(a, b, atol=1e-4), (a - b).abs().max()


def test_add_rmsnorm_train_unused_normed():
    """Dropping the normed output must not crash the backward (dnormed
    is None): the op degrades to a pass-through residual add."""
    import torch

    from neuronx_distributed_amd import ops

    d = torch.randn(4, 32, requires_grad=True)
    r = torch.randn(4, 32, requires_grad=True)
    w = torch.randn(32, requires_grad=True)
    h, _ = ops.add_rmsnorm_train(d, r, w, 1e-5)
    h.sum().backward()
    assert torch.allclose(d.grad, torch.ones_like(d))
    assert torch.allclose(r.grad, torch.ones_like(r))
    assert w.grad is None or torch.all(w.grad == 0)
