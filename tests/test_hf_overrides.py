"""HF `transformers` integration (overrides/): the HIP flash kernel as a
registered HF attention implementation, and HF<->native llama state-dict
conversion — an HF llama checkpoint must load into the native model and
reproduce the HF model's logits (reference overrides/transformer_overrides.py
capability, realized via AttentionInterface instead of monkey-patching)."""

import pytest
import torch

from dist_utils import run_distributed

transformers = pytest.importorskip("transformers")


def _tiny_hf_llama():
    from transformers import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, vocab_size=256,
                      max_position_embeddings=128, rms_norm_eps=1e-5,
                      rope_theta=10000.0, attention_bias=False,
                      tie_word_embeddings=False)
    torch.manual_seed(0)
    return LlamaForCausalLM(cfg)


def test_hf_attention_impl_matches_sdpa():
    """HF llama running attn_implementation="nxda_fused" (CPU: composed
    fp32 reference path) must match the stock sdpa implementation."""
    from neuronx_distributed_amd.overrides import register_flash_attention

    impl = register_flash_attention()
    m = _tiny_hf_llama().eval()
    x = torch.randint(0, 256, (2, 16))
    with torch.no_grad():
        ref = m(x).logits
        m.set_attn_implementation(impl)
        out = m(x).logits
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()


def _hf_to_native_worker(rank, world):
    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.overrides import (
        convert_hf_llama_state_dict, convert_to_hf_llama_state_dict)
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    hf = _tiny_hf_llama().eval()
    native_sd = convert_hf_llama_state_dict(hf.state_dict())

    cfg = get_config("tiny")  # same shape as _tiny_hf_llama
    m = LlamaForCausalLM(cfg)
    missing, unexpected = m.load_state_dict(native_sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing  # only rope buffers

    m.eval()
    x = torch.randint(0, 256, (2, 16))
    with torch.no_grad():
        ref = hf(x).logits
        out = m(x)
    assert torch.allclose(out, ref, atol=2e-4), (out - ref).abs().max()

    # round-trip back to HF names and reload into a fresh HF model
    back = convert_to_hf_llama_state_dict(
        {k: v for k, v in m.state_dict().items() if "rope_" not in k})
    hf2 = _tiny_hf_llama()
    missing2, unexpected2 = hf2.load_state_dict(back, strict=False)
    assert not unexpected2, unexpected2
    with torch.no_grad():
        out2 = hf2.eval()(x).logits
    assert torch.allclose(out2, ref, atol=1e-5)
    return float(out.sum())


def test_hf_checkpoint_to_native_logits_parity():
    run_distributed(_hf_to_native_worker, world_size=1)
