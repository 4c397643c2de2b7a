"""Property-based tests (hypothesis) for serialization-critical paths:
packed-quantization round-trips, Karmarkar-Karp bin balance, and the
xser flatten/unflatten structure preservation."""

import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings


@settings(max_examples=40, deadline=None)
@given(st.integers(1, 7), st.integers(4, 64))
def test_pack_unpack_x4_roundtrip_fp8(rows, cols_quads):
    from neuronx_distributed_amd.quantization.quantization_utils import (
        QuantizedDtype, pack_x4, unpack_x4)

    cols = cols_quads  # packed width; unpacked = 4x
    torch.manual_seed(rows * 100 + cols)
    bits = torch.randint(0, 256, (rows, cols * 4), dtype=torch.uint8)
    for dt in (QuantizedDtype.F8E4M3FN_X4, QuantizedDtype.F8E5M2_X4):
        q = bits.view(dt.unpacked.torch_dtype)
        packed = pack_x4(q, dt)
        assert packed.shape == (rows, cols)
        out = unpack_x4(packed, dt)
        assert out.dtype == dt.unpacked.torch_dtype
        assert out.view(torch.uint8).equal(bits)  # exact bit round-trip


@settings(max_examples=30, deadline=None)
@given(st.lists(st.integers(1, 5000), min_size=1, max_size=40),
       st.integers(1, 8))
def test_kk_bins_cover_and_balance(sizes, nbins):
    from neuronx_distributed_amd.trainer.checkpoint import (
        assign_tensors_to_bins)

    tensors = [torch.empty(s, dtype=torch.int8) for s in sizes]
    bins = assign_tensors_to_bins(tensors, nbins)
    assert len(bins) == nbins
    seen = sorted(i for b in bins for i in b)
    assert seen == list(range(len(tensors)))  # exact cover, no dupes
    if len(sizes) >= nbins:
        loads = sorted(sum(sizes[i] for i in b) for b in bins)
        # greedy/KK guarantee: max bin <= mean + max element
        assert loads[-1] <= sum(sizes) / nbins + max(sizes) + 1


@settings(max_examples=25, deadline=None)
@given(st.recursive(
    st.one_of(st.integers(-5, 5), st.text(max_size=4),
              st.booleans(), st.none(),
              st.integers(1, 4).map(lambda n: torch.arange(n).float())),
    lambda children: st.one_of(
        st.lists(children, max_size=3),
        st.dictionaries(st.text(max_size=3), children, max_size=3)),
    max_leaves=12))
def test_xser_flatten_unflatten_roundtrip(obj):
    from neuronx_distributed_amd.trainer.checkpoint import (
        _xser_flatten, _xser_unflatten)

    tensors = []
    skeleton = _xser_flatten(obj, tensors)
    restored = _xser_unflatten(skeleton, lambda i: tensors[i])

    def eq(a, b):
        if isinstance(a, torch.Tensor):
            return isinstance(b, torch.Tensor) and torch.equal(a, b)
        if isinstance(a, dict):
            return (isinstance(b, dict) and a.keys() == b.keys()
                    and all(eq(a[k], b[k]) for k in a))
        if isinstance(a, list):
            return (isinstance(b, list) and len(a) == len(b)
                    and all(eq(x, y) for x, y in zip(a, b)))
        return a == b

    assert eq(obj, restored)


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 4), st.integers(1, 4), st.integers(1, 6))
def test_hf_state_dict_converters_roundtrip(layers, heads, i_mult):
    """HF -> native -> HF llama state-dict conversion is lossless for
    arbitrary layer counts / widths (fused gate_up split included)."""
    from neuronx_distributed_amd.overrides import (
        convert_hf_llama_state_dict, convert_to_hf_llama_state_dict)

    H, inter = heads * 8, i_mult * 8
    hf = {"model.embed_tokens.weight": torch.randn(16, H),
          "model.norm.weight": torch.randn(H),
          "lm_head.weight": torch.randn(16, H)}
    for n in range(layers):
        p = f"model.layers.{n}."
        hf[p + "self_attn.q_proj.weight"] = torch.randn(H, H)
        hf[p + "self_attn.k_proj.weight"] = torch.randn(H // 2, H)
        hf[p + "self_attn.v_proj.weight"] = torch.randn(H // 2, H)
        hf[p + "self_attn.o_proj.weight"] = torch.randn(H, H)
        hf[p + "mlp.gate_proj.weight"] = torch.randn(inter, H)
        hf[p + "mlp.up_proj.weight"] = torch.randn(inter, H)
        hf[p + "mlp.down_proj.weight"] = torch.randn(H, inter)
        hf[p + "input_layernorm.weight"] = torch.randn(H)
        hf[p + "post_attention_layernorm.weight"] = torch.randn(H)

    native = convert_hf_llama_state_dict(hf)
    assert not any(k.endswith(("gate_proj.weight", "up_proj.weight"))
                   and not k.endswith("gate_up_proj.weight")
                   for k in native)
    back = convert_to_hf_llama_state_dict(native)
    assert back.keys() == hf.keys()
    for k in hf:
        assert torch.equal(back[k], hf[k]), k


@settings(max_examples=25, deadline=None)
@given(st.recursive(
    st.one_of(st.integers(-9, 9), st.text(max_size=3), st.none(),
              st.tuples(st.integers(1, 3), st.integers(1, 3)).map(
                  lambda s: torch.randn(*s))),
    lambda ch: st.one_of(st.lists(ch, max_size=3),
                         st.tuples(ch, ch),
                         st.dictionaries(st.text(max_size=2), ch,
                                         max_size=3)),
    max_leaves=10))
def test_pipeline_serialization_roundtrip(obj):
    """SerializationManager (the PP wire format): arbitrary nested
    structures split into (skeleton, metas, tensors) and rebuild exactly
    — tuples stay tuples, tensor identity by index."""
    from neuronx_distributed_amd.utils.serialization import (
        SerializationManager)

    m = SerializationManager()
    skel, metas, tensors = m.serialize(obj)
    rebuilt = m.deserialize(skel, list(tensors))

    def eq(a, b):
        if isinstance(a, torch.Tensor):
            return isinstance(b, torch.Tensor) and torch.equal(a, b)
        if type(a) is not type(b):
            return False
        if isinstance(a, dict):
            return a.keys() == b.keys() and all(eq(a[k], b[k]) for k in a)
        if isinstance(a, (list, tuple)):
            return len(a) == len(b) and all(eq(x, y) for x, y in zip(a, b))
        return a == b

    assert eq(obj, rebuilt)
    assert len(metas) == len(tensors)
    for mt, t in zip(metas, tensors):
        assert mt.shape == tuple(t.shape) and mt.dtype == t.dtype


@settings(max_examples=40, deadline=None)
@given(st.sampled_from([1, 2, 4, 8]), st.sampled_from([1, 2]),
       st.integers(1, 4), st.integers(0, 1))
def test_create_local_weight_shards_reassemble(tp, stride, blocks, dim):
    """Stride-aware TP sharding (the fused gate-up/QKV layout): slicing
    every rank's shard and reassembling by stride sub-blocks recovers
    the full weight exactly, for any tp/stride/width/partition dim."""
    from neuronx_distributed_amd.parallel.utils import create_local_weight

    full_dim = tp * stride * blocks  # smallest legal multiple times blocks
    shape = [full_dim, 3] if dim == 0 else [3, full_dim]
    torch.manual_seed(full_dim + dim)
    full = torch.randn(*shape)
    per = full_dim // tp
    shards = [create_local_weight(full, dim, per, stride, rank=r,
                                  world_size=tp) for r in range(tp)]
    assert all(s.shape[dim] == per for s in shards)
    # reassemble: for each stride sub-block b, ranks contribute their
    # b-th sub-slice in rank order
    sub = per // stride
    parts = []
    for b in range(stride):
        for r in range(tp):
            parts.append(shards[r].narrow(dim, b * sub, sub))
    assert torch.equal(torch.cat(parts, dim=dim), full)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 3), st.integers(1, 16))
def test_lse_merge_associative(B, S):
    """The (O, lse) online-softmax merge (ring attention / split-KV
    decode) must be associative and order-independent — the property the
    ring schedule relies on when blocks arrive in rotation order."""
    from neuronx_distributed_amd.kernels.ring_attn import _merge

    torch.manual_seed(B * 100 + S)
    parts = []
    for _ in range(3):
        o = torch.randn(B, 2, S, 8)
        l = torch.randn(B, 2, S) * 3
        parts.append((o, l))
    (o1, l1), (o2, l2), (o3, l3) = parts
    a_o, a_l = _merge(*_merge(o1, l1, o2, l2), o3, l3)
    b_o, b_l = _merge(o1, l1, *_merge(o2, l2, o3, l3))
    c_o, c_l = _merge(o3, l3, *_merge(o2, l2, o1, l1))
    assert torch.allclose(a_l, b_l, atol=1e-5)
    assert torch.allclose(a_o, b_o, atol=1e-5)
    assert torch.allclose(a_l, c_l, atol=1e-5)
    assert torch.allclose(a_o, c_o, atol=1e-5)


@settings(max_examples=50, deadline=None)
@given(st.integers(1, 6), st.integers(1, 8), st.integers(1, 40))
def test_ring_window_block_classification_exact(cp, C, W):
    """_block_kind covers the sliding-window attention EXACTLY: skip
    blocks contain no in-window (q, k) pair, full blocks are entirely
    in-window, and the union of non-skip blocks covers every in-window
    causal pair — for arbitrary cp / block size / window."""
    from neuronx_distributed_amd.kernels.ring_attn import _block_kind

    S = cp * C
    for i in range(cp):
        for blk in range(cp):
            kind = _block_kind(i, blk, C, W, causal=True)
            pairs = [(q, k) for q in range(i * C, (i + 1) * C)
                     for k in range(blk * C, (blk + 1) * C)]
            in_win = [(q, k) for q, k in pairs if k <= q and k > q - W]
            if kind == "skip":
                assert not in_win, (i, blk, C, W)
            elif kind == "full":
                causal_pairs = [(q, k) for q, k in pairs if k <= q]
                assert in_win == causal_pairs and in_win, (i, blk, C, W)
            elif kind == "diag":
                assert blk == i
            else:
                assert kind == "partial"
                assert in_win, (i, blk, C, W)  # partial is never empty
