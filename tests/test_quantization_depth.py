"""Round-2 quantization depth: packed *_X4 storage dtypes and the
HF-quantized-checkpoint state-dict adaptor (reference
quantization_config.py:100-125, quantization_layers.py:356)."""

import pytest
import torch

from neuronx_distributed_amd.quantization.quantization_config import (
    QuantizationConfig, QuantizationType, QuantizedDtype)
from neuronx_distributed_amd.quantization.quantization_utils import (
    _fp4_decode, _fp4_encode, dequantize, pack_x4, quantize_symmetric,
    unpack_x4)


def test_fp8_x4_pack_roundtrip():
    torch.manual_seed(0)
    w = torch.randn(8, 16)
    q = (w * 10).clamp(-448, 448).to(torch.float8_e4m3fn)
    packed = pack_x4(q, QuantizedDtype.F8E4M3FN_X4)
    assert packed.dtype == torch.uint32 and packed.shape == (8, 4)
    back = unpack_x4(packed, QuantizedDtype.F8E4M3FN_X4)
    assert back.dtype == torch.float8_e4m3fn
    assert torch.equal(back.view(torch.uint8), q.view(torch.uint8))


def test_fp4_grid_roundtrip():
    vals = torch.tensor([0.0, 0.5, -1.0, 1.5, -2.0, 3.0, 4.0, -6.0])
    codes = _fp4_encode(vals)
    assert torch.allclose(_fp4_decode(codes), vals)
    # nearest-value rounding
    assert float(_fp4_decode(_fp4_encode(torch.tensor([2.4])))) == 2.0
    assert float(_fp4_decode(_fp4_encode(torch.tensor([-5.9])))) == -6.0


@pytest.mark.parametrize("dt", [QuantizedDtype.F8E4M3FN_X4,
                                QuantizedDtype.F8E5M2_X4,
                                QuantizedDtype.F4E2M1FN_X4])
def test_packed_quantize_dequantize(dt):
    torch.manual_seed(1)
    w = torch.randn(16, 32)
    cfg = QuantizationConfig(quantized_dtype=dt)
    q, s = quantize_symmetric(w, cfg)
    assert q.dtype == dt.torch_dtype
    assert q.shape == (16, 32 // 4)
    back = dequantize(q, s, torch.float32, quantized_dtype=dt)
    assert back.shape == w.shape
    rel = (back - w).norm() / w.norm()
    limit = 0.30 if dt == QuantizedDtype.F4E2M1FN_X4 else 0.06
    assert rel < limit, rel


def test_packed_quantized_layer_forward():
    from dist_utils import run_distributed

    run_distributed(_packed_layer_worker, world_size=1)


def _packed_layer_worker(rank, world):
    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.parallel.layers import ColumnParallelLinear
    from neuronx_distributed_amd.quantization.quantization_layers import \
        QuantizedColumnParallel

    ps.initialize_model_parallel(tensor_model_parallel_size=1)
    torch.manual_seed(0)
    fl = ColumnParallelLinear(64, 32, bias=False, gather_output=False,
                              dtype=torch.float32)
    cfg = QuantizationConfig(quantized_dtype=QuantizedDtype.F8E4M3FN_X4)
    ql = QuantizedColumnParallel.from_float(fl, cfg)
    assert ql.weight.dtype == torch.uint32
    x = torch.randn(8, 64)
    with torch.no_grad():
        out = ql(x)
        ref = fl(x)
    rel = (out - ref).norm() / ref.norm()
    assert rel < 0.06, rel
    return 0.0


def test_state_dict_adaptor_plain_and_packed():
    from neuronx_distributed_amd.quantization.quantization_layers import \
        QuantizedParallelLinearLayerStateDictAdaptor as A

    torch.manual_seed(0)
    w = torch.randn(4, 8)
    # plain entry
    sd = {"lin.weight": w, "lin.weight_scale": torch.tensor([0.1])}
    assert torch.equal(A.get_weight_from_state_dict("lin.", sd), w)
    assert float(A.get_scale_from_state_dict("lin.", sd)) == \
        pytest.approx(0.1)
    A.set_weight_to_state_dict("lin.", w * 2, sd)
    assert torch.equal(sd["lin.weight"], w * 2)
    assert A.get_bias_from_state_dict("lin.", sd) is None

    # torch.ao packed_params entry (HF dynamic-int8 export shape)
    qw = torch.quantize_per_tensor(w, scale=0.05, zero_point=0,
                                   dtype=torch.qint8)
    bias = torch.randn(4)
    sd2 = {"lin._packed_params.dtype": torch.qint8,
           "lin._packed_params._packed_params": (qw, bias)}
    got = A.get_weight_from_state_dict("lin.", sd2)
    assert got.dtype == torch.int8
    assert torch.equal(got, torch.int_repr(qw))
    s = A.get_scale_from_state_dict("lin.", sd2)
    assert float(s) == pytest.approx(0.05)
    assert torch.equal(A.get_bias_from_state_dict("lin.", sd2), bias)
    A.set_weight_to_state_dict("lin.", torch.int_repr(qw)[:2], sd2)
    assert sd2["lin._packed_params._packed_params"][0].shape[0] == 2
