"""fp8 vs bf16 at the actual training fwd GEMM shapes, incl. the full
quantize+scaled_mm path used by NXDA_FP8_LINEAR."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t_ms(fn, n=10, w=3):
    for _ in range(w): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3

def main():
    from neuronx_distributed_amd.quantization.quantization_config import (
        QuantizationConfig, QuantizedDtype)
    from neuronx_distributed_amd.quantization.quantization_utils import (
        fp8_scaled_linear, quantize_symmetric)
    dev = "cuda"
    cfg = QuantizationConfig(quantized_dtype=QuantizedDtype.F8E4M3,
                             quantize_activation=True)
    for (M, N, K, tag) in [(32768, 6144, 4096, "qkv"),
                           (32768, 4096, 4096, "o"),
                           (32768, 22016, 4096, "gateup"),
                           (32768, 4096, 11008, "down")]:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        bf = t_ms(lambda: torch.nn.functional.linear(x, w))
        qw, ws = quantize_symmetric(w, cfg)
        f8 = t_ms(lambda: fp8_scaled_linear(x, qw, ws))
        def full():
            q2, s2 = quantize_symmetric(w, cfg)
            return fp8_scaled_linear(x, q2, s2)
        f8full = t_ms(full)
        fl = 2 * M * N * K
        print(f"{tag:7s} bf16 {bf:7.2f} ms ({fl/bf/1e9:5.0f} TF)  "
              f"fp8(pre-q) {f8:7.2f} ({fl/f8/1e9:5.0f} TF)  "
              f"fp8(full) {f8full:7.2f}")



def tensorwise():
    dev = "cuda"
    for (M, N, K, tag) in [(32768, 6144, 4096, "qkv"),
                           (32768, 4096, 4096, "o"),
                           (32768, 22016, 4096, "gateup"),
                           (32768, 4096, 11008, "down")]:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        bf = t_ms(lambda: torch.nn.functional.linear(x, w))
        def full_tw():
            ws = (w.abs().amax() / 448.0).clamp(min=1e-8).float()
            qw = (w / ws).clamp(-448, 448).to(torch.float8_e4m3fn)
            xs = (x.abs().amax() / 448.0).clamp(min=1e-8).float()
            qx = (x / xs).clamp(-448, 448).to(torch.float8_e4m3fn)
            return torch._scaled_mm(qx, qw.t(), scale_a=xs, scale_b=ws,
                                    out_dtype=torch.bfloat16)
        tw = t_ms(full_tw)
        fl = 2 * M * N * K
        print(f"{tag:7s} bf16 {bf:7.2f} ({fl/bf/1e9:5.0f} TF)  "
              f"fp8 tensorwise FULL {tw:7.2f} ({fl/tw/1e9:5.0f} TF)")

if __name__ == "__main__":
    main()
    tensorwise()
