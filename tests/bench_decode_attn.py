"""Microbench decode_attn at llama3-8b decode shapes."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from neuronx_distributed_amd import ops

def main():
    B, Hq, Hkv, D = 32, 32, 8, 128
    pos = 1024
    Smax = pos + 64
    dev = "cuda"
    torch.manual_seed(0)
    cos, sin = ops.precompute_rope_freqs(Smax, D, device=dev)
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    q2 = torch.randn(B, Hq * D, dtype=torch.bfloat16, device=dev)
    k2 = torch.randn(B, Hkv * D, dtype=torch.bfloat16, device=dev)
    v2 = torch.randn(B, Hkv * D, dtype=torch.bfloat16, device=dev)
    pos_t = torch.tensor([pos], dtype=torch.int64, device=dev)
    f = lambda: ops.decode_attn_step(q2, k2, v2, kc, vc, cos, sin, pos_t,
                                     Hq, Hkv, 1.0 / 11.3137)
    for _ in range(20): f()
    torch.cuda.synchronize()
    n = 200
    t0 = time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / n * 1e6
    bytes_read = B * Hkv * pos * D * 2 * 2
    print(f"decode_attn: {us:.1f} us  ({bytes_read/us/1e3:.2f} TB/s effective)")

if __name__ == "__main__":
    main()
